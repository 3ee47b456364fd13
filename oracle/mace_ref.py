"""CPU oracle for the MACE energy+force forward (full graph, no
partitions).

Straight-line restatement of the computation the reference orchestrates
in implementations/mace/models.py:45-220 (ScaleShiftMACE_Dist.dist_forward
collapsed to one partition) — per-edge radial embedding and spherical
harmonics, two RealAgnosticResidualInteractionBlock-style interactions
with uvu tensor products and per-edge radial weights, correlation-3
symmetric-contraction product blocks, per-layer readouts, scale/shift,
e0 atomic energies, forces from autograd on the interaction energy
(mace get_outputs; forces = -dE/dpos).

Independence from the product (distmlip_amd/mace_ops.py):
  * the symmetric contraction is evaluated through the DENSE
    symmetric-basis tensors U = so3.symmetric_basis (the product uses the
    tree factorization — a different algorithm whose equality is itself
    pinned in tests);
  * the tensor-product messages are single dense einsums per path;
  * radial embedding / envelope / linear applications are restated here
    (the envelope polynomial via oracle/basis_ref.envelope_coeffs — the
    solved, not hard-coded, coefficients).
The shared so3 coupling constants are pinned against sympy/scipy in
tests/test_so3.py.

TEST INFRASTRUCTURE ONLY — see oracle/__init__.py.
"""
from __future__ import annotations

import math

import numpy as np
import torch

from distmlip_amd import so3
from distmlip_amd.mace_model import MACECore
from oracle.basis_ref import envelope_coeffs


def _bessel_env(r: torch.Tensor, r_max: float, n: int, p: int):
    freqs = torch.arange(1, n + 1, dtype=r.dtype) * math.pi
    d = r.unsqueeze(-1)
    bes = math.sqrt(2.0 / r_max) * torch.sin(freqs * d / r_max) / d
    a, b, c = envelope_coeffs(p)
    u = r / r_max
    env = 1.0 + a * u ** p + b * u ** (p + 1) + c * u ** (p + 2)
    env = torch.where(r <= r_max, env, torch.zeros_like(r))
    return bes * env.unsqueeze(-1)


def _lin(x, w):
    """per-l o3.Linear application, 1/sqrt(C) normalized."""
    return torch.einsum("ncd,mc->nmd", x, w) / math.sqrt(w.shape[1])


def _dense_U(nu: int, lo: int, dtype):
    return torch.tensor(so3.symmetric_basis(nu, lo), dtype=dtype)


def mace_oracle_forward(core: MACECore, structure, src, dst, offsets,
                        dtype: torch.dtype = torch.float64,
                        compute_forces: bool = True,
                        compute_stress: bool = False):
    cfg = core.config
    core = core.to(dtype)
    C = cfg.channels

    lat0 = torch.tensor(np.asarray(structure.lattice), dtype=dtype)
    strain = torch.zeros(3, 3, dtype=dtype)
    if compute_stress:
        strain.requires_grad_(True)
    lattice = lat0 @ (torch.eye(3, dtype=dtype) + strain)
    frac = torch.tensor(np.asarray(structure.frac_coords), dtype=dtype)
    pos = frac @ lattice
    if compute_forces:
        if not pos.requires_grad:
            pos.requires_grad_(True)
        pos.retain_grad()

    species = torch.tensor(np.asarray(structure.species), dtype=torch.long)
    src_t = torch.tensor(np.asarray(src), dtype=torch.long)
    dst_t = torch.tensor(np.asarray(dst), dtype=torch.long)
    off_t = torch.tensor(np.asarray(offsets), dtype=dtype)

    # mace get_edge_vectors_and_lengths: sender = edge_index[0],
    # receiver = edge_index[1]; vectors = pos[receiver] - pos[sender]
    # + shifts (mace_utils.py:74-78 builds shifts = unit_shifts @ cell)
    vectors = pos[dst_t] + off_t @ lattice - pos[src_t]
    lengths = torch.linalg.norm(vectors, dim=1)

    Y = so3.real_sh(vectors)                                  # [E, 16]
    edge_feats = _bessel_env(lengths, cfg.r_max, cfg.num_bessel,
                             cfg.cutoff_p)

    N = len(species)
    x = {0: core.node_embedding[species].unsqueeze(-1)}       # [N, C, 1]
    e0 = core.atomic_energies[species].sum()

    node_es_layers = []
    for i, (inter, prod) in enumerate(zip(core.interactions, core.products)):
        # linear_up
        x_up = {l: _lin(x[l], inter.linear_up.w[str(l)]) for l in inter.in_ls}
        # per-edge path weights
        tp_w = edge_feats
        for k, lin in enumerate(inter.radial.layers):
            tp_w = tp_w @ lin.weight.t()
            if k < len(inter.radial.layers) - 1:
                tp_w = tp_w * torch.sigmoid(tp_w)             # silu
        tp_w = tp_w.view(-1, len(inter.paths), C)
        # uvu TP messages + scatter to receivers, one dense einsum/path
        msg = {l3: torch.zeros(N, C, 2 * l3 + 1, dtype=dtype)
               for l3 in inter.target_ls}
        for p, (l1, l2, l3) in enumerate(inter.paths):
            CG = torch.tensor(so3.real_cg(l1, l2, l3), dtype=dtype)
            o2, d2 = so3.L_OFF[l2], so3.L_DIMS[l2]
            m = torch.einsum("ecd,ef,dfg,ec->ecg",
                             x_up[l1][src_t], Y[:, o2:o2 + d2], CG,
                             tp_w[:, p, :])
            msg[l3] = msg[l3].index_add(0, dst_t, m)
        m_lin = {l: _lin(msg[l], inter.linear_post.w[str(l)])
                 / cfg.avg_num_neighbors for l in inter.target_ls}
        # sc skip (per-element channel mix on shared irreps)
        sc = {}
        for l in inter.skip_ls:
            W = inter.skip[str(l)][species]                   # [N, C, C]
            sc[l] = torch.einsum("ncd,nmc->nmd", x[l], W) / math.sqrt(C)
        # product basis: dense symmetric-basis contraction
        y = {}
        for lo in prod.out_ls:
            acc = torch.zeros(N, C, 2 * lo + 1, dtype=dtype)
            xs = torch.cat([m_lin[l] for l in sorted(m_lin)], dim=-1)  # [N,C,16]
            for nu in range(1, cfg.correlation + 1):
                key = f"{lo}_{nu}"
                if key not in prod.weights:
                    continue
                U = _dense_U(nu, lo, dtype)          # [16]*nu + [do, P]
                w = prod.weights[key][species]       # [N, P, C]
                # explicit nu-step dense contraction (shapes differ per nu)
                if nu == 1:
                    phi = torch.einsum("ncI,IoP->ncoP", xs, U)
                elif nu == 2:
                    phi = torch.einsum("ncI,ncJ,IJoP->ncoP", xs, xs, U)
                else:
                    t1 = torch.einsum("IJKoP,ncK->ncIJoP", U, xs)
                    t2 = torch.einsum("ncIJoP,ncJ->ncIoP", t1, xs)
                    phi = torch.einsum("ncIoP,ncI->ncoP", t2, xs)
                acc = acc + torch.einsum("ncoP,nPc->nco", phi, w)
            y[lo] = acc
        y = {l: _lin(y[l], prod.linear.w[str(l)]) for l in prod.out_ls}
        for l in y:
            if l in sc:
                y[l] = y[l] + sc[l]
        x = y
        # readout
        if i < cfg.num_interactions - 1:
            es = torch.einsum("nc,c->n", x[0][:, :, 0], core.readout_linear[i])
        else:
            h = x[0][:, :, 0] @ core.readout_mlp1.t()
            h = h * torch.sigmoid(h)                          # silu
            es = h @ core.readout_mlp2
        node_es_layers.append(es)

    node_inter_es = core.scale * sum(node_es_layers) + core.shift
    inter_e = node_inter_es.sum()
    total_e = e0 + inter_e

    out = {"energy": total_e, "interaction_energy": inter_e.detach(),
           "node_inter_es": node_inter_es.detach()}
    if compute_forces:
        grads = [pos, strain] if compute_stress else [pos]
        gv = torch.autograd.grad(inter_e, grads)
        out["forces"] = -gv[0]
        if compute_stress:
            volume = abs(np.linalg.det(np.asarray(structure.lattice)))
            out["stress"] = -gv[1] / volume * -160.21766208
    return out
