"""CPU oracle for the UMA/eSCN E+F forward (full graph, no partitions).

Straight-line restatement of the computation the reference orchestrates
in implementations/uma/escn_md.py:249-523 collapsed to one partition:
per-edge z-aligned Wigner rotations, edge scalars, edge-degree
embedding, num_layers [norm -> rotated SO(2) messages -> S2 activation
-> scatter -> FFN] blocks, final norm, scalar energy head, forces via
autograd.

Independence from the product (distmlip_amd/uma_ops.py): the grid
projection uses torch.linalg.pinv of the sampled harmonics (equal to the
product's quadrature-weight form only because the Gauss-Legendre x
uniform grid integrates the band-limited products exactly); the SO(2)
mixing, norms and head are restated with separate einsums.  The shared
so3 rotation machinery is pinned independently in tests/test_so3.py.

TEST INFRASTRUCTURE ONLY — see oracle/__init__.py.
"""
from __future__ import annotations

import math

import numpy as np
import torch

from distmlip_amd import so3
from distmlip_amd.uma_model import (UMACore, gaussian_basis, m_indices,
                                    s2_grid)


def _pinv_grids(cfg, dtype):
    """Independent grid construction: own GL x uniform points and a
    WEIGHTED pseudo-inverse (sqrt(w)-scaled lstsq).  The weighting is
    essential: an unweighted pinv projects in the discrete sample inner
    product, which is NOT rotation-invariant (GL points cluster at the
    poles) — found as a 7%-level equivariance break during bring-up."""
    nt, nph = cfg.grid_theta, cfg.grid_phi
    ct, wt = np.polynomial.legendre.leggauss(nt)
    phi = np.arange(nph) * (2 * np.pi / nph)
    ctg, phig = np.meshgrid(ct, phi, indexing="ij")
    st = np.sqrt(1 - ctg ** 2)
    pts = np.stack([st * np.cos(phig), st * np.sin(phig), ctg],
                   axis=-1).reshape(-1, 3)
    w = np.repeat(wt, nph) * (2 * np.pi / nph) / (4 * np.pi)
    Y = so3.real_sh(torch.tensor(pts), normalize=False)[
        :, :cfg.S].to(dtype)
    sw = torch.tensor(np.sqrt(w)).to(dtype).unsqueeze(1)
    from_g = torch.linalg.pinv(Y * sw) * sw.t()
    return Y, from_g


def _norm(x, scale, lmax, eps=1e-6):
    outs = []
    for l in range(lmax + 1):
        o, d = l * l, 2 * l + 1
        blk = x[:, o:o + d, :]
        rms = (blk.pow(2).mean((1, 2), keepdim=True) + eps).sqrt()
        outs.append(blk / rms * scale[l])
    return torch.cat(outs, 1)


def uma_oracle_forward(core: UMACore, structure, src, dst, offsets,
                       dtype: torch.dtype = torch.float64,
                       compute_forces: bool = True,
                       compute_stress: bool = False):
    cfg = core.config
    core = core.to(dtype)
    C, S, lmax = cfg.sphere_channels, cfg.S, cfg.lmax

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

    vectors = pos[dst_t] + off_t @ lattice - pos[src_t]
    lengths = torch.linalg.norm(vectors, dim=1)
    N, E = len(species), len(src_t)

    R = so3.edge_align_rotation(vectors)
    D = so3.wigner_D_batch(R, lmax)
    Dinv = D.transpose(-1, -2)

    g = gaussian_basis(lengths, cfg.cutoff, cfg.num_gauss)
    xe = torch.cat([g, core.source_embedding[species[src_t]],
                    core.target_embedding[species[dst_t]]], 1)
    x_edge = torch.nn.functional.silu(xe @ core.edge_proj.t())

    x = torch.zeros(N, S, C, dtype=dtype)
    x[:, 0, :] = core.sphere_embedding[species]

    m0, pm = m_indices(lmax)
    to_g, from_g = _pinv_grids(cfg, dtype)

    def s2act(t):
        f = torch.nn.functional.silu(torch.einsum("gs,nsc->ngc", to_g, t))
        return torch.einsum("sg,ngc->nsc", from_g, f)

    # edge-degree embedding
    w = (x_edge @ core.edge_degree.t()).view(E, lmax + 1, C)
    med = torch.zeros(E, S, C, dtype=dtype)
    med[:, m0, :] = w
    med = torch.einsum("est,etc->esc", Dinv, med)
    x = x + torch.zeros_like(x).index_add_(0, dst_t, med) / cfg.avg_degree

    for blk in core.blocks:
        h = _norm(x, blk.norm1, lmax)
        xt = torch.cat([torch.einsum("est,etc->esc", D, h[src_t]),
                        torch.einsum("est,etc->esc", D, h[dst_t])], 2)
        ge = blk.edge_mlp(x_edge)
        gate = torch.sigmoid(ge[:, :lmax + 1])
        mt = torch.zeros(E, S, C, dtype=dtype)
        x0 = xt[:, m0, :].reshape(E, -1)
        mt[:, m0, :] = (x0 @ blk.msg.w0.t()).view(E, lmax + 1, C) \
            * gate[:, 0].view(E, 1, 1)
        for m in range(1, lmax + 1):
            plus, minus = pm[m - 1]
            xp = xt[:, plus, :].reshape(E, -1)
            xm = xt[:, minus, :].reshape(E, -1)
            wr, wi = blk.msg.wr[m - 1], blk.msg.wi[m - 1]
            gm = gate[:, m].view(E, 1, 1)
            mt[:, plus, :] = (xp @ wr.t() - xm @ wi.t()).view(
                E, len(plus), C) * gm
            mt[:, minus, :] = (xp @ wi.t() + xm @ wr.t()).view(
                E, len(plus), C) * gm
        msg = torch.einsum("est,etc->esc", Dinv, mt)
        x = x + torch.zeros_like(x).index_add_(0, dst_t, msg) \
            / cfg.avg_degree
        h2 = _norm(x, blk.norm2, lmax)
        h2 = torch.einsum("nsc,dc->nsd", h2, blk.ffn1)
        h2 = s2act(h2)
        x = x + torch.einsum("nsc,dc->nsd", h2, blk.ffn2)

    h = _norm(x, core.final_norm, lmax)
    s = torch.nn.functional.silu(h[:, 0, :] @ core.head1.t())
    es = core.scale * (s @ core.head2) + core.shift
    total_e = es.sum()

    out = {"energy": total_e, "node_energies": es.detach()}
    if compute_forces:
        grads = [pos, strain] if compute_stress else [pos]
        gv = torch.autograd.grad(total_e, grads)
        out["forces"] = -gv[0]
        if compute_stress:
            volume = abs(np.linalg.det(np.asarray(structure.lattice)))
            out["stress"] = -gv[1] / volume * -160.21766208
    return out
