"""CHGNet_Dist — the distributed CHGNet forward (the drop-in model class).

API mirror of the reference adapter
implementations/matgl/models/chgnet.py (CHGNet_Dist): `from_existing`
(chgnet.py:551-560), `enable_distributed_mode` (chgnet.py:455-549),
`potential_forward_dist` (chgnet.py:21-206) returning
(node_types, positions, strain, (total_E, site_props)), and
`dist_forward` (chgnet.py:208-453) — same layer order, feature routing and
halo points, re-implemented over this build's ops backend instead of
DGL/matgl message passing.

Single-process mode (this file): a list of per-partition devices exactly
like the reference (`enable_distributed_mode(gpus)`, "cpu" entries
allowed, chgnet.py:465-469); halo = differentiable slice copies
(dist.py:356 semantics).  SPMD one-process-per-GPU mode lives in
distmlip_amd/runtime.py and reuses the same per-partition compute via
`partition_forward` below.

Arithmetic: the gated-MLP first layers are algebraically split
(cat(v_s,v_d,e) @ W == v_s@Ws + v_d@Wd + e@We) so the hot path is
{per-node GEMM -> fused gather-add -> per-edge GEMM -> gated combine ->
segmented scatter-add}; identical math to the oracle restatement up to
fp summation order.
"""
from __future__ import annotations

import os
from copy import deepcopy
from typing import List, Optional

import numpy as np
import torch
from torch import nn

import distmlip_amd
from distmlip_amd.model import (
    CHGNetConfig,
    CHGNetCore,
    GatedMLP,
    bond_expansion_from_dist,
    compute_theta,
    fourier_expansion,
)
from distmlip_amd.ops_base import default_ops_factory


def _split_first(lin: nn.Linear, parts: List[int]):
    """Split a Linear over a concatenated input into weight blocks."""
    ws = []
    off = 0
    for p in parts:
        ws.append(lin.weight[:, off:off + p])
        off += p
    assert off == lin.weight.shape[1]
    return ws, lin.bias


def _packed_weights(mlp: GatedMLP):
    """Per-MLP packed weights, cached on the module when params are frozen
    (they are in SpmdEngine/CHGNet_Dist inference) so the cat/stack/
    transpose launches run once, not every step:

      wcg [2h, in], bcg [2h] — [core1 ; gate1] stacked first layer (one
      GEMM feeds both branches; the input is read once);
      w2 [2, d, d], b2 [2, 1, d] — core2|gate2 as a batched GEMM over the
      packed hidden halves;
      fusedT [d, 2h] — the per-edge/per-line input block's weight
      (columns 2d:3d of wcg) transposed for the fused edge-MLP kernel.
    """
    frozen = not any(p.requires_grad for p in mlp.parameters())
    # _version tracks in-place mutation (load_state_dict copy_ bumps it):
    # a frozen model whose weights are reloaded must not reuse stale packs
    key = (mlp.core1.weight.device, mlp.core1.weight.dtype,
           tuple(p._version for p in mlp.parameters()))
    if frozen:
        cached = getattr(mlp, "_dm_packed", None)
        if cached is not None and cached[0] == key:
            return cached[1]
    wcg = torch.cat([mlp.core1.weight, mlp.gate1.weight], dim=0)
    bcg = torch.cat([mlp.core1.bias, mlp.gate1.bias], dim=0)
    w2 = torch.stack([mlp.core2.weight.t(), mlp.gate2.weight.t()])
    b2 = torch.stack([mlp.core2.bias.unsqueeze(0), mlp.gate2.bias.unsqueeze(0)])
    d = mlp.core2.weight.shape[0]
    fusedT = wcg[:, 2 * d:3 * d].t().contiguous()
    packed = (wcg, bcg, w2.contiguous(), b2, fusedT)
    if frozen:
        mlp._dm_packed = (key, packed)
    return packed


class _SecondLayer(torch.autograd.Function):
    """cg [2,*,d] = baddbmm over the packed hidden halves of h [*,2d].
    Hand-written backward: two GEMMs writing STRIDED into one row-major
    dh (ldc = 2d), so the [*,2d] re-pack copy autograd's view+transpose
    backward would insert never happens.  w2/b2 are frozen packs
    (_packed_weights) — no weight grads."""

    @staticmethod
    def forward(ctx, h, w2, b2):
        d = w2.shape[1]
        hb = h.view(-1, 2, d).transpose(0, 1)
        ctx.save_for_backward(w2)
        return torch.baddbmm(b2, hb, w2)

    @staticmethod
    def backward(ctx, dcg):
        (w2,) = ctx.saved_tensors
        d = w2.shape[1]
        dcg = dcg.contiguous()
        dh = torch.empty(dcg.shape[1], 2 * d, dtype=dcg.dtype,
                         device=dcg.device)
        torch.mm(dcg[0], w2[0].t(), out=dh[:, :d])
        torch.mm(dcg[1], w2[1].t(), out=dh[:, d:])
        return dh, None, None


def _second_layer_packed(h, w2, b2, d: int):
    if h.requires_grad and not w2.requires_grad:
        return _SecondLayer.apply(h, w2, b2)
    hb = h.view(-1, 2, d).transpose(0, 1)
    return torch.baddbmm(b2, hb, w2)


_FUSED_BIG_ROWS = 40_000_000


def _use_fused(rows: int) -> bool:
    """Fused first-layer kernel policy.  Measured on li100k (same box,
    run 25): in the GRAD-RECORDING path the fused kernel loses ~8%
    step time to rocBLAS+gather-add (172 vs 187 ms), but in NO-GRAD
    passes (checkpoint outer forward, inference) it skips the z save and
    the [E,128] GEMM round-trip entirely (~4 GB vs ~9 GB of HBM per MLP),
    so: fused iff grad is off — plus, for HUGE graphs (rows > 40M, where
    a single [E,128] fp32 tensor is >20 GB), fused in the recording path
    too: it materializes one fewer such transient per MLP, which is what
    decides whether 2M atoms fit in 288 GB.  DM_FUSED_MLP=1 /
    DM_NO_FUSED_MLP=1 force it on / off for A/B runs."""
    if os.environ.get("DM_NO_FUSED_MLP", "0") == "1":
        return False
    if os.environ.get("DM_FUSED_MLP", "0") == "1":
        return True
    return not torch.is_grad_enabled() or rows > _FUSED_BIG_ROWS


def gated_mlp_split3(mlp: GatedMLP, v, e, pd, ops, d: int, w=None, base=None):
    """GatedMLP over cat(v[src], v[dst], e) via split-linear + one 2h-wide
    gather_add3, finished by the fused gated-combine epilogue:
    returns base + silu(core2(silu(z_c))) * sigmoid(gate2(silu(z_g))) * w."""
    wcg, bcg, w2, b2, fusedT = _packed_weights(mlp)
    ws, wd = wcg[:, :d], wcg[:, d:2 * d]
    if (d == 64 and not wcg.requires_grad and _use_fused(e.shape[0])
            and hasattr(ops, "edge_mlp3_act")):
        # per-edge GEMM fused into the gather kernel (no [E,2d] transient)
        h = ops.edge_mlp3_act(e, fusedT, bcg, v @ ws.t(), v @ wd.t(), pd)
    else:
        we = wcg[:, 2 * d:]
        h = ops.gather_add3_act(v @ ws.t(), v @ wd.t(),
                                torch.addmm(bcg, e, we.t()), pd)
    cg = _second_layer_packed(h, w2, b2, d)
    return ops.gated_combine_packed(cg, w, base)


def gated_mlp_split4(mlp: GatedMLP, n, a, v, pd, ops, d: int, w=None,
                     base=None):
    """GatedMLP over cat(n[l_src], n[l_dst], a, v[center]), same structure."""
    wcg, bcg, w2p, b2p, fusedT = _packed_weights(mlp)
    w1, w2, wv = wcg[:, :d], wcg[:, d:2 * d], wcg[:, 3 * d:]
    if (d == 64 and not wcg.requires_grad and _use_fused(a.shape[0])
            and hasattr(ops, "edge_mlp4_act")):
        h = ops.edge_mlp4_act(a, fusedT, bcg, n @ w1.t(), n @ w2.t(),
                              v @ wv.t(), pd)
    else:
        wa = wcg[:, 2 * d:3 * d]
        h = ops.gather_add4_act(n @ w1.t(), n @ w2.t(),
                                torch.addmm(bcg, a, wa.t()), v @ wv.t(), pd)
    cg = _second_layer_packed(h, w2p, b2p, d)
    return ops.gated_combine_packed(cg, w, base)


class PartitionData:
    """Static per-partition index tensors, prepared once per forward.

    On a HIP device indices are int32 and carry the builder's CSR layouts
    (dst-sorted edge row_ptr + permutation CSRs) for the kernel path; on
    CPU they are int64 for torch indexing (test backends)."""

    def __init__(self, dist_info, p: int, device, use_bond_graph: bool):
        dev = torch.device(device)
        self.device = dev
        gpu = dev.type == "cuda"
        it = torch.int32 if gpu else torch.long
        self.n_atoms = dist_info.num_atoms(p)
        self.n_owned = dist_info.num_owned_atoms(p)

        def T(a):
            return torch.as_tensor(np.asarray(a)).to(dtype=it).to(dev)

        self.src = T(dist_info.src_nodes[p])
        self.dst = T(dist_info.dst_nodes[p])
        csr = dist_info.csr_parts[p] if getattr(dist_info, "csr_parts", None) else None
        if gpu and csr is None:
            raise RuntimeError(
                "HIP path needs the native builder's CSR layouts "
                "(Distributed.create_distributed with the default backend)")
        if csr is not None:
            self.row_ptr = T(csr["row_ptr"])
            self.src_perm = T(csr["src_perm"])
            self.src_row_ptr = T(csr["src_row_ptr"])
        if use_bond_graph:
            self.n_bonds = dist_info.num_bonds(p)
            self.l_src = T(dist_info.line_src_nodes[p])
            self.l_dst = T(dist_info.line_dst_nodes[p])
            self.center = T(dist_info.local_center_atom_indices_list[p])
            self.map_de = torch.as_tensor(
                np.asarray(dist_info.bond_mapping_DE_list[p]),
                dtype=torch.long).to(dev)
            self.map_ude = torch.as_tensor(
                np.asarray(dist_info.bond_mapping_UDE_list[p]),
                dtype=torch.long).to(dev)
            if csr is not None:
                self.line_row_ptr = T(csr["line_row_ptr"])
                self.line_src_perm = T(csr["line_src_perm"])
                self.line_src_row_ptr = T(csr["line_src_row_ptr"])
                self.center_perm = T(csr["center_perm"])
                self.center_row_ptr = T(csr["center_row_ptr"])

    @property
    def line_src_csr(self):
        if hasattr(self, "line_src_perm"):
            return (self.line_src_perm, self.line_src_row_ptr)
        return None

    @property
    def line_dst_csr(self):
        # lines are l_dst-sorted: the backward scatter needs no permutation
        if hasattr(self, "line_row_ptr"):
            return (None, self.line_row_ptr)
        return None


class CHGNet_Dist(nn.Module):
    """Distributed CHGNet model (drop-in for the reference CHGNet_Dist)."""

    __version__ = 1

    def __init__(self, core: CHGNetCore):
        super().__init__()
        self.core = core
        self.config = core.config
        self.dist_enabled = False
        self.use_bond_graph = core.config.use_bond_graph
        self.cutoff = core.config.cutoff
        self.three_body_cutoff = core.config.three_body_cutoff

    # -- reference surface -------------------------------------------------

    @classmethod
    def from_existing(cls, model, dtype=None):
        """Wrap an existing CHGNetCore (reference chgnet.py:551-560).

        INTEGRATION.md documents the weight mapping a matgl CHGNet would
        convert through; in this container models come from
        CHGNetCore.seeded().
        """
        core = model.core if isinstance(model, CHGNet_Dist) else model
        m = cls(deepcopy(core))
        if dtype is not None:
            m.core = m.core.to(dtype)
        return m

    def enable_distributed_mode(self, gpus, ops_factory=None):
        """Replicate weights per device (reference chgnet.py:455-549).

        `gpus` is a list of ints (cuda ordinals) and/or "cpu" strings,
        exactly like the reference (chgnet.py:465-469).  `ops_factory`
        maps device -> OpsBackend; the default is the HIP product backend
        which refuses non-GPU devices.
        """
        if self.dist_enabled:
            raise Exception("Current model already has distributed mode enabled.")
        self.gpus = []
        for g in gpus:
            self.gpus.append("cpu" if g == "cpu" else f"cuda:{g}")
        factory = ops_factory or default_ops_factory
        self.cores = [deepcopy(self.core).to(dev).eval() for dev in self.gpus]
        self.ops = [factory(torch.device(dev)) for dev in self.gpus]
        self.dist_enabled = True

    # -- forward -----------------------------------------------------------

    def potential_forward_dist(self, dist_info, structure, lattice_matrix,
                               calc_stresses, calc_forces, calc_hessian,
                               state_attr=None):
        """Reference chgnet.py:21-206 flow (single-process, P partitions)."""
        assert self.dist_enabled
        float_th = next(self.core.parameters()).dtype
        root = self.gpus[0]
        P = dist_info.num_partitions

        lattice0 = torch.tensor(np.asarray(lattice_matrix), dtype=float_th, device=root)
        strain = lattice0.new_zeros(3, 3)
        if calc_stresses:
            strain.requires_grad_(True)
        lattice = lattice0 @ (torch.eye(3, device=root, dtype=float_th) + strain)

        frac = torch.tensor(np.asarray(structure.frac_coords), dtype=float_th,
                            device=root)
        pos_global = frac @ lattice                      # chgnet.py:58-61
        if calc_forces:
            pos_global.requires_grad_(True)
            pos_global.retain_grad()

        node_types = torch.tensor(np.asarray(structure.species), dtype=torch.long,
                                  device=root)

        parts = [PartitionData(dist_info, p, self.gpus[p], self.use_bond_graph)
                 for p in range(P)]

        # global bond geometry on the root device (reference chgnet.py:96-100;
        # the SPMD path computes this per-rank instead)
        idx1 = torch.as_tensor(dist_info.py_index_1, dtype=torch.long).to(root)
        idx2 = torch.as_tensor(dist_info.py_index_2, dtype=torch.long).to(root)
        off = torch.tensor(np.asarray(dist_info.py_offsets), dtype=float_th,
                           device=root)
        offshift = off @ lattice
        big_bond_vec = pos_global[idx2] + offshift - pos_global[idx1]
        big_bond_dist = torch.linalg.norm(big_bond_vec, dim=1)

        bond_vec, bond_dist, bond_expansion = [], [], []
        for p in range(P):
            bv = dist_info.global_to_local_edges(big_bond_vec, p, self.gpus[p])
            bd = dist_info.global_to_local_edges(big_bond_dist, p, self.gpus[p])
            bond_vec.append(bv)
            bond_dist.append(bd)
            cp = self.cores[p]
            bond_expansion.append(bond_expansion_from_dist(
                bd, cp.rbf_freq_atom, self.config.cutoff,
                self.config.cutoff_exponent))         # chgnet.py:115-124

        bond_graphs = None
        if self.use_bond_graph:
            # bond-node geometry: owned slots from local edges, ghosts via
            # bond_transfer (reference chgnet.py:129-164)
            nd_dist = [dist_info.edge_to_bond(bond_dist[p], p, self.gpus[p])
                       for p in range(P)]
            nd_vec = [dist_info.edge_to_bond(bond_vec[p], p, self.gpus[p])
                      for p in range(P)]
            dist_info.bond_transfer(nd_dist)
            dist_info.bond_transfer(nd_vec)

            bond_graphs = []
            for p in range(P):
                cp, pd = self.cores[p], parts[p]
                exp3 = bond_expansion_from_dist(
                    nd_dist[p], cp.rbf_freq_bond, self.config.three_body_cutoff,
                    self.config.cutoff_exponent)      # chgnet.py:170-181
                theta = compute_theta(
                    self.ops[p].gather(nd_vec[p], pd.l_src, csr=pd.line_src_csr),
                    self.ops[p].gather(nd_vec[p], pd.l_dst, csr=pd.line_dst_csr))
                angle_exp = fourier_expansion(theta, cp.angle_freq)
                bond_graphs.append({
                    "bond_expansion3": exp3,
                    "angle_expansion": angle_exp,
                })

        return (
            node_types,
            pos_global,
            strain,
            self.dist_forward(parts, bond_expansion, bond_graphs, dist_info),
        )

    def dist_forward(self, parts, bond_expansion, bond_graphs, dist_info):
        """Reference chgnet.py:208-453 flow over the ops backend."""
        cfg = self.config
        P = dist_info.num_partitions
        d = cfg.dim

        v_list, e_list, a_list, n_list = [], [], [], []
        for p in range(P):
            cp = self.cores[p]
            v_list.append(cp.atom_embedding(self._local_species[p]))
            e_list.append(cp.bond_embedding(bond_expansion[p]))
            if self.use_bond_graph:
                a_list.append(cp.angle_embedding(bond_graphs[p]["angle_expansion"]))

        if self.use_bond_graph:
            for p in range(P):
                n_list.append(dist_info.edge_to_bond(
                    e_list[p], p, parts[p].device))    # chgnet.py:255-261
            dist_info.bond_transfer(n_list)

        # shared message weights (chgnet.py:272-294)
        w_ab = [self.cores[p].atom_bond_weights(bond_expansion[p]) for p in range(P)]
        w_bb = [self.cores[p].bond_bond_weights(bond_expansion[p]) for p in range(P)]
        w_3b = None
        if self.use_bond_graph:
            w_3b = [self.cores[p].threebody_bond_weights(
                bond_graphs[p]["bond_expansion3"]) for p in range(P)]

        def atom_conv(layer_i):
            for p in range(P):
                blk, pd, ops = self.cores[p].atom_convs[layer_i], parts[p], self.ops[p]
                e_list[p] = gated_mlp_split3(
                    blk.edge_mlp, v_list[p], e_list[p], pd, ops, d,
                    w=w_bb[p], base=e_list[p])
                msg = gated_mlp_split3(
                    blk.node_mlp, v_list[p], e_list[p], pd, ops, d, w=w_ab[p])
                v_list[p] = ops.scatter_edges(msg, pd, base=v_list[p])

        for layer_i in range(cfg.n_blocks - 1):          # chgnet.py:296-368
            atom_conv(layer_i)
            if self.use_bond_graph:
                for p in range(P):
                    dist_info.edge_to_bond(e_list, p, inplace=True,
                                           bond_node_features=n_list)
                dist_info.bond_transfer(n_list)
                dist_info.atom_transfer(v_list)

                for p in range(P):
                    blk, pd, ops = self.cores[p].bond_convs[layer_i], parts[p], self.ops[p]
                    msg = gated_mlp_split4(
                        blk.bond_mlp, n_list[p], a_list[p], v_list[p],
                        pd, ops, d,
                        w=ops.gather(w_3b[p], pd.l_src, csr=pd.line_src_csr))
                    n_list[p] = ops.scatter_lines(msg, pd, base=n_list[p])
                    dist_info.bond_to_edge(n_list, e_list, p)

                if layer_i < cfg.n_blocks - 2:
                    # the last bond block's angle update is dead compute
                    # (the reference still runs it, chgnet.py:353-368;
                    # numerics are identical without it) — skip
                    dist_info.bond_transfer(n_list)
                    for p in range(P):                   # angle pass chgnet.py:353-368
                        blk, pd, ops = self.cores[p].bond_convs[layer_i], parts[p], self.ops[p]
                        a_list[p] = gated_mlp_split4(
                            blk.angle_mlp, n_list[p], a_list[p], v_list[p],
                            pd, ops, d, base=a_list[p])
            else:
                dist_info.atom_transfer(v_list)

        site_props = [self.cores[p].sitewise_readout(v_list[p]) for p in range(P)]
        site_agg = dist_info.aggregate(site_props, self.gpus[0])  # chgnet.py:391-398

        atom_conv(-1)                                    # chgnet.py:400-419
        dist_info.atom_transfer(v_list)

        final = [self.cores[p].final_layer(v_list[p]) for p in range(P)]
        combined = dist_info.aggregate(final, self.gpus[0])       # chgnet.py:429-433
        total_e = combined.sum()

        return total_e, site_agg

    # helper: species per partition, set by the potential layer
    def set_local_species(self, dist_info, species: np.ndarray):
        self._local_species = []
        for p in range(dist_info.num_partitions):
            ids = np.asarray(dist_info.global_ids[p])
            self._local_species.append(
                torch.as_tensor(np.asarray(species)[ids], dtype=torch.long)
                .to(self.gpus[p]))
