"""CPU oracle for the CHGNet energy+force forward (full graph, no partitions).

An INDEPENDENT plain-torch restatement of the computation the reference
orchestrates in implementations/matgl/models/chgnet.py:21-453 +
implementations/matgl/pes.py:50-146, written straight-line over the full
(unpartitioned) graph:

  positions (chgnet.py:44-64) -> bond_vec/bond_dist (chgnet.py:96-100)
  -> RBF * polynomial_cutoff applied to the RBF output (chgnet.py:115-124)
  -> bond-graph geometry + theta + Fourier (chgnet.py:129-197)
  -> embeddings (chgnet.py:231-251) -> shared weights (chgnet.py:272-294)
  -> (n_blocks-1) x [atom conv; edge_to_bond; bond-node conv; bond_to_edge;
     angle conv] (chgnet.py:296-368)
  -> sitewise readout (chgnet.py:391-398) -> final atom block
  (chgnet.py:400-419) -> final_layer + sum readout (chgnet.py:422-440)
  -> scale/shift + element refs (pes.py:109-113) -> forces = -dE/dpos
  (pes.py:121-124), stress via strain gradient (pes.py:140-145).

Model-arithmetic parity vs upstream matgl is UNPINNED (oracle/__init__.py);
this file IS the executable definition the product must match.  Tests use
it in fp64 (tight partition-invariance) and fp32 (GPU parity tolerance).

Independence: every FORMULA here is oracle-local (oracle/basis_ref.py —
radial Bessel / cutoff envelope / Fourier / theta / gated-MLP / linear
application), restated separately from the product's copies in
distmlip_amd/model.py; `CHGNetCore` is imported ONLY as the weight
container (no product forward() code runs here), so a transcription error
in the product's formulas cannot validate itself.

TEST INFRASTRUCTURE ONLY — see oracle/__init__.py.
"""
from __future__ import annotations

import numpy as np
import torch

from distmlip_amd.model import CHGNetCore  # weights container only
from oracle.basis_ref import (
    bond_expansion_from_dist_t,
    compute_theta_t,
    fourier_expansion_t,
    gated_mlp,
    linear,
    silu,
)


def build_full_line_graph(src: np.ndarray, dst: np.ndarray,
                          within_idx: np.ndarray):
    """Full-graph bond (line) graph.

    Bond-graph NODES are the directed atom edges within the three-body
    cutoff (`within_idx` into the global edge arrays).  Line EDGES connect
    bond b1=(a->b) to bond b2=(b->c) whenever dst(b1)==src(b2) and
    dst(b2)!=src(b1) (the backtrack skip, utils.c:727-729 — note the skip
    compares ATOM ids only, so a->b->a via a different periodic image is
    also skipped, a reference quirk we preserve).  The center atom of the
    line is src(b2)=dst(b1)=b (utils.c:733).

    Returns (line_src, line_dst, center_atom): int64 arrays of local bond
    ids (indices into within_idx) and global atom ids.
    """
    b_src = np.asarray(src)[within_idx]
    b_dst = np.asarray(dst)[within_idx]
    nb = len(b_src)
    order = np.argsort(b_src, kind="stable")         # bonds grouped by src atom
    sorted_src = b_src[order]
    n_atoms = int(max(b_src.max(initial=-1), b_dst.max(initial=-1)) + 1) if nb else 0
    grp_start = np.searchsorted(sorted_src, np.arange(n_atoms), side="left")
    grp_end = np.searchsorted(sorted_src, np.arange(n_atoms), side="right")

    counts = (grp_end - grp_start)[b_dst] if nb else np.zeros(0, np.int64)
    l_src = np.repeat(np.arange(nb, dtype=np.int64), counts)
    # for each e1, the e2 candidates are order[grp_start[dst(e1)] : grp_end[...]]
    offs = np.concatenate([np.arange(c) for c in counts]) if nb and counts.sum() else \
        np.zeros(0, np.int64)
    l_dst = order[np.repeat(grp_start[b_dst], counts) + offs] if nb else \
        np.zeros(0, np.int64)
    keep = b_dst[l_dst] != b_src[l_src]               # backtrack skip
    l_src, l_dst = l_src[keep], l_dst[keep]
    center = b_src[l_dst]
    return l_src.astype(np.int64), l_dst.astype(np.int64), center.astype(np.int64)


def oracle_forward(core: CHGNetCore, structure, src, dst, offsets, within_idx,
                   dtype: torch.dtype = torch.float64,
                   compute_forces: bool = True,
                   compute_stress: bool = False):
    """Full-graph CHGNet E+F forward on CPU.  Returns a dict."""
    cfg = core.config
    core = core.to(dtype)
    lat0 = torch.tensor(np.asarray(structure.lattice), dtype=dtype)
    strain = torch.zeros(3, 3, dtype=dtype)
    if compute_stress:
        strain.requires_grad_(True)
    lattice = lat0 @ (torch.eye(3, dtype=dtype) + strain)   # chgnet.py:41-43

    frac = torch.tensor(np.asarray(structure.frac_coords), dtype=dtype)
    pos = frac @ lattice                                     # chgnet.py:58-61
    if compute_forces:
        # reference keeps positions CONNECTED to strain (chgnet.py:63-64:
        # requires_grad_ + retain_grad on the non-leaf product), so the
        # stress gradient flows through both positions and offshift
        if not pos.requires_grad:
            pos.requires_grad_(True)
        pos.retain_grad()

    species = torch.tensor(np.asarray(structure.species), dtype=torch.long)
    src_t = torch.tensor(np.asarray(src), dtype=torch.long)
    dst_t = torch.tensor(np.asarray(dst), dtype=torch.long)
    off_t = torch.tensor(np.asarray(offsets), dtype=dtype)

    offshift = off_t @ lattice                               # chgnet.py:55-57
    bond_vec = pos[dst_t] + offshift - pos[src_t]            # chgnet.py:96-99
    bond_dist = torch.linalg.norm(bond_vec, dim=1)           # chgnet.py:100

    bond_expansion = bond_expansion_from_dist_t(
        bond_dist, core.rbf_freq_atom, cfg.cutoff, cfg.cutoff_exponent)

    v = core.atom_embedding.weight[species]
    e = linear(core.bond_embedding, bond_expansion)

    w_ab = linear(core.atom_bond_weights, bond_expansion)
    w_bb = linear(core.bond_bond_weights, bond_expansion)

    use_bg = cfg.use_bond_graph and len(within_idx) > 0
    if use_bg:
        within_t = torch.tensor(np.asarray(within_idx), dtype=torch.long)
        l_src_np, l_dst_np, center_np = build_full_line_graph(src, dst, within_idx)
        l_src = torch.tensor(l_src_np, dtype=torch.long)
        l_dst = torch.tensor(l_dst_np, dtype=torch.long)
        center = torch.tensor(center_np, dtype=torch.long)

        bond_expansion3 = bond_expansion_from_dist_t(
            bond_dist[within_t], core.rbf_freq_bond, cfg.three_body_cutoff,
            cfg.cutoff_exponent)
        w_3b = linear(core.threebody_bond_weights, bond_expansion3)

        bv_b = bond_vec[within_t]
        theta = compute_theta_t(bv_b[l_src], bv_b[l_dst])    # chgnet.py:190-194
        a = linear(core.angle_embedding,
                   fourier_expansion_t(theta, core.angle_freq))
        n = e[within_t]                                      # chgnet.py:255-261

    n_atoms = len(species)

    def atom_conv(block, v, e):
        """edge update then node update (node sees updated edges)."""
        x = torch.cat([v[src_t], v[dst_t], e], dim=1)
        e = e + gated_mlp(block.edge_mlp, x) * w_bb
        x = torch.cat([v[src_t], v[dst_t], e], dim=1)
        msg = gated_mlp(block.node_mlp, x) * w_ab
        v = v + torch.zeros_like(v).index_add_(0, dst_t, msg)
        return v, e

    for layer_i in range(cfg.n_blocks - 1):                  # chgnet.py:296-368
        v, e = atom_conv(core.atom_convs[layer_i], v, e)
        if use_bg:
            n = e[within_t]                                  # edge_to_bond
            blk = core.bond_convs[layer_i]
            x = torch.cat([n[l_src], n[l_dst], a, v[center]], dim=1)
            msg = gated_mlp(blk.bond_mlp, x) * w_3b[l_src]
            n = n + torch.zeros_like(n).index_add_(0, l_dst, msg)
            e = e.index_put((within_t,), n)                  # bond_to_edge
            x = torch.cat([n[l_src], n[l_dst], a, v[center]], dim=1)
            a = a + gated_mlp(blk.angle_mlp, x)

    site_props = linear(core.sitewise_readout, v)            # chgnet.py:391-398

    v, e = atom_conv(core.atom_convs[-1], v, e)              # chgnet.py:400-419

    # final_layer = Linear-SiLU-Linear-SiLU-Linear (chgnet.py:422-440),
    # applied with oracle-local arithmetic
    fl = core.final_layer
    atom_energies = linear(fl[4], silu(linear(fl[2], silu(linear(fl[0], v)))))
    total_e = atom_energies.sum()

    total_e = core.data_std * total_e + core.data_mean       # pes.py:109
    total_e = total_e + core.element_refs[species].sum()     # pes.py:111-113

    out = {"energy": total_e, "site_props": site_props,
           "atom_energies": atom_energies.detach()}
    if compute_forces:
        grads = [pos, strain] if compute_stress else [pos]
        gv = torch.autograd.grad(total_e, grads, retain_graph=False)
        out["forces"] = -gv[0]                               # pes.py:121-124
        if compute_stress:
            volume = abs(np.linalg.det(np.asarray(structure.lattice)))
            out["stress"] = -gv[1] / volume * -160.21766208  # pes.py:140-145
    return out


# ---------------------------------------------------------------------------
# CPU ops backend for the DISTRIBUTED orchestration (test injection only).
# Implements the distmlip_amd.ops_base.OpsBackend protocol in plain torch so
# the product's partition/halo orchestration can run on CPU in tests.  The
# product default backend is the HIP one, which refuses to run without the
# extension — this class must never be reachable from product code.
# ---------------------------------------------------------------------------

class CpuRefOps:
    """Plain-torch implementation of the ops protocol (tests only);
    see distmlip_amd.ops_base.OpsBackend for the contract."""

    is_reference = True

    def gather(self, x, idx, csr=None):
        return x[idx]

    def gather_add3_act(self, zs, zd, ze, pd):
        return torch.nn.functional.silu(zs[pd.src] + zd[pd.dst] + ze)

    def gather_add4_act(self, z1, z2, za, zv, pd):
        return torch.nn.functional.silu(
            z1[pd.l_src] + z2[pd.l_dst] + za + zv[pd.center])

    def _scatter(self, msg, idx, n_out, base):
        out = torch.zeros((n_out,) + tuple(msg.shape[1:]), dtype=msg.dtype,
                          device=msg.device).index_add(0, idx, msg)
        return out if base is None else base + out

    def scatter_edges(self, msg, pd, base=None):
        return self._scatter(msg, pd.dst, pd.n_atoms, base)

    def scatter_rows(self, msg, dst_rel, row_ptr_rel, n_rows):
        return self._scatter(msg, dst_rel.long(), n_rows, None)

    def scatter_lines(self, msg, pd, base=None):
        return self._scatter(msg, pd.l_dst, pd.n_bonds, base)

    def gated_combine(self, c, g, w=None, base=None):
        out = torch.nn.functional.silu(c) * torch.sigmoid(g)
        if w is not None:
            out = out * w
        return out if base is None else base + out

    def gated_combine_packed(self, cg, w=None, base=None):
        return self.gated_combine(cg[0], cg[1], w, base)

    def edge_mlp3_act(self, erow, wt, bias, zs, zd, pd):
        return torch.nn.functional.silu(
            erow @ wt + bias + zs[pd.src] + zd[pd.dst])

    def edge_mlp4_act(self, arow, wt, bias, z1, z2, zv, pd):
        return torch.nn.functional.silu(
            arow @ wt + bias + z1[pd.l_src] + z2[pd.l_dst] + zv[pd.center])

    # raw primitives for the hand-sequenced conv backward (ops_base)

    def r_gather_add3(self, zs, zd, ze, pd):
        z = zs[pd.src] + zd[pd.dst] + ze
        return z, torch.nn.functional.silu(z)

    def r_combine_fwd(self, cg, w, base):
        out = torch.nn.functional.silu(cg[0]) * torch.sigmoid(cg[1])
        if w is not None:
            out = out * w
        return out if base is None else base + out

    def r_combine_bwd(self, go, cg, w):
        c, g = cg[0], cg[1]
        sc = torch.sigmoid(c)
        silu_c = c * sc
        sg = torch.sigmoid(g)
        gw = go if w is None else go * w
        dc = gw * sg * (sc * (1 + c * (1 - sc)))
        dg = gw * silu_c * sg * (1 - sg)
        dw = go * silu_c * sg if w is not None else None
        return torch.stack([dc, dg]), dw

    def r_silu_bwd(self, go_h, z):
        s = torch.sigmoid(z)
        return go_h * (s * (1 + z * (1 - s)))

    def r_gather_dst(self, x, pd):
        return x[pd.dst]

    def r_seg_dst(self, msg, pd, base=None):
        out = torch.zeros((pd.n_atoms,) + tuple(msg.shape[1:]),
                          dtype=msg.dtype, device=msg.device
                          ).index_add_(0, pd.dst, msg)
        return out if base is None else base + out

    def r_seg_src(self, msg, pd):
        return torch.zeros((pd.n_atoms,) + tuple(msg.shape[1:]),
                           dtype=msg.dtype, device=msg.device
                           ).index_add_(0, pd.src, msg)

    def _seg(self, msg, idx, n, base=None):
        out = torch.zeros((n,) + tuple(msg.shape[1:]), dtype=msg.dtype,
                          device=msg.device).index_add_(0, idx, msg)
        return out if base is None else base + out

    def r_gather_add4(self, z1, z2, za, zv, pd):
        z = z1[pd.l_src] + z2[pd.l_dst] + za + zv[pd.center]
        return z, torch.nn.functional.silu(z)

    def r_gather_lsrc(self, x, pd):
        return x[pd.l_src]

    def r_gather_ldst(self, x, pd):
        return x[pd.l_dst]

    def r_seg_ldst(self, msg, pd, base=None):
        return self._seg(msg, pd.l_dst, pd.n_bonds, base)

    def r_seg_lsrc(self, msg, pd):
        return self._seg(msg, pd.l_src, pd.n_bonds)

    def r_seg_center(self, msg, pd):
        return self._seg(msg, pd.center, pd.n_atoms)

    def edge_geom_rbf(self, pos, offshift, freqs, cutoff, pexp, pd):
        from distmlip_amd.model import bond_expansion_from_dist
        bv = pos[pd.dst] + offshift - pos[pd.src]
        bd = torch.linalg.norm(bv, dim=1)
        return bv, bd, bond_expansion_from_dist(bd, freqs, cutoff, pexp)

    def rbf_env(self, d, freqs, cutoff, pexp):
        from distmlip_amd.model import bond_expansion_from_dist
        return bond_expansion_from_dist(d, freqs, cutoff, pexp)
