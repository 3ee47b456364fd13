"""Ops backend protocol — the seam between the model orchestration and the
hot kernels.

The product backend (distmlip_amd.ops.HipOps) implements every primitive
with hand-written HIP kernels through the C-ABI extension
(include/distmlip_hip.h) and REFUSES to run if the extension is missing or
the tensors are not on a HIP device.  Tests may inject a plain-torch
reference backend (oracle.chgnet_ref.CpuRefOps) to exercise the same
orchestration on CPU; nothing in this package imports it.

All primitives are differentiable.  `pd` is the per-partition index bundle
(distmlip_amd.chgnet.PartitionData) carrying the dst-sorted edge CSR and
the permutation CSRs the graph builder emits; `csr` is an optional
(perm, row_ptr) pair giving the deterministic backward of a gather.

    gather(x, idx, csr)                  -> x[idx]
    gather_add3_act(zs, zd, ze, pd)      -> silu(zs[pd.src] + zd[pd.dst] + ze)
    gather_add4_act(z1, z2, za, zv, pd)  -> silu(z1[pd.l_src] + z2[pd.l_dst]
                                                 + za + zv[pd.center])
    edge_mlp3_act(erow, wt, bias, zs, zd, pd)
                                         -> silu(erow @ wt + bias
                                                 + zs[pd.src] + zd[pd.dst])
                                            (per-edge GEMM fused into the
                                            gather kernel; frozen wt/bias)
    edge_mlp4_act(arow, wt, bias, z1, z2, zv, pd)
                                         -> the 3-gather line-graph form
    scatter_edges(msg, pd, base)         -> base + segment-sum of the
                                            dst-sorted msg rows per node

  RAW primitives (prefix r_): NON-differentiable single kernels used by
  the hand-sequenced conv backward (distmlip_amd.conv._AtomConvFn),
  which replaces autograd's per-tensor gradient accumulation for the
  atom-conv blocks.  Both backends implement them so the CPU fp64
  exactness tests cover the product's hand-written reverse pass:

    r_gather_add3(zs, zd, ze, pd)        -> (z, silu(z)) raw forward
    r_combine_fwd(cg, w, base)           -> base + silu(cg0)*sigmoid(cg1)*w
    r_combine_bwd(go, cg, w)             -> (dcg [2,*,D], dw)
    r_silu_bwd(go_h, z)                  -> go_h * silu'(z)
    r_gather_dst(x, pd)                  -> x[pd.dst]
    r_seg_dst(msg, pd, base)             -> segment-sum over the dst CSR
    r_seg_src(msg, pd)                   -> permuted segment-sum (src CSR)
    scatter_lines(msg, pd, base)         -> same over the line CSR per bond
    gated_combine(c, g, w, base)         -> base + silu(c)*sigmoid(g)*w
    gated_combine_packed(cg, w, base)    -> same over PACKED cg [2,*,D]
                                            (cg[0]=c, cg[1]=g; lets the
                                            second-layer core|gate GEMMs run
                                            as one bmm with no slicing)
    edge_geom_rbf(pos, off, freqs, c, p, pd) -> (bond_vec, bond_dist, rbf*env)
    rbf_env(d, freqs, cutoff, pexp)      -> rbf*env
"""
from __future__ import annotations

from typing import Optional, Protocol, Tuple

import torch


class OpsBackend(Protocol):
    def gather(self, x: torch.Tensor, idx: torch.Tensor,
               csr: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
               ) -> torch.Tensor: ...

    def gather_add3_act(self, zs, zd, ze, pd) -> torch.Tensor: ...

    def gather_add4_act(self, z1, z2, za, zv, pd) -> torch.Tensor: ...

    def edge_mlp3_act(self, erow, wt, bias, zs, zd, pd) -> torch.Tensor: ...

    def edge_mlp4_act(self, arow, wt, bias, z1, z2, zv, pd) -> torch.Tensor: ...

    def gated_combine(self, c, g, w=None, base=None) -> torch.Tensor: ...

    def gated_combine_packed(self, cg, w=None, base=None) -> torch.Tensor: ...

    def edge_geom_rbf(self, pos, offshift, freqs, cutoff, pexp, pd): ...

    def rbf_env(self, d, freqs, cutoff, pexp) -> torch.Tensor: ...

    def scatter_edges(self, msg, pd, base=None) -> torch.Tensor: ...

    def scatter_rows(self, msg, dst_rel, row_ptr_rel,
                     n_rows) -> torch.Tensor: ...

    def r_gather_add3(self, zs, zd, ze, pd): ...

    def r_combine_fwd(self, cg, w, base) -> torch.Tensor: ...

    def r_combine_bwd(self, go, cg, w): ...

    def r_silu_bwd(self, go_h, z) -> torch.Tensor: ...

    def r_gather_dst(self, x, pd) -> torch.Tensor: ...

    def r_seg_dst(self, msg, pd, base=None) -> torch.Tensor: ...

    def r_seg_src(self, msg, pd) -> torch.Tensor: ...

    def scatter_lines(self, msg, pd, base=None) -> torch.Tensor: ...


def default_ops_factory(device: torch.device):
    """Product backend resolution: HIP on GPU, loud failure elsewhere."""
    device = torch.device(device)
    if device.type == "cuda":
        from distmlip_amd.ops import HipOps
        return HipOps()
    raise RuntimeError(
        f"distmlip_amd has no product compute path for device {device!r}: "
        "the hot path is the HIP extension (gfx950). CPU execution exists "
        "only in tests via an injected reference backend."
    )
