"""Ops backend protocol — the seam between the model orchestration and the
hot kernels.

The product backend (distmlip_amd.ops.HipOps) implements every primitive
with hand-written HIP kernels through the C-ABI extension and REFUSES to
run if the extension is missing or the tensors are not on a HIP device.
Tests may inject a plain-torch reference backend (oracle.chgnet_ref.
CpuRefOps) to exercise the same orchestration on CPU; nothing in this
package imports it.

Primitives (all differentiable):
    gather(x, idx)                      -> x[idx]
    gather_add3(zs, zd, ze, src, dst)   -> zs[src] + zd[dst] + ze
    gather_add4(zb1, zb2, za, zv,
                l_src, l_dst, center)   -> zb1[l_src] + zb2[l_dst] + za + zv[center]
    scatter_sum(msg, idx, n_out)        -> zeros(n_out,...).index_add_(0, idx, msg)
    edge_geom(pos, src, dst, offshift)  -> (bond_vec, bond_dist)
"""
from __future__ import annotations

from typing import Protocol

import torch


class OpsBackend(Protocol):
    def gather(self, x: torch.Tensor, idx: torch.Tensor) -> torch.Tensor: ...

    def gather_add3(self, zs, zd, ze, src, dst) -> torch.Tensor: ...

    def gather_add4(self, zb1, zb2, za, zv, l_src, l_dst, center) -> torch.Tensor: ...

    def scatter_sum(self, msg, idx, n_out: int) -> torch.Tensor: ...

    def edge_geom(self, pos, src, dst, offshift): ...


class ComposedMixin:
    """Default compositions for backends that only define the core three."""

    def gather_add3(self, zs, zd, ze, src, dst):
        return self.gather(zs, src) + self.gather(zd, dst) + ze

    def gather_add4(self, zb1, zb2, za, zv, l_src, l_dst, center):
        return (self.gather(zb1, l_src) + self.gather(zb2, l_dst) + za
                + self.gather(zv, center))


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
