"""Product-side eSCN/UMA math (torch, GPU-ready, autograd-capable).

Primitives the UMA runtime composes per partition; the oracle
(oracle/uma_ref.py) restates them with its own einsums/pinv-based grid.
"""
from __future__ import annotations

import math
from functools import lru_cache
from typing import Dict

import torch

from distmlip_amd import so3
from distmlip_amd.uma_model import (UMABlock, UMAConfig, UMACore, SO2Conv,
                                    gaussian_basis, m_indices, s2_grid)


def _grids(cfg: UMAConfig, like: torch.Tensor):
    return s2_grid(cfg.lmax, cfg.grid_theta, cfg.grid_phi,
                   str(like.dtype).split(".")[-1], str(like.device))


def rms_norm(x: torch.Tensor, scale: torch.Tensor, lmax: int,
             eps: float = 1e-6) -> torch.Tensor:
    """Equivariant RMS norm per (node, l) over the (m, channel) block
    jointly (the eSCN norm family's shape); learned scale per
    (l, channel).  Normalizing per channel over only the 2l+1
    m-components makes near-zero channels explode through rsqrt —
    measured as ~1e5-magnitude forces at the full shape."""
    outs = []
    for l in range(lmax + 1):
        o, d = l * l, 2 * l + 1
        blk = x[:, o:o + d, :]
        ms = blk.pow(2).mean(dim=(1, 2), keepdim=True)
        outs.append(blk * torch.rsqrt(ms + eps) * scale[l])
    return torch.cat(outs, dim=1)


def so2_conv(conv: SO2Conv, xt: torch.Tensor, gate: torch.Tensor,
             C: int) -> torch.Tensor:
    """xt [E, S, in_mult*C] rotated-frame features; gate [E, lmax+1]
    per-m modulation; returns [E, S, C]."""
    E = xt.shape[0]
    lmax = conv.lmax
    m0, pm = m_indices(lmax)
    out = xt.new_zeros(E, (lmax + 1) ** 2, C)
    x0 = xt[:, m0, :].reshape(E, -1)
    out[:, m0, :] = (x0 @ conv.w0.t()).view(E, lmax + 1, C) \
        * gate[:, 0].view(E, 1, 1)
    for m in range(1, lmax + 1):
        plus, minus = pm[m - 1]
        xp = xt[:, plus, :].reshape(E, -1)
        xm = xt[:, minus, :].reshape(E, -1)
        wr, wi = conv.wr[m - 1], conv.wi[m - 1]
        op = (xp @ wr.t() - xm @ wi.t()).view(E, len(plus), C)
        om = (xp @ wi.t() + xm @ wr.t()).view(E, len(plus), C)
        g = gate[:, m].view(E, 1, 1)
        out[:, plus, :] = op * g
        out[:, minus, :] = om * g
    return out


def s2_act(x: torch.Tensor, cfg: UMAConfig,
           chunk: int = 65536) -> torch.Tensor:
    """Pointwise silu on the sphere: to grid, silu, band-limited
    projection back (the eSCN grid activation).  Chunked over nodes —
    the [N, G, C] grid tensor at 238k atoms x 512 points x 128 channels
    is 62 GB fp32, so it is never materialized whole."""
    to_g, from_g = _grids(cfg, x)

    def one(t):
        # tensordot keeps these as ONE tall GEMM ([N*C, S] x [S, G]);
        # the einsum forms lower to batch-N (S x G) bmm tiles (the
        # rocBLAS ~1%-of-peak shape)
        f = torch.tensordot(t, to_g, dims=([1], [1]))      # [N, C, G]
        f = torch.nn.functional.silu(f)
        out = torch.tensordot(f, from_g, dims=([2], [1]))  # [N, C, S]
        return out.permute(0, 2, 1).contiguous()

    if x.shape[0] <= chunk:
        return one(x)
    outs = []
    for i in range(0, x.shape[0], chunk):
        xc = x[i:i + chunk]
        if torch.is_grad_enabled() and x.requires_grad:
            # per-chunk checkpoint: silu saves its [N_c, G, C] grid input
            # otherwise — ~8 GB per chunk held across the whole backward
            outs.append(torch.utils.checkpoint.checkpoint(
                one, xc, use_reentrant=False))
        else:
            outs.append(one(xc))
    return torch.cat(outs, dim=0)


def edge_scalars(core: UMACore, lengths: torch.Tensor, spec_src,
                 spec_dst) -> torch.Tensor:
    cfg = core.config
    g = gaussian_basis(lengths, cfg.cutoff, cfg.num_gauss)
    x = torch.cat([g, core.source_embedding[spec_src],
                   core.target_embedding[spec_dst]], dim=1)
    return torch.nn.functional.silu(x @ core.edge_proj.t())


def edge_degree_embed(core: UMACore, x_edge: torch.Tensor,
                      Dinv: torch.Tensor) -> torch.Tensor:
    """Per-edge radial profile on the m=0 rows of the edge frame,
    rotated back (escn EdgeDegreeEmbedding analog): returns per-edge
    [E, S, C] to be scattered to receivers / avg_degree."""
    cfg = core.config
    C = cfg.sphere_channels
    E = x_edge.shape[0]
    w = (x_edge @ core.edge_degree.t()).view(E, cfg.lmax + 1, C)
    m0, _ = m_indices(cfg.lmax)
    m_t = x_edge.new_zeros(E, cfg.S, C)
    m_t[:, m0, :] = w
    return torch.einsum("est,etc->esc", Dinv, m_t)


def block_message(blk: UMABlock, cfg: UMAConfig, x_src, x_dst, x_edge,
                  D, Dinv):
    """Rotated SO(2) message for one block: [E, S, C] in the NODE frame,
    ready for scatter to receivers."""
    ge = blk.edge_mlp(x_edge)
    gate = torch.sigmoid(ge[:, :cfg.lmax + 1])
    xt = torch.cat([torch.einsum("est,etc->esc", D, x_src),
                    torch.einsum("est,etc->esc", D, x_dst)], dim=2)
    mt = so2_conv(blk.msg, xt, gate, cfg.sphere_channels)
    # message nonlinearity = the per-m sigmoid gate (exactly equivariant);
    # the grid silu is applied per NODE in the FFN — per-edge grids would
    # cost E x G x C memory and add azimuthal aliasing per edge
    return torch.einsum("est,etc->esc", Dinv, mt)


def node_ffn(blk: UMABlock, cfg: UMAConfig, x):
    h = rms_norm(x, blk.norm2, cfg.lmax)
    h = torch.einsum("nsc,dc->nsd", h, blk.ffn1)
    h = s2_act(h, cfg)
    return torch.einsum("nsc,dc->nsd", h, blk.ffn2)


def energy_head(core: UMACore, x):
    h = rms_norm(x, core.final_norm, core.config.lmax)
    s = h[:, 0, :]
    s = torch.nn.functional.silu(s @ core.head1.t())
    return core.scale * (s @ core.head2) + core.shift


# ---------------------------------------------------------------------------
# fused HIP rotation family (round 2): dm_rot_{gather,scatter,dD}_f32 —
# replaces the per-edge Wigner bmms (62% of the uma250k step as rocBLAS
# tiny-tile batched GEMMs, profiles/r2_uma_kernel_stats.csv)
# ---------------------------------------------------------------------------

def _rot_gather_raw(h, idx, D, trans):
    from distmlip_amd.ops import _check, _fp, _ip, _stream, hip_lib
    E = D.shape[0]
    C = h.shape[-1]
    out = torch.empty(E, 9, C, dtype=h.dtype, device=h.device)
    _check(hip_lib().dm_rot_gather_f32(
        _fp(h), _ip(idx) if idx is not None else None, _fp(D), trans,
        _fp(out), E, C, _stream()), "dm_rot_gather_f32")
    return out


def _rot_dD_raw(go_e, h, idx, trans):
    from distmlip_amd.ops import _check, _fp, _ip, _stream, hip_lib
    E = go_e.shape[0]
    C = go_e.shape[-1]
    dD = torch.empty(E, 9, 9, dtype=go_e.dtype, device=go_e.device)
    _check(hip_lib().dm_rot_dD_f32(
        _fp(go_e), _fp(h), _ip(idx) if idx is not None else None, trans,
        _fp(dD.view(E, 81)), E, C, _stream()), "dm_rot_dD_f32")
    return dD


class _RotGather(torch.autograd.Function):
    """out[e] = D_e . h[idx[e]]   (h [N,9,C] f32, D [E,9,9] f32)."""

    @staticmethod
    def forward(ctx, h, D, idx):
        h = h.contiguous()
        D = D.contiguous()
        out = _rot_gather_raw(h, idx, D.view(-1, 81), 0)
        ctx.save_for_backward(h, D, idx)
        return out

    @staticmethod
    def backward(ctx, go):
        h, D, idx = ctx.saved_tensors
        go = go.contiguous()
        # dh: scatter of D^T . go to the gathered rows
        tmp = _rot_gather_raw(go, None, D.view(-1, 81), 1)
        dh = torch.zeros_like(h).index_add_(0, idx.long(), tmp)
        dD = _rot_dD_raw(go, h, idx, 0)
        return dh, dD, None


class _RotScatter(torch.autograd.Function):
    """out[n] = sum_{e in row n} D_e^T . mt[e]  over a contiguous
    (dst-sorted) row_ptr covering rows [0, N_local)."""

    @staticmethod
    def forward(ctx, mt, D, row_ptr, dst_local):
        from distmlip_amd.ops import _check, _fp, _ip, _stream, hip_lib
        mt = mt.contiguous()
        D = D.contiguous()
        N = row_ptr.shape[0] - 1
        C = mt.shape[-1]
        out = torch.empty(N, 9, C, dtype=mt.dtype, device=mt.device)
        _check(hip_lib().dm_rot_scatter_f32(
            _fp(mt), _fp(D.view(-1, 81)), 1, None, _ip(row_ptr), None,
            _fp(out), N, C, _stream()), "dm_rot_scatter_f32")
        ctx.save_for_backward(mt, D, dst_local)
        return out

    @staticmethod
    def backward(ctx, go):
        mt, D, dst_local = ctx.saved_tensors
        go = go.contiguous()
        dmt = _rot_gather_raw(go, dst_local, D.view(-1, 81), 0)
        dD = _rot_dD_raw(mt, go, dst_local, 0)
        return dmt, dD, None, None


def _so2_split(conv: SO2Conv):
    """Per-side (src/dst) weight splits of the concatenated SO(2) mixes
    (the cat over the channel dim becomes two tall GEMMs); cached on the
    module while frozen."""
    key = tuple(p._version for p in conv.parameters())
    cached = getattr(conv, "_dm_split", None)
    if cached is not None and cached[0] == key:
        return cached[1]
    lmax = conv.lmax
    n0 = lmax + 1
    C = conv.w0.shape[0] // n0
    def split(w, nl):
        v = w.view(w.shape[0], nl, 2, C)
        return (v[:, :, 0, :].reshape(w.shape[0], nl * C).contiguous(),
                v[:, :, 1, :].reshape(w.shape[0], nl * C).contiguous())
    packs = {"w0": split(conv.w0, n0)}
    for m in range(1, lmax + 1):
        nl = lmax + 1 - m
        packs[f"wr{m}"] = split(conv.wr[m - 1], nl)
        packs[f"wi{m}"] = split(conv.wi[m - 1], nl)
    if not any(p.requires_grad for p in conv.parameters()):
        conv._dm_split = (key, packs)
    return packs


def so2_conv_split(conv: SO2Conv, xs: torch.Tensor, xd: torch.Tensor,
                   gate: torch.Tensor, C: int) -> torch.Tensor:
    """so2_conv over separate rotated src/dst tensors [E,9,C] (no cat
    materialization; the concatenated GEMM splits exactly)."""
    E = xs.shape[0]
    lmax = conv.lmax
    m0, pm = m_indices(lmax)
    sp = _so2_split(conv)
    ws, wd = sp["w0"]
    a0 = xs[:, m0, :].reshape(E, -1) @ ws.t() \
        + xd[:, m0, :].reshape(E, -1) @ wd.t()
    # under bf16 autocast the GEMMs come back bf16: allocate the output
    # in THAT dtype (indexed writes require matching dtypes)
    out = xs.new_zeros(E, (lmax + 1) ** 2, C, dtype=a0.dtype)
    out[:, m0, :] = a0.view(E, lmax + 1, C) * gate[:, 0].view(E, 1, 1)
    for m in range(1, lmax + 1):
        plus, minus = pm[m - 1]
        xps = xs[:, plus, :].reshape(E, -1)
        xpd = xd[:, plus, :].reshape(E, -1)
        xms = xs[:, minus, :].reshape(E, -1)
        xmd = xd[:, minus, :].reshape(E, -1)
        wrs, wrd = sp[f"wr{m}"]
        wis, wid_ = sp[f"wi{m}"]
        op = (xps @ wrs.t() + xpd @ wrd.t()
              - xms @ wis.t() - xmd @ wid_.t())
        om = (xps @ wis.t() + xpd @ wid_.t()
              + xms @ wrs.t() + xmd @ wrd.t())
        g = gate[:, m].view(E, 1, 1)
        out[:, plus, :] = op.view(E, len(plus), C) * g
        out[:, minus, :] = om.view(E, len(plus), C) * g
    return out


def rot_kernels_available(x_like: torch.Tensor, lmax: int) -> bool:
    import os
    return (os.environ.get("DM_UMA_ROT", "hip") == "hip"
            and x_like.is_cuda and lmax == 2
            and x_like.shape[-1] in (64, 128))
