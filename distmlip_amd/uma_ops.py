"""Product-side eSCN/UMA math (torch, GPU-ready, autograd-capable).

Primitives the UMA runtime composes per partition; the oracle
(oracle/uma_ref.py) restates them with its own einsums/pinv-based grid.
"""
from __future__ import annotations

import math
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
        f = torch.nn.functional.silu(torch.einsum("gs,nsc->ngc", to_g, t))
        return torch.einsum("sg,ngc->nsc", from_g, f)

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
