"""UMA / eSCN-MD backbone restatement — weights container + S2 grid.

The reference's UMA adapter monkey-patches a fairchem eSCN-MD backbone
(implementations/uma/escn_md.py:249-523: per-edge Wigner rotations,
edge-degree embedding, num_layers message blocks with per-layer
atom_transfer halos, final norm) — fairchem-core is not installable
(reference pin, pyproject.toml:34) and no checkpoint is fetchable, so the
backbone internals are restated from the published eSCN architecture at a
fixed shape:

  lmax 2 ((lmax+1)^2 = 9 spherical coefficients), 128 sphere channels,
  4 layers, cutoff 6 A, 64-Gaussian distance basis, per-edge z-aligned
  frames (so3.edge_align_rotation + so3.wigner_D_batch), SO(2)
  convolutions per |m| with the complex (w_r, w_i) structure, S2
  grid-sampled pointwise activation (Gauss-Legendre x uniform-azimuth
  quadrature), equivariant RMS norms, edge-degree embedding, scalar
  energy head; forces via autograd (the MD variant's path).

Known property preserved from eSCN: the grid activation is band-limited,
so equivariance is exact only up to the quadrature truncation — the
rotation-invariance test tolerance reflects that (tests/test_uma.py).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from functools import lru_cache
from typing import Tuple

import numpy as np
import torch
from torch import nn

from distmlip_amd import so3


@dataclass
class UMAConfig:
    n_elements: int = 4
    lmax: int = 2
    sphere_channels: int = 128
    num_layers: int = 4
    cutoff: float = 6.0
    num_gauss: int = 64
    spec_emb: int = 32
    edge_ch: int = 128                 # projected x_edge width
    grid_theta: int = 16
    grid_phi: int = 32
    avg_degree: float = 40.0
    energy_scale: float = 1.0
    energy_shift: float = 0.0

    @property
    def S(self) -> int:
        return (self.lmax + 1) ** 2


def l_of_row(lmax: int) -> np.ndarray:
    return np.concatenate([[l] * (2 * l + 1) for l in range(lmax + 1)])


@lru_cache(maxsize=None)
def s2_grid(lmax: int, n_theta: int, n_phi: int, dtype_str: str,
            device_str: str) -> Tuple[torch.Tensor, torch.Tensor]:
    """(to_grid [G, S], from_grid [S, G]): Gauss-Legendre x uniform
    azimuth quadrature; from_grid integrates f*Y/(4 pi) (component
    normalization).  Exact for band-limited integrands up to the
    quadrature degree; the silu tail beyond it is the (known) eSCN
    approximation."""
    ct, wt = np.polynomial.legendre.leggauss(n_theta)
    phi = np.arange(n_phi) * (2 * np.pi / n_phi)
    ctg, phig = np.meshgrid(ct, phi, indexing="ij")
    st = np.sqrt(1 - ctg ** 2)
    pts = np.stack([st * np.cos(phig), st * np.sin(phig), ctg],
                   axis=-1).reshape(-1, 3)
    w = np.repeat(wt, n_phi) * (2 * np.pi / n_phi)       # sums to 4 pi
    Y = so3.real_sh(torch.tensor(pts), normalize=False).numpy()
    S = (lmax + 1) ** 2
    Y = Y[:, :S]
    to_grid = Y
    from_grid = (Y * w[:, None]).T / (4 * np.pi)
    dt = getattr(torch, dtype_str)
    dev = torch.device(device_str)
    return (torch.tensor(to_grid, dtype=dt, device=dev),
            torch.tensor(from_grid, dtype=dt, device=dev))


class SO2Conv(nn.Module):
    """SO(2)-equivariant edge convolution in the z-aligned edge frame:
    independent mixing per |m| with the complex (w_r, w_i) structure for
    m > 0 — gauge covariance of the azimuthal frame choice, the eSCN
    core op."""

    def __init__(self, cfg: UMAConfig, in_mult: int):
        super().__init__()
        C = cfg.sphere_channels
        self.lmax = cfg.lmax
        # m = 0: rows (l, 0), l = 0..lmax
        n0 = cfg.lmax + 1
        self.w0 = nn.Parameter(torch.empty(n0 * C, in_mult * n0 * C))
        # m > 0: rows (l, +-m), l = m..lmax
        self.wr = nn.ParameterList()
        self.wi = nn.ParameterList()
        for m in range(1, cfg.lmax + 1):
            nl = cfg.lmax + 1 - m
            self.wr.append(nn.Parameter(torch.empty(nl * C,
                                                    in_mult * nl * C)))
            self.wi.append(nn.Parameter(torch.empty(nl * C,
                                                    in_mult * nl * C)))
        # per-edge, per-m modulation from x_edge
        self.rad = nn.Parameter(torch.empty(cfg.lmax + 1, cfg.edge_ch))


@lru_cache(maxsize=None)
def m_indices(lmax: int):
    """Row indices per m: (m0_rows, [(plus_rows, minus_rows)] for m>0).
    Within each l block (offset sum of (2k+1)), m column is l+m."""
    offs = [l * l for l in range(lmax + 1)]          # block offsets
    m0 = [offs[l] + l for l in range(lmax + 1)]
    pm = []
    for m in range(1, lmax + 1):
        plus = [offs[l] + l + m for l in range(m, lmax + 1)]
        minus = [offs[l] + l - m for l in range(m, lmax + 1)]
        pm.append((plus, minus))
    return m0, pm


class UMABlock(nn.Module):
    """One eSCN message-passing block: equivariant norm -> rotated SO(2)
    edge messages with S2 activation -> scatter / avg_degree -> residual;
    then norm -> per-node S2 grid MLP (FFN) -> residual."""

    def __init__(self, cfg: UMAConfig):
        super().__init__()
        C = cfg.sphere_channels
        self.msg = SO2Conv(cfg, in_mult=2)           # src || dst
        self.norm1 = nn.Parameter(torch.ones(cfg.lmax + 1, C))
        self.norm2 = nn.Parameter(torch.ones(cfg.lmax + 1, C))
        self.ffn1 = nn.Parameter(torch.empty(C, C))
        self.ffn2 = nn.Parameter(torch.empty(C, C))
        self.edge_mlp = nn.Sequential(
            nn.Linear(cfg.edge_ch, cfg.edge_ch), nn.SiLU(),
            nn.Linear(cfg.edge_ch, cfg.edge_ch))


class UMACore(nn.Module):
    """All learnable state of the eSCN-MD restatement."""

    def __init__(self, config: UMAConfig | None = None):
        super().__init__()
        cfg = config or UMAConfig()
        self.config = cfg
        C = cfg.sphere_channels

        self.sphere_embedding = nn.Parameter(torch.empty(cfg.n_elements, C))
        self.source_embedding = nn.Parameter(
            torch.empty(cfg.n_elements, cfg.spec_emb))
        self.target_embedding = nn.Parameter(
            torch.empty(cfg.n_elements, cfg.spec_emb))
        self.edge_proj = nn.Parameter(torch.empty(
            cfg.edge_ch, cfg.num_gauss + 2 * cfg.spec_emb))
        # edge-degree embedding: x_edge -> per-l radial profile on the
        # m=0 rows of the edge frame (escn EdgeDegreeEmbedding analog)
        self.edge_degree = nn.Parameter(torch.empty(
            (cfg.lmax + 1) * C, cfg.edge_ch))

        self.blocks = nn.ModuleList(
            [UMABlock(cfg) for _ in range(cfg.num_layers)])
        self.final_norm = nn.Parameter(torch.ones(cfg.lmax + 1, C))
        self.head1 = nn.Parameter(torch.empty(C, C))
        self.head2 = nn.Parameter(torch.empty(C))
        self.register_buffer("scale", torch.tensor(cfg.energy_scale))
        self.register_buffer("shift", torch.tensor(cfg.energy_shift))

    @classmethod
    def seeded(cls, config: UMAConfig | None = None, seed: int = 0,
               dtype: torch.dtype = torch.float32) -> "UMACore":
        g = torch.Generator().manual_seed(seed)
        core = cls(config)
        with torch.no_grad():
            for p in core.parameters():
                if p.dim() == 0:
                    continue
                std = 1.0 / math.sqrt(p.shape[-1]) if p.dim() > 1 else 1.0
                p.copy_(torch.empty_like(p).normal_(0.0, std, generator=g))
            for b in core.blocks:
                b.norm1.fill_(1.0)
                b.norm2.fill_(1.0)
            core.final_norm.fill_(1.0)
        return core.to(dtype)


def gaussian_basis(r: torch.Tensor, cutoff: float, n: int) -> torch.Tensor:
    centers = torch.linspace(0, cutoff, n, dtype=r.dtype, device=r.device)
    width = cutoff / n
    return torch.exp(-((r.unsqueeze(-1) - centers) / width) ** 2)
