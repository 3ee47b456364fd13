"""CHGNet parameter container + config.

The reference delegates all model arithmetic to matgl (pinned @5171392,
/root/reference/pyproject.toml:27; imports at
implementations/matgl/models/chgnet.py:8-14), which is not installed in
this container and ships no numeric tests — so the arithmetic here is OUR
restatement of the CHGNet architecture (Deng et al., CHGNet; matgl
implementation symbols `CHGNet`, `CHGNetGraphConv`, `CHGNetLineGraphConv`,
`RadialBesselFunction`, `FourierExpansion`, `polynomial_cutoff`) with the
feature dimensions SURVEY.md §8 records for the reference's default model
(node/edge/angle dim 64, RBF 9, Fourier basis 21, n_blocks 4).  Parity at
the matgl boundary is *unpinned* (oracle/__init__.py); the executable
contract is: oracle (oracle/chgnet_ref.py) == distributed product forward,
bit-for-bit in the graph layer and within stated fp tolerance in the
model layer.

This module holds ONLY parameters/config (shared by oracle and product);
no graph ops, no device logic.
"""
from __future__ import annotations

from dataclasses import dataclass, field

import math
import torch
from torch import nn

N_ELEMENTS_DEFAULT = 89  # matgl DEFAULT_ELEMENTS table size


@dataclass
class CHGNetConfig:
    n_elements: int = N_ELEMENTS_DEFAULT
    dim: int = 64                 # atom/bond/angle feature dim
    rbf_n: int = 9                # radial Bessel functions (max_n)
    max_f: int = 10               # Fourier frequencies -> 2*max_f+1 = 21 basis
    n_blocks: int = 4             # atom-graph conv blocks (bond blocks = n_blocks-1)
    cutoff: float = 6.0
    three_body_cutoff: float = 3.0
    cutoff_exponent: int = 5
    use_bond_graph: bool = True
    data_mean: float = 0.0
    data_std: float = 1.0
    readout_field: str = "atom_feat"
    readout_operation: str = "sum"

    @property
    def n_fourier(self) -> int:
        return 2 * self.max_f + 1


class GatedMLP(nn.Module):
    """x -> silu(L2(silu(L1(x)))) * sigmoid(G2(silu(G1(x)))).

    Restatement of matgl's GatedMLP as used by CHGNetGraphConv /
    CHGNetLineGraphConv (imported at reference chgnet.py:8,
    chgnet_layers.py:1): a core branch with SiLU on hidden and output, and
    a gate branch ending in a sigmoid.
    """

    def __init__(self, in_dim: int, hidden: int, out_dim: int):
        super().__init__()
        self.core1 = nn.Linear(in_dim, hidden)
        self.core2 = nn.Linear(hidden, out_dim)
        self.gate1 = nn.Linear(in_dim, hidden)
        self.gate2 = nn.Linear(hidden, out_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        core = torch.nn.functional.silu(self.core2(torch.nn.functional.silu(self.core1(x))))
        gate = torch.sigmoid(self.gate2(torch.nn.functional.silu(self.gate1(x))))
        return core * gate


class AtomConv(nn.Module):
    """One CHGNet atom-graph block (matgl CHGNetAtomGraphBlock analog).

    edge update : e' = e + edge_mlp(cat(v_src, v_dst, e)) * w_bb
    node update : v' = v + sum_{e: dst=i} node_mlp(cat(v_src, v_dst, e')) * w_ab
    (edge update first, node update sees updated edges — matgl
    CHGNetGraphConv.forward order.)
    """

    def __init__(self, dim: int):
        super().__init__()
        self.edge_mlp = GatedMLP(3 * dim, dim, dim)
        self.node_mlp = GatedMLP(3 * dim, dim, dim)


class BondConv(nn.Module):
    """One CHGNet bond-graph (line-graph) block (CHGNetBondGraphBlock analog).

    Line edge l = (b1 -> b2) with center atom c and angle feature a_l:
      bond update  : n'[b2] = n[b2] + sum_l bond_mlp(cat(n_b1, n_b2, a_l, v_c)) * w_3b[b1]
      angle update : a'     = a     + angle_mlp(cat(n'_b1, n'_b2, a_l, v_c))
    (node pass then edge pass, as the reference drives it with
    convolution_type="node" / "edge" — chgnet.py:326-368.)
    """

    def __init__(self, dim: int):
        super().__init__()
        self.bond_mlp = GatedMLP(4 * dim, dim, dim)
        self.angle_mlp = GatedMLP(4 * dim, dim, dim)


class CHGNetCore(nn.Module):
    """All learnable state of the CHGNet restatement (shared oracle/product)."""

    def __init__(self, config: CHGNetConfig | None = None):
        super().__init__()
        cfg = config or CHGNetConfig()
        self.config = cfg
        d, nr, nf = cfg.dim, cfg.rbf_n, cfg.n_fourier

        self.atom_embedding = nn.Embedding(cfg.n_elements, d)
        self.bond_embedding = nn.Linear(nr, d, bias=False)
        self.angle_embedding = nn.Linear(nf, d, bias=False)

        # Learnable radial Bessel frequencies, init n*pi (matgl
        # RadialBesselFunction(learnable=True)).
        self.rbf_freq_atom = nn.Parameter(
            torch.arange(1, nr + 1, dtype=torch.get_default_dtype()) * math.pi)
        self.rbf_freq_bond = nn.Parameter(
            torch.arange(1, nr + 1, dtype=torch.get_default_dtype()) * math.pi)
        # Learnable Fourier frequencies, init 1..max_f (matgl FourierExpansion).
        self.angle_freq = nn.Parameter(
            torch.arange(1, cfg.max_f + 1, dtype=torch.get_default_dtype()))

        # Shared message weights (reference chgnet.py:272-294:
        # atom_bond_weights / bond_bond_weights / threebody_bond_weights,
        # each a bias-free Linear rbf -> dim).
        self.atom_bond_weights = nn.Linear(nr, d, bias=False)
        self.bond_bond_weights = nn.Linear(nr, d, bias=False)
        self.threebody_bond_weights = nn.Linear(nr, d, bias=False)

        self.atom_convs = nn.ModuleList([AtomConv(d) for _ in range(cfg.n_blocks)])
        self.bond_convs = nn.ModuleList(
            [BondConv(d) for _ in range(cfg.n_blocks - 1)]) if cfg.use_bond_graph else None

        self.sitewise_readout = nn.Linear(d, 1)
        self.final_layer = nn.Sequential(
            nn.Linear(d, d), nn.SiLU(), nn.Linear(d, d), nn.SiLU(), nn.Linear(d, 1))

        self.register_buffer("data_mean", torch.tensor(cfg.data_mean))
        self.register_buffer("data_std", torch.tensor(cfg.data_std))
        # Per-element energy reference offsets (matgl AtomRef analog;
        # applied at reference pes.py:111-113).
        self.element_refs = nn.Parameter(torch.zeros(cfg.n_elements))

    @classmethod
    def seeded(cls, config: CHGNetConfig | None = None, seed: int = 0,
               dtype: torch.dtype = torch.float32) -> "CHGNetCore":
        """Deterministic random-init weights (the bench/parity model)."""
        g = torch.Generator().manual_seed(seed)
        core = cls(config)
        with torch.no_grad():
            for p in core.parameters():
                if p.dim() >= 1 and p is not core.rbf_freq_atom \
                        and p is not core.rbf_freq_bond and p is not core.angle_freq:
                    p.copy_(torch.empty_like(p).normal_(0.0, 0.2, generator=g))
            core.element_refs.normal_(0.0, 0.1, generator=g)
        return core.to(dtype)


# ---- basis functions (shared restatements) --------------------------------

def radial_bessel(dist: torch.Tensor, freqs: torch.Tensor, cutoff: float) -> torch.Tensor:
    """sqrt(2/c) * sin(f_n * r / c) / r  — matgl RadialBesselFunction."""
    d = dist.unsqueeze(-1)
    return math.sqrt(2.0 / cutoff) * torch.sin(freqs * d / cutoff) / d


def polynomial_cutoff(x: torch.Tensor, cutoff: float, exponent: int = 5) -> torch.Tensor:
    """matgl polynomial envelope (utils/cutoff.py restatement).

    NOTE the reference applies this to the RBF OUTPUT, not the distance
    (chgnet.py:119-121 passes `bond_expansion` as r) — we replicate that
    call pattern literally in both oracle and product.
    """
    e = exponent
    ratio = x / cutoff
    c1 = -(e + 1) * (e + 2) / 2.0
    c2 = float(e * (e + 2))
    c3 = -e * (e + 1) / 2.0
    env = 1.0 + c1 * ratio ** e + c2 * ratio ** (e + 1) + c3 * ratio ** (e + 2)
    return torch.where(x <= cutoff, env, torch.zeros((), dtype=x.dtype, device=x.device))


def bond_expansion_from_dist(dist: torch.Tensor, freqs: torch.Tensor,
                             cutoff: float, exponent: int) -> torch.Tensor:
    """RBF then the reference's literal smooth-cutoff application
    (chgnet.py:115-124): expansion * polynomial_cutoff(expansion)."""
    rbf = radial_bessel(dist, freqs, cutoff)
    return polynomial_cutoff(rbf, cutoff, exponent) * rbf


def fourier_expansion(theta: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    """[1/2, cos(f_k * theta), sin(f_k * theta)] -> 2*max_f+1 dims
    (matgl FourierExpansion restatement; learnable freqs init 1..max_f)."""
    t = theta.unsqueeze(-1) * freqs
    half = torch.full_like(theta.unsqueeze(-1), 0.5)
    return torch.cat([half, torch.cos(t), torch.sin(t)], dim=-1)


def compute_theta(src_bond_vec: torch.Tensor, dst_bond_vec: torch.Tensor,
                  eps: float = 1e-7) -> torch.Tensor:
    """Angle at the center atom between (-u) and v.

    Restates matgl compute_theta as driven by the reference: line edge
    (b1=(a->b) -> b2=(b->c)), u = bond_vec(b1) = b-a, v = bond_vec(b2) =
    c-b; src_bond_sign = -1 for all bonds (chgnet.py:190-194), so the
    angle is between (b->a) and (b->c).
    """
    u = -src_bond_vec
    v = dst_bond_vec
    cos = (u * v).sum(-1) / (
        torch.linalg.norm(u, dim=-1) * torch.linalg.norm(v, dim=-1))
    return torch.acos(cos.clamp(-1.0 + eps, 1.0 - eps))
