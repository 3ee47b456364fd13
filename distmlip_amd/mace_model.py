"""MACE model restatement — weights container + architecture metadata.

The reference distributes an upstream mace-torch ScaleShiftMACE
(implementations/mace/models.py:40-263); mace-torch/e3nn are not
installable here and no checkpoint can be fetched (no network), so — as
with CHGNet (distmlip_amd/model.py) — this is a from-scratch restatement
of the published architecture at the BASELINE config #4 shape
("MACE-MP-0 medium"):

  r_max 6.0, 8 Bessel radial basis x polynomial cutoff (p=5), SH l<=3
  ('component' normalization), num_interactions 2, hidden irreps
  128x0e + 128x1o, correlation 3, readout MLP 16x0e, silu gates,
  RealAgnosticResidualInteractionBlock-style interactions, per-element
  symmetric-contraction weights, energy scale/shift + per-element E0s.

Weight-basis caveat: the symmetric-contraction path
basis spans the identical equivariant function space as e3nn's U-matrix
basis (ranks pinned by character theory in tests/test_so3.py) but in a
different orthonormal orientation, so pretrained mace-torch checkpoints
are not bit-loadable — with seeded random weights (all this environment
can run, reference-equally) the model families coincide.

Layer shapes per interaction i (mace.modules.models.MACE.__init__):
  i=0: node feats 128x0e -> interaction target 128x(0e+1o+2e+3o)
       -> product (corr 3) -> hidden 128x(0e+1o), residual sc
  i=1 (last): node 128x(0e+1o) -> target 128x(0e+1o+2e+3o)
       -> product -> 128x0e only, residual sc; nonlinear readout
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Tuple

import numpy as np
import torch
from torch import nn

from distmlip_amd import so3


@dataclass
class MACEConfig:
    n_elements: int = 4
    channels: int = 128
    l_max: int = 3                    # SH order of edge attrs
    hidden_ls: Tuple[int, ...] = (0, 1)   # 128x0e + 128x1o
    num_interactions: int = 2
    correlation: int = 3
    r_max: float = 6.0
    num_bessel: int = 8
    cutoff_p: int = 5
    radial_mlp: Tuple[int, ...] = (64, 64, 64)
    readout_mlp_dim: int = 16
    avg_num_neighbors: float = 40.0
    atomic_inter_scale: float = 1.0
    atomic_inter_shift: float = 0.0


def tp_paths(in_ls, l_max: int) -> List[Tuple[int, int, int]]:
    """uvu tensor-product paths (l_node, l_sh, l_target) with the O(3)
    parity selection (feature parity (-1)^l throughout, as in MACE where
    hidden/interaction irreps are the proper 0e/1o/2e/3o tower)."""
    out = []
    for l1 in in_ls:
        for l2 in range(l_max + 1):
            for l3 in range(l_max + 1):
                if (l1 + l2 + l3) % 2 == 0 and so3.cg_nonzero(l1, l2, l3):
                    out.append((l1, l2, l3))
    return out


class RadialMLP(nn.Module):
    """[num_bessel] -> hidden... -> n_out, silu between layers, no biases
    (e3nn FullyConnectedNet shape)."""

    def __init__(self, n_in: int, hidden, n_out: int):
        super().__init__()
        dims = [n_in] + list(hidden) + [n_out]
        self.layers = nn.ModuleList(
            [nn.Linear(dims[i], dims[i + 1], bias=False)
             for i in range(len(dims) - 1)])

    def forward(self, x):
        for i, lin in enumerate(self.layers):
            x = lin(x)
            if i < len(self.layers) - 1:
                x = torch.nn.functional.silu(x)
        return x


class IrrepsLinear(nn.Module):
    """o3.Linear restatement: independent channel-mixing weight per l."""

    def __init__(self, ls, channels: int):
        super().__init__()
        self.ls = tuple(ls)
        self.w = nn.ParameterDict({
            str(l): nn.Parameter(torch.empty(channels, channels))
            for l in self.ls})


class Interaction(nn.Module):
    """RealAgnosticResidualInteractionBlock restatement (weights only;
    forwards live in the runtime/oracle orchestrations).

    linear_up (o3.Linear on input irreps) -> conv_tp (uvu, per-edge
    weights from the radial MLP) -> scatter to receivers -> linear_post /
    avg_num_neighbors; sc = skip_tp(node_feats, node_attrs) — a
    per-element channel mix on the irreps shared with hidden_out."""

    def __init__(self, cfg: MACEConfig, in_ls, out_hidden_ls):
        super().__init__()
        C = cfg.channels
        self.in_ls = tuple(in_ls)
        self.target_ls = tuple(range(cfg.l_max + 1))
        self.out_hidden_ls = tuple(out_hidden_ls)
        self.paths = tp_paths(self.in_ls, cfg.l_max)
        self.linear_up = IrrepsLinear(self.in_ls, C)
        self.radial = RadialMLP(cfg.num_bessel, cfg.radial_mlp,
                                len(self.paths) * C)
        self.linear_post = IrrepsLinear(self.target_ls, C)
        # skip_tp: FullyConnectedTensorProduct(in_irreps, Ex0e, hidden_out)
        self.skip_ls = tuple(l for l in self.in_ls if l in self.out_hidden_ls)
        self.skip = nn.ParameterDict({
            str(l): nn.Parameter(torch.empty(cfg.n_elements, C, C))
            for l in self.skip_ls})


class ProductBasis(nn.Module):
    """EquivariantProductBasisBlock restatement: per-element weights for
    the nu = 1..correlation symmetric-contraction paths of each target l,
    followed by an o3.Linear; residual sc added by the caller."""

    def __init__(self, cfg: MACEConfig, out_ls):
        super().__init__()
        C = cfg.channels
        self.out_ls = tuple(out_ls)
        self.weights = nn.ParameterDict()
        for lo in self.out_ls:
            for nu in range(1, cfg.correlation + 1):
                P = so3.symmetric_basis(nu, lo).shape[-1]
                if P:
                    self.weights[f"{lo}_{nu}"] = nn.Parameter(
                        torch.empty(cfg.n_elements, P, C))
        self.linear = IrrepsLinear(self.out_ls, C)


class MACECore(nn.Module):
    """All learnable state of the MACE restatement (config #4 shape)."""

    def __init__(self, config: MACEConfig | None = None):
        super().__init__()
        cfg = config or MACEConfig()
        self.config = cfg
        C = cfg.channels

        self.node_embedding = nn.Parameter(torch.empty(cfg.n_elements, C))
        self.atomic_energies = nn.Parameter(torch.zeros(cfg.n_elements))

        self.interactions = nn.ModuleList()
        self.products = nn.ModuleList()
        in_ls = (0,)
        for i in range(cfg.num_interactions):
            last = i == cfg.num_interactions - 1
            out_ls = (0,) if last else cfg.hidden_ls
            self.interactions.append(Interaction(cfg, in_ls, out_ls))
            self.products.append(ProductBasis(cfg, out_ls))
            in_ls = out_ls

        # readouts: LinearReadoutBlock for all but last (scalar part ->
        # 1x0e), NonLinearReadoutBlock for the last (C -> 16 -> silu -> 1)
        self.readout_linear = nn.ParameterList([
            nn.Parameter(torch.empty(C))
            for _ in range(cfg.num_interactions - 1)])
        self.readout_mlp1 = nn.Parameter(
            torch.empty(cfg.readout_mlp_dim, C))
        self.readout_mlp2 = nn.Parameter(torch.empty(cfg.readout_mlp_dim))

        self.register_buffer("scale", torch.tensor(cfg.atomic_inter_scale))
        self.register_buffer("shift", torch.tensor(cfg.atomic_inter_shift))

    @classmethod
    def seeded(cls, config: MACEConfig | None = None, seed: int = 0,
               dtype: torch.dtype = torch.float32) -> "MACECore":
        """Deterministic random weights, fan-in-scaled so activations stay
        O(1) through both layers and the correlation-3 products."""
        g = torch.Generator().manual_seed(seed)
        core = cls(config)
        C = core.config.channels
        with torch.no_grad():
            for name, p in core.named_parameters():
                if p.dim() == 0:
                    continue
                fan_in = p.shape[-1]
                std = 1.0 / math.sqrt(fan_in)
                p.copy_(torch.empty_like(p).normal_(0.0, std, generator=g))
            core.atomic_energies.normal_(0.0, 1.0, generator=g)
        return core.to(dtype)
