"""MACE drop-in surface — mirror of the reference's
`ScaleShiftMACE_Dist` (implementations/mace/models.py:40-263): the same
`from_existing` / `enable_distributed_mode(gpus)` / `dist_forward(data,
dist_info, ...)` flow, single process driving one partition per device
with `Distributed.atom_transfer` halos between layers (models.py:165) —
re-implemented over this package's from-scratch MACE restatement
(mace_model / mace_ops) instead of mace-torch/e3nn.

The SPMD production engine (one process per GPU over RCCL) is
distmlip_amd/mace_runtime.MaceSpmdEngine; this class is the
reference-shaped plugin boundary.
"""
from __future__ import annotations

import math
from copy import deepcopy
from typing import Dict, List, Optional

import numpy as np
import torch

from distmlip_amd import mace_ops, so3
from distmlip_amd.dist import Distributed
from distmlip_amd.mace_model import MACECore


def get_neighborhood_dist(positions: np.ndarray, cutoff: float,
                          pbc=None, cell: Optional[np.ndarray] = None,
                          true_self_interaction: bool = False,
                          num_partitions: int = 2):
    """Mirror of the reference's neighbor-list swap-in
    (mace_utils.py:25-78): builds the Distributed partition and returns
    (edge_index [2,E], shifts [E,3], unit_shifts [E,3], cell, dist_info)
    with D = positions[j] - positions[i] + unit_shifts @ cell."""
    import os
    assert not true_self_interaction, \
        "Distributed mode does not support self-edges with zero distance"
    if pbc is None:
        pbc = (False, False, False)
    if cell is None:
        cell = np.identity(3, dtype=float)
    cell = np.asarray(cell, dtype=float).copy()
    identity = np.identity(3, dtype=float)
    max_positions = np.max(np.absolute(positions)) + 1
    for ax in range(3):
        if not pbc[ax]:
            # extend the cell in non-periodic directions (mace_utils
            # 5*cutoff margin, mace_utils.py:50-57)
            cell[ax, :] = max_positions * 5 * cutoff * identity[ax, :]
    pbc_arr = np.array(pbc, dtype=np.int64)
    frac = Distributed.cartesian_to_wrapped_fractional(positions, cell,
                                                       pbc_arr)
    dist_info = Distributed.create_distributed(
        cart_coords=positions, frac_coords=frac, lattice_matrix=cell,
        num_partitions=num_partitions, pbc=pbc_arr, cutoff=cutoff,
        three_body_cutoff=0.0, use_bond_graph=False,
        num_threads=int(os.environ.get("DISTMLIP_NUM_THREADS", 8)))
    edge_index = np.concatenate((np.asarray(dist_info.py_index_1)[None, :],
                                 np.asarray(dist_info.py_index_2)[None, :]),
                                axis=0)
    unit_shifts = np.asarray(dist_info.py_offsets)
    shifts = np.dot(unit_shifts, cell)
    return edge_index, shifts, unit_shifts, cell, dist_info


class MACE_Dist:
    """Reference surface: models.py:224-263 (`enable_distributed_mode`
    replicates blocks per device; `from_existing` adopts an existing
    model's state)."""

    def __init__(self, core: MACECore):
        self.core = core
        self.dist_enabled = False
        self.gpus: List[torch.device] = []

    @classmethod
    def from_existing(cls, model: MACECore) -> "MACE_Dist":
        # reference: model.to("cpu"); copy __dict__ (models.py:255-263)
        return cls(deepcopy(model).to("cpu"))

    def enable_distributed_mode(self, gpus) -> None:
        # reference maps int -> "cuda:i", keeps "cpu" (models.py:224-232)
        self.gpus = [torch.device("cpu") if g == "cpu"
                     else torch.device(f"cuda:{g}") for g in gpus]
        # replicate the model per device (reference deepcopies each block
        # list, models.py:236-252; one deepcopy of the core per device is
        # the same replication)
        self.core_dist = [deepcopy(self.core).to(d).eval()
                          for d in self.gpus]
        for c in self.core_dist:
            c.requires_grad_(False)
        self.dist_enabled = True

    # -- reference dist_forward (models.py:45-220), E+F essentials -------

    def dist_forward(self, data: Dict[str, torch.Tensor], dist_info:
                     Distributed, training: bool = False,
                     compute_force: bool = True,
                     compute_stress: bool = False
                     ) -> Dict[str, Optional[torch.Tensor]]:
        """data: {"positions" [N,3] (grad leaf), "species" [N] long,
        "shifts" [E,3]} — the subset of the reference's AtomicData the
        E+F path reads; edge indices come from dist_info exactly as the
        reference takes them (models.py:97-100)."""
        P = len(self.gpus)
        cfg = self.core.config
        C = cfg.channels
        dev0 = self.gpus[0]

        positions = data["positions"]
        species = data["species"]
        shifts = data["shifts"]
        ft = positions.dtype

        src = torch.as_tensor(np.asarray(dist_info.py_index_1),
                              dtype=torch.long)
        dst = torch.as_tensor(np.asarray(dist_info.py_index_2),
                              dtype=torch.long)
        # vectors on the full graph, under autograd (reference
        # prepare_graph/get_edge_vectors — models.py:60-78)
        vectors = positions[dst] + shifts.to(ft) - positions[src]
        lengths = torch.linalg.norm(vectors, dim=1)

        e0 = self.core_dist[0].atomic_energies.to(dev0)[
            species.to(dev0)].sum()

        vec_d = dist_info.distribute_edge_features(vectors, self.gpus)
        len_d = dist_info.distribute_edge_features(
            lengths.unsqueeze(1), self.gpus)
        spec_d = [torch.as_tensor(
            np.asarray(species)[np.asarray(dist_info.global_ids[p])],
            dtype=torch.long, device=self.gpus[p]) for p in range(P)]
        src_local = [torch.as_tensor(dist_info.src_nodes[p],
                                     dtype=torch.long, device=self.gpus[p])
                     for p in range(P)]
        dst_local = [torch.as_tensor(dist_info.dst_nodes[p],
                                     dtype=torch.long, device=self.gpus[p])
                     for p in range(P)]

        Y_d = [so3.real_sh(vec_d[p]) for p in range(P)]
        ef_d = [mace_ops.bessel_cutoff(len_d[p].squeeze(1), cfg.r_max,
                                       cfg.num_bessel, cfg.cutoff_p)
                for p in range(P)]

        x_d = [{0: self.core_dist[p].node_embedding[spec_d[p]].unsqueeze(-1)}
               for p in range(P)]
        node_es_layers = []

        for i in range(cfg.num_interactions):
            es_d = [None] * P
            for p in range(P):
                core = self.core_dist[p]
                inter, prod = core.interactions[i], core.products[i]
                xd = x_d[p]
                x_up = mace_ops.irreps_linear(inter.linear_up, xd)
                tp_w = inter.radial(ef_d[p]).view(-1, len(inter.paths), C)
                gathered = {l: x_up[l][src_local[p]] for l in x_up}
                msgs = mace_ops.conv_tp_messages(inter, gathered, Y_d[p],
                                                 tp_w)
                n_p = dist_info.num_atoms(p)
                m = {l3: torch.zeros(n_p, C, 2 * l3 + 1, dtype=ft,
                                     device=self.gpus[p]).index_add_(
                        0, dst_local[p], msgs[l3]) for l3 in msgs}
                m = mace_ops.irreps_linear(inter.linear_post, m)
                m = {l: t / cfg.avg_num_neighbors for l, t in m.items()}
                sc = mace_ops.skip_tp(inter, xd, spec_d[p])
                y = mace_ops.symmetric_contract(prod, m, spec_d[p],
                                                cfg.correlation)
                y = mace_ops.irreps_linear(prod.linear, y)
                for l in y:
                    if l in sc:
                        y[l] = y[l] + sc[l]
                x_d[p] = y
                # readout before the transfer (models.py:158-160)
                if i < cfg.num_interactions - 1:
                    es_d[p] = torch.einsum("nc,c->n", y[0][:, :, 0],
                                           core.readout_linear[i])
                else:
                    h = y[0][:, :, 0] @ core.readout_mlp1.t()
                    h = h * torch.sigmoid(h)
                    es_d[p] = h @ core.readout_mlp2

            # atom_transfer between layers (models.py:165): concatenated
            # per-l blocks, one transfer, matching the reference's single
            # node_feats tensor
            if i < cfg.num_interactions - 1:
                ls = sorted(x_d[0])
                flat = [torch.cat([x_d[p][l].reshape(len(x_d[p][l]), -1)
                                   for l in ls], dim=1) for p in range(P)]
                flat = dist_info.atom_transfer(flat)
                for p in range(P):
                    o, xn = 0, {}
                    for l in ls:
                        d = 2 * l + 1
                        xn[l] = flat[p][:, o:o + C * d].view(-1, C, d)
                        o += C * d
                    x_d[p] = xn

            node_es_layers.append(dist_info.aggregate(es_d, dev0))

        core0 = self.core_dist[0]
        node_inter_es = core0.scale * sum(node_es_layers) + core0.shift
        inter_e = node_inter_es.sum()
        total_energy = e0 + inter_e

        out = {"energy": total_energy, "interaction_energy": inter_e,
               "node_energy": node_inter_es.detach(),
               "forces": None, "stress": None}
        if compute_force:
            gv = torch.autograd.grad(inter_e, positions,
                                     retain_graph=compute_stress)
            out["forces"] = -gv[0]
        return out
