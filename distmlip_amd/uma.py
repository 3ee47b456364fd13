"""UMA drop-in surface — mirror of the reference's monkey-patched
eSCN-MD backbone (implementations/uma/escn_md.py:525-570:
`from_existing` attaches a distributed `forward` +
`enable_distributed_mode(gpus)`; the forward takes a data dict and runs
one partition per device with `Distributed.atom_transfer` halos between
layers) — re-implemented over this package's from-scratch eSCN
restatement (uma_model / uma_ops).

The reference parallelizes partitions with a ThreadPoolExecutor + CUDA
events (escn_md.py:442-500); this mirror relies on async kernel launch
per device within the partition loop — the MI355X-native production mode
is one PROCESS per GPU (uma_runtime.UmaSpmdEngine), where the overlap
question disappears.
"""
from __future__ import annotations

from copy import deepcopy
from typing import Dict, List, Optional

import numpy as np
import torch

from distmlip_amd import so3, uma_ops
from distmlip_amd.dist import Distributed
from distmlip_amd.uma_model import UMACore


class UMA_Dist:
    """Reference surface: from_existing / enable_distributed_mode /
    forward(data_dict) (escn_md.py:249-523, 525-570)."""

    def __init__(self, core: UMACore):
        self.core = core
        self.dist_enabled = False
        self.gpus: List[torch.device] = []

    @classmethod
    def from_existing(cls, model: UMACore) -> "UMA_Dist":
        return cls(deepcopy(model).to("cpu"))

    def enable_distributed_mode(self, gpus) -> None:
        assert not self.dist_enabled, \
            "Distributed mode already enabled. Create a new UMA model " \
            "if you wish to change the GPUs."      # escn_md.py:526-527
        self.gpus = [torch.device("cpu") if g == "cpu"
                     else torch.device(f"cuda:{g}") for g in gpus]
        self.core_dist = [deepcopy(self.core).to(d).eval()
                          for d in self.gpus]
        for c in self.core_dist:
            c.requires_grad_(False)
        self.dist_enabled = True

    def forward(self, data: Dict[str, torch.Tensor],
                dist_info: Distributed,
                compute_force: bool = True) -> Dict[str, torch.Tensor]:
        """data: {"positions" [N,3] (grad leaf), "species" [N] long,
        "shifts" [E,3]}; edge indices come from dist_info (the
        _generate_graph dist branch, compute.py:74-92)."""
        P = len(self.gpus)
        cfg = self.core.config
        C, S = cfg.sphere_channels, cfg.S
        dev0 = self.gpus[0]

        positions = data["positions"]
        species = data["species"]
        shifts = data["shifts"]
        ft = positions.dtype

        src = torch.as_tensor(np.asarray(dist_info.py_index_1),
                              dtype=torch.long)
        dst = torch.as_tensor(np.asarray(dist_info.py_index_2),
                              dtype=torch.long)
        vectors = positions[dst] + shifts.to(ft) - positions[src]
        lengths = torch.linalg.norm(vectors, dim=1)

        # per-partition wigner matrices (escn_md.py:283-291 pattern)
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

        D_d, Dinv_d, xe_d, x_d = [], [], [], []
        for p in range(P):
            core = self.core_dist[p]
            R = so3.edge_align_rotation(vec_d[p])
            D = so3.wigner_D_batch(R, cfg.lmax)
            D_d.append(D)
            Dinv_d.append(D.transpose(-1, -2))
            xe_d.append(uma_ops.edge_scalars(
                core, len_d[p].squeeze(1), spec_d[p][src_local[p]],
                spec_d[p][dst_local[p]]))
            x = vec_d[p].new_zeros(len(spec_d[p]), S, C)
            x[:, 0, :] = core.sphere_embedding[spec_d[p]]
            med = uma_ops.edge_degree_embed(core, xe_d[p], Dinv_d[p])
            n_p = dist_info.num_atoms(p)
            x = x + torch.zeros_like(x).index_add_(
                0, dst_local[p], med) / cfg.avg_degree
            x_d.append(x)
        x_d = self._transfer(dist_info, x_d, S, C)     # escn_md.py:416

        for li in range(cfg.num_layers):
            for p in range(P):
                core = self.core_dist[p]
                blk = core.blocks[li]
                h = uma_ops.rms_norm(x_d[p], blk.norm1, cfg.lmax)
                msg = uma_ops.block_message(
                    blk, cfg, h[src_local[p]], h[dst_local[p]], xe_d[p],
                    D_d[p], Dinv_d[p])
                x2 = x_d[p] + torch.zeros_like(x_d[p]).index_add_(
                    0, dst_local[p], msg) / cfg.avg_degree
                x_d[p] = x2 + uma_ops.node_ffn(blk, cfg, x2)
            if li < cfg.num_layers - 1:
                x_d = self._transfer(dist_info, x_d, S, C)

        # aggregate then final head on device 0 (escn_md.py:500-503)
        x = dist_info.aggregate(
            [x_d[p].reshape(len(x_d[p]), -1) for p in range(P)],
            dev0).view(-1, S, C)
        es = uma_ops.energy_head(self.core_dist[0], x)
        total = es.sum()
        out = {"energy": total, "node_energy": es.detach(),
               "forces": None}
        if compute_force:
            gv = torch.autograd.grad(total, positions)
            out["forces"] = -gv[0]
        return out

    @staticmethod
    def _transfer(dist_info, x_d, S, C):
        flat = [x.reshape(len(x), -1) for x in x_d]
        flat = dist_info.atom_transfer(flat)
        return [f.view(-1, S, C) for f in flat]
