"""Potential_Dist — energy/force/stress assembly over CHGNet_Dist.

API mirror of the reference implementations/matgl/pes.py (Potential_Dist,
pes.py:14-146): per-call graph build via Distributed.create_distributed
(pes.py:75-85), model forward, energy scale/shift + element references
(pes.py:109-113), forces via torch.autograd.backward (pes.py:121-124),
stresses via the strain gradient with the ASE stress scale -160.21766208
eV/A^3 (pes.py:140-145).  Thread count priority: ctor arg, then
DISTMLIP_NUM_THREADS, default 8 (pes.py:66).

Takes distmlip_amd.structures.Structure (or anything with frac_coords /
lattice / species / pbc); an ase.Atoms adapter is Structure.from_ase.
"""
from __future__ import annotations

import os

import numpy as np
import torch

from distmlip_amd.dist import Distributed


class Potential_Dist(torch.nn.Module):
    __version__ = 2

    def __init__(self, model, calc_forces: bool = True, calc_stresses: bool = False,
                 calc_hessian: bool = False, num_threads=None, debug_mode: bool = False,
                 graph_backend=None):
        super().__init__()
        self.model = model
        assert self.model.dist_enabled, "Distributed mode must be enabled"
        assert hasattr(self.model, "gpus"), "Model should have gpus attribute"
        self.calc_forces = calc_forces
        self.calc_stresses = calc_stresses
        if calc_hessian:
            print("Warning: turning off calc_hessian as it is not implemented "
                  "within distributed inference.")
        self.calc_hessian = False
        self.num_threads = num_threads
        self.debug_mode = debug_mode
        self.graph_backend = graph_backend

    def forward(self, structure, state_attr=None, tol: float = 1.0e-8):
        num_threads = self.num_threads if self.num_threads else int(
            os.environ.get("DISTMLIP_NUM_THREADS", 8))

        lattice_matrix = np.asarray(structure.lattice, dtype=float)
        frac_coords = np.asarray(structure.frac_coords, dtype=float)
        cart_coords = frac_coords @ lattice_matrix
        pbc = np.asarray(structure.pbc, dtype=np.int64)
        num_partitions = len(self.model.gpus)

        dist_info = Distributed.create_distributed(
            cart_coords=cart_coords, frac_coords=frac_coords,
            lattice_matrix=lattice_matrix, num_partitions=num_partitions,
            pbc=pbc, use_bond_graph=self.model.use_bond_graph,
            cutoff=float(self.model.cutoff),
            three_body_cutoff=float(self.model.three_body_cutoff),
            tol=tol, num_threads=num_threads, backend=self.graph_backend,
        )
        return self.forward_with_graph(structure, dist_info, lattice_matrix)

    def forward_with_graph(self, structure, dist_info, lattice_matrix=None):
        """Forward on a pre-built Distributed graph (MD-step reuse hook)."""
        if lattice_matrix is None:
            lattice_matrix = np.asarray(structure.lattice, dtype=float)

        self.model.set_local_species(dist_info, structure.species)
        model_out = self.model.potential_forward_dist(
            dist_info, structure, lattice_matrix,
            self.calc_stresses, self.calc_forces, self.calc_hessian, None)

        if self.debug_mode:
            return model_out[-1]

        node_types, positions, strain, predictions = model_out
        total_energies, site_wise = predictions

        core = self.model.core
        total_energies = core.data_std.to(total_energies.device) * total_energies \
            + core.data_mean.to(total_energies.device)          # pes.py:109
        refs = self.model.cores[0].element_refs
        total_energies = total_energies + refs[node_types].sum()  # pes.py:111-113

        forces = None
        stresses = None
        hessian = None
        if self.calc_forces:
            torch.autograd.backward(total_energies)              # pes.py:121-124
            forces = -positions.grad
        if self.calc_stresses:
            volume = np.abs(np.linalg.det(lattice_matrix))
            sts = -strain.grad
            stresses = sts * (1.0 / volume * -160.21766208)      # pes.py:140-145
        return total_energies, forces, stresses, hessian
