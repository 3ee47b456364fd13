"""Distributed graph object — the partition/halo contract of the build.

API mirror of the reference `DistMLIP/distributed/dist.py` (class
Distributed, dist.py:8-721): same constructor fields, same marker
semantics (dist.py:44-51), same `transfer_nodes` / `atom_transfer` /
`bond_transfer` / `aggregate` / `global_to_local_*` / `edge_to_bond` /
`bond_to_edge` behavior — re-implemented from the documented contract, not
translated.  Excluded on purpose: the dead/broken
`aggregate_bond_node` (reference dist.py:421-460) and the unreliable
`G2L_DE_mapping_list` (dist.py:91-92).

Marker layout, per partition (reference dist.py:44-51 + utils.c:1102-1154):
    [0, end_pure, end_to_0, ..., end_to_{P-1}, end_from_0, ..., end_from_{P-1}, total]
    length 2P+2 after the Python layer appends the total (dist.py:234-249).
    Nodes are ordered [pure | to_0.. | from_0..]; pure+to = OWNED nodes,
    from_* = GHOSTS owned by other partitions.  `transfer_nodes` copies the
    "to q" slice of partition p into the "from p" slice of partition q
    (dist.py:344-356).  Edges are owned by the partition of their dst node
    (utils.c:206, 235).

Two transfer transports:
  * single-process (list of per-partition tensors, any devices): direct
    slice copies, exactly the reference's mechanism (dist.py:356);
  * SPMD (one process per GPU over torch.distributed / RCCL): see
    distmlip_amd/runtime.py (HaloExchange autograd function).
"""
from __future__ import annotations

from typing import List, Optional, Union

import numpy as np
import torch


class Distributed:
    """Distributed graph for parallelized MLIP inference."""

    def __init__(
        self,
        src_nodes: List[np.ndarray],
        dst_nodes: List[np.ndarray],
        markers: List[np.ndarray],
        local_coords: List[np.ndarray],
        global_ids: List[np.ndarray],
        py_index_1: np.ndarray,
        py_index_2: np.ndarray,
        py_offsets: np.ndarray,
        py_distances: np.ndarray,
        line_src_nodes: List[np.ndarray],
        line_dst_nodes: List[np.ndarray],
        within_r_indices: np.ndarray,
        line_markers: Optional[List[np.ndarray]],
        num_UDEs_per_partition: List[int],
        bond_mapping_DE_list: List[np.ndarray],
        bond_mapping_UDE_list: List[np.ndarray],
        L2G_DE_mapping_list: List[np.ndarray],
        local_center_atom_indices_list: List[np.ndarray],
        use_bond_graph: bool,
        total_num_nodes: int,
        bde_global_edge_list: Optional[List[np.ndarray]] = None,
    ) -> None:
        self.src_nodes = src_nodes
        self.dst_nodes = dst_nodes
        self.markers = markers
        self.local_coords = local_coords
        self.global_ids = global_ids
        self.py_index_1 = py_index_1
        self.py_index_2 = py_index_2
        self.py_offsets = py_offsets
        self.py_distances = py_distances
        self.line_src_nodes = line_src_nodes
        self.line_dst_nodes = line_dst_nodes
        self.within_r_indices = within_r_indices
        self.line_markers = line_markers
        self.num_UDEs_per_partition = num_UDEs_per_partition
        self.bond_mapping_DE_list = bond_mapping_DE_list
        self.bond_mapping_UDE_list = bond_mapping_UDE_list
        self.L2G_DE_mapping_list = L2G_DE_mapping_list
        self.local_center_atom_indices_list = local_center_atom_indices_list
        self.use_bond_graph = use_bond_graph

        # Build extension (not in the reference): per-BDE global atom-edge
        # index, incl. ghost BDEs — lets each rank compute ghost bond
        # GEOMETRY locally instead of transferring it (SURVEY §7 hard
        # part (d): removes the reference's GPU0-centralized bond_vec).
        self.bde_global_edge_list = bde_global_edge_list

        self.num_partitions = len(src_nodes)
        self.total_num_edges = len(py_index_1)
        self.total_num_nodes = total_num_nodes

    # -- construction -----------------------------------------------------

    @staticmethod
    def cartesian_to_wrapped_fractional(positions_cartesian, lattice, pbc):
        """Cartesian -> fractional, wrapped on periodic axes (reference
        dist.py:129-157).  Divergence: the reference short-circuits the
        fully non-periodic case by returning CARTESIAN coordinates; we
        always convert properly (the huge padded cell the MACE adapter
        builds makes the reference's shortcut approximately work — ours
        is exact for the same inputs)."""
        lattice = np.asarray(lattice, dtype=float)
        frac = np.linalg.solve(lattice.T,
                               np.transpose(positions_cartesian)).T
        for i, periodic in enumerate(np.asarray(pbc).astype(bool)):
            if periodic:
                frac[:, i] %= 1.0
                frac[:, i] %= 1.0
        return frac

    @classmethod
    def create_distributed(
        cls,
        cart_coords: np.ndarray,
        frac_coords: np.ndarray,
        lattice_matrix: np.ndarray,
        num_partitions: int,
        pbc: np.ndarray,
        cutoff: float,
        three_body_cutoff: float = 0,
        tol: float = 1e-8,
        use_bond_graph: bool = False,
        num_threads: int = 1,
        backend=None,
        focus_partition: int = -1,
    ) -> "Distributed":
        """Partition a periodic structure and build the Distributed graph.

        Signature mirror of reference dist.py:158-232.  `backend` is a
        callable with the get_subgraphs_fast signature (fast.c:102-113);
        default = the build's native C++ builder (distmlip_amd.capi).
        Unlike the reference (utils.c:48-52), num_partitions == 1 is
        allowed: it yields a single partition with all nodes pure and no
        halo (the 1-GPU path).
        """
        cart_coords = np.ascontiguousarray(cart_coords, dtype=float)
        frac_coords = np.ascontiguousarray(frac_coords, dtype=float)
        lattice_matrix = np.ascontiguousarray(lattice_matrix, dtype=float)

        csr_parts = None
        if backend is None:
            from distmlip_amd import capi
            backend = capi.get_subgraphs_fast
            # native builder: keep the per-partition CSR layouts (scatter-add
            # row pointers + permutation CSRs) for the HIP kernel path
            out = backend(
                cart_coords, float(cutoff), np.asarray(pbc, dtype=np.int64),
                lattice_matrix, int(num_partitions), float(three_body_cutoff),
                float(tol), int(num_threads), bool(use_bond_graph), frac_coords,
                return_csr=True, focus=int(focus_partition),
            )
            out, csr_parts = out
        else:
            out = backend(
                cart_coords, float(cutoff), np.asarray(pbc, dtype=np.int64),
                lattice_matrix, int(num_partitions), float(three_body_cutoff),
                float(tol), int(num_threads), bool(use_bond_graph), frac_coords,
            )
        (src_nodes, dst_nodes, markers, local_coords, global_ids,
         py_index_1, py_index_2, py_offsets, py_distances,
         line_src_nodes, line_dst_nodes, within_r_indices, line_markers,
         num_UDEs_per_partition, bond_mapping_DE_list, bond_mapping_UDE_list,
         L2G_DE_mapping_list, _G2L_unused, local_center_atom_indices_list) = out[:19]
        bde_global_edge_list = out[19] if len(out) > 19 else None

        # Append the total count to each marker array (reference
        # dist.py:234-249): final marker length 2P+2.
        markers = [np.append(m, len(local_coords[i])) for i, m in enumerate(markers)]
        if use_bond_graph:
            line_markers = [
                np.append(line_markers[i], num_UDEs_per_partition[i])
                for i in range(len(line_markers))
            ]
        else:
            line_markers = None

        obj = cls(
            src_nodes, dst_nodes, markers, local_coords, global_ids,
            py_index_1, py_index_2, py_offsets, py_distances,
            line_src_nodes, line_dst_nodes, within_r_indices, line_markers,
            num_UDEs_per_partition, bond_mapping_DE_list,
            bond_mapping_UDE_list, L2G_DE_mapping_list,
            local_center_atom_indices_list, use_bond_graph, len(cart_coords),
            bde_global_edge_list=bde_global_edge_list,
        )
        obj.csr_parts = csr_parts
        return obj

    # -- size queries (reference dist.py:462-551) -------------------------

    def num_atoms(self, partition: int) -> int:
        return len(self.local_coords[partition])

    def num_atom_edges(self, partition: int) -> int:
        return len(self.src_nodes[partition])

    def num_bonds(self, partition: int) -> int:
        assert self.use_bond_graph
        return int(self.line_markers[partition][-1])

    def num_bond_edges(self, partition: int) -> int:
        assert self.use_bond_graph
        return len(self.line_src_nodes[partition])

    def num_owned_atoms(self, partition: int) -> int:
        """pure + to regions (the nodes this partition computes)."""
        return int(self.markers[partition][1 + self.num_partitions])

    def num_owned_bonds(self, partition: int) -> int:
        assert self.use_bond_graph
        return int(self.line_markers[partition][1 + self.num_partitions])

    def num_atom_border_nodes(self, partition: int) -> int:
        return self.num_atoms(partition) - self.num_owned_atoms(partition)

    def num_bond_border_nodes(self, partition: int) -> int:
        return self.num_bonds(partition) - self.num_owned_bonds(partition)

    # -- single-process transfers (reference dist.py:277-388) -------------

    def aggregate(
        self,
        features_to_aggregate: List[torch.Tensor],
        device: Union[str, torch.device] = "cpu",
        aggregate_dim: Optional[int] = None,
    ) -> torch.Tensor:
        """Gather OWNED-node features (pure+to) into a global tensor."""
        if not aggregate_dim:
            aggregate_dim = self.total_num_nodes
        combined = torch.empty(
            (aggregate_dim,) + tuple(features_to_aggregate[0].shape[1:]),
            device=device, dtype=features_to_aggregate[0].dtype,
        )
        for p in range(self.num_partitions):
            cut = self.num_owned_atoms(p)
            ids = torch.as_tensor(self.global_ids[p][:cut], dtype=torch.long,
                                  device=device)
            combined[ids] = features_to_aggregate[p][:cut].to(device)
        return combined

    def transfer_nodes(self, features: List[torch.Tensor],
                       markers: List[np.ndarray]) -> List[torch.Tensor]:
        """Copy each partition's "to q" slice into q's "from p" slice.

        In-place, differentiable through torch's slice-copy autograd —
        exactly the reference mechanism (dist.py:344-356).
        """
        assert len(features) == self.num_partitions
        P = self.num_partitions
        src_slices = [[] for _ in range(P)]  # gather sources first: all-old-values
        for curr in range(P):
            for to in range(P):
                if curr == to:
                    continue
                fs, fe = int(markers[curr][1 + to]), int(markers[curr][1 + to + 1])
                ts, te = int(markers[to][1 + P + curr]), int(markers[to][1 + P + curr + 1])
                if fs != fe:
                    src_slices[to].append((ts, te, features[curr][fs:fe]))
        for to in range(P):
            for ts, te, sl in src_slices[to]:
                idx = torch.arange(ts, te, device=features[to].device)
                # out-of-place index_copy keeps autograd versioning clean while
                # preserving the reference's slice-copy semantics (dist.py:356)
                features[to] = features[to].index_copy(0, idx, sl.to(features[to].device))
        return features

    def atom_transfer(self, features: List[torch.Tensor]) -> List[torch.Tensor]:
        return self.transfer_nodes(features, self.markers)

    def bond_transfer(self, features: List[torch.Tensor]) -> List[torch.Tensor]:
        assert self.use_bond_graph, \
            "Cannot transfer border nodes if use_bond_graph is False"
        return self.transfer_nodes(features, self.line_markers)

    # -- global <-> local distribution (reference dist.py:553-633) --------

    def global_to_local_nodes(self, global_node_features: torch.Tensor,
                              partition: int,
                              device: Union[str, torch.device] = "cpu") -> torch.Tensor:
        ids = torch.as_tensor(self.global_ids[partition], dtype=torch.long,
                              device=global_node_features.device)
        return global_node_features[ids].to(device)

    def global_to_local_edges(self, global_edge_features: torch.Tensor,
                              partition: int,
                              device: Union[str, torch.device] = "cpu") -> torch.Tensor:
        ids = torch.as_tensor(self.L2G_DE_mapping_list[partition], dtype=torch.long,
                              device=global_edge_features.device)
        return global_edge_features[ids].to(device)

    def distribute_node_features(self, global_node_features, devices):
        assert len(devices) == self.num_partitions
        return [self.global_to_local_nodes(global_node_features, i, devices[i])
                for i in range(len(devices))]

    def distribute_edge_features(self, global_edges_features, devices):
        assert len(devices) == self.num_partitions
        return [self.global_to_local_edges(global_edges_features, i, devices[i])
                for i in range(len(devices))]

    # -- atom-edge <-> bond-node remaps (reference dist.py:635-702) -------

    def edge_to_bond(self, edge_features, partition: int,
                     device: Union[str, torch.device] = "cpu",
                     inplace: bool = False, bond_node_features=None):
        """Scatter local atom-edge features into the owned bond-node slots."""
        ude = torch.as_tensor(self.bond_mapping_UDE_list[partition], dtype=torch.long)
        de = torch.as_tensor(self.bond_mapping_DE_list[partition], dtype=torch.long)
        if inplace:
            # "inplace" keeps the reference's call shape (dist.py:662-669) but
            # swaps the list entry for an out-of-place index_copy result.
            assert bond_node_features is not None
            dev = bond_node_features[partition].device
            src = edge_features[partition][de.to(edge_features[partition].device)].to(dev)
            bond_node_features[partition] = \
                bond_node_features[partition].index_copy(0, ude.to(dev), src)
            return None
        bond_features = torch.zeros(
            (self.num_bonds(partition),) + tuple(edge_features.shape[1:]),
            device=device, dtype=edge_features.dtype)
        src = edge_features[de.to(edge_features.device)].to(device)
        return bond_features.index_copy(0, ude.to(device), src)

    def bond_to_edge(self, bond_node_features, atom_edge_features,
                     partition: int) -> None:
        ude = torch.as_tensor(self.bond_mapping_UDE_list[partition], dtype=torch.long)
        de = torch.as_tensor(self.bond_mapping_DE_list[partition], dtype=torch.long)
        dev = atom_edge_features[partition].device
        src = bond_node_features[partition][
            ude.to(bond_node_features[partition].device)].to(dev)
        atom_edge_features[partition] = \
            atom_edge_features[partition].index_copy(0, de.to(dev), src)

    def __repr__(self):
        val = (f"Distributed:\n    Total num atoms: {self.total_num_nodes}\n"
               f"    Total num edges: {self.total_num_edges}\n"
               f"    Bond graph exists: {self.use_bond_graph}\n")
        for i in range(self.num_partitions):
            val += (f"Partition {i}:\n\t# atom nodes: {self.num_atoms(i)} "
                    f"({self.num_atom_border_nodes(i)} border)\n"
                    f"\t# atom edges: {len(self.src_nodes[i])}\n")
            if self.use_bond_graph:
                val += (f"\t# bond nodes: {self.num_bonds(i)} "
                        f"({self.num_bond_border_nodes(i)} border)\n"
                        f"\t# bond edges: {self.num_bond_edges(i)}\n")
        return val
