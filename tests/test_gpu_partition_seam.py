"""Seam test for the GPU SPMD slab build (CPU-runnable).

`gpu_graph.assemble_partition` is the device-agnostic torch restatement of
the native builder's partition assembly (graph_build.cpp:build_partitions).
Fed the native builder's own GLOBAL edge arrays (same emission order), its
per-rank outputs must be BIT-EQUAL to the native build's partition arrays —
markers, global ids, dst-sorted local edges + CSRs, BDE buckets, line
graph, permutation CSRs.  (Focused == full partition equality is already
pinned by test_focus_build.py; the GPU path swaps only the NL source,
whose per-dst rows are pinned by test_gpu_graph_matches_cpu_builder.)
"""
import numpy as np
import pytest
import torch

from distmlip_amd import gpu_graph
from distmlip_amd.dist import Distributed
from distmlip_amd.structures import diamond_si


def _global_arrays(d):
    src = torch.as_tensor(np.asarray(d.py_index_1), dtype=torch.long)
    dst = torch.as_tensor(np.asarray(d.py_index_2), dtype=torch.long)
    off = torch.as_tensor(np.asarray(d.py_offsets), dtype=torch.float64)
    off_i8 = off.to(torch.int8)
    bond_flag = torch.zeros(len(src), dtype=torch.uint8)
    bond_flag[torch.as_tensor(np.asarray(d.within_r_indices),
                              dtype=torch.long)] = 1
    return src, dst, off_i8, bond_flag


@pytest.mark.parametrize("P", [2, 3, 4])
def test_assemble_partition_bit_equal_native(P):
    reps = (16, 2, 2) if P >= 4 else (12, 2, 2)
    s = diamond_si(reps, jitter=0.12, seed=2)
    d = Distributed.create_distributed(
        s.cart_coords, s.frac_coords, s.lattice, P, s.pbc, 6.0, 3.0,
        use_bond_graph=True, num_threads=4)
    src, dst, off_i8, bond_flag = _global_arrays(d)
    _, _, home = gpu_graph.compute_walls_home(
        s.frac_coords, s.lattice, P, 6.0, 3.0, True)

    for r in range(P):
        pd = gpu_graph.assemble_partition(src, dst, off_i8, bond_flag,
                                          home, P, r, True, s.num_atoms)
        csr = d.csr_parts[r]
        assert np.array_equal(pd.markers, np.asarray(d.markers[r])), r
        assert np.array_equal(pd.global_ids, np.asarray(d.global_ids[r])), r
        assert pd.n_owned == d.num_owned_atoms(r)
        assert np.array_equal(pd.src.numpy(),
                              np.asarray(d.src_nodes[r], dtype=np.int32)), r
        assert np.array_equal(pd.dst.numpy(),
                              np.asarray(d.dst_nodes[r], dtype=np.int32)), r
        assert np.array_equal(pd.row_ptr.numpy(), csr["row_ptr"]), r
        assert np.array_equal(pd.src_perm.numpy(), csr["src_perm"]), r
        assert np.array_equal(pd.src_row_ptr.numpy(), csr["src_row_ptr"]), r
        assert np.array_equal(pd.off_i8.numpy(), csr["offsets_i8"]), r
        # bond layer
        assert np.array_equal(pd.line_markers,
                              np.asarray(d.line_markers[r])), r
        assert pd.n_bonds == d.num_bonds(r)
        assert np.array_equal(pd.map_de.numpy(),
                              np.asarray(d.bond_mapping_DE_list[r])), r
        assert np.array_equal(pd.map_ude.numpy(),
                              np.asarray(d.bond_mapping_UDE_list[r])), r
        assert np.array_equal(pd.l_src.numpy(),
                              np.asarray(d.line_src_nodes[r],
                                         dtype=np.int32)), r
        assert np.array_equal(pd.l_dst.numpy(),
                              np.asarray(d.line_dst_nodes[r],
                                         dtype=np.int32)), r
        assert np.array_equal(
            pd.center.numpy(),
            np.asarray(d.local_center_atom_indices_list[r],
                       dtype=np.int32)), r
        assert np.array_equal(pd.line_row_ptr.numpy(),
                              csr["line_row_ptr"]), r
        assert np.array_equal(pd.line_src_perm.numpy(),
                              csr["line_src_perm"]), r
        assert np.array_equal(pd.line_src_row_ptr.numpy(),
                              csr["line_src_row_ptr"]), r
        assert np.array_equal(pd.center_perm.numpy(), csr["center_perm"]), r
        assert np.array_equal(pd.center_row_ptr.numpy(),
                              csr["center_row_ptr"]), r


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
@pytest.mark.parametrize("P", [2, 4])
def test_gpu_build_partition_matches_native(P):
    """On-device SPMD build vs the native CPU focused build: regions and
    markers bit-equal (set-determined); edge/bond/line layers equal as
    sets (the GPU NL's per-dst row order may differ from the CPU
    builder's global emission order — physics-identical, and
    rank-consistent since every rank builds from the same full NL).

    Cells must satisfy gpu_graph.supported (>= 3 cells of >= cutoff per
    dim) — the engines guard this; a first run of this test with 11 A
    transverse dims silently under-binned the cell list."""
    reps = (16, 4, 4) if P >= 4 else (12, 4, 4)
    s = diamond_si(reps, jitter=0.12, seed=2)
    assert gpu_graph.supported(s, 6.0)
    dev = torch.device("cuda:0")
    for r in range(P):
        pd = gpu_graph.build_partition(s, P, r, 6.0, 3.0, 1e-8, True, dev)
        d = Distributed.create_distributed(
            s.cart_coords, s.frac_coords, s.lattice, P, s.pbc, 6.0, 3.0,
            use_bond_graph=True, num_threads=4, focus_partition=r)
        assert np.array_equal(pd.markers, np.asarray(d.markers[r]))
        assert np.array_equal(pd.line_markers, np.asarray(d.line_markers[r]))
        assert np.array_equal(pd.global_ids, np.asarray(d.global_ids[r]))
        assert pd.n_owned == d.num_owned_atoms(r)
        assert pd.n_bonds == d.num_bonds(r)

        def ekeys(src, dst, off):
            return set(zip(np.asarray(src, dtype=np.int64).tolist(),
                           np.asarray(dst, dtype=np.int64).tolist(),
                           *(np.asarray(off, dtype=np.int64).T.tolist())))
        k_gpu = ekeys(pd.src.cpu(), pd.dst.cpu(), pd.off_i8.cpu())
        k_cpu = ekeys(d.src_nodes[r], d.dst_nodes[r],
                      d.csr_parts[r]["offsets_i8"])
        assert k_gpu == k_cpu
        # owned-bond edge-key sets through map_de
        g_src, g_dst = pd.src.cpu().numpy(), pd.dst.cpu().numpy()
        g_off = pd.off_i8.cpu().numpy()
        mde = pd.map_de.cpu().numpy()
        b_gpu = ekeys(g_src[mde], g_dst[mde], g_off[mde])
        csrc = np.asarray(d.src_nodes[r])
        cdst = np.asarray(d.dst_nodes[r])
        coff = d.csr_parts[r]["offsets_i8"]
        cmde = np.asarray(d.bond_mapping_DE_list[r])
        b_cpu = ekeys(csrc[cmde], cdst[cmde], coff[cmde])
        assert b_gpu == b_cpu


def test_compute_walls_home_matches_builder_regions():
    """Walls/home replication sanity on a no-bond-graph build (the MACE
    path): markers and global ids reproduce for P=2."""
    s = diamond_si((10, 2, 2), jitter=0.1, seed=5)
    P = 2
    d = Distributed.create_distributed(
        s.cart_coords, s.frac_coords, s.lattice, P, s.pbc, 6.0, 0.0,
        use_bond_graph=False, num_threads=4)
    src, dst, off_i8, _ = _global_arrays(d)
    bond_flag = torch.zeros(len(src), dtype=torch.uint8)
    _, _, home = gpu_graph.compute_walls_home(
        s.frac_coords, s.lattice, P, 6.0, 0.0, False)
    for r in range(P):
        pd = gpu_graph.assemble_partition(src, dst, off_i8, bond_flag,
                                          home, P, r, False, s.num_atoms)
        assert np.array_equal(pd.markers, np.asarray(d.markers[r]))
        assert np.array_equal(pd.global_ids, np.asarray(d.global_ids[r]))
        assert np.array_equal(pd.src.numpy(),
                              np.asarray(d.src_nodes[r], dtype=np.int32))
        assert np.array_equal(pd.dst.numpy(),
                              np.asarray(d.dst_nodes[r], dtype=np.int32))
