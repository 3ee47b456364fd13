"""ctypes bindings over the build's C-ABI shared libraries.

Two libraries, both built in-tree by distmlip_amd/csrc/Makefile (the .so
files are gitignored but travel with the gpurun snapshot):

  libdistmlip_graph.so — native graph builder (CPU/OpenMP); ABI in
      include/distmlip_graph.h.  `get_subgraphs_fast` below mirrors the
      reference's Python entry point (subgraph_creation_fast.c:92-453
      19-tuple) plus one build extension (per-BDE global edge ids).
  libdistmlip_hip.so  — HIP kernels for gfx950; ABI in
      include/distmlip_hip.h; bound in distmlip_amd/ops.py.

Everything fails LOUDLY if a library is missing — there is no fallback.
"""
from __future__ import annotations

import ctypes
import os
from ctypes import (POINTER, c_char_p, c_double, c_int32, c_int64, c_void_p)

import numpy as np

_CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")


def _load(name: str) -> ctypes.CDLL:
    path = os.path.join(_CSRC, name)
    if not os.path.exists(path):
        raise RuntimeError(
            f"{path} not built. Run `make -C distmlip_amd/csrc` (or "
            f"__graft_entry__.build()). The product path has no fallback.")
    return ctypes.CDLL(path)


_graph_lib = None


def graph_lib() -> ctypes.CDLL:
    global _graph_lib
    if _graph_lib is None:
        lib = _load("libdistmlip_graph.so")
        lib.dm_graph_build.restype = c_int32
        lib.dm_graph_build.argtypes = [
            POINTER(c_double), POINTER(c_double), POINTER(c_int64), c_int64,
            c_double, c_double, c_double, c_int32, c_int32, c_int32,
            POINTER(c_void_p)]
        lib.dm_graph_build_focus.restype = c_int32
        lib.dm_graph_build_focus.argtypes = [
            POINTER(c_double), POINTER(c_double), POINTER(c_int64), c_int64,
            c_double, c_double, c_double, c_int32, c_int32, c_int32, c_int32,
            POINTER(c_void_p)]
        lib.dm_graph_global_view.restype = c_int32
        lib.dm_graph_partition_view.restype = c_int32
        lib.dm_graph_free.argtypes = [c_void_p]
        lib.dm_last_error.restype = c_char_p
        _graph_lib = lib
    return _graph_lib


class _GlobalView(ctypes.Structure):
    _fields_ = [("n_atoms", c_int64), ("n_edges", c_int64),
                ("n_within", c_int64), ("num_partitions", c_int32),
                ("src", POINTER(c_int64)), ("dst", POINTER(c_int64)),
                ("offsets", POINTER(c_double)), ("dist", POINTER(c_double)),
                ("within", POINTER(c_int64))]


class _PartView(ctypes.Structure):
    _fields_ = [("n_nodes", c_int64), ("n_owned", c_int64),
                ("n_edges", c_int64), ("n_bonds", c_int64),
                ("n_owned_bonds", c_int64), ("n_lines", c_int64),
                ("n_mapping", c_int64),
                ("markers", POINTER(c_int64)), ("global_ids", POINTER(c_int64)),
                ("src_local", POINTER(c_int32)), ("dst_local", POINTER(c_int32)),
                ("edge_gids", POINTER(c_int64)),
                ("line_markers", POINTER(c_int64)),
                ("line_src", POINTER(c_int32)), ("line_dst", POINTER(c_int32)),
                ("line_center", POINTER(c_int32)),
                ("map_de", POINTER(c_int64)), ("map_ude", POINTER(c_int64)),
                ("bde_edge_gids", POINTER(c_int64)),
                ("row_ptr", POINTER(c_int32)),
                ("src_perm", POINTER(c_int32)),
                ("src_row_ptr", POINTER(c_int32)),
                ("line_row_ptr", POINTER(c_int32)),
                ("line_src_perm", POINTER(c_int32)),
                ("line_src_row_ptr", POINTER(c_int32)),
                ("center_perm", POINTER(c_int32)),
                ("center_row_ptr", POINTER(c_int32)),
                ("offsets_i8", POINTER(ctypes.c_int8))]


class _Owner:
    """Keeps the C++ graph handle alive for as long as any exported numpy
    array references it (chained through the ctypes buffer objects)."""

    def __init__(self, handle):
        self._h = handle

    def __del__(self):
        try:
            graph_lib().dm_graph_free(self._h)
        except Exception:
            pass


_CTYPE_OF = {np.int64: c_int64, np.float64: c_double,
             np.int32: c_int32, np.int8: ctypes.c_int8}


def _as_np(ptr, count, nptype, owner, shape=None):
    if count == 0 or not ptr:
        return np.zeros(shape if shape else (0,), dtype=nptype)
    ctype = _CTYPE_OF[nptype]
    buf = (ctype * count).from_address(
        ctypes.cast(ptr, c_void_p).value)
    buf._owner = owner  # keepalive chain (ctypes instances allow attributes)
    arr = np.frombuffer(buf, dtype=nptype)
    return arr.reshape(shape) if shape else arr


def build_graph(frac_coords, lattice, pbc, cutoff, bond_cutoff, tol,
                num_partitions, num_threads, use_bond_graph, focus=-1):
    """Low-level build; returns (owner, global dict, [partition dicts]).
    focus >= 0 = the SPMD per-rank slab build (see dm_graph_build_focus)."""
    lib = graph_lib()
    frac = np.ascontiguousarray(frac_coords, dtype=np.float64)
    lat = np.ascontiguousarray(lattice, dtype=np.float64)
    pbc = np.ascontiguousarray(pbc, dtype=np.int64)
    n = len(frac)
    h = c_void_p()
    rc = lib.dm_graph_build_focus(
        frac.ctypes.data_as(POINTER(c_double)),
        lat.ctypes.data_as(POINTER(c_double)),
        pbc.ctypes.data_as(POINTER(c_int64)),
        c_int64(n), c_double(cutoff), c_double(bond_cutoff), c_double(tol),
        c_int32(num_partitions), c_int32(num_threads),
        c_int32(1 if use_bond_graph else 0), c_int32(focus), ctypes.byref(h))
    if rc != 0:
        msg = lib.dm_last_error().decode()
        if rc == -4:
            raise RuntimeError(f"Partition walls are too close together. {msg}")
        raise RuntimeError(f"dm_graph_build failed ({rc}): {msg}")
    owner = _Owner(h)

    gv = _GlobalView()
    lib.dm_graph_global_view(h, ctypes.byref(gv))
    g = {
        "src": _as_np(gv.src, gv.n_edges, np.int64, owner),
        "dst": _as_np(gv.dst, gv.n_edges, np.int64, owner),
        "offsets": _as_np(gv.offsets, 3 * gv.n_edges, np.float64, owner,
                          (gv.n_edges, 3)),
        "dist": _as_np(gv.dist, gv.n_edges, np.float64, owner),
        "within": _as_np(gv.within, gv.n_within, np.int64, owner),
        "n_atoms": gv.n_atoms,
    }
    parts = []
    for p in range(num_partitions):
        pv = _PartView()
        lib.dm_graph_partition_view(h, c_int32(p), ctypes.byref(pv))
        parts.append({
            "markers": _as_np(pv.markers, 2 * num_partitions + 1, np.int64, owner),
            "global_ids": _as_np(pv.global_ids, pv.n_nodes, np.int64, owner),
            "src_local": _as_np(pv.src_local, pv.n_edges, np.int32, owner),
            "dst_local": _as_np(pv.dst_local, pv.n_edges, np.int32, owner),
            "edge_gids": _as_np(pv.edge_gids, pv.n_edges, np.int64, owner),
            "line_markers": _as_np(pv.line_markers, 2 * num_partitions + 1,
                                   np.int64, owner) if use_bond_graph else None,
            "line_src": _as_np(pv.line_src, pv.n_lines, np.int32, owner),
            "line_dst": _as_np(pv.line_dst, pv.n_lines, np.int32, owner),
            "line_center": _as_np(pv.line_center, pv.n_lines, np.int32, owner),
            "map_de": _as_np(pv.map_de, pv.n_mapping, np.int64, owner),
            "map_ude": _as_np(pv.map_ude, pv.n_mapping, np.int64, owner),
            "bde_edge_gids": _as_np(pv.bde_edge_gids, pv.n_bonds, np.int64, owner),
            "n_bonds": pv.n_bonds,
            "n_owned_bonds": pv.n_owned_bonds,
            "row_ptr": _as_np(pv.row_ptr, pv.n_nodes + 1, np.int32, owner),
            "src_perm": _as_np(pv.src_perm, pv.n_edges, np.int32, owner),
            "src_row_ptr": _as_np(pv.src_row_ptr, pv.n_nodes + 1, np.int32, owner),
            "line_row_ptr": _as_np(pv.line_row_ptr, pv.n_bonds + 1, np.int32, owner),
            "line_src_perm": _as_np(pv.line_src_perm, pv.n_lines, np.int32, owner),
            "line_src_row_ptr": _as_np(pv.line_src_row_ptr, pv.n_bonds + 1,
                                       np.int32, owner),
            "center_perm": _as_np(pv.center_perm, pv.n_lines, np.int32, owner),
            "center_row_ptr": _as_np(pv.center_row_ptr, pv.n_nodes + 1,
                                     np.int32, owner),
            "offsets_i8": _as_np(pv.offsets_i8, 3 * pv.n_edges, np.int8, owner,
                                 (pv.n_edges, 3)),
        })
    return owner, g, parts


def get_subgraphs_fast(cart_coords, cutoff, pbc, lattice, num_partitions,
                       bond_cutoff, tol, num_threads, use_bond_graph,
                       frac_coords, return_csr=False, focus=-1):
    """Reference-compatible entry (subgraph_creation_fast.c:92-453 tuple
    + one extension element: per-partition per-BDE global edge ids).
    With return_csr=True also returns the per-partition CSR dicts the HIP
    kernel path consumes."""
    _owner, g, parts = build_graph(frac_coords, lattice, pbc, cutoff,
                                   bond_cutoff, tol, num_partitions,
                                   num_threads, use_bond_graph, focus=focus)
    frac = np.ascontiguousarray(frac_coords, dtype=np.float64)
    lat = np.ascontiguousarray(lattice, dtype=np.float64)
    wrapped_cart = frac @ lat

    local_coords = [wrapped_cart[p["global_ids"]] for p in parts]

    ret = (
        [p["src_local"] for p in parts],
        [p["dst_local"] for p in parts],
        [p["markers"] for p in parts],
        local_coords,
        [p["global_ids"] for p in parts],
        g["src"], g["dst"], g["offsets"], g["dist"],
        [p["line_src"] for p in parts],
        [p["line_dst"] for p in parts],
        g["within"],
        [p["line_markers"] for p in parts] if use_bond_graph else [],
        [int(p["n_bonds"]) for p in parts],
        [p["map_de"] for p in parts],
        [p["map_ude"] for p in parts],
        [p["edge_gids"] for p in parts],
        [None for _ in parts],                  # G2L (excluded: unreliable)
        [p["line_center"] for p in parts],
        [p["bde_edge_gids"] for p in parts],    # build extension
    )
    if return_csr:
        return ret, parts
    return ret
