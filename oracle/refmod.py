"""Loader for the reference's compiled graph module (the exact graph oracle).

The module is built by `oracle/Makefile` from the reference's own sources
(/root/reference/DistMLIP/distributed/*.c, recipe restated from the
reference setup.py:4-13) into oracle/_ref/.  /root/reference itself does
NOT exist on the GPU box — only the prebuilt .so travels with the repo
snapshot, so nothing here reads /root/reference at run time.

Call signature restated from subgraph_creation_fast.c:92-118 (get_subgraphs)
and the Python wrapper distributed/dist.py:158-275.
"""
from __future__ import annotations

import importlib.util
import os
import subprocess
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
_REF_DIR = os.path.join(_HERE, "_ref")

_mod = None


def available() -> bool:
    return _find_so() is not None


def _find_so():
    if not os.path.isdir(_REF_DIR):
        return None
    for f in os.listdir(_REF_DIR):
        if f.startswith("subgraph_creation_fast") and f.endswith(".so"):
            return os.path.join(_REF_DIR, f)
    return None


def build_if_possible() -> bool:
    """Build oracle/_ref from /root/reference if present (build container only)."""
    if available():
        return True
    if not os.path.isdir("/root/reference/DistMLIP/distributed"):
        return False
    subprocess.run(["make", "-C", _HERE], check=True, capture_output=True)
    return available()


def load():
    """Import the compiled reference module; raises if not built."""
    global _mod
    if _mod is not None:
        return _mod
    so = _find_so()
    if so is None:
        raise RuntimeError(
            "oracle/_ref/subgraph_creation_fast*.so not built; run `make -C oracle` "
            "in the build container (needs /root/reference)."
        )
    spec = importlib.util.spec_from_file_location("subgraph_creation_fast", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _mod = mod
    return mod


def get_subgraphs_fast(cart_coords, cutoff, pbc, lattice, num_partitions,
                       bond_cutoff, tol, num_threads, use_bond_graph, frac_coords):
    """Direct call into the reference C module (arg order: fast.c:102-113)."""
    import numpy as np
    m = load()
    return m.get_subgraphs_fast(
        np.ascontiguousarray(cart_coords, dtype=np.float64),
        float(cutoff),
        np.ascontiguousarray(pbc, dtype=np.int64),
        np.ascontiguousarray(lattice, dtype=np.float64),
        int(num_partitions),
        float(bond_cutoff),
        float(tol),
        int(num_threads),
        bool(use_bond_graph),
        np.ascontiguousarray(frac_coords, dtype=np.float64),
    )
