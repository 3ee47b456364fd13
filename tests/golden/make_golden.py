"""Generates tests/golden/golden.json — committed known-answer vectors.

Run in the BUILD container (needs /root/reference for the exact graph
oracle): python tests/golden/make_golden.py
The fixtures pin: graph shape (N, E, B, L per partition), marker arrays,
and the fp64 oracle energy/forces for the seeded si_slab structure, so any
later drift in the restatements is caught without /root/reference.
"""
import json
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from distmlip_amd.model import CHGNetCore  # noqa: E402
from distmlip_amd.structures import diamond_si  # noqa: E402
from oracle import refmod  # noqa: E402
from oracle.chgnet_ref import build_full_line_graph, oracle_forward  # noqa: E402
from oracle.graph_ref import brute_force_neighbors  # noqa: E402


def main():
    s = diamond_si((12, 2, 2), jitter=0.12, seed=2)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    l_src, l_dst, center = build_full_line_graph(g["src"], g["dst"],
                                                 g["within_bond_r"])
    core = CHGNetCore.seeded(seed=0).double()
    out = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    F = out["forces"].numpy()

    golden = {
        "structure": {"kind": "diamond_si", "reps": [12, 2, 2], "jitter": 0.12,
                      "seed": 2, "n_atoms": int(s.num_atoms)},
        "graph": {"n_edges": int(len(g["src"])),
                  "n_bonds": int(len(g["within_bond_r"])),
                  "n_lines": int(len(l_src)),
                  "dist_sum": float(g["dist"].sum()),
                  "line_center_sum": int(center.sum())},
        "model": {"seed": 0,
                  "energy": float(out["energy"].item()),
                  "forces_abs_sum": float(np.abs(F).sum()),
                  "forces_first3": F[:3].tolist(),
                  "site_props_sum": float(out["site_props"].sum().item())},
    }

    if refmod.available() or refmod.build_if_possible():
        ref = refmod.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice,
                                        2, 3.0, 1e-8, 4, True, s.frac_coords)
        golden["reference_partition_P2"] = {
            "markers": [np.asarray(m).tolist() for m in ref[2]],
            "line_markers": [np.asarray(m).tolist() for m in ref[12]],
            "num_UDEs": [int(x) for x in ref[13]],
            "edges_per_partition": [int(len(np.asarray(e))) for e in ref[0]],
            "lines_per_partition": [int(len(np.asarray(e))) for e in ref[9]],
        }

    path = os.path.join(os.path.dirname(__file__), "golden.json")
    with open(path, "w") as f:
        json.dump(golden, f, indent=1)
    print("wrote", path)


if __name__ == "__main__":
    main()
