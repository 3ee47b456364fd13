"""Generate committed golden vectors for the MACE and UMA oracle
restatements (round 2) — run in the build container; the fixtures pin
the model arithmetic against drift on any box, without /root/reference.
"""
import json
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from distmlip_amd.structures import diamond_si          # noqa: E402
from distmlip_amd.mace_model import MACEConfig, MACECore  # noqa: E402
from distmlip_amd.uma_model import UMAConfig, UMACore   # noqa: E402
from oracle.graph_ref import brute_force_neighbors      # noqa: E402
from oracle.mace_ref import mace_oracle_forward         # noqa: E402
from oracle.uma_ref import uma_oracle_forward           # noqa: E402


def main():
    out = {}
    s = diamond_si((6, 2, 2), jitter=0.1, seed=2)
    s.species = np.asarray(s.species) % 3
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)

    mc = MACECore.seeded(MACEConfig(n_elements=3, channels=16,
                                    avg_num_neighbors=20.0,
                                    atomic_inter_scale=0.7,
                                    atomic_inter_shift=0.1),
                         seed=9).double()
    rm = mace_oracle_forward(mc, s, g["src"], g["dst"], g["offsets"])
    out["mace"] = {
        "structure": {"reps": [6, 2, 2], "jitter": 0.1, "seed": 2,
                      "species_mod": 3},
        "energy": rm["energy"].item(),
        "forces_head": rm["forces"][:5].tolist(),
        "forces_absmax": rm["forces"].abs().max().item(),
    }

    uc = UMACore.seeded(UMAConfig(n_elements=3, sphere_channels=16,
                                  num_layers=2, edge_ch=32, num_gauss=16,
                                  spec_emb=8, avg_degree=20.0),
                        seed=9).double()
    ru = uma_oracle_forward(uc, s, g["src"], g["dst"], g["offsets"])
    out["uma"] = {
        "structure": {"reps": [6, 2, 2], "jitter": 0.1, "seed": 2,
                      "species_mod": 3},
        "energy": ru["energy"].item(),
        "forces_head": ru["forces"][:5].tolist(),
        "forces_absmax": ru["forces"].abs().max().item(),
    }
    path = os.path.join(os.path.dirname(__file__), "golden_models.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print("wrote", path)


if __name__ == "__main__":
    main()
