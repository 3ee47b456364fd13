"""Verlet-skin graph reuse: a superset graph built at
cutoff+skin, masked per step by the true cutoffs, must reproduce the
fresh-build oracle exactly while actually REUSING the graph across small
displacements."""
import copy

import numpy as np
import torch

from distmlip_amd.model import CHGNetCore
from distmlip_amd.runtime import SpmdEngine
from distmlip_amd.structures import diamond_si
from oracle.chgnet_ref import CpuRefOps, oracle_forward
from oracle.graph_ref import brute_force_neighbors


def test_step_verlet_matches_fresh_oracle():
    rng = np.random.default_rng(5)
    s = diamond_si((12, 2, 2), jitter=0.10, seed=2)
    core = CHGNetCore.seeded(seed=0).double()
    eng = SpmdEngine(core, world=1, threads=2, device="cpu", ops=CpuRefOps())

    for step_i in range(4):
        out = eng.step_verlet(s, skin=1.0)
        g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
        ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                             g["within_bond_r"], dtype=torch.float64)
        dE = abs(out["energy"].item() - ref["energy"].item())
        F = np.zeros((s.num_atoms, 3))
        F[out["global_ids_owned"]] = out["forces_owned"].numpy()
        dF = np.abs(F - ref["forces"].numpy()).max()
        assert dE < 1e-9, (step_i, dE)
        assert dF < 1e-10, (step_i, dF)
        # random walk, 0.08 A/step cartesian — stays within skin/2 for
        # several steps
        s = copy.deepcopy(s)
        cart = s.frac_coords @ s.lattice
        cart += rng.normal(0, 0.05, size=cart.shape).clip(-0.08, 0.08)
        f = cart @ np.linalg.inv(s.lattice)
        s.frac_coords = np.mod(np.mod(f, 1.0), 1.0)

    assert eng._vcache["rebuilds"] == 1, \
        f"graph was rebuilt {eng._vcache['rebuilds']}x; reuse never engaged"
