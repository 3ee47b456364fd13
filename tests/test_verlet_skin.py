"""Verlet-skin graph reuse: a superset graph built at
cutoff+skin, masked per step by the true cutoffs, must reproduce the
fresh-build oracle exactly while actually REUSING the graph across small
displacements."""
import copy

import numpy as np
import torch

from distmlip_amd.model import CHGNetCore
from distmlip_amd.runtime import SpmdEngine
from distmlip_amd.structures import Structure, diamond_si, random_cell
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


def test_step_verlet_skewed_cell_with_stress():
    """Masked-superset reuse on a SKEWED cell, with stress, vs the fresh
    oracle (fp64 exact)."""
    s = random_cell(260, a=13.5, n_species=3, seed=7, skew=0.07)
    core = CHGNetCore.seeded(seed=1).double()
    eng = SpmdEngine(core, world=1, threads=2, device="cpu", ops=CpuRefOps())
    out = eng.step_verlet(s, skin=0.8, calc_stresses=True)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64,
                         compute_stress=True)
    assert abs(out["energy"].item() - ref["energy"].item()) < 1e-9
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].numpy()
    # forces reach ~5e2 here; 2e-9 absolute is ~1e-12 relative — fp64
    # rounding between the product's formula sequences and the oracle's
    # INDEPENDENT restatements (oracle/basis_ref.py)
    assert np.abs(F - ref["forces"].numpy()).max() < 2e-9
    assert np.abs(out["stress"].numpy() - ref["stress"].numpy()).max() < 1e-8


def test_verlet_mask_selection_boundary():
    """Regression for the run-34 bug class: members sitting EXACTLY on a
    selection boundary must get the same include/exclude decision from
    the verlet mask as from a fresh fp64 build.

    Atoms are placed on exact binary fractions of a 12 A cell so the
    boundary distances (3.0 for the bond graph) are fp64-exact; both
    sides must apply the builder's `d^2 < r^2 + tol` decision to them.
    Before the fix the mask re-derived distances from fp32 model tensors
    with a strict `<`, which flips such members (0.83 eV/A force error
    at li100k scale, run 34)."""
    a = 12.0
    frac = np.array([
        [0.0, 0.0, 0.0],
        [0.25, 0.0, 0.0],      # exactly 3.0 A from atom 0 (boundary)
        [0.0, 0.171875, 0.0],  # 2.0625 A: inside the bond cutoff
        [0.5, 0.5, 0.5],
        [0.5, 0.5, 0.171875],  # pair deep in the cell
    ])
    s = Structure(frac_coords=frac, lattice=np.eye(3) * a,
                  species=np.zeros(len(frac), dtype=np.int64),
                  pbc=np.ones(3, dtype=np.int64))
    core = CHGNetCore.seeded(seed=0).double()
    eng = SpmdEngine(core, world=1, threads=2, device="cpu", ops=CpuRefOps())
    out = eng.step_verlet(s, skin=1.0)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    # the reference selects by d^2 < r^2 + tol (fpis.c:760-764), so the
    # exactly-on-boundary pair IS in the fresh three-body set — the mask
    # must reach the same decision
    within_d = np.linalg.norm(
        (s.frac_coords[g["dst"]] + g["offsets"] - s.frac_coords[g["src"]])
        @ s.lattice, axis=1)[g["within_bond_r"]]
    assert np.any(np.isclose(within_d, 3.0, atol=1e-12))
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    assert abs(out["energy"].item() - ref["energy"].item()) < 1e-9
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].numpy()
    assert np.abs(F - ref["forces"].numpy()).max() < 1e-10


def test_hand_sequenced_conv_matches_default():
    """DM_FUSED_CONV=1 (hand-sequenced atom-conv reverse pass,
    distmlip_amd/conv.py) must match the op-by-op autograd path exactly
    in fp64 — pins the hand-written backward against the default."""
    import os

    s = diamond_si((10, 2, 2), jitter=0.1, seed=3)
    core = CHGNetCore.seeded(seed=0).double()
    for verlet in (False, True):       # True: bond masks through the
        outs = {}                      # hand-sequenced bond conv
        for flag in ("0", "1"):
            os.environ["DM_FUSED_CONV"] = flag
            try:
                eng = SpmdEngine(core, world=1, threads=2, device="cpu",
                                 ops=CpuRefOps())
                outs[flag] = (eng.step_verlet(s, skin=0.8, calc_stresses=True)
                              if verlet else eng.step(s, calc_stresses=True))
            finally:
                os.environ.pop("DM_FUSED_CONV", None)
        dE = abs(outs["0"]["energy"].item() - outs["1"]["energy"].item())
        assert dE < 1e-11, (verlet, dE)
        dF = (outs["0"]["forces_owned"] -
              outs["1"]["forces_owned"]).abs().max()
        assert dF.item() < 1e-11, (verlet, dF)
        dS = (outs["0"]["stress"] - outs["1"]["stress"]).abs().max()
        assert dS.item() < 1e-9, (verlet, dS)
