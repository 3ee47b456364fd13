"""The central KAT: partitioned distributed forward == independent full-graph
oracle (same seeded weights), fp64 to machine precision, fp32 to the
north-star tolerance (forces 1e-4 eV/A)."""
import numpy as np
import pytest
import torch

from distmlip_amd.chgnet import CHGNet_Dist
from distmlip_amd.model import CHGNetConfig, CHGNetCore
from distmlip_amd.pes import Potential_Dist
from oracle.chgnet_ref import CpuRefOps, oracle_forward


def _dist_EF(core, s, backend, P, dtype, calc_stresses=False):
    model = CHGNet_Dist.from_existing(core, dtype=dtype)
    model.enable_distributed_mode(["cpu"] * P, ops_factory=lambda dev: CpuRefOps())
    pot = Potential_Dist(model, calc_forces=True, calc_stresses=calc_stresses,
                         graph_backend=backend)
    return pot.forward(s)


@pytest.mark.parametrize("P", [2, 3])
def test_invariance_fp64(core64, si_slab, si_slab_graph, ref_graph_backend, P):
    g = si_slab_graph
    ref = oracle_forward(core64, si_slab, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    E, F, _, _ = _dist_EF(core64, si_slab, ref_graph_backend, P, torch.float64)
    assert abs(E.item() - ref["energy"].item()) < 1e-10
    assert (F - ref["forces"]).abs().max().item() < 1e-11


def test_invariance_fp32(core64, si_slab, si_slab_graph, ref_graph_backend):
    g = si_slab_graph
    ref = oracle_forward(core64, si_slab, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    E, F, _, _ = _dist_EF(core64.float(), si_slab, ref_graph_backend, 2,
                          torch.float32)
    assert abs(E.item() - ref["energy"].item()) < 1e-2 * max(1.0, abs(ref["energy"].item()))
    # north-star force tolerance: 1e-4 eV/A
    assert (F.double() - ref["forces"]).abs().max().item() < 1e-4


def test_invariance_no_bond_graph(si_slab, si_slab_graph, ref_graph_backend):
    cfg = CHGNetConfig(use_bond_graph=False)
    core = CHGNetCore.seeded(cfg, seed=1).double()
    g = si_slab_graph
    ref = oracle_forward(core, si_slab, g["src"], g["dst"], g["offsets"],
                         np.zeros(0, dtype=np.int64), dtype=torch.float64)
    E, F, _, _ = _dist_EF(core, si_slab, ref_graph_backend, 2, torch.float64)
    assert abs(E.item() - ref["energy"].item()) < 1e-10
    assert (F - ref["forces"]).abs().max().item() < 1e-11


def test_stress_invariance_fp64(core64, si_slab, si_slab_graph, ref_graph_backend):
    g = si_slab_graph
    ref = oracle_forward(core64, si_slab, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64,
                         compute_stress=True)
    E, F, S, _ = _dist_EF(core64, si_slab, ref_graph_backend, 2, torch.float64,
                          calc_stresses=True)
    assert S is not None
    assert (S - ref["stress"]).abs().max().item() < 1e-10


@pytest.mark.parametrize("seed", range(3))
def test_engine_vs_oracle_randomized(seed):
    """Randomized engine-vs-oracle breadth on CPU fp64: random skewed
    cells with mixed species through SpmdEngine world=1."""
    import torch

    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import random_cell
    from oracle.chgnet_ref import CpuRefOps, oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    rng = np.random.default_rng(4000 + seed)
    s = random_cell(int(rng.integers(60, 160)),
                    a=float(rng.uniform(12.0, 16.0)), n_species=4,
                    seed=int(rng.integers(1 << 30)),
                    skew=float(rng.uniform(0, 0.08)))
    core = CHGNetCore.seeded(seed=seed).double()
    eng = SpmdEngine(core, world=1, threads=2, device="cpu", ops=CpuRefOps())
    out = eng.step(s, calc_stresses=True)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64,
                         compute_stress=True)
    assert abs(out["energy"].item() - ref["energy"].item()) < 1e-9
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].numpy()
    assert np.abs(F - ref["forces"].numpy()).max() < 1e-10, seed
    assert np.abs(out["stress"].numpy() - ref["stress"].numpy()).max() < 1e-8
