"""MACE path on GPU: HIP-backend engine vs the fp64 CPU oracle."""
import numpy as np
import pytest
import torch

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")
pytestmark = pytest.mark.gpu


@requires_gpu
def test_mace_engine_gpu_vs_oracle():
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si
    from oracle.graph_ref import brute_force_neighbors
    from oracle.mace_ref import mace_oracle_forward

    s = diamond_si((8, 2, 2), jitter=0.1, seed=2)
    s.species = np.asarray(s.species) % 3
    cfg = MACEConfig(n_elements=3, channels=128, avg_num_neighbors=40.0,
                     atomic_inter_scale=0.7, atomic_inter_shift=0.1)
    core = MACECore.seeded(cfg, seed=0)

    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)
    ref = mace_oracle_forward(core.double(), s, g["src"], g["dst"],
                              g["offsets"], dtype=torch.float64,
                              compute_stress=True)

    eng = MaceSpmdEngine(core.float(), world=1, threads=4)
    out = eng.step(s, calc_stresses=True)

    assert abs(out["energy"].item() - ref["energy"].item()) < 5e-3 * max(
        1.0, abs(ref["energy"].item()))
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].double().cpu().numpy()
    dF = np.abs(F - ref["forces"].numpy()).max()
    assert dF < 1e-4, f"MACE GPU force error {dF} exceeds 1e-4 eV/A"
    dS = np.abs(out["stress"].cpu().numpy() - ref["stress"].numpy()).max()
    assert dS < 1e-3, dS


@requires_gpu
def test_mace_engine_gpu_checkpointed_matches():
    """checkpoint='on' (the big-workload path) must match checkpoint-off
    bit-for-bit on the same inputs (same kernels, same order)."""
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si

    s = diamond_si((6, 2, 2), jitter=0.1, seed=3)
    s.species = np.asarray(s.species) % 3
    cfg = MACEConfig(n_elements=3, channels=64)
    core = MACECore.seeded(cfg, seed=1).float()
    outs = {}
    for ck in ("off", "on"):
        eng = MaceSpmdEngine(core, world=1, threads=4, checkpoint=ck)
        outs[ck] = eng.step(s)
    assert outs["off"]["energy"].item() == outs["on"]["energy"].item()
    assert torch.equal(outs["off"]["forces_owned"],
                       outs["on"]["forces_owned"])
