"""MACE and UMA SPMD engines on GPU hardware: world=2, both ranks on
cuda:0 via the gloo-staged halo (RCCL refuses duplicate devices —
profiles/r2_rccl_probe.log).  Exercises per-rank GPU partition builds,
HIP kernels, HaloExchange with real halo content and the reverse force
halo for the round-2 model families."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")
pytestmark = pytest.mark.gpu


def _worker(rank, world, init_file, out_dir, family):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    os.environ["DM_HALO_GLOO"] = "1"

    from distmlip_amd.structures import diamond_si

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        torch.cuda.set_device(0)
        s = diamond_si((12, 4, 4), jitter=0.12, seed=2)
        s.species = np.asarray(s.species) % 3
        if family == "mace":
            from distmlip_amd.mace_model import MACEConfig, MACECore
            from distmlip_amd.mace_runtime import MaceSpmdEngine
            core = MACECore.seeded(
                MACEConfig(n_elements=3, channels=64,
                           avg_num_neighbors=40.0), seed=0).float()
            eng = MaceSpmdEngine(core, world, threads=2, device="cuda:0")
        else:
            from distmlip_amd.uma_model import UMAConfig, UMACore
            from distmlip_amd.uma_runtime import UmaSpmdEngine
            core = UMACore.seeded(
                UMAConfig(n_elements=3, sphere_channels=64, num_layers=2),
                seed=0).float()
            eng = UmaSpmdEngine(core, world, threads=2, device="cuda:0")
        out = eng.step(s)
        np.save(os.path.join(out_dir, f"E_{rank}.npy"),
                np.array([out["energy"].item()]))
        np.save(os.path.join(out_dir, f"F_{rank}.npy"),
                out["forces_owned"].double().cpu().numpy())
        np.save(os.path.join(out_dir, f"gids_{rank}.npy"),
                out["global_ids_owned"])
    finally:
        dist.destroy_process_group()


@requires_gpu
@pytest.mark.parametrize("family", [
    "mace",
    pytest.param("uma", marks=pytest.mark.xfail(
        reason="torch-ROCm concurrency bug in the 2-ranks-on-one-GPU "
               "topology (test-only; production is one rank per GPU over "
               "RCCL): border-row grads corrupt under overlapping kernel "
               "launches; AMD_SERIALIZE_KERNEL=3 (set below for the "
               "spawned workers) makes the run exact on most boxes, so "
               "this usually XPASSes with the numerics genuinely "
               "verified.  Full bisection: DESIGN.md §13 addendum 2.",
        strict=False)),
])
def test_spmd_world2_one_gpu_models(family, tmp_path):
    from distmlip_amd.structures import diamond_si
    from oracle.graph_ref import brute_force_neighbors

    world = 2
    # AMD_SERIALIZE_KERNEL must be in the environment when libamdhip64
    # LOADS in the spawned children (setting it inside the worker is too
    # late), so export it around the spawn for the uma family.
    ser = None
    if family == "uma":
        ser = os.environ.get("AMD_SERIALIZE_KERNEL")
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
    try:
        mp.spawn(_worker, args=(world, str(tmp_path / "pg"),
                                str(tmp_path), family), nprocs=world,
                 join=True)
    finally:
        if family == "uma":
            if ser is None:
                os.environ.pop("AMD_SERIALIZE_KERNEL", None)
            else:
                os.environ["AMD_SERIALIZE_KERNEL"] = ser

    s = diamond_si((12, 4, 4), jitter=0.12, seed=2)
    s.species = np.asarray(s.species) % 3
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)
    if family == "mace":
        from distmlip_amd.mace_model import MACEConfig, MACECore
        from oracle.mace_ref import mace_oracle_forward
        core = MACECore.seeded(
            MACEConfig(n_elements=3, channels=64,
                       avg_num_neighbors=40.0), seed=0).double()
        ref = mace_oracle_forward(core, s, g["src"], g["dst"],
                                  g["offsets"])
    else:
        from distmlip_amd.uma_model import UMAConfig, UMACore
        from oracle.uma_ref import uma_oracle_forward
        core = UMACore.seeded(
            UMAConfig(n_elements=3, sphere_channels=64, num_layers=2),
            seed=0).double()
        ref = uma_oracle_forward(core, s, g["src"], g["dst"],
                                 g["offsets"])

    F = np.zeros((s.num_atoms, 3))
    covered = np.zeros(s.num_atoms, dtype=bool)
    for r in range(world):
        E_r = np.load(f"{tmp_path}/E_{r}.npy")[0]
        assert abs(E_r - ref["energy"].item()) < 5e-3 * max(
            1.0, abs(ref["energy"].item()))
        gids = np.load(f"{tmp_path}/gids_{r}.npy")
        F[gids] = np.load(f"{tmp_path}/F_{r}.npy")
        assert not covered[gids].any()
        covered[gids] = True
    assert covered.all()
    fscale = max(1.0, np.abs(ref["forces"].numpy()).max())
    dF = np.abs(F - ref["forces"].numpy()).max()
    assert dF < 2e-3 * fscale, (family, dF, fscale)
