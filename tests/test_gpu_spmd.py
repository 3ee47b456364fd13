"""SPMD engine on GPU hardware: world=2, both ranks on cuda:0.

RCCL categorically refuses two ranks on one device ("Duplicate GPU
detected", rccl 2.26.6 — probe log committed at
profiles/r2_rccl_probe.log), so the RCCL transport itself cannot execute
inside a 1-GPU lease.  This test exercises EVERYTHING ELSE of the
multi-GPU path on real hardware: two SPMD ranks with per-rank focused
slab builds, HIP kernels per partition, the HaloExchange autograd
Function with real halo traffic (staged via the DM_HALO_GLOO path), the
reverse force halo-add, and the final energy all-reduce — asserting the
gloo-test numerics (fp64 oracle to fp32-GPU tolerance, reference
dist.py:323-358 contract).  The nccl-backend leg differs ONLY in the
process-group backend handed to init_process_group.
"""
import os
import tempfile

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")
pytestmark = pytest.mark.gpu


def _worker(rank, world, init_file, out_dir):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    os.environ["DM_HALO_GLOO"] = "1"      # stage halos via host (1-GPU box)
    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import diamond_si

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        torch.cuda.set_device(0)          # both ranks share the one GPU
        s = diamond_si((12, 2, 2), jitter=0.12, seed=2)
        core = CHGNetCore.seeded(seed=0).float()
        eng = SpmdEngine(core, world, threads=2, device="cuda:0")
        out = eng.step(s, calc_stresses=True)
        np.save(os.path.join(out_dir, f"E_{rank}.npy"),
                np.array([out["energy"].item()]))
        np.save(os.path.join(out_dir, f"F_{rank}.npy"),
                out["forces_owned"].double().cpu().numpy())
        np.save(os.path.join(out_dir, f"gids_{rank}.npy"),
                out["global_ids_owned"])
        np.save(os.path.join(out_dir, f"S_{rank}.npy"),
                out["stress"].cpu().numpy())
    finally:
        dist.destroy_process_group()


@requires_gpu
def test_spmd_world2_one_gpu_vs_oracle(tmp_path):
    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.structures import diamond_si
    from oracle.chgnet_ref import oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    world = 2
    init_file = str(tmp_path / "pg_init")
    mp.spawn(_worker, args=(world, init_file, str(tmp_path)),
             nprocs=world, join=True)

    s = diamond_si((12, 2, 2), jitter=0.12, seed=2)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    core = CHGNetCore.seeded(seed=0).double()
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64,
                         compute_stress=True)

    F = np.zeros((s.num_atoms, 3))
    covered = np.zeros(s.num_atoms, dtype=bool)
    for r in range(world):
        E_r = np.load(f"{tmp_path}/E_{r}.npy")[0]
        assert abs(E_r - ref["energy"].item()) < 5e-3 * max(
            1.0, abs(ref["energy"].item()))
        gids = np.load(f"{tmp_path}/gids_{r}.npy")
        F[gids] = np.load(f"{tmp_path}/F_{r}.npy")
        assert not covered[gids].any(), "owned sets overlap"
        covered[gids] = True
    assert covered.all(), "owned sets do not cover all atoms"
    dF = np.abs(F - ref["forces"].numpy()).max()
    assert dF < 1e-4, f"SPMD GPU force error {dF} exceeds 1e-4 eV/A"
    S = np.load(f"{tmp_path}/S_0.npy")
    assert np.abs(S - ref["stress"].numpy()).max() < 1e-3
