import os, sys
sys.path.insert(0, "/root/repo")
import numpy as np, torch
import torch.distributed as dist
import torch.multiprocessing as mp

def worker(rank, world, init_file, out_dir, dev_mode, env):
    sys.path.insert(0, "/root/repo")
    os.environ["DM_HALO_GLOO"] = "1"
    os.environ["DM_NO_GPU_BUILD"] = "1"     # identical CPU partitions
    os.environ["DM_UMA_ROT"] = "torch"
    os.environ["DM_UMA_DUMP_DIR"] = out_dir + "/" + dev_mode
    for k, v in env.items():
        os.environ[k] = v
    os.makedirs(os.environ["DM_UMA_DUMP_DIR"], exist_ok=True)
    from distmlip_amd.structures import diamond_si
    from distmlip_amd.uma_model import UMAConfig, UMACore
    from distmlip_amd.uma_runtime import UmaSpmdEngine
    from oracle.chgnet_ref import CpuRefOps
    if world > 1:
        dist.init_process_group("gloo", init_method=f"file://{init_file}",
                                rank=rank, world_size=world)
    try:
        s = diamond_si((12, 4, 4), jitter=0.12, seed=2)
        s.species = np.asarray(s.species) % 3
        nl = int(os.environ.get("DM_NLAYERS", "2"))
        core = UMACore.seeded(UMAConfig(n_elements=3, sphere_channels=64,
                                        num_layers=nl), seed=0).float()
        if dev_mode.startswith("gpu"):
            torch.cuda.set_device(0)
            ops = CpuRefOps() if "refops" in dev_mode else None
            eng = UmaSpmdEngine(core, world, threads=2, device="cuda:0",
                                ops=ops)
        else:
            eng = UmaSpmdEngine(core, world, threads=2, device="cpu",
                                ops=CpuRefOps())
        eng.step(s)
    finally:
        if world > 1:
            dist.destroy_process_group()

if __name__ == "__main__":
    # NOTE: the DM_ABL graph-ablation levers used mid-investigation were
    # stripped from the engine once the bisection concluded (DESIGN.md
    # §13 addendum 2); the remaining levers are the dump hooks below
    # plus DM_FORCE_HALO_NODE / DM_HALO_DUMP_DIR in the runtimes and
    # AMD_SERIALIZE_KERNEL=3 (which makes the run exact).
    out = os.environ.get("DM_PREHALO_OUT", "/tmp/umaph")
    os.makedirs(out, exist_ok=True)
    TRIALS = [
        ("arg", {"DM_UMA_DUMP_ARG": "1"}),
        ("hv", {"DM_UMA_DUMP_STAGES": "vectors"}),
        ("hD", {"DM_UMA_DUMP_STAGES": "D"}),
        ("hx", {"DM_UMA_DUMP_STAGES": "xedge"}),
        ("serial", {"AMD_SERIALIZE_KERNEL": "3"}),
    ]
    for abl, env in TRIALS:
        cn, gn = f"cpu_{abl}", f"gpu_{abl}"
        mp.spawn(worker, args=(2, f"{out}/pg_{cn}", out + "_" + abl,
                               cn, env), nprocs=2, join=True)
        mp.spawn(worker, args=(2, f"{out}/pg_{gn}", out + "_" + abl,
                               gn, env), nprocs=2, join=True)
        base = out + "_" + abl
        ds, sc = [], 0.0
        for r in range(2):
            a = np.load(f"{base}/{gn}/prehalo_{r}.npy")
            b = np.load(f"{base}/{cn}/prehalo_{r}.npy")
            ds.append(np.abs(a - b).max())
            sc = max(sc, np.abs(b).max())
        print(f"abl={abl or '(none)':<9} prehalo max diffs "
              f"{ds[0]:.4f} {ds[1]:.4f} (scale {sc:.1f})", flush=True)
