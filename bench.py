"""bench.py — atom-steps/s for the CHGNet E+F distributed forward.

Contract: `python bench.py --gpus N --steps K --warmup W` runs the flagship
workload; for N>1 the driver launches one rank per GPU via
torch.distributed.run.  One JSON line from rank 0.

A "step" is one full energy+force forward exactly as the reference does it
per MD step (implementations/matgl/pes.py:50-146): neighbor-list +
partition build (native C++ builder, CPU) + the HIP-kernel model forward +
the autograd force backward.  `value` = whole-job atom-steps/s over all
ranks; config.breakdown reports graph-build vs model time separately.

Workloads (BASELINE.json configs; distmlip_amd/structures.py):
  si1m (default)  : diamond-Si 1,000,000 atoms fixed — THE config the
      BASELINE metric is quoted on ("CHGNet 1M-atom supercell,
      1/2/4/8 MI355X"); strong scaling, fits one GPU (checkpointed)
  li100k          : BCC-Li ~100k atoms PER GPU (weak scaling; N=1 is
      exactly config #2, 101,306 atoms)
  si1k            : 1,000-atom plumbing config (#1)
  mace62k/mace500k: config #4 (MACE-MP-0-medium shape) — 500k/8 atoms
      per GPU weak unit / 512k fixed
  uma250k/uma2m   : config #5 (UMA eSCN shape, bf16 autocast) — 2M/8
      atoms per GPU weak unit / 2.0M fixed
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

# avoid allocator fragmentation at the 1M-atom workload (10+ GiB blocks)
os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")


def _install_stderr_filter(patterns):
    """tunableop's hipBLASLt probing floods stderr with thousands of
    'matrix and stride size must be positive' lines — rejected candidate
    solutions for unusual GEMM shapes, not errors of this program.  Replace
    fd 2 with a pipe whose reader thread forwards every line EXCEPT those
    to the real stderr, so driver tails stay readable."""
    import threading
    real = os.dup(2)
    r, w = os.pipe()
    os.dup2(w, 2)
    os.close(w)

    def pump():
        buf = b""
        while True:
            try:
                chunk = os.read(r, 65536)
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            while b"\n" in buf:
                line, buf = buf.split(b"\n", 1)
                if not any(p in line for p in patterns):
                    os.write(real, line + b"\n")
        if buf:
            os.write(real, buf)

    threading.Thread(target=pump, daemon=True).start()

    def restore():
        # put the real stderr back before interpreter teardown: tools
        # that print at process finalization (rocprofv3 stats, C library
        # destructors) must not write into the dying pipe
        import time
        time.sleep(0.05)          # let the pump drain in-flight lines
        os.dup2(real, 2)
        os.close(r)

    import atexit
    atexit.register(restore)


_install_stderr_filter([b"matrix and stride size must be positive"])

import torch  # noqa: E402

HBM_PEAK_BYTES = 8.0e12  # MI355X spec peak (MI355X_MICROARCH.md); measured
                         # achievable ~6.3 TB/s


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--workload", type=str, default="si1m",
                    choices=["li100k", "si1m", "si2m", "si1k",
                             "mace62k", "mace500k", "uma250k", "uma2m"])
    ap.add_argument("--threads", type=int, default=0,
                    help="graph-builder threads (0 = cpu_count/world)")
    ap.add_argument("--verlet", type=float, default=0.0, metavar="SKIN",
                    help="MD mode: random-walk the structure each step and "
                         "reuse the graph via Verlet-skin SKIN (A); the "
                         "headline line always uses per-step rebuilds")
    ap.add_argument("--no-bond-graph", action="store_true")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--breakdown", action="store_true",
                    help="print per-stage timings (graph/H2D+fwd/bwd)")
    return ap.parse_args()


class SegSumTimer:
    """HIP-event timing of the judged scatter-add kernel (D=64 launches
    for CHGNet; any D for the MACE message scatters)."""

    def __init__(self, D: int = 64):
        self.events = []     # (start, stop, E_rows, N_rows, D, has_base)
        self.enabled = False
        self.D = D

    def wrap(self):
        import distmlip_amd.ops as ops
        orig = ops.raw_seg_sum
        timer = self

        def timed(msg, row_ptr, n_rows, base=None):
            if not timer.enabled or (timer.D and msg.shape[-1] != timer.D):
                return orig(msg, row_ptr, n_rows, base)
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            out = orig(msg, row_ptr, n_rows, base)
            e.record()
            timer.events.append((s, e, msg.shape[0], n_rows, msg.shape[1],
                                 base is not None))
            return out

        ops.raw_seg_sum = timed
        # the autograd Functions call through the module-level name
        return self

    def summary(self):
        if not self.events:
            return None
        torch.cuda.synchronize()
        total_ms, total_bytes = 0.0, 0
        for s, e, E, N, D, has_base in self.events:
            ms = s.elapsed_time(e)
            total_ms += ms
            b = E * D * 4 + N * D * 4 + (N * D * 4 if has_base else 0)
            total_bytes += b
        n = len(self.events)
        avg_s = total_ms / 1e3 / n
        achieved = total_bytes / n / avg_s
        # PMC-measured HBM traffic of the big forward seg_sum launches at
        # li100k (profiles/r06_pmc_*): FETCH_SIZE 627.1 MB/launch, doubled
        # per the gfx950 calibration (MI355X_MICROARCH.md §HBM: wide
        # coalesced reads tallied at half) + WRITE_SIZE 73.5 MB as
        # reported = 1.328 GB/launch vs 1.340 GB algorithmic — traffic ==
        # algorithmic within 1%, so we report the measured ratio applied
        # to this run's algorithmic bytes.
        measured_traffic_ratio = 1.328 / 1.340
        return {
            "bound": "hbm",
            "achieved": achieved / 1e9,          # GB/s
            "peak": HBM_PEAK_BYTES / 1e9,
            "unit": "GB/s",
            "frac": achieved / HBM_PEAK_BYTES,
            "traffic": total_bytes / n * measured_traffic_ratio,
            "traffic_note": "PMC-derived (profiles/r06_pmc_*.csv); gfx950 "
                            "FETCH_SIZE doubled per MI355X_MICROARCH.md",
            "kernel": f"dm_seg_sum_f32(D={self.D or 'any'})",
            "launches": n,
            "avg_ms": total_ms / n,
            "algorithmic_bytes_per_launch": total_bytes / n,
        }


def cpu_baseline_leg(workload_name, threads):
    """Time the ORACLE (CPU restatement, kind='port') on a bounded sample of
    the same workload family and scale to atom-steps/s."""
    import numpy as _np

    from distmlip_amd.dist import Distributed
    from distmlip_amd.structures import diamond_si

    torch.set_num_threads(threads)
    if workload_name.startswith("uma"):
        from distmlip_amd.uma_model import UMACore
        from oracle.uma_ref import uma_oracle_forward

        s = diamond_si(4, jitter=0.35, seed=0)   # 512-atom bounded sample
        rng = _np.random.default_rng(99)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(_np.int64)
        t0 = time.time()
        d = Distributed.create_distributed(
            s.cart_coords, s.frac_coords, s.lattice, 1, s.pbc, 6.0, 0.0,
            use_bond_graph=False, num_threads=threads)
        core = UMACore.seeded(seed=0).float()
        uma_oracle_forward(core, s, d.py_index_1, d.py_index_2,
                           d.py_offsets, dtype=torch.float32)
        dt = time.time() - t0
        return {"value": s.num_atoms / dt, "unit": "atom_steps_per_s",
                "cores": threads, "kind": "port",
                "sample": f"diamond-Si {s.num_atoms} atoms, 1 full UMA "
                          f"E+F forward (graph build + oracle "
                          f"restatement), {dt:.2f}s"}

    if workload_name.startswith("mace"):
        from distmlip_amd.mace_model import MACECore
        from oracle.mace_ref import mace_oracle_forward

        s = diamond_si(4, jitter=0.1, seed=0)    # 512-atom bounded sample
        rng = _np.random.default_rng(77)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(_np.int64)
        t0 = time.time()
        d = Distributed.create_distributed(
            s.cart_coords, s.frac_coords, s.lattice, 1, s.pbc, 6.0, 0.0,
            use_bond_graph=False, num_threads=threads)
        core = MACECore.seeded(seed=0).float()
        mace_oracle_forward(core, s, d.py_index_1, d.py_index_2,
                            d.py_offsets, dtype=torch.float32)
        dt = time.time() - t0
        return {"value": s.num_atoms / dt, "unit": "atom_steps_per_s",
                "cores": threads, "kind": "port",
                "sample": f"diamond-Si {s.num_atoms} atoms, 1 full MACE "
                          f"E+F forward (graph build + dense oracle "
                          f"restatement), {dt:.2f}s"}

    from distmlip_amd.model import CHGNetCore
    from oracle.chgnet_ref import oracle_forward

    s = diamond_si(6, jitter=0.1, seed=0)        # 1,728-atom bounded sample
    t0 = time.time()
    d = Distributed.create_distributed(
        s.cart_coords, s.frac_coords, s.lattice, 1, s.pbc, 6.0, 3.0,
        use_bond_graph=True, num_threads=threads)
    core = CHGNetCore.seeded(seed=0).float()
    out = oracle_forward(core, s, d.py_index_1, d.py_index_2, d.py_offsets,
                         d.within_r_indices, dtype=torch.float32)
    dt = time.time() - t0
    return {
        "value": s.num_atoms / dt,
        "unit": "atom_steps_per_s",
        "cores": threads,
        "kind": "port",
        "sample": f"diamond-Si {s.num_atoms} atoms, 1 full E+F forward "
                  f"(graph build + oracle CPU restatement), {dt:.2f}s",
    }


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    n_gpus = max(args.gpus, world)

    use_bg = not args.no_bond_graph
    # builder scaling saturates ~16-32 threads; oversubscribing a 256-thread
    # host regresses 5x (NUMA + team spawn) — cap unless told otherwise
    threads = args.threads or max(1, min(32, (os.cpu_count() or 8) // max(world, 1)))

    from distmlip_amd.structures import workload

    is_mace = args.workload.startswith("mace")
    is_uma = args.workload.startswith("uma")
    weak = args.workload in ("li100k", "mace62k", "uma250k")
    s = workload(args.workload, n_gpus=n_gpus if weak else 1)

    if world > 1:
        torch.distributed.init_process_group("nccl")
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local_rank)

    if is_uma:
        from distmlip_amd.uma_model import UMACore
        from distmlip_amd.uma_runtime import UmaSpmdEngine
        core = UMACore.seeded(seed=0).float()
        engine = UmaSpmdEngine(core, world, threads=threads,
                               autocast_bf16=True)
    elif is_mace:
        from distmlip_amd.mace_model import MACECore
        from distmlip_amd.mace_runtime import MaceSpmdEngine
        core = MACECore.seeded(seed=0).float()
        engine = MaceSpmdEngine(core, world, threads=threads)
    else:
        from distmlip_amd.model import CHGNetCore
        from distmlip_amd.runtime import SpmdEngine
        core = CHGNetCore.seeded(seed=0).float()
        if not use_bg:
            core.config.use_bond_graph = False
        engine = SpmdEngine(core.float(), world, threads=threads,
                            use_bond_graph=use_bg)

    def step():
        return engine.step(s)

    if args.verlet > 0:
        walk_rng = np.random.default_rng(1234)  # SAME walk on every rank
        inv_lat = np.linalg.inv(s.lattice)

        def step():  # noqa: F811
            # MD-style random walk (~0.05 A/step) + Verlet-skin graph reuse
            cart = s.frac_coords @ s.lattice
            cart += walk_rng.normal(0, 0.03, size=cart.shape)
            f = cart @ inv_lat
            s.frac_coords = np.mod(np.mod(f, 1.0), 1.0)
            return engine.step_verlet(s, skin=args.verlet)

    if world == 1 and args.breakdown:
        stages = {}

        def mark(name, t0):
            torch.cuda.synchronize()
            stages.setdefault(name, []).append(time.time() - t0)
            return time.time()

        def step():  # noqa: F811
            t = time.time()
            d = engine.build_graph(s)
            t = mark("graph_cpu", t)
            out = engine.step(s, dist_info=d)
            t = mark("engine_step", t)
            return out

        import atexit

        def report():
            for k, v in stages.items():
                print(f"# stage {k}: {1e3*sum(v)/len(v):.1f} ms avg over "
                      f"{len(v)} calls", file=sys.stderr)
        atexit.register(report)

    timer = SegSumTimer(D=0 if (is_mace or is_uma) else 64).wrap()

    graph_ms, model_ms = [], []
    for _ in range(args.warmup):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()

    timer.enabled = True
    t0 = time.time()
    for _ in range(args.steps):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    t1 = time.time()
    timer.enabled = False

    elapsed = t1 - t0
    if world > 1:
        t = torch.tensor([elapsed], device="cuda")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    total_atoms = s.num_atoms  # whole-job atoms per step (weak: grown with N)
    value = total_atoms * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0 and os.environ.get("DM_TUNABLEOP_WRITE"):
        # persist GEMM algorithm choices tuned during warmup so the
        # shipped tunableop_gfx950.csv can be extended offline
        try:
            torch.cuda.tunable.write_file(os.environ["DM_TUNABLEOP_WRITE"])
        except Exception as e:
            print(f"# tunableop write failed: {e}", file=sys.stderr)

    if rank == 0:
        roofline = timer.summary()
        cpu_b = None
        if not args.skip_cpu_baseline and n_gpus == 1:
            # cpu_baseline leg runs at N=1 only (tier contract)
            cpu_b = cpu_baseline_leg(args.workload, threads)
        line = {
            "metric": "atom_steps_per_s",
            "value": value,
            "unit": "atom_steps/s (full E+F forward incl per-step graph build)",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak" if weak else "strong",
            "vs_baseline": None,
            "dtype": "bf16-autocast" if is_uma else "f32",
            "data": "synthetic",
            "config": {
                "workload": args.workload,
                "verlet_skin": args.verlet or None,
                "verlet_rebuilds": (getattr(engine, "_vcache", None) or
                                    {}).get("rebuilds"),
                "n_atoms": int(total_atoms),
                "model": ("uma-escn-shape" if is_uma else
                          "mace-mp-0-medium-shape" if is_mace
                          else "chgnet-shape"),
                "cutoff": 6.0,
                "three_body_cutoff": None if is_mace else 3.0,
                "use_bond_graph": False if is_mace else use_bg,
                "builder_threads": threads,
                "parallelism": f"graph-parallel slab dp{n_gpus}",
            },
            "roofline": roofline,
            "cpu_baseline": cpu_b,
        }
        print(json.dumps(line))


if __name__ == "__main__":
    main()
