"""Microbench (not a test): within-process A/B of the seg-sum variants on
the li100k shape.  Run on a GPU box:
    python tests/perf_seg_sum.py
"""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import distmlip_amd.ops as ops
    lib = ops.hip_lib()
    dev = torch.device("cuda:0")
    E, N, D = 5_031_446, 101_306, 64
    rng = np.random.default_rng(0)
    deg = rng.poisson(E / N, N)
    deg = np.maximum(deg, 1)
    scale = E / deg.sum()
    rp = np.round(np.concatenate([[0], np.cumsum(deg * scale)])).astype(np.int64)
    E_eff = int(rp[-1])
    rp_t = torch.tensor(rp, dtype=torch.int32, device=dev)
    msg = torch.randn(E_eff, D, device=dev)
    base = torch.randn(N, D, device=dev)

    # DM_SEG_VARIANT is latched at library load: run this script once per
    # variant (the gpurun command does both)
    _ = lib
    torch.cuda.synchronize()
    evs = []
    for _ in range(50):
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        ops.raw_seg_sum(msg, rp_t, N, base)
        e.record()
        evs.append((s, e))
    torch.cuda.synchronize()
    ts = sorted(s.elapsed_time(e) for s, e in evs)
    ms = ts[len(ts) // 2]
    bytes_ = (E_eff + 2 * N) * D * 4
    print(f"variant={os.environ.get('DM_SEG_VARIANT', '1')} median {ms:.4f} ms"
          f"  {bytes_ / ms * 1e3 / 1e12:.2f} TB/s "
          f"({bytes_ / ms * 1e3 / 8e12 * 100:.1f}% of 8 TB/s spec)")


if __name__ == "__main__":
    main()
