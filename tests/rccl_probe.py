"""One-GPU RCCL probe (run under torchrun --nproc-per-node 2).

Tries the actual nccl(=RCCL) backend with both ranks on the same device
(cuda:0), exercising the exact p2p pattern the SPMD halo uses
(dist.batch_isend_irecv of contiguous slices).  Prints PROBE_OK or the
failure, so we learn whether a 1-GPU lease can host a world=2 RCCL test.
"""
import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)          # both ranks share the one GPU
    dist.init_process_group("nccl")
    dev = torch.device("cuda:0")
    try:
        send = torch.full((4, 8), float(rank + 1), device=dev)
        recv = torch.empty(4, 8, device=dev)
        peer = 1 - rank
        ops = [dist.P2POp(dist.irecv, recv, peer),
               dist.P2POp(dist.isend, send, peer)]
        for r in dist.batch_isend_irecv(ops):
            r.wait()
        torch.cuda.synchronize()
        expect = float(peer + 1)
        ok = bool((recv == expect).all().item())
        print(f"rank {rank}: p2p {'PROBE_OK' if ok else 'WRONG_DATA'}",
              flush=True)
        t = torch.ones(1, device=dev)
        dist.all_reduce(t)
        print(f"rank {rank}: allreduce={t.item()} "
              f"{'PROBE_OK' if t.item() == world else 'WRONG'}", flush=True)
    except Exception as e:
        print(f"rank {rank}: PROBE_FAIL {type(e).__name__}: {e}", flush=True)
        sys.exit(1)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
