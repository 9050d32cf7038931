"""World-2 RCCL on ONE physical GPU (both ranks on cuda:0) — attempted.

MEASURED RESULT (kept for the record): RCCL 2.26.6 refuses with
"Duplicate GPU detected: rank 1 and rank 0 both on CUDA device" at
communicator init, exactly like NCCL. World > 1 RCCL therefore cannot
be exercised on a 1-GPU lease by any arrangement; the deepest
single-GPU proof remains world-1 RCCL (init + device-buffer collectives
+ graph capture around a live all-reduce, tests/test_gpu_dist.py), with
world 2-4 semantics covered on gloo.
Run under torchrun --nproc-per-node 2 with HIP_VISIBLE_DEVICES=0.
"""
import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=world)
    t = torch.full((1024,), float(rank + 1), device="cuda:0")
    dist.all_reduce(t)
    expect = sum(range(1, world + 1))
    ok1 = bool((t == expect).all())
    b = torch.full((256,), 7.0 if rank == 0 else 0.0, device="cuda:0")
    dist.broadcast(b, src=0)
    ok2 = bool((b == 7.0).all())
    # the engine's fused-moments message shape
    m = torch.randn(64, 325, device="cuda:0")
    s0 = m.sum().item()
    dist.all_reduce(m)
    ok3 = abs(m.sum().item()) < abs(s0) * world * 10 + 1e3
    dist.barrier()
    print(f"rank {rank}: all_reduce={ok1} broadcast={ok2} moments={ok3}",
          flush=True)
    dist.destroy_process_group()
    sys.exit(0 if (ok1 and ok2 and ok3) else 1)


if __name__ == "__main__":
    main()
