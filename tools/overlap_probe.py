#!/usr/bin/env python3
"""Does an async RCCL collective overlap compute on this box?

Times (1) N GEMMs alone, (2) N all-gathers alone, (3) interleaved
issue-async-gather-then-GEMM — if (3) ~= max(1,2) the comm stream
overlaps compute; if (3) ~= (1)+(2) something serializes (event scope,
record-streams bookkeeping, or queue contention).  Run at world=1 (RCCL
single-rank: gathers are device copies on the comm stream) or under
torchrun at world>1 for real ring kernels.
"""
import os
import time

import torch
import torch.distributed as dist


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e3


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29561")
    dev = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}")
    torch.cuda.set_device(dev)
    dist.init_process_group("nccl", rank=rank, world_size=world,
                            device_id=dev)

    M = 8192
    a = torch.randn(M, M, device=dev, dtype=torch.bfloat16)
    b = torch.randn(M, M, device=dev, dtype=torch.bfloat16)
    c = torch.empty(M, M, device=dev, dtype=torch.bfloat16)
    n = 2 * 1024 * 1024 * 1024 // 2 // world * world  # 2 GB bf16 total
    out = torch.empty(n, device=dev, dtype=torch.bfloat16)
    shard = torch.randn(n // world, device=dev, dtype=torch.bfloat16)

    def gemm(k=4):
        for _ in range(k):
            torch.matmul(a, b, out=c)

    def gather():
        dist.all_gather_into_tensor(out, shard)

    def both():
        w = dist.all_gather_into_tensor(out, shard, async_op=True)
        gemm()
        w.wait()

    # engine-mimicking variant: the gather output's storage is freed and
    # re-grown each round (FSDP reshard/unshard) — exposes allocator
    # cross-stream bookkeeping costs
    def both_resize():
        out.untyped_storage().resize_(0)
        out.untyped_storage().resize_(n * 2)
        w = dist.all_gather_into_tensor(out, shard, async_op=True)
        gemm()
        w.wait()

    t_g = timeit(gemm)
    t_c = timeit(gather)
    t_b = timeit(both)
    t_r = timeit(both_resize)
    if rank == 0:
        print(f"gemm {t_g:.2f} ms  gather {t_c:.2f} ms  "
              f"interleaved {t_b:.2f} ms  resize-interleaved {t_r:.2f} ms  "
              f"(serial {t_g + t_c:.2f}, perfect {max(t_g, t_c):.2f})")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
