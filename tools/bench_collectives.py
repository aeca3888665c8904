#!/usr/bin/env python3
"""RCCL/xGMI collective microbenchmark at the framework's message shapes.

Measures the three collective shapes the engines actually issue
(SURVEY.md §2b): DDP bucket all-reduce, FSDP flat-param all-gather /
fp32 grad reduce-scatter, and TP/SP activation all-gather/reduce-scatter —
plus broadcast and barrier.  One JSON line per (op, size) with algbw and
busbw (nccl-tests convention), so ring-vs-direct algorithm choices and
bucket/unit granularity can be derived from measurements instead of
NVLink-era defaults (reference tunables: bucket_cap_mb=500 at
/root/reference/02-distributed-data-parallel/train_llm.py:67, per-decoder
FSDP units at 04:88-89).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 tools/bench_collectives.py
    python tools/bench_collectives.py            # N=1 API self-check

busbw factors (B = bytes of the FULL tensor, n = world):
  all_reduce:      algbw * 2(n-1)/n
  all_gather:      algbw * (n-1)/n      (B = gathered output bytes)
  reduce_scatter:  algbw * (n-1)/n      (B = input bytes)
  broadcast:       algbw
"""
import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

MB = 1024 * 1024

# (name, bytes of the full message, dtype) — engine shapes at the
# BASELINE configs; see module docstring.
DEFAULT_SHAPES = [
    ("tp_act_shard_b16_s1024_h4096_tp8", 16 * MB, torch.bfloat16),
    ("tp_act_full_b16_s1024_h4096", 128 * MB, torch.bfloat16),
    ("ddp_bucket_default", 128 * MB, torch.bfloat16),
    ("fsdp_unit_llama8b_bf16", 436 * MB, torch.bfloat16),
    ("fsdp_grad_llama8b_fp32", 872 * MB, torch.float32),
    ("ddp_bucket_500mb_ref", 500 * MB, torch.bfloat16),
]


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--sizes-mb", type=float, nargs="*", default=None,
                   help="override: sweep these sizes (bf16) for every op")
    p.add_argument("--device", default=None)
    return p.parse_args()


def _time_op(fn, device, iters, warmup):
    for _ in range(warmup):
        fn()
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    t0 = time.time()
    for _ in range(iters):
        fn()
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    t1 = time.time()
    # max over ranks
    t = torch.tensor([(t1 - t0) / iters], dtype=torch.float64,
                     device=device if device.type == "cuda" else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t.item()


def bench_one(name, nbytes, dtype, device, world, rank, iters, warmup):
    esz = torch.tensor([], dtype=dtype).element_size()
    numel = nbytes // esz // world * world  # divisible by world
    full = torch.rand(numel, device=device).to(dtype)
    shard = torch.rand(numel // world, device=device).to(dtype)
    out_shard = torch.empty_like(shard)
    is_cuda = device.type == "cuda"
    results = []

    def rec(op, sec, bw_factor):
        b = numel * esz
        alg = b / sec / 1e9
        results.append({
            "op": op, "shape": name, "bytes": b, "dtype": str(dtype),
            "world": world, "us": sec * 1e6, "algbw_gbs": alg,
            "busbw_gbs": alg * bw_factor,
        })

    rec("all_reduce",
        _time_op(lambda: dist.all_reduce(full), device, iters, warmup),
        2 * (world - 1) / world if world > 1 else 1.0)

    if is_cuda:
        def ag():
            dist.all_gather_into_tensor(full, shard)
    else:
        chunks = list(full.chunk(world))

        def ag():
            dist.all_gather(chunks, shard)
    rec("all_gather", _time_op(ag, device, iters, warmup),
        (world - 1) / world if world > 1 else 1.0)

    if is_cuda:
        def rs():
            dist.reduce_scatter_tensor(out_shard, full)
    else:  # gloo: all-reduce + slice (same fallback the engines use)
        def rs():
            dist.all_reduce(full)
            out_shard.copy_(full[rank * shard.numel():
                                 (rank + 1) * shard.numel()])
    rec("reduce_scatter", _time_op(rs, device, iters, warmup),
        (world - 1) / world if world > 1 else 1.0)

    rec("broadcast",
        _time_op(lambda: dist.broadcast(full, group_src=0), device, iters,
                 warmup), 1.0)

    # direct (fully-connected xGMI) algorithms — parallel/xgmi.py; ring
    # vs direct is the measured choice DTGA_XGMI_ALGO selects
    from distributed_training_guide_amd.parallel import xgmi

    rec("direct_all_reduce",
        _time_op(lambda: xgmi.direct_all_reduce(full), device, iters,
                 warmup), 2 * (world - 1) / world if world > 1 else 1.0)
    rec("direct_all_gather",
        _time_op(lambda: xgmi.direct_all_gather_into(full, shard), device,
                 iters, warmup), (world - 1) / world if world > 1 else 1.0)
    rec("direct_reduce_scatter",
        _time_op(lambda: xgmi.direct_reduce_scatter(out_shard, full),
                 device, iters, warmup),
        (world - 1) / world if world > 1 else 1.0)
    del full, shard, out_shard
    return results


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    from distributed_training_guide_amd.parallel.pg import init_distributed

    init_distributed(device)

    shapes = DEFAULT_SHAPES
    if args.sizes_mb:
        shapes = [(f"{mb:g}MB", int(mb * MB), torch.bfloat16)
                  for mb in args.sizes_mb]

    all_results = []
    for name, nbytes, dtype in shapes:
        all_results += bench_one(name, nbytes, dtype, device, world, rank,
                                 args.iters, args.warmup)

    # barrier latency
    t = _time_op(lambda: dist.barrier(), device, args.iters, args.warmup)
    all_results.append({"op": "barrier", "shape": "-", "bytes": 0,
                        "dtype": "-", "world": world, "us": t * 1e6,
                        "algbw_gbs": None, "busbw_gbs": None})

    if rank == 0:
        for r in all_results:
            print(json.dumps(r))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
