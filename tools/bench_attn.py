#!/usr/bin/env python3
"""Micro-benchmark for the attention kernels (run on a GPU box).

Reports achieved TFLOP/s for forward and backward at training shapes,
causal-masked FLOP accounting (guide §5.4: random data, within-run timing).
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--S", type=int, default=1024)
    p.add_argument("--Hq", type=int, default=32)
    p.add_argument("--Hkv", type=int, default=8)
    p.add_argument("--D", type=int, default=128)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    from distributed_training_guide_amd._ext import ext
    from distributed_training_guide_amd.ops import reference as R

    torch.manual_seed(0)
    B, S, Hq, Hkv, D = args.B, args.S, args.Hq, args.Hkv, args.D
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    scale = D ** -0.5

    # causal flop accounting: 2 gemms fwd, 5 gemms bwd, x0.5 causal
    flops_fwd = 4 * B * Hq * S * S * D * 0.5
    flops_bwd = 10 * B * Hq * S * S * D * 0.5

    o, lse = ext().attn_fwd(q, k, v, scale)
    if args.check:
        orf = R.attention_ref(q.float(), k.float(), v.float(), scale)
        err = (o.float() - orf).norm() / orf.norm()
        print(f"fwd rel err: {err.item():.4f}")
        assert err < 3e-2

    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.iters):
        o, lse = ext().attn_fwd(q, k, v, scale)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / args.iters
    print(f"fwd: {dt * 1e3:.3f} ms  {flops_fwd / dt / 1e12:.1f} TF/s")

    do = torch.randn_like(o)
    dq, dk, dv = ext().attn_bwd(do, q, k, v, o, lse, scale)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.iters):
        dq, dk, dv = ext().attn_bwd(do, q, k, v, o, lse, scale)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / args.iters
    print(f"bwd: {dt * 1e3:.3f} ms  {flops_bwd / dt / 1e12:.1f} TF/s")


if __name__ == "__main__":
    main()
