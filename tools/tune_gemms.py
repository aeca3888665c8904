#!/usr/bin/env python3
"""Offline hipBLASLt/rocBLAS TunableOp search over the training-step GEMM
shapes (worst first: the TN wgrad family runs 989-1225 TF/s vs 1300-1645
for fwd — tools/bench_gemm_shapes.py).  Writes a result file and prints
tuned vs untuned per shape.

Finding (round 2, profiles/gemm_study_r02.md): the search's internal
timings do NOT reproduce — loading the results (API or env) leaves every
shape and the end-to-end step bit-identical to the Default picks.  Kept
as tooling for future ROCm stacks; the default Tensile selections are
already at the random-operand wall on this one.

    python tools/tune_gemms.py --out gpurun_out/tunableop_llama-3-8b.csv
"""
import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

REPO = Path(__file__).resolve().parent.parent


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="gpurun_out/tunableop_llama-3-8b.csv")
    p.add_argument("--max-ms", type=int, default=20000,
                   help="tuning budget per solution set")
    p.add_argument("--iters", type=int, default=10)
    return p.parse_args()


def main():
    args = parse_args()
    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    import torch
    import torch.cuda.tunable as tunable

    from bench_gemm_shapes import SHAPES, bench  # noqa: E402

    # untuned baseline
    base = {}
    for name, m, n, k, a_t, b_t in SHAPES:
        base[name] = bench(m, n, k, a_t, b_t, iters=args.iters)

    tunable.enable(True)
    tunable.tuning_enable(True)
    tunable.set_filename(args.out)
    tunable.set_max_tuning_duration(args.max_ms)
    # tune worst-first (wgrad family first)
    order = sorted(SHAPES, key=lambda s: base[s[0]][0])
    t0 = time.time()
    for name, m, n, k, a_t, b_t in order:
        bench(m, n, k, a_t, b_t, iters=1)
        print(f"tuned {name} ({time.time() - t0:.0f}s elapsed)", flush=True)
    # results are flushed to the filename incrementally / at exit;
    # torch 2.10 has no tunable.write_file()
    tunable.tuning_enable(False)

    print(f"{'shape':14s} {'untuned':>9s} {'tuned':>9s} {'gain':>7s}")
    total_gain = 0.0
    for name, m, n, k, a_t, b_t in SHAPES:
        tf0, ms0 = base[name]
        tf1, ms1 = bench(m, n, k, a_t, b_t, iters=args.iters)
        layers = 1 if name.startswith("lm_head") else 32
        total_gain += (ms0 - ms1) * layers
        print(f"{name:14s} {tf0:7.0f}TF {tf1:7.0f}TF {100*(tf1/tf0-1):6.1f}%")
    print(f"total step-GEMM time saved: {total_gain:.1f} ms")


if __name__ == "__main__":
    main()
