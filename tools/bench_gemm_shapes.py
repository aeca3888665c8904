#!/usr/bin/env python3
"""Per-shape hipBLASLt GEMM throughput on the exact training-step shapes
(fwd / dgrad / wgrad of every projection at bs24 s1024, llama-3-8b),
random bf16 operands.  Reference point: the best hand-written HIP GEMM on
this chip reaches ~1320-1340 TF on uniform-random operands (clock/power
wall — guide §5); shapes far below that are the ones worth attacking."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

M = 24576  # bs24 * s1024

# (name, m, n, k, a_t, b_t): y[m,n] = a @ b with layouts as in the step
SHAPES = []
for name, n, k in [("qkv", 6144, 4096), ("o", 4096, 4096),
                   ("gate_up", 28672, 4096), ("down", 4096, 14336),
                   ("lm_head", 128256, 4096)]:
    SHAPES.append((f"{name}.fwd", M, n, k, False, True))    # x @ W^T
    SHAPES.append((f"{name}.dgrad", M, k, n, False, False))  # dy @ W
    SHAPES.append((f"{name}.wgrad", n, k, M, True, False))   # dy^T @ x


def bench(m, n, k, a_t, b_t, iters=10):
    a = torch.randn(((k, m) if a_t else (m, k)), device="cuda",
                    dtype=torch.bfloat16)
    b = torch.randn(((n, k) if b_t else (k, n)), device="cuda",
                    dtype=torch.bfloat16)
    av = a.t() if a_t else a
    bv = b.t() if b_t else b
    out = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        torch.matmul(av, bv, out=out)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        torch.matmul(av, bv, out=out)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    return 2 * m * n * k / dt / 1e12, dt * 1e3


def main():
    total_tf = 0.0
    total_ms = 0.0
    for name, m, n, k, a_t, b_t in SHAPES:
        tf, ms = bench(m, n, k, a_t, b_t)
        layers = 1 if name.startswith("lm_head") else 32
        total_ms += ms * layers
        print(f"{name:14s} m={m:6d} n={n:6d} k={k:6d} "
              f"{'T' if a_t else 'N'}{'T' if b_t else 'N'}  "
              f"{tf:7.0f} TF/s  {ms:6.2f} ms x{layers}")
    print(f"sum over step (32 layers + lm_head): {total_ms:.0f} ms")


if __name__ == "__main__":
    main()
