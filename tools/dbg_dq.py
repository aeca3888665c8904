#!/usr/bin/env python3
"""Localize dq mismatches (debug aid for attention_bwd)."""
import math
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

from distributed_training_guide_amd import ops
from distributed_training_guide_amd.ops import reference as R

torch.manual_seed(0)
B, S, Hq, Hkv, D = 2, 128, 4, 4, 64
q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
scale = 1 / math.sqrt(D)
o = ops.flash_attention(q, k, v, scale)
qr = q.detach().float().requires_grad_(True)
kr = k.detach().float().requires_grad_(True)
vr = v.detach().float().requires_grad_(True)
orf = R.attention_ref(qr, kr, vr, scale)
do = torch.randn_like(o)
o.backward(do)
orf.backward(do.float())

err = (q.grad.float() - qr.grad).abs()  # [B,S,Hq,D]
tol = 0.05 * qr.grad.abs().mean()
bad = err > err.mean() * 10 + 0.1
print("total bad:", bad.sum().item(), "of", bad.numel())
print("dq rel err", ((q.grad.float()-qr.grad).norm()/qr.grad.norm()).item())
idx = bad.nonzero()
if len(idx):
    rows = sorted(set((int(i[1]) for i in idx)))
    print("bad qrows:", rows[:40], "..." if len(rows) > 40 else "")
    ds = sorted(set((int(i[3]) for i in idx)))
    print("bad d:", ds[:40], "..." if len(ds) > 40 else "")
    bs = sorted(set((int(i[0]), int(i[2])) for i in idx))
    print("bad (b,h):", bs[:10])
    # per-row counts for first bad rows
    for r in rows[:6]:
        cnt = bad[:, r].sum().item()
        print(f"  row {r}: {cnt} bad elems")
# also check dk/dv
for name, g, gr in [("dk", k.grad, kr.grad), ("dv", v.grad, vr.grad)]:
    rel = (g.float() - gr).norm() / gr.norm()
    print(name, "rel err", rel.item())

# exact coordinates for b=0,h=0
g = q.grad.float()[0, :, 0, :]
gr_ = qr.grad[0, :, 0, :]
e2 = (g - gr_).abs()
bad2 = (e2 > 0.08).nonzero()
print("b0h0 exact bad (row,d):", [tuple(map(int, x)) for x in bad2[:50]])
for r_, d_ in [tuple(map(int, x)) for x in bad2[:8]]:
    print(f"  ({r_},{d_}): got {g[r_,d_].item():.4f} want {gr_[r_,d_].item():.4f}")

# rel err per (16-row block x 16-d block), b0h0
print("relerr grid (rows x dblocks):")
for rb in range(0, S, 16):
    line = []
    for db in range(0, D, 16):
        gg = g[rb:rb+16, db:db+16]
        rr = gr_[rb:rb+16, db:db+16]
        line.append(f"{((gg-rr).norm()/ (rr.norm()+1e-9)).item():.3f}")
    print(f"  rows {rb:3d}+: " + " ".join(line))
