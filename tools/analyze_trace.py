#!/usr/bin/env python3
"""Analyze a rocprofv3 kernel-trace SQLite database (.db).

    rocprofv3 --kernel-trace --stats -d out/ -- python bench.py --steps 3
    python tools/analyze_trace.py out/*/NNN_results.db [--steps 5]

Reports, from rocpd_kernel_dispatch / rocpd_info_kernel_symbol:
  * per-family busy time (gemm / attention / fused ops / torch fallback)
  * GPU idle gaps > threshold and which kernels bracket them
  * per-stream concurrency (comm/compute overlap, as in
    profiles/overlap_study_r02.md)
"""
import argparse
import collections
import glob
import sqlite3

FAMS = ("attn_fwd", "attn_dkdv", "attn_dq", "attn_delta", "adamw",
        "add_rmsnorm", "rmsnorm", "ce_", "qkv_rope", "silu_mul", "embed",
        "ln_", "gelu", "elementwise", "reduce", "copy", "fill", "cat")


def fam(name):
    if "Cijk" in name:
        return "gemm (hipBLASLt)"
    for key in FAMS:
        if key in name:
            return key
    return "other"


def merge(iv):
    out = []
    for s, e in sorted(iv):
        if out and s <= out[-1][1]:
            out[-1][1] = max(out[-1][1], e)
        else:
            out.append([s, e])
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db", help="path or glob to *_results.db")
    p.add_argument("--steps", type=int, default=1,
                   help="divide totals by this (bench steps+warmup)")
    p.add_argument("--gap-us", type=float, default=30.0)
    args = p.parse_args()
    path = (glob.glob(args.db) or [args.db])[0]
    db = sqlite3.connect(path)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE "
        "'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch_", "")
    rows = list(cur.execute(f"""
        SELECT k.start, k.end, k.stream_id, ks.display_name
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        ORDER BY k.start"""))
    if not rows:
        print("no kernel dispatches in trace")
        return
    n = args.steps

    agg = collections.Counter()
    cnt = collections.Counter()
    for s, e, _, name in rows:
        agg[fam(name)] += e - s
        cnt[fam(name)] += 1
    total = sum(agg.values())
    merged = merge([(s, e) for s, e, _, _ in rows])
    busy = sum(e - s for s, e in merged)
    span = rows[-1][1] - rows[0][0]
    print(f"span {span/1e6:.1f} ms, kernel-busy {busy/1e6:.1f} ms "
          f"({100*busy/span:.1f}%), /{n}: busy {busy/n/1e6:.1f} ms")
    print("\nper family (/steps):")
    for k, v in agg.most_common(16):
        print(f"  {k:24s} {v/n/1e6:9.2f} ms ({100*v/total:5.1f}%) "
              f"x{cnt[k]//n}")

    print(f"\ngaps > {args.gap_us} us (whole trace):")
    shown = 0
    for i in range(1, len(rows)):
        gap = rows[i][0] - rows[i - 1][1]
        if gap > args.gap_us * 1e3 and shown < 12:
            print(f"  {gap/1e3:9.1f} us after [{rows[i-1][3][:44]}] "
                  f"before [{rows[i][3][:44]}]")
            shown += 1

    streams = sorted({r[2] for r in rows})
    if len(streams) > 1:
        print("\nper-stream busy + overlap with stream "
              f"{streams[0]} (compute):")
        m0 = merge([(s, e) for s, e, st, _ in rows if st == streams[0]])
        for st in streams[1:]:
            mi = merge([(s, e) for s, e, stt, _ in rows if stt == st])
            bi = sum(e - s for s, e in mi)
            i = j = ov = 0
            while i < len(m0) and j < len(mi):
                lo = max(m0[i][0], mi[j][0])
                hi = min(m0[i][1], mi[j][1])
                if hi > lo:
                    ov += hi - lo
                if m0[i][1] < mi[j][1]:
                    i += 1
                else:
                    j += 1
            print(f"  stream {st}: busy {bi/1e6:8.2f} ms, overlapped "
                  f"{ov/1e6:8.2f} ms ({100*ov/max(bi,1):.0f}%)")


if __name__ == "__main__":
    main()
