#!/usr/bin/env python3
"""`top` for an MI355X cluster — counterpart of the reference's
nvidia-smi-based poller (/root/reference/top-cluster.py:17-94), rebuilt on
rocm-smi's JSON interface.

Polls every host over ssh for per-GPU utilization, power draw/cap, VRAM
use and process count, aggregates per node and cluster-wide, and prints a
table every --poll-freq milliseconds.

    python top-cluster.py hosts            # hosts = file with one host/line
    python top-cluster.py --local          # just this node (no ssh)
"""
import argparse
import json
import subprocess
import time
from datetime import datetime

SMI_CMD = ("rocm-smi --showuse --showpower --showmemuse --showmeminfo vram "
           "--showpids --json")


def query_host(host: str, local: bool = False):
    if local:
        cmd = ["bash", "-c", SMI_CMD]
    else:
        cmd = ["ssh", "-o", "ConnectTimeout=5", host, SMI_CMD]
    try:
        out = subprocess.run(cmd, capture_output=True, text=True,
                             timeout=15).stdout
        # rocm-smi --json prints one json object (possibly after warnings)
        start = out.find("{")
        data = json.loads(out[start:]) if start >= 0 else {}
    except (subprocess.TimeoutExpired, json.JSONDecodeError):
        return None
    gpus = []
    nproc = 0
    for key, card in data.items():
        if not key.startswith("card"):
            if key == "system":
                pids = card.get("PIDs using GPU", "") if isinstance(card, dict) else ""
                nproc = len([p for p in str(pids).split(",") if p.strip()])
            continue
        def num(*names, default=0.0):
            for n in names:
                v = card.get(n)
                if v is None:
                    continue
                try:
                    return float(str(v).replace("W", "").strip())
                except ValueError:
                    continue
            return default
        util = num("GPU use (%)")
        power = num("Average Graphics Package Power (W)",
                    "Current Socket Graphics Package Power (W)")
        cap = num("Max Graphics Package Power (W)", default=1000.0)
        vram_used = num("VRAM Total Used Memory (B)")
        vram_total = num("VRAM Total Memory (B)", default=1.0)
        gpus.append(dict(util=util, power=power, cap=cap,
                         mem=100.0 * vram_used / max(vram_total, 1.0)))
    return dict(gpus=gpus, nproc=nproc)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("hosts", nargs="?", default=None,
                   help="file listing hosts (one per line)")
    p.add_argument("--local", action="store_true", help="poll this node only")
    p.add_argument("--poll-freq", type=int, default=5000, help="ms")
    p.add_argument("--iterations", type=int, default=0,
                   help="stop after N polls (0 = run forever)")
    args = p.parse_args()

    if args.local or args.hosts is None:
        hosts = ["localhost"]
        local = True
    else:
        hosts = [h.strip() for h in open(args.hosts) if h.strip()]
        local = False

    n_iter = 0
    while True:
        rows = []
        cl_util = cl_pow = cl_cap = cl_mem = cl_np = n_gpu = 0
        for h in hosts:
            r = query_host(h, local=local)
            if r is None or not r["gpus"]:
                rows.append((h, "UNREACHABLE", "", "", ""))
                continue
            g = r["gpus"]
            util = sum(x["util"] for x in g) / len(g)
            powp = 100 * sum(x["power"] for x in g) / max(
                sum(x["cap"] for x in g), 1)
            mem = sum(x["mem"] for x in g) / len(g)
            rows.append((h, f"{util:5.1f}%", f"{powp:5.1f}%",
                         f"{mem:5.1f}%", str(r["nproc"])))
            cl_util += sum(x["util"] for x in g)
            cl_pow += sum(x["power"] for x in g)
            cl_cap += sum(x["cap"] for x in g)
            cl_mem += sum(x["mem"] for x in g)
            cl_np += r["nproc"]
            n_gpu += len(g)
        print(f"\n=== {datetime.now().isoformat(timespec='seconds')} "
              f"({n_gpu} GPUs) ===")
        print(f"{'node':24s} {'util':>7s} {'power':>7s} {'mem':>7s} "
              f"{'nproc':>6s}")
        for r in rows:
            print(f"{r[0]:24s} {r[1]:>7s} {r[2]:>7s} {r[3]:>7s} {r[4]:>6s}")
        if n_gpu:
            print(f"{'cluster':24s} {cl_util / n_gpu:6.1f}% "
                  f"{100 * cl_pow / max(cl_cap, 1):6.1f}% "
                  f"{cl_mem / n_gpu:6.1f}% {cl_np:>6d}")
        n_iter += 1
        if args.iterations and n_iter >= args.iterations:
            break
        time.sleep(args.poll_freq / 1000.0)


if __name__ == "__main__":
    main()
