#!/usr/bin/env python3
"""Flagship training benchmark — driver contract (BASELINE.json metric).

Measures tokens/sec for a Llama-style causal LM training step on N MI355X
GPUs: forward + backward + fused-AdamW update on synthetic data /
random-init weights, bf16 compute, through this repo's HIP kernel path and
its RCCL/xGMI parallel engines.

    python bench.py --gpus 1 --steps 10 --warmup 3
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bench.py --gpus 8 --steps 10 --warmup 3

Default parallelism is the headline config (single at N=1, DDP at N>1,
Llama-3-8B bs24 s1024).  --parallelism selects the other BASELINE.json
configs, one command each (defaults follow the reference chapter configs,
/root/reference/04-fully-sharded-data-parallel/train_llm.py:83-95,
06-tensor-parallel/train_llm.py:79-121, 07-2d-parallel/train_llm.py:47-53):

    bench.py --parallelism ddp  --model llama-2-7b        # ch 2
    bench.py --parallelism fsdp                           # ch 4 (8B FSDP)
    bench.py --parallelism tp                             # ch 6 (TP=N)
    bench.py --parallelism 2d  --tensor-parallel 2        # ch 7 (70B at N>=4)

Rank 0 prints exactly one JSON line: whole-job tokens/s (aggregate over all
N GPUs), ms_per_step = MAX over ranks, peak HBM, config.  vs_baseline is
null: the reference's only numeric figure (~137 tok/s/GPU) is its
Llama-405B / 64xH100 row — a different model/config, not this
metric's config.
"""
import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import torch

PARALLELISMS = ("auto", "single", "ddp", "fsdp", "tp", "2d")

# per-parallelism defaults: (model, per-dp-replica batch)
_DEFAULTS = {
    "single": ("llama-3-8b", 24),
    "ddp": ("llama-3-8b", 24),
    "fsdp": ("llama-3-8b", 24),
    "tp": ("llama-3-8b", 16),    # reference ch6: 8B TP=8 bs16 s1024
    "2d": ("llama-3-70b", 4),    # reference ch7: 70B-class, 288 GB sizing
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--parallelism", default="auto", choices=PARALLELISMS,
                   help="auto = single at N=1, ddp at N>1 (headline config)")
    p.add_argument("--model", default=None,
                   help="default depends on --parallelism (8B; 70B for 2d)")
    p.add_argument("--batch-size", type=int, default=None,
                   help="per DP replica; default depends on --parallelism "
                        "(24 for dp/fsdp: 204 GB at bs24 + DDP buckets "
                        "still fits 288 GB at N=8)")
    p.add_argument("--seq-length", type=int, default=1024)
    p.add_argument("--bucket-cap-mb", type=int, default=128)
    p.add_argument("--tensor-parallel", type=int, default=0,
                   help="tp degree for tp/2d (0 = N for tp, 2 for 2d)")
    p.add_argument("--checkpoint-activations", action="store_true")
    p.add_argument("--hip-graphs", action="store_true",
                   help="capture the whole training step in one hipGraph "
                        "and replay it (single-GPU; grads kept allocated; "
                        "AdamW bias-correction step frozen at capture — "
                        "a launch-overhead experiment, not the default "
                        "training path)")
    p.add_argument("--zero1", action="store_true",
                   help="ZeRO-1 optimizer sharding under ddp (off by "
                        "default: on one 288GB-HBM node the replicated "
                        "fused AdamW avoids the un-overlapped post-step "
                        "shard all-gather)")
    p.add_argument("--device", default=None)
    return p.parse_args()


def _maybe_load_tunableop(model_name: str, device):
    """Load a committed hipBLASLt/rocBLAS TunableOp result file (offline
    GEMM autotuning for this model's shapes) if one exists in profiles/.
    Tuning itself is done offline (see profiles/README); here we only READ
    the chosen solutions — no runtime tuning overhead."""
    if device.type != "cuda" or os.environ.get("PYTORCH_TUNABLEOP_TUNING"):
        return
    f = Path(__file__).parent / "profiles" / f"tunableop_{model_name}.csv"
    if not f.exists():
        return
    try:
        import torch.cuda.tunable as tunable

        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(str(f))
    except Exception as e:  # noqa: BLE001 - perf feature, never fatal
        print(f"tunableop load failed: {e}", file=sys.stderr)


def build(par, args, config, device, world):
    """Build (model, optimizer, dp_size, parallelism_label) for one of the
    BASELINE.json parallelism configs on this repo's own engines."""
    from distributed_training_guide_amd.ops import FusedAdamW

    if par in ("single", "ddp"):
        from distributed_training_guide_amd.models import build_model

        model = build_model(config, device=device, dtype=torch.bfloat16)
        if args.checkpoint_activations:
            from distributed_training_guide_amd.parallel.fsdp import \
                apply_activation_checkpointing

            apply_activation_checkpointing(model)
        label = "single"
        if world > 1:
            from distributed_training_guide_amd.parallel.ddp import \
                DistributedDataParallel

            model = DistributedDataParallel(model,
                                            bucket_cap_mb=args.bucket_cap_mb)
            label = f"dp{world}"
        if world > 1 and args.zero1:
            from distributed_training_guide_amd.parallel.zero1 import \
                ZeroRedundancyOptimizer

            opt = ZeroRedundancyOptimizer(model.parameters(),
                                          optimizer_class=FusedAdamW, lr=3e-5)
        else:
            opt = FusedAdamW(model.parameters(), lr=3e-5)
        return model, opt, world, label

    if par == "fsdp":
        from distributed_training_guide_amd.models import build_model
        from distributed_training_guide_amd.parallel.fsdp import (
            FSDP, apply_activation_checkpointing)

        # meta init -> materialize unit by unit (reference 04:74-95 flow)
        with torch.device("meta"):
            model = build_model(config, dtype=torch.bfloat16)
        if args.checkpoint_activations:
            apply_activation_checkpointing(model)
        model = FSDP(model, device=device, reduce_dtype=torch.float32)
        opt = FusedAdamW(model.parameters(), lr=3e-5)
        return model, opt, world, f"fsdp{world}"

    # tp / 2d share the TP model over a 2-D mesh
    from distributed_training_guide_amd.parallel.mesh import DeviceMesh2D
    from distributed_training_guide_amd.parallel.tp import (
        TPLlamaDecoderLayer, TPLlamaForCausalLM)

    tp = args.tensor_parallel or (world if par == "tp" else min(2, world))
    mesh = DeviceMesh2D(tp_size=tp)
    model = TPLlamaForCausalLM(config, mesh, device=device,
                               dtype=torch.bfloat16, loss_parallel=True)
    if par == "tp":
        if args.checkpoint_activations:
            from distributed_training_guide_amd.parallel.fsdp import \
                apply_activation_checkpointing

            apply_activation_checkpointing(model,
                                           layer_cls=TPLlamaDecoderLayer)
        if mesh.dp_size > 1:
            from distributed_training_guide_amd.parallel.ddp import \
                DistributedDataParallel

            model = DistributedDataParallel(
                model, bucket_cap_mb=args.bucket_cap_mb,
                process_group=mesh.dp_group)
        opt = FusedAdamW(model.parameters(), lr=3e-5)
        label = f"tp{tp}" + (f"_dp{mesh.dp_size}" if mesh.dp_size > 1 else "")
        return model, opt, mesh.dp_size, label

    # 2d: FSDP over the dp mesh dim of TP-sharded layers (07:121-123)
    from distributed_training_guide_amd.parallel.fsdp import (
        FSDP, apply_activation_checkpointing)

    if args.checkpoint_activations:
        apply_activation_checkpointing(model, layer_cls=TPLlamaDecoderLayer)
    model = FSDP(model, layer_cls=TPLlamaDecoderLayer,
                 process_group=mesh.dp_group, device=device,
                 reduce_dtype=torch.float32)
    opt = FusedAdamW(model.parameters(), lr=3e-5)
    return model, opt, mesh.dp_size, f"2d_fsdp{mesh.dp_size}_tp{tp}"


def main():
    args = parse_args()
    import torch.distributed as dist

    from distributed_training_guide_amd.models import get_config

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

    par = args.parallelism
    if par == "auto":
        par = "single" if world == 1 else "ddp"
    if par == "2d" and args.model is None and world < 4:
        # 70B weights+moments need >= 4-way FSDP sharding to fit 288 GB
        print(f"2d at N={world}: defaulting to llama-3-8b "
              "(llama-3-70b needs N>=4)", file=sys.stderr)
        args.model = "llama-3-8b"
    model_name = args.model or _DEFAULTS[par][0]
    B = args.batch_size or _DEFAULTS[par][1]
    S = args.seq_length

    if world > 1 or par in ("tp", "2d", "fsdp"):
        from distributed_training_guide_amd.parallel.pg import init_distributed

        init_distributed(device)

    _maybe_load_tunableop(model_name, device)

    # model init seeded IDENTICALLY on every rank (sharded strategies
    # slice rank-locally from same-seed full tensors); data per-rank
    torch.manual_seed(1234)
    config = get_config(model_name)
    model, opt, dp_size, label = build(par, args, config, device, world)

    torch.manual_seed(5678 + rank)
    ids = torch.randint(0, config.vocab_size, (B, S), device=device)

    def step():
        out = model(input_ids=ids, labels=ids)
        out.loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
        return out.loss

    if args.hip_graphs:
        if world > 1:
            raise SystemExit("--hip-graphs is single-GPU only")

        def step_graphable():
            opt.zero_grad(set_to_none=False)  # stable grad storage
            out = model(input_ids=ids, labels=ids)
            out.loss.backward()
            opt.step()
            return out.loss

        # warmup on a side stream (torch graph-capture protocol), then
        # capture one full step; replay is a single hipGraphLaunch
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                step_graphable()
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static_loss = step_graphable()

        def step():  # noqa: F811 - graph-replay step
            graph.replay()
            return static_loss

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    def barrier():
        if world > 1:
            dist.barrier()

    try:
        for _ in range(args.warmup):
            step()
    except torch.OutOfMemoryError:
        print(f"OOM during warmup at batch-size {B} (model {model_name}, "
              f"parallelism {par}): try a smaller --batch-size, "
              "--checkpoint-activations, or DTGA_CE_MODE=fused",
              file=sys.stderr)
        raise
    sync()          # drain this rank's device work before aligning hosts
    barrier()
    sync()          # RCCL barrier enqueues device work; drain it too
    t0 = time.time()
    for _ in range(args.steps):
        step()
    barrier()
    sync()
    t1 = time.time()

    ms_per_step = (t1 - t0) / args.steps * 1000.0
    if world > 1:
        t = torch.tensor([ms_per_step], device=device if device.type == "cuda"
                         else "cpu", dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        ms_per_step = t.item()

    tokens_per_s = dp_size * B * S / (ms_per_step / 1000.0)
    peak_gb = (torch.cuda.max_memory_allocated(device) / 1e9
               if device.type == "cuda" else 0.0)

    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec/GPU for Llama-style causal LM at 1/2/4/8 "
                      "MI355X; peak HBM per GPU",
            "value": tokens_per_s,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong" if par == "tp" and world > 1 else "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "peak_hbm_gb": peak_gb,
            "config": {
                "model": model_name,
                "global_batch": B * dp_size,
                "seq_len": S,
                "parallelism": label,
            },
        }))
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
