#!/usr/bin/env python3
"""Prefetch model weights to node-local storage before a multi-node launch.

MI355X counterpart of /root/reference/05-training-llama-405b/download.py:6-21.
The reference pulls Llama-3.1-405B from the HuggingFace hub into $HF_HOME on
every node (config+tokenizer everywhere, full weights only where needed via
--skip-model) because loading 764 GB from shared NFS takes ~50 min vs ~3 min
node-local (reference 05-.../README.md:48-55).

This environment has no network, so "prefetch" here means materializing the
random-init weights once into a node-local safetensors shard directory that
`--broadcast-init` / FSDPStrategy can load from disk on rank 0.  Run it on
every node; pass --skip-model on nodes that only need config metadata.

    python 05-training-llama-405b/download.py -m llama-3-8b --dest /tmp/weights
    python 05-training-llama-405b/download.py -m llama-3-8b --dest /tmp/weights --skip-model
"""
import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("-m", "--model-name", required=True)
    p.add_argument("--dest", default="/tmp/dtga-weights",
                   help="node-local directory (NOT shared NFS)")
    p.add_argument("--skip-model", action="store_true", default=False,
                   help="only write config (for non-0 nodes)")
    p.add_argument("--shard-gb", default=4.0, type=float,
                   help="max safetensors shard size")
    args = p.parse_args(argv)

    from distributed_training_guide_amd.models import get_config

    import dataclasses

    config = get_config(args.model_name)
    dest = Path(args.dest) / args.model_name
    dest.mkdir(parents=True, exist_ok=True)
    (dest / "config.json").write_text(
        json.dumps(dataclasses.asdict(config), indent=2))
    print(f"wrote {dest/'config.json'}")
    if args.skip_model:
        return

    import torch
    from safetensors.torch import save_file

    from distributed_training_guide_amd.models import build_model

    torch.manual_seed(0)
    model = build_model(config, device=torch.device("cpu"),
                        dtype=torch.bfloat16)
    limit = int(args.shard_gb * (1 << 30))
    shard, shard_bytes, n_shard, index = {}, 0, 0, {}
    # drop aliases of already-seen storages (tied embeddings): the loader
    # re-ties on load, and safetensors refuses shared tensors
    names, seen = [], {}
    for name, t in model.state_dict().items():
        key = (t.untyped_storage().data_ptr(), t.stride(), t.shape)
        if key in seen:
            continue
        seen[key] = name
        names.append((name, t))

    def flush():
        nonlocal shard, shard_bytes, n_shard
        if not shard:
            return
        fname = f"model-{n_shard:05d}.safetensors"
        save_file(shard, str(dest / fname))
        for k in shard:
            index[k] = fname
        print(f"wrote {dest/fname} ({shard_bytes/(1<<30):.2f} GB)")
        shard, shard_bytes, n_shard = {}, 0, n_shard + 1

    for name, t in names:
        nbytes = t.numel() * t.element_size()
        if shard_bytes + nbytes > limit:
            flush()
        shard[name] = t.contiguous()
        shard_bytes += nbytes
    flush()
    (dest / "model.safetensors.index.json").write_text(
        json.dumps({"weight_map": index}, indent=2))
    print(f"wrote index ({len(index)} tensors, {n_shard} shards)")


if __name__ == "__main__":
    main()
