#!/usr/bin/env python3
"""Chapter 4 — fully sharded data parallelism (FSDP).

MI355X-native counterpart of
/root/reference/04-fully-sharded-data-parallel/train_llm.py: meta-device
init, per-decoder-layer + root flat-param sharding with
reshard_after_forward, bf16 params / fp32 gradient reduce-scatter, optional
CPU offload, model.unshard() prefetch, sharded per-rank checkpoint files
with metadata and reshard-on-load — on OUR flat-param engine over
RCCL/xGMI (parallel/fsdp.py), not torch FSDP.

    torchrun --standalone --nproc-per-node 8 \
        04-fully-sharded-data-parallel/train_llm.py -m llama-3-8b -d synthetic
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from torch.distributed.elastic.multiprocessing.errors import record

from distributed_training_guide_amd.parallel.fsdp_strategy import FSDPStrategy
from distributed_training_guide_amd.parallel.pg import destroy
from distributed_training_guide_amd.trainer import get_parser, run_training


def build_parser():
    p = get_parser()
    p.add_argument("--cpu-offload", action="store_true",
                   help="shards + optimizer state + update on host "
                        "(reference 04:384)")
    return p


@record
def main(argv=None):
    args = build_parser().parse_args(argv)
    strategy = FSDPStrategy(args)
    try:
        return run_training(args, strategy)
    finally:
        destroy()


if __name__ == "__main__":
    main()
