#!/usr/bin/env python3
"""Chapter 5 — very-large-model FSDP (the reference's Llama-405B chapter,
/root/reference/05-training-llama-405b/train_llm.py).

Everything from chapter 4 plus the large-model options: activation
checkpointing (05:165-178), explicit prefetch depth (05:148-161), CPU
offload with per-rank thread tuning (05:69-72), broadcast-from-rank-0 full
state init (05:118-126).  There is no network in this environment, so the
weights are random-init at the named architecture (BASELINE.json: synthetic
data / random-init); --broadcast-init exercises the rank-0 broadcast path.

    torchrun --standalone --nproc-per-node 8 05-training-llama-405b/train_llm.py \
        -m llama-3-70b -d synthetic --checkpoint-activations -s 4096 -b 1
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
    p.add_argument("--cpu-offload", action="store_true")
    p.add_argument("--checkpoint-activations", action="store_true")
    p.add_argument("--prefetch-layers", default=1, type=int,
                   help="how many layer all-gathers to prefetch ahead")
    p.add_argument("--broadcast-init", action="store_true",
                   help="init via rank-0 full state dict broadcast "
                        "(the pretrained-weights path of the reference)")
    p.add_argument("--weights-dir", default=None,
                   help="node-local safetensors dir written by download.py; "
                        "rank 0 loads on CPU and broadcasts shards")
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
