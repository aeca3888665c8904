#!/usr/bin/env python3
"""Chapter 7 — 2D parallelism: FSDP(dp) x TP(tp) on one 8-GPU node.

MI355X-native counterpart of /root/reference/07-2d-parallel/train_llm.py:
the chapter-6 TP plan across the inner mesh dim plus the flat-param FSDP
engine sharding each (tp-local) decoder layer across the outer dp dim —
two communicator sets on one node, sized for 288 GB HBM per GPU
(BASELINE.json: Llama-3-70B FSDP(4) x TP(2)).

    torchrun --standalone --nproc-per-node 8 07-2d-parallel/train_llm.py \
        -m llama-3-70b -d synthetic --tensor-parallel 2
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from torch.distributed.elastic.multiprocessing.errors import record

from distributed_training_guide_amd.parallel.pg import destroy
from distributed_training_guide_amd.parallel.tp_strategy import TwoDStrategy
from distributed_training_guide_amd.trainer import get_parser, run_training


def build_parser():
    p = get_parser()
    p.add_argument("-tp", "--tensor-parallel", default=8, type=int)
    p.add_argument("--loss-parallel", action="store_true")
    p.add_argument("--checkpoint-activations", action="store_true")
    p.add_argument("--cpu-offload", action="store_true")
    return p


@record
def main(argv=None):
    args = build_parser().parse_args(argv)
    strategy = TwoDStrategy(args)
    try:
        return run_training(args, strategy)
    finally:
        destroy()


if __name__ == "__main__":
    main()
