#!/usr/bin/env python3
"""Chapter 1 — single-GPU (or CPU) causal-LM training.

MI355X-native counterpart of
/root/reference/01-single-gpu/train_llm.py: same CLI, timers, log dict,
checkpoint files {model.pt, optimizer.pt, lr_scheduler.pt, state.json} and
resume-with-batch-skip behavior; the model/ops stack is this repo's HIP
kernel path (see distributed_training_guide_amd/).

    python 01-single-gpu/train_llm.py -m gpt2 -d synthetic -s 1024 -b 8
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from distributed_training_guide_amd.parallel.single import SingleDeviceStrategy
from distributed_training_guide_amd.trainer import get_parser, run_training


def main(argv=None):
    args = get_parser().parse_args(argv)
    strategy = SingleDeviceStrategy(args)
    return run_training(args, strategy)


if __name__ == "__main__":
    main()
