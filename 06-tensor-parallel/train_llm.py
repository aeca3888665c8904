#!/usr/bin/env python3
"""Chapter 6 — tensor + sequence parallelism over xGMI.

MI355X-native counterpart of
/root/reference/06-tensor-parallel/train_llm.py: the full TP plan
(vocab-parallel embedding -> per layer: SP norms, seq all-gather into
column-sharded qkv, row-sharded o_proj with seq reduce-scatter, same for
the MLP -> SP final norm -> column-sharded lm_head with replicate or
loss-parallel output) hand-placed on RCCL collectives
(parallel/tp.py), no DTensor.

    torchrun --standalone --nproc-per-node 8 \
        06-tensor-parallel/train_llm.py -m llama-3-8b -d synthetic -b 16
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from torch.distributed.elastic.multiprocessing.errors import record

from distributed_training_guide_amd.parallel.pg import destroy
from distributed_training_guide_amd.parallel.tp_strategy import TPStrategy
from distributed_training_guide_amd.trainer import get_parser, run_training


def build_parser():
    p = get_parser()
    p.add_argument("-tp", "--tensor-parallel", default=0, type=int,
                   help="tp degree (default: whole world)")
    p.add_argument("--loss-parallel", action="store_true",
                   help="fused CE over vocab-sharded logits "
                        "(06-.../README.md:243-271)")
    p.add_argument("--bucket-cap-mb", default=128, type=int)
    return p


@record
def main(argv=None):
    args = build_parser().parse_args(argv)
    strategy = TPStrategy(args)
    try:
        return run_training(args, strategy)
    finally:
        destroy()


if __name__ == "__main__":
    main()
