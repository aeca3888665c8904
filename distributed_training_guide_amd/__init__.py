"""MI355X-native distributed training framework.

A from-scratch rebuild of the capabilities of
LambdaLabsML/distributed-training-guide designed for AMD
MI355X (CDNA4 / gfx950): the transformer hot path is hand-written HIP with
MFMA/LDS tiling (``_hip/``), and the distributed mechanics (DDP gradient
buckets, ZeRO-1, FSDP flat-param sharding, tensor+sequence parallelism, 2D
FSDP x TP) are implemented here on RCCL-over-xGMI collectives — not on
torch's DDP/FSDP/DTensor wrappers.

Layout:
    ops/       HIP-kernel-backed autograd ops (+ fp32 eager references)
    models/    Llama / GPT-2 model families built on ops/
    parallel/  process bootstrap, device mesh, DDP, ZeRO-1, FSDP, TP/SP, 2D
    data/      synthetic + on-disk data pipelines, distributed sampler
    utils/     timers, memory stats, checkpointing, rank logging, elastic
    trainer.py shared trainer loop behind every chapter entrypoint
"""

__version__ = "0.2.0"
