"""Rank-aware logging — the reference prefixes every record with [rank=N]
(/root/reference/02-distributed-data-parallel/train_llm.py:43-46)."""
import logging
import os


def setup_logging(rank: int | None = None, level=logging.INFO):
    if rank is None:
        rank = int(os.environ.get("RANK", 0))
    logging.basicConfig(
        format=f"[rank={rank}] [%(asctime)s] %(levelname)s:%(message)s",
        level=level,
        force=True,
    )
