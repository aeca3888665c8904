"""Data loading entry point — counterpart of the reference's
_load_and_preprocess_data (/root/reference/01-single-gpu/train_llm.py:192-245).

`--dataset-name synthetic` (the offline default here) builds the synthetic
dataset; any other name goes through HF `datasets` + `tokenizers`, which
works only when the dataset/tokenizer are already cached on disk (this
environment has no network).

Packing strategy (ours, not the reference's run_clm map(group_texts)
boilerplate): tokenized documents are concatenated into ONE flat int32
token stream with an EOS separator, and the dataset serves fixed-length
row views of that stream — a single pass over the corpus, O(1) row
access, and the same {input_ids, attention_mask, labels} row shape the
trainer expects.
"""
import logging

import torch
from torch.utils.data import Dataset

LOGGER = logging.getLogger(__name__)


def load_and_preprocess_data(args, config):
    name = getattr(args, "dataset_name", None) or "synthetic"
    if name == "synthetic":
        from .synthetic import SyntheticTextDataset

        return SyntheticTextDataset(
            vocab_size=config.vocab_size,
            seq_length=args.seq_length,
            num_samples=getattr(args, "num_samples", 4096),
            seed=args.seed,
        )
    return _load_hf_dataset(args, config)


class PackedTokenDataset(Dataset):
    """Fixed-length rows over one flat token stream (int32 on host; rows
    come out int64 as the CE kernel expects)."""

    def __init__(self, stream: torch.Tensor, seq_length: int):
        self.stream = stream
        self.seq_length = seq_length
        self.n = stream.numel() // seq_length

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        L = self.seq_length
        ids = self.stream[idx * L: (idx + 1) * L].long()
        return {
            "input_ids": ids,
            "attention_mask": torch.ones(L, dtype=torch.long),
            "labels": ids.clone(),
        }


def _load_hf_dataset(args, config):
    """Tokenize an on-disk HF dataset and pack it into a flat stream.
    All ranks may call this concurrently; wrap in rank0_first at the call
    site so rank 0 populates the tokenizer/dataset cache before the rest
    read it (reference 02:72-73)."""
    import numpy as np

    import datasets
    from transformers import AutoTokenizer

    tokenizer = AutoTokenizer.from_pretrained(args.model_name)
    ds = datasets.load_dataset(args.dataset_name, args.dataset_subset)
    ds = ds["train"] if "train" in ds else next(iter(ds.values()))
    text_col = "text" if "text" in ds.column_names else ds.column_names[0]

    import multiprocessing

    tokenized = ds.map(
        lambda batch: {"ids": tokenizer(batch[text_col])["input_ids"]},
        batched=True, remove_columns=ds.column_names,
        num_proc=max(1, multiprocessing.cpu_count() // 2),
        load_from_cache_file=True, desc="tokenize")

    seq_length = args.seq_length or tokenizer.model_max_length
    seq_length = min(seq_length, config.max_position_embeddings)

    eos = tokenizer.eos_token_id
    pieces = []
    for row in tokenized["ids"]:
        pieces.append(np.asarray(row, dtype=np.int32))
        if eos is not None and (len(row) == 0 or row[-1] != eos):
            pieces.append(np.asarray([eos], dtype=np.int32))
    stream = torch.from_numpy(np.concatenate(pieces)) if pieces else \
        torch.empty(0, dtype=torch.int32)
    n_rows = stream.numel() // seq_length
    LOGGER.info(f"packed {stream.numel()} tokens into {n_rows} rows of "
                f"{seq_length}")
    return PackedTokenDataset(stream, seq_length)
