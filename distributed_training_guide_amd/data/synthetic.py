"""Synthetic causal-LM dataset: deterministic random token rows of the same
shape the reference's pipeline produces ({input_ids, attention_mask, labels},
labels = input_ids — /root/reference/01-single-gpu/train_llm.py:221-235).

BASELINE.json mandates synthetic data / random-init weights (no network for
datasets or checkpoints); rows are reproducible per (seed, index) so resume
and determinism tests behave like a real on-disk dataset.
"""
import torch
from torch.utils.data import Dataset


class SyntheticTextDataset(Dataset):
    def __init__(self, vocab_size: int, seq_length: int,
                 num_samples: int = 4096, seed: int = 0):
        self.vocab_size = vocab_size
        self.seq_length = seq_length
        self.num_samples = num_samples
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + idx)
        ids = torch.randint(0, self.vocab_size, (self.seq_length,),
                            generator=g, dtype=torch.long)
        return {
            "input_ids": ids,
            "attention_mask": torch.ones(self.seq_length, dtype=torch.long),
            "labels": ids.clone(),
        }


def default_collate(batch):
    """Stack dict rows — the reference uses transformers'
    default_data_collator (01:69); same behavior for our fixed-length rows."""
    out = {}
    for k in batch[0]:
        out[k] = torch.stack([row[k] for row in batch])
    return out
