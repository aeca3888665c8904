"""Data loading entry point — counterpart of the reference's
_load_and_preprocess_data (/root/reference/01-single-gpu/train_llm.py:192-245).

`--dataset-name synthetic` (the offline default here) builds the synthetic
dataset; any other name goes through HF `datasets` + `tokenizers` with the
same tokenize -> concat -> chunk flow as the reference, which works only when
the dataset/tokenizer are already cached on disk (this environment has no
network).
"""
import logging
from itertools import chain

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


def _load_hf_dataset(args, config):
    import multiprocessing

    import datasets
    from transformers import AutoTokenizer

    tokenizer = AutoTokenizer.from_pretrained(args.model_name)
    data = datasets.load_dataset(args.dataset_name, args.dataset_subset)
    column_names = data["train"].column_names
    text_column = "text" if "text" in column_names else column_names[0]

    def tokenize_function(examples):
        return tokenizer(examples[text_column])

    tokenized = data.map(
        tokenize_function, batched=True, remove_columns=column_names,
        num_proc=multiprocessing.cpu_count(), load_from_cache_file=True,
        desc="Running tokenizer on dataset")

    seq_length = args.seq_length or tokenizer.model_max_length
    if seq_length > config.max_position_embeddings:
        seq_length = min(1024, config.max_position_embeddings)

    def group_texts(examples):
        concatenated = {k: list(chain(*examples[k])) for k in examples}
        total = len(concatenated[list(examples.keys())[0]])
        if total > seq_length:
            total = (total // seq_length) * seq_length
        result = {
            k: [t[i: i + seq_length] for i in range(0, total, seq_length)]
            for k, t in concatenated.items()
        }
        result["labels"] = result["input_ids"].copy()
        return result

    lm = tokenized.map(
        group_texts, batched=True, num_proc=multiprocessing.cpu_count(),
        load_from_cache_file=True,
        desc=f"Grouping texts in chunks of {seq_length}")
    return lm["train"]
