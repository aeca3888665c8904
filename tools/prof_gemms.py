#!/usr/bin/env python3
"""Per-shape GEMM timing via torch.profiler (find inefficient GEMMs)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

from distributed_training_guide_amd.models import build_model, get_config
from distributed_training_guide_amd.ops import FusedAdamW

config = get_config("llama-3-8b")
model = build_model(config, device=torch.device("cuda"), dtype=torch.bfloat16)
opt = FusedAdamW(model.parameters(), lr=3e-5)
ids = torch.randint(0, config.vocab_size, (16, 1024), device="cuda")

def step():
    out = model(input_ids=ids, labels=ids)
    out.loss.backward()
    opt.step()
    opt.zero_grad(set_to_none=True)

for _ in range(2):
    step()
torch.cuda.synchronize()
from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CUDA], record_shapes=True) as prof:
    step()
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True).table(
    sort_by="cuda_time_total", row_limit=25, max_name_column_width=40,
    max_shapes_column_width=60))
