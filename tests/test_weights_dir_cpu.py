"""download.py safetensors prefetch -> --weights-dir broadcast init
(reference 05: download.py:6-21 + from_pretrained/broadcast 05:76-126)."""
import importlib.util
from pathlib import Path

import torch

from utils_dist import run_dist

REPO = Path(__file__).resolve().parent.parent


def _write_weights(tmpdir):
    spec = importlib.util.spec_from_file_location(
        "dl", REPO / "05-training-llama-405b" / "download.py")
    dl = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(dl)
    dl.main(["-m", "llama-debug", "--dest", tmpdir, "--shard-gb", "0.01"])


def _worker(rank, world, tmpdir):
    from distributed_training_guide_amd.models import build_model
    from distributed_training_guide_amd.parallel.fsdp import FSDP
    from distributed_training_guide_amd.parallel.fsdp_strategy import \
        _load_safetensors_dir

    with torch.device("meta"):
        model = build_model("llama-debug", dtype=torch.float32)
    model = FSDP(model, device=torch.device("cpu"))
    full = {}
    if rank == 0:
        full = _load_safetensors_dir(Path(tmpdir), "llama-debug")
    model.load_full_state_dict(full, broadcast_from_rank0=True)

    got = model.full_state_dict(rank0_only=False, offload_to_cpu=True)
    want = _load_safetensors_dir(Path(tmpdir), "llama-debug")
    for n, t in want.items():
        assert torch.equal(got[n].float(), t.float()), n


def test_weights_dir_roundtrip(tmp_path):
    _write_weights(str(tmp_path))
    assert (tmp_path / "llama-debug" /
            "model.safetensors.index.json").exists()
    run_dist(_worker, world_size=2, args=(str(tmp_path),))
