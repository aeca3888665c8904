"""Resume bitwise-determinism: --deterministic saves/restores numpy/random/
torch RNG state at checkpoints (the determinism recipe,
reference related-topics/determinism/README.md:46-68), so 4+2 steps with a
resume is BITWISE identical to 6 straight steps."""
import importlib.util
from pathlib import Path

import torch

REPO = Path(__file__).resolve().parent.parent


def _run(tmp_path, extra):
    spec = importlib.util.spec_from_file_location(
        "ch1", REPO / "01-single-gpu" / "train_llm.py")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    args = ["-m", "llama-debug", "-d", "synthetic", "-b", "2", "-s", "32",
            "--num-samples", "16", "--save-dir", str(tmp_path),
            "--device", "cpu", "--lr", "1e-2", "--deterministic"] + extra
    return mod.main(args)


def test_resume_bitwise_determinism(tmp_path):
    torch.manual_seed(0)
    _run(tmp_path, ["--max-steps", "6", "-e", "straight", "--ckpt-freq", "6"])
    sd_straight = torch.load(tmp_path / "straight" / "model.pt",
                             weights_only=True)

    torch.manual_seed(0)
    _run(tmp_path, ["--max-steps", "4", "-e", "resumed", "--ckpt-freq", "4"])
    assert (tmp_path / "resumed" / "rng_rank0.pt").exists(), \
        "--deterministic must checkpoint RNG state"
    _run(tmp_path, ["--max-steps", "6", "-e", "resumed", "--ckpt-freq", "2"])
    sd_resumed = torch.load(tmp_path / "resumed" / "model.pt",
                            weights_only=True)

    for k in sd_straight:
        assert torch.equal(sd_straight[k], sd_resumed[k]), \
            f"{k} differs after resume (not bitwise deterministic)"
