"""Phase timers + memory stats — same observable semantics as the
reference's LocalTimer / get_mem_stats
(/root/reference/01-single-gpu/train_llm.py:248-285): device-synchronized
enter/exit fences so the four phases (data/forward/backward/update) are
comparable with the reference's published numbers."""
import time

import torch


class LocalTimer:
    def __init__(self, device: torch.device):
        if device.type == "cuda":
            self.synchronize = lambda: torch.cuda.synchronize(device=device)
        else:
            self.synchronize = lambda: None
        self.measurements = []
        self.start_time = None

    def __enter__(self):
        self.synchronize()
        self.start_time = time.time()
        return self

    def __exit__(self, exc_type, value, tb):
        if tb is None:
            self.synchronize()
            self.measurements.append(time.time() - self.start_time)
        self.start_time = None

    def avg_elapsed_ms(self):
        if not self.measurements:
            return 0.0
        return 1000 * (sum(self.measurements) / len(self.measurements))

    def reset(self):
        self.measurements = []
        self.start_time = None


def get_mem_stats(device=None):
    """HBM stats dict with the reference's keys (01:248-257)."""
    if not torch.cuda.is_available():
        return {"total_gb": 0.0, "curr_alloc_gb": 0.0, "peak_alloc_gb": 0.0,
                "curr_resv_gb": 0.0, "peak_resv_gb": 0.0}
    mem = torch.cuda.memory_stats(device)
    props = torch.cuda.get_device_properties(device)
    return {
        "total_gb": 1e-9 * props.total_memory,
        "curr_alloc_gb": 1e-9 * mem["allocated_bytes.all.current"],
        "peak_alloc_gb": 1e-9 * mem["allocated_bytes.all.peak"],
        "curr_resv_gb": 1e-9 * mem["reserved_bytes.all.current"],
        "peak_resv_gb": 1e-9 * mem["reserved_bytes.all.peak"],
    }


def reset_peak_memory_stats(device=None):
    if torch.cuda.is_available():
        torch.cuda.reset_peak_memory_stats(device)
