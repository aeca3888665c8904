from .timers import LocalTimer, get_mem_stats, reset_peak_memory_stats
from .logging import setup_logging

__all__ = ["LocalTimer", "get_mem_stats", "reset_peak_memory_stats",
           "setup_logging"]
