"""Memory accounting (reference include/memory_info.h:28
MemoryInfo::getMaxMemoryUsage + memory_use.cu tests)."""

from __future__ import annotations

import resource

import torch


class MemoryInfo:
    @staticmethod
    def get_max_memory_usage() -> dict:
        """Peak device + host memory of this process in MiB."""
        out = {"host_peak_mib":
               resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0}
        if torch.cuda.is_available():
            out["device_peak_mib"] = \
                torch.cuda.max_memory_allocated() / (1024.0 ** 2)
            out["device_reserved_mib"] = \
                torch.cuda.max_memory_reserved() / (1024.0 ** 2)
        return out

    @staticmethod
    def reset_peaks():
        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
