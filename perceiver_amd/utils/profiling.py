"""Profiling helpers (the tracing/observability subsystem the reference lacks,
SURVEY.md §5): CUDA-event step timers, a torch.profiler wrapper for HIP kernel
traces, and the rocprofv3 invocation recipe for PMC counters."""
from __future__ import annotations

import contextlib
import json
import time
from typing import Dict, List, Optional

import torch


class StepTimer:
    """Per-step wall/GPU timing with CUDA events; summary() gives mean/min ms."""

    def __init__(self, device: Optional[torch.device] = None):
        self.use_cuda = torch.cuda.is_available() if device is None else device.type == "cuda"
        self.samples_ms: List[float] = []
        self._start = None

    def __enter__(self):
        if self.use_cuda:
            self._ev0 = torch.cuda.Event(enable_timing=True)
            self._ev1 = torch.cuda.Event(enable_timing=True)
            self._ev0.record()
        self._start = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.use_cuda:
            self._ev1.record()
            self._ev1.synchronize()
            self.samples_ms.append(self._ev0.elapsed_time(self._ev1))
        else:
            self.samples_ms.append((time.perf_counter() - self._start) * 1e3)

    def summary(self) -> Dict[str, float]:
        if not self.samples_ms:
            return {}
        return {
            "mean_ms": sum(self.samples_ms) / len(self.samples_ms),
            "min_ms": min(self.samples_ms),
            "max_ms": max(self.samples_ms),
            "n": len(self.samples_ms),
        }


@contextlib.contextmanager
def kernel_trace(out_path: str, wait: int = 0, warmup: int = 1, active: int = 3):
    """torch.profiler wrapper: records HIP kernel activity and writes a chrome
    trace; use rocprofv3 (see tools/) for hardware PMC counters."""
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=activities) as prof:
        yield prof
    prof.export_chrome_trace(out_path)


def kernel_table(prof, row_limit: int = 25) -> str:
    return prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=row_limit)


ROCPROF_RECIPE = """\
# per-kernel time:  (run from /tmp with TMPDIR=/tmp)
rocprofv3 --kernel-trace --stats -d OUT -- python bench.py --steps 3 --warmup 1
# PMC counters (separate run; never combined with trace domains):
rocprofv3 --pmc SQ_LDS_BANK_CONFLICT,SQ_INSTS_MFMA,SQ_WAVE_CYCLES -d OUT -- <cmd>
"""


def log_summary(path: str, record: dict):
    with open(path, "a") as f:
        f.write(json.dumps(record) + "\n")
