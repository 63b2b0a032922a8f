"""Profiling helpers: rocprofv3 integration + step timing.

The reference has no tracing; here rocprof is first-class:
`rocprof_cmd` builds the rocprofv3 command line used to profile any
framework entrypoint, and `StepTimer` gives device-synchronized per-step
wall times for the metrics logger.
"""

from __future__ import annotations

import time

import torch


def rocprof_cmd(cmd: str, out_dir: str = "gpurun_out/prof",
                stats: bool = True, kernel_trace: bool = True,
                pmc: list[str] | None = None) -> str:
    """Build a rocprofv3 command. NOTE: --pmc must NOT be combined with
    sys/runtime/hip/hsa trace domains (pool rule); collect counters in a
    stats-only run."""
    parts = ["rocprofv3"]
    if pmc:
        parts += ["--pmc", ",".join(pmc)]
        if stats:
            parts += ["--stats"]
    else:
        if kernel_trace:
            parts += ["--kernel-trace"]
        if stats:
            parts += ["--stats"]
    parts += ["-d", out_dir, "--", cmd]
    return " ".join(parts)


class StepTimer:
    """Device-synchronized step timer (max-over-ranks is applied by the
    bench harness)."""

    def __init__(self, device: torch.device):
        self.device = device
        self._t0 = None

    def start(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self._t0 = time.perf_counter()

    def stop(self) -> float:
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        return time.perf_counter() - self._t0
