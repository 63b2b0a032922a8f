"""Structured per-step metrics (the reference logs only
slf4j phase lines; quantitative results were offline CSVs — here metrics
are in-process, structured and timed)."""

from __future__ import annotations

import json
import logging
import time
from pathlib import Path
from typing import Optional

log = logging.getLogger("gan_deeplearning4j_amd")


class MetricsLogger:
    def __init__(self, out_dir: Optional[str] = None, print_every: int = 1,
                 is_main: bool = True):
        self.print_every = print_every
        self.is_main = is_main
        self._file = None
        if out_dir is not None and is_main:
            p = Path(out_dir)
            p.mkdir(parents=True, exist_ok=True)
            self._file = open(p / "metrics.jsonl", "a")
        self._t_last = time.perf_counter()

    def step(self, it: int, **metrics):
        now = time.perf_counter()
        metrics["step_time_s"] = round(now - self._t_last, 6)
        self._t_last = now
        metrics["iter"] = it
        if self._file is not None:
            self._file.write(json.dumps(metrics) + "\n")
            self._file.flush()
        if self.is_main and it % self.print_every == 0:
            pretty = " ".join(
                f"{k}={v:.5g}" if isinstance(v, float) else f"{k}={v}"
                for k, v in metrics.items()
            )
            log.info("step %s", pretty)

    def close(self):
        if self._file is not None:
            self._file.close()
