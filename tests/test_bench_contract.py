"""Driver-contract tests for bench.py.

Runs the EXACT launch shapes the driver uses: single-process, and
torchrun --nnodes=1 --nproc-per-node 2 with gloo on CPU (world_size>1
rendezvous on 127.0.0.1).
"""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _check_json_line(out: str, n_gpus_expected: int):
    lines = [l for l in out.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{out[-2000:]}"
    d = json.loads(lines[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["value"] > 0
    assert d["n_gpus"] == n_gpus_expected
    assert d["data"] == "synthetic"
    assert d["scaling"] == "weak"
    assert "global_batch" in d["config"]
    return d


def test_bench_single_process_cpu():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--batch", "8", "--arch", "dcgan28"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    d = _check_json_line(r.stdout, 1)
    assert d["config"]["parallelism"] == "dp1"


def test_bench_torchrun_2proc_cpu():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29571", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch", "8", "--arch", "dcgan28"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=900,
    )
    assert r.returncode == 0, r.stderr[-3000:]
    d = _check_json_line(r.stdout, 2)
    # whole-job value over both ranks; weak scaling => global batch 16
    assert d["config"]["global_batch"] == 16
    assert d["config"]["parallelism"] == "dp2"
