"""utils/profiling.py: rocprof command construction + StepTimer (CPU)."""

import time

import torch

from gan_deeplearning4j_amd.utils.profiling import StepTimer, rocprof_cmd


def test_rocprof_cmd_trace_mode():
    c = rocprof_cmd("python bench.py --steps 3", out_dir="gpurun_out/x")
    assert c.startswith("rocprofv3 ")
    assert "--kernel-trace" in c and "--stats" in c
    assert "-d gpurun_out/x" in c
    assert c.endswith("-- python bench.py --steps 3")
    assert "--pmc" not in c


def test_rocprof_cmd_pmc_mode_excludes_trace_domains():
    # pool rule: --pmc must never be combined with trace domains
    c = rocprof_cmd("python x.py", pmc=["SQ_ACTIVE_INST_ANY", "SQ_WAIT_ANY"])
    assert "--pmc SQ_ACTIVE_INST_ANY,SQ_WAIT_ANY" in c
    assert "--kernel-trace" not in c
    assert "-trace" not in c


def test_step_timer_cpu():
    t = StepTimer(torch.device("cpu"))
    t.start()
    time.sleep(0.01)
    dt = t.stop()
    assert 0.005 < dt < 1.0
