import sys, torch
sys.path.insert(0, "/root/repo")
from gan_deeplearning4j_amd.ops.backend import hip_ext
e = hip_ext()
DEV = "cuda:0"

# 1. large-tensor byte equality: delayed(first=True) vs exact
g = torch.Generator().manual_seed(0)
x = (torch.randn(6_000_000 * 2, generator=g) * 2).to(DEV, torch.bfloat16)
q_ref, s_ref, i_ref = e.fp8_quantize(x)
scale = torch.ones(1, device=DEV); inv = torch.ones(1, device=DEV)
amax = torch.zeros(1, dtype=torch.int32, device=DEV)
q_d = e.fp8_quantize_delayed(x, scale, inv, amax, True)
torch.cuda.synchronize()
print("bytes equal:", bool((q_ref == q_d).all()),
      "scale ref/delayed:", float(s_ref), float(scale),
      "inv:", float(i_ref), float(inv))

# 2. second call on a same-scale tensor: bytes should match exact quant
x2 = (torch.randn(6_000_000 * 2, generator=g) * 2).to(DEV, torch.bfloat16)
q2_d = e.fp8_quantize_delayed(x2, scale, inv, amax, False)
q2_ref, s2_ref, i2_ref = e.fp8_quantize(x2)
torch.cuda.synchronize()
mism = (q2_d != q2_ref).float().mean()
print("call2 mismatch frac:", float(mism), "scales:", float(scale), float(s2_ref))

# 3. trainer with per-layer tracing
from gan_deeplearning4j_amd.ops import gpu_ops
orig = e.conv_fwd_implicit_fp8
calls = [0]
def traced(*a):
    y = orig(*a)
    torch.cuda.synchronize()
    xq, wq = a[0], a[1]
    ix, iw = a[3], a[4]
    fin = bool(torch.isfinite(y.float()).all())
    calls[0] += 1
    print(f"call{calls[0]}: mode={a[-1]} M={y.shape[0]} N={y.shape[1]} "
          f"ix={float(ix):.4g} iw={float(iw):.4g} finite={fin} "
          f"ymax={float(y.float().abs().max()):.4g}", flush=True)
    return y
e.conv_fwd_implicit_fp8 = traced
import gan_deeplearning4j_amd.ops.gpu_ops as go
from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer
import os
os.environ["GDLJ_FP8_CONVT"] = "0"
os.environ["GDLJ_FP8_BWD"] = "0"
cfg = preset("dcgan128")
gen, dis = build_dcgan(cfg)
tr = GanTrainer(gen, dis, cfg, device=torch.device(DEV),
                dtype=torch.bfloat16, capture=False)
torch.manual_seed(0)
real = (torch.rand(8, 3, 128, 128, device=DEV, dtype=torch.bfloat16) * 2 - 1)
out = tr.step(real)
torch.cuda.synchronize()
print("losses:", {k: float(v) for k, v in out.items()})
