"""PMC probe: run the 8p TN kernel on square-4k for counter collection."""
import os, sys, torch
os.environ.setdefault("GDLJ_8P_MINM", "0")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from gan_deeplearning4j_amd.ops.backend import hip_ext
e = hip_ext()
A = torch.randn(4096, 4096, device="cuda").to(torch.bfloat16)
B = torch.randn(4096, 4096, device="cuda").to(torch.bfloat16)
for _ in range(30):
    e.gemm_tn(A, B, None, 0, 0.0, False)
torch.cuda.synchronize()
print("done")
