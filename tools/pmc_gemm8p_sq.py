import sys, torch
sys.path.insert(0, "/root/repo")
from gan_deeplearning4j_amd.ops.backend import hip_ext
e = hip_ext()
A = torch.randn(4096, 4096, device="cuda").to(torch.bfloat16)
B = torch.randn(4096, 4096, device="cuda").to(torch.bfloat16)
for _ in range(30):
    e.gemm_tn(A, B, None, 0, 0.0, False)
torch.cuda.synchronize()
print("done")
