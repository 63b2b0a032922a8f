"""FetchSize probe for the conv1-class forward gather.

conv1 dcgan64 at bench scale: x [N,8,64,64] NHWC (C8=8), K=5x5x8=200
-> pad 256, Cout=64. Logical staged A bytes per call = M*K*2; unique
image bytes = N*64*64*8*2. The FETCH_SIZE counter (x1024 B, ~2x
undercount on wide streams per the guide) tells how much HBM re-read
the L2 actually absorbs.
"""
import sys
import torch

sys.path.insert(0, "/root/repo")
from gan_deeplearning4j_amd.ops import gpu_ops

N = 4096
x = torch.randn(N, 8, 64, 64, device="cuda").to(torch.bfloat16)
w = (torch.randn(64, 8, 5, 5, device="cuda") * 0.1).to(torch.bfloat16)
for _ in range(5):
    y = gpu_ops.conv2d(x, w, None, 2, 2, "lrelu", 0.2)
torch.cuda.synchronize()
M = N * 32 * 32
print("logical A bytes/call:", M * 256 * 2 / 1e9, "GB; unique image:",
      N * 64 * 64 * 8 * 2 / 1e9, "GB; out:", M * 64 * 2 / 1e9, "GB")
