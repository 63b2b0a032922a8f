"""gan_deeplearning4j_amd — an MI355X-native GAN training framework.

A from-scratch re-design of the capabilities of `hamaadshah/gan_deeplearning4j`
(reference: Java/src/main/java/org/deeplearning4j/dl4jGANComputerVision.java)
for AMD Instinct MI355X (gfx950, CDNA4):

- PyTorch-ROCm as the tensor/autograd substrate.
- Hand-written HIP/CDNA4 kernels (MFMA im2col-GEMM convs, BatchNorm,
  activations, BCE-with-logits, fused Adam/RMSProp) in `ops/hip/`.
- RCCL over xGMI data parallelism (one process per GPU) in `parallel/`.
- DL4J-style ComputationGraph API (`graph/`): GraphBuilder, fit()/output(),
  get_param()/set_param(), TransferLearning, ModelSerializer-compatible
  .zip checkpoints.
- The reference's alternating D/G/classifier GAN protocol (`train/`),
  including label softening, frozen-copy weight sync and parameter-averaging
  semantics (reference Java:408-621).
"""

__version__ = "1.0.0"

from . import config  # noqa: F401

__all__ = ["config", "__version__"]
