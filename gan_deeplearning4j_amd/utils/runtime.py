"""Device-runtime configuration (the reference's CudaEnvironment block,
Java:103-115: float dtype, allowMultiGPU, device cache limit, cross-device
access, verbose, auto-GC window — re-expressed for ROCm/HIP).

On MI355X the analogous knobs are the HIP caching-allocator settings, the
per-process visible devices, and deterministic/verbose toggles.
"""

from __future__ import annotations

import logging
import os

import torch

log = logging.getLogger("gan_deeplearning4j_amd")


def configure_runtime(
    use_gpu: bool = True,
    allocator_max_split_mb: int | None = None,
    garbage_collection_threshold: float | None = None,
    verbose: bool = False,
    deterministic: bool = False,
) -> torch.device:
    """Configure the HIP runtime/allocator; returns the selected device.

    - allocator_max_split_mb / garbage_collection_threshold map to
      PYTORCH_HIP_ALLOC_CONF (the reference's setMaximumDeviceCache /
      auto-GC window analogs). Must run before the first allocation.
    - Multi-GPU visibility is per-process (one rank per GPU over RCCL),
      so the reference's allowMultiGPU/allowCrossDeviceAccess toggles
      map to process-group membership + dmabuf IPC
      (HSA_ENABLE_IPC_MODE_LEGACY=0), not an in-process flag.
    """
    conf = []
    if allocator_max_split_mb is not None:
        conf.append(f"max_split_size_mb:{allocator_max_split_mb}")
    if garbage_collection_threshold is not None:
        conf.append(f"garbage_collection_threshold:{garbage_collection_threshold}")
    if conf:
        os.environ.setdefault("PYTORCH_HIP_ALLOC_CONF", ",".join(conf))
    if verbose:
        os.environ.setdefault("AMD_LOG_LEVEL", "3")
        log.setLevel(logging.DEBUG)
    if deterministic:
        torch.use_deterministic_algorithms(True, warn_only=True)
    if use_gpu and torch.cuda.is_available():
        dev = torch.device("cuda", 0)
        if verbose:
            props = torch.cuda.get_device_properties(dev)
            log.info("device: %s, %d CUs, %.0f GB HBM", props.name,
                     props.multi_processor_count,
                     props.total_memory / 2**30)
        return dev
    return torch.device("cpu")


def memory_summary(device=None) -> dict:
    """Allocator stats (the reference's verbose memory manager analog)."""
    if not torch.cuda.is_available():
        return {}
    return {
        "allocated_gb": torch.cuda.memory_allocated(device) / 2**30,
        "reserved_gb": torch.cuda.memory_reserved(device) / 2**30,
        "max_allocated_gb": torch.cuda.max_memory_allocated(device) / 2**30,
    }
