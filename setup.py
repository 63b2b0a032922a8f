"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Kernels (.hip) are compiled directly by hipcc for gfx950; ext.cpp (torch
bindings) goes through torch.utils.cpp_extension so it links against the
running PyTorch. The resulting _C*.so lives inside the package directory
(it travels with repo snapshots; no JIT cache involvement).
"""

import os
import subprocess
import sys
from pathlib import Path

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils import cpp_extension  # noqa: E402

ROOT = Path(__file__).resolve().parent
HIP_DIR = ROOT / "gan_deeplearning4j_amd" / "ops" / "hip"
OBJ_DIR = ROOT / "build" / "hip_obj"

HIP_SOURCES = sorted(HIP_DIR.glob("*.hip"))
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
HIP_FLAGS = [
    "--offload-arch=gfx950",
    "-O3",
    "-std=c++17",
    "-fPIC",
    "-fgpu-rdc" if False else "-fno-gpu-rdc",
    "-c",
]


def compile_hip_objects() -> list[str]:
    OBJ_DIR.mkdir(parents=True, exist_ok=True)
    objs = []
    for src in HIP_SOURCES:
        obj = OBJ_DIR / (src.stem + ".o")
        if not obj.exists() or obj.stat().st_mtime < src.stat().st_mtime or \
                obj.stat().st_mtime < (HIP_DIR / "common.h").stat().st_mtime:
            cmd = [HIPCC, *HIP_FLAGS, str(src), "-o", str(obj)]
            print("[hipcc]", " ".join(cmd))
            subprocess.run(cmd, check=True)
        objs.append(str(obj))
    return objs


ext = cpp_extension.CUDAExtension(
    name="gan_deeplearning4j_amd._C",
    sources=[str(HIP_DIR / "ext.cpp"), str(HIP_DIR / "csv_loader.cpp")],
    extra_objects=compile_hip_objects(),
    extra_compile_args={"cxx": ["-O2"], "nvcc": ["-O2"]},
)

setup(
    name="gan_deeplearning4j_amd",
    version="0.1.0",
    packages=["gan_deeplearning4j_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
