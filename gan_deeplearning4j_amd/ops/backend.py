"""HIP extension loader.

The extension is built IN-TREE (setup.py build_ext --inplace) so the .so
travels with the repo snapshot to GPU machines.  On a GPU box the compute
path must be the HIP kernels: if a CUDA tensor reaches an op and the
extension is absent, we raise instead of silently falling back to eager.
"""

from __future__ import annotations

import glob
import os

_ext = None
_tried = False


def _find_ext_path() -> str | None:
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    hits = glob.glob(os.path.join(here, "_C*.so"))
    return hits[0] if hits else None


def hip_ext():
    """Return the loaded HIP extension module, raising if unavailable."""
    global _ext, _tried
    if _ext is not None:
        return _ext
    if not _tried:
        _tried = True
        path = _find_ext_path()
        if path is not None:
            import importlib.util

            spec = importlib.util.spec_from_file_location(
                "gan_deeplearning4j_amd._C", path
            )
            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)  # type: ignore[union-attr]
            _ext = mod
            return _ext
    if _ext is None:
        raise RuntimeError(
            "gan_deeplearning4j_amd HIP extension (_C*.so) is not built. "
            "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            "GPU ops refuse to run without their native kernels."
        )
    return _ext


def has_hip_ext() -> bool:
    try:
        hip_ext()
        return True
    except RuntimeError:
        return False
