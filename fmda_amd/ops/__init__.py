"""HIP/CDNA4 ops for fmda_amd.

The compiled extension (`fmda_amd/ops/_fmda_hip*.so`, built in-tree by
`setup.py build_ext --inplace` with PYTORCH_ROCM_ARCH=gfx950) is REQUIRED on
a GPU box: `load_extension()` raises if a CUDA device is visible and the
extension is missing, so GPU tests can never silently pass on an eager
PyTorch fallback.
"""
import glob
import os

import torch

_ext = None
_ext_err = None


def load_extension():
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    try:
        from . import _fmda_hip  # built in-tree
        _ext = _fmda_hip
        return _ext
    except ImportError as e:
        _ext_err = e
    if torch.cuda.is_available():
        here = os.path.dirname(__file__)
        built = glob.glob(os.path.join(here, "_fmda_hip*.so"))
        raise RuntimeError(
            "fmda_amd HIP extension is required on a GPU but could not be "
            f"imported (found .so files: {built}). Build it in-tree with "
            "`python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_ext_err}")
    return None


def extension_available() -> bool:
    try:
        return load_extension() is not None
    except RuntimeError:
        return False
