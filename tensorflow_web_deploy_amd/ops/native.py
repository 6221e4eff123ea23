"""Loader for the in-tree HIP extension (_hipops).

Fails LOUDLY when running on a GPU box without the compiled extension:
the native CDNA4 kernels ARE the compute path — a silent PyTorch
fallback would invalidate every benchmark (and the driver checks which
.so the GPU processes actually load).
"""
from __future__ import annotations

import torch

_ext = None
_err = None
try:
    from tensorflow_web_deploy_amd import _hipops as _ext  # type: ignore
except Exception as e:  # pragma: no cover
    _err = e


def have_native() -> bool:
    return _ext is not None


def ext():
    if _ext is None:
        raise RuntimeError(
            "tensorflow_web_deploy_amd._hipops is not built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {_err})")
    return _ext


def require_native_on_gpu() -> None:
    if torch.cuda.is_available() and _ext is None:
        raise RuntimeError(
            "GPU present but the HIP extension is missing — refusing to fall "
            f"back to eager PyTorch. Build _hipops first. ({_err})")
