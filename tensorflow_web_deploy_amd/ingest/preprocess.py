"""Image decode + preprocess (correctness path).

Reference behavior: PIL open -> resize -> np.array -> normalize -> NHWC
batch (reference class: `preprocess(path) -> ndarray` feeding
`sess.run`). This module is the CPU/PIL correctness path; the
performance path (HIP resize/normalize kernels on a side stream, fed by
CPU JPEG entropy decode) lives in ingest/gpu_pipeline.py and is
validated against this one.
"""
from __future__ import annotations

import io
from typing import Tuple

import numpy as np

try:
    from PIL import Image
    _HAVE_PIL = True
except Exception:  # pragma: no cover
    _HAVE_PIL = False


def decode_image(data: bytes, size: Tuple[int, int], channels: int,
                 normalize: str = "unit") -> np.ndarray:
    """bytes -> float32 HWC array resized to `size`.

    normalize: 'unit' -> [0,1); 'inception' -> [-1,1); 'raw' -> [0,255].
    """
    if not _HAVE_PIL:
        raise RuntimeError("PIL unavailable; cannot decode uploaded images")
    img = Image.open(io.BytesIO(data))
    img = img.convert("L" if channels == 1 else "RGB")
    img = img.resize(size, Image.BILINEAR)
    arr = np.asarray(img, dtype=np.float32)
    if channels == 1:
        arr = arr[:, :, None]
    if normalize == "unit":
        arr = arr / 255.0
    elif normalize == "inception":
        arr = arr / 127.5 - 1.0
    return arr


def batch_images(arrays) -> np.ndarray:
    return np.stack(arrays, axis=0)
