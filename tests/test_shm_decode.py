"""ShmDecodePool: persistent shm-ring decode workers, CPU-only.

The pool replaces ProcessPoolExecutor for serving-path decode because
pool.map's dispatch alone measured 20-40 ms per 64 tasks; the ring's
per-batch cost is 2 pipe bytes per worker (see gpu_pipeline.py)."""
import io

import numpy as np
import pytest
from PIL import Image

from tensorflow_web_deploy_amd.ingest.gpu_pipeline import (ShmDecodePool,
                                                           _decode_worker)


def _jpeg(h, w):
    arr = (np.random.rand(h, w, 3) * 255).astype("uint8")
    buf = io.BytesIO()
    Image.fromarray(arr, "RGB").save(buf, "JPEG", quality=90)
    return buf.getvalue()


@pytest.fixture(scope="module")
def pool():
    p = ShmDecodePool(3, workers=2, slot_bytes=1 << 19, slots=64)
    p.warm()
    yield p
    p.close()


def test_decode_matches_reference(pool):
    blobs = [_jpeg(96, 128), _jpeg(224, 224), _jpeg(31, 57)]
    meta = pool.decode_batch(blobs)
    assert meta is not None
    # meta offsets are relative to the active (double-buffered) half
    base = pool.hdr + pool.in_bytes + pool.last_window_off
    for i, b in enumerate(blobs):
        ref = _decode_worker((b, 3))
        off, h, w, c = meta[i]
        assert (h, w, c) == ref.shape
        got = pool.buf[base + off:base + off + ref.nbytes]
        np.testing.assert_array_equal(got.reshape(ref.shape), ref)


def test_oversized_falls_back(pool):
    # 600*600*3 > 512 KiB slot -> whole batch defers to the pipe path
    assert pool.decode_batch([_jpeg(600, 600)]) is None


def test_bad_image_raises_with_index(pool):
    with pytest.raises(ValueError, match="image 1"):
        pool.decode_batch([_jpeg(32, 32), b"garbage", _jpeg(32, 32)])


def test_batch_too_large_returns_none(pool):
    assert pool.decode_batch([_jpeg(8, 8)] * 64) is None  # slots-1 max


def test_reuse_across_batches(pool):
    for _ in range(3):
        meta = pool.decode_batch([_jpeg(50, 40) for _ in range(8)])
        assert meta is not None and len(meta) == 8
        assert all(tuple(m[1:]) == (50, 40, 3) for m in meta)


def test_corrupt_entropy_does_not_kill_workers(pool):
    """ADVICE round-1 medium: a corrupt-but-wellformed JPEG raised
    RuntimeError past the worker's ValueError-only catch, killing the
    worker and permanently disabling the shm fast path. The worker must
    fall through to PIL (bad image -> per-index error) and stay alive."""
    good = _jpeg(32, 32)
    data = bytearray(_jpeg(64, 64))
    sos = data.find(b"\xff\xda")
    corrupt = None
    from tensorflow_web_deploy_amd import _jpegc
    for off in range(sos + 20, min(sos + 220, len(data) - 2)):
        trial = bytes(data[:off]) + bytes([data[off] ^ 0xFF]) + bytes(
            data[off + 1:])
        try:
            _jpegc.parse(trial)
        except RuntimeError:
            corrupt = trial
            break
        except ValueError:
            continue
    if corrupt is None:
        pytest.skip("could not craft a RuntimeError-raising JPEG")
    # PIL may decode it (sloppily) or reject it; either way the pool
    # must survive and keep serving subsequent batches.
    try:
        pool.decode_batch([good, corrupt, good])
    except ValueError:
        pass
    assert not pool._dead
    meta = pool.decode_batch([good, good])
    assert meta is not None
