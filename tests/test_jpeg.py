"""Baseline-JPEG entropy decoder (_jpegc) vs PIL, CPU-only.

The parser emits quantized DCT coefficients; this test dequantizes +
IDCTs them in numpy (scipy idctn, ortho norm == JPEG spec) and applies
libjpeg's triangle chroma upsample, then compares against PIL's full
decode of the same bytes. Agreement within 2 LSB everywhere validates
marker parse, Huffman tables, DC prediction, restart intervals and
sampling handling. The HIP kernels (jpeg_gpu.hip) implement the same
dequant/IDCT/upsample math on the GPU."""
import io

import numpy as np
import pytest
from PIL import Image

from tensorflow_web_deploy_amd import _jpegc


def _ref_decode(info):
    from scipy.fft import idctn
    W, H = info["width"], info["height"]
    planes = []
    hmax = max(h for h, v in info["sampling"])
    vmax = max(v for h, v in info["sampling"])

    def tri2x(p, axis):
        p = np.moveaxis(p, axis, 0)
        prev = np.concatenate([p[:1], p[:-1]], 0)
        nxt = np.concatenate([p[1:], p[-1:]], 0)
        out = np.empty((2 * p.shape[0],) + p.shape[1:], p.dtype)
        out[0::2] = 0.75 * p + 0.25 * prev
        out[1::2] = 0.75 * p + 0.25 * nxt
        return np.moveaxis(out, 0, axis)

    for ci in range(info["ncomp"]):
        bh, bw = info["blocks"][ci]
        coef = info["coef"][ci].reshape(bh, bw, 8, 8).astype(np.float64)
        q = info["qt"][ci].reshape(8, 8).astype(np.float64)
        pix = idctn(coef * q, axes=(2, 3), norm="ortho") + 128.0
        plane = pix.transpose(0, 2, 1, 3).reshape(bh * 8, bw * 8)
        h, v = info["sampling"][ci]
        if vmax // v == 2:
            plane = tri2x(plane, 0)
        if hmax // h == 2:
            plane = tri2x(plane, 1)
        planes.append(plane[:H, :W])
    if info["ncomp"] == 1:
        return np.clip(np.round(planes[0]), 0, 255).astype(np.uint8)[..., None]
    Y, Cb, Cr = planes
    R = Y + 1.402 * (Cr - 128)
    G = Y - 0.344136 * (Cb - 128) - 0.714136 * (Cr - 128)
    B = Y + 1.772 * (Cb - 128)
    return np.clip(np.round(np.stack([R, G, B], -1)), 0, 255).astype(np.uint8)


def _jpeg(arr, mode, **kw):
    b = io.BytesIO()
    Image.fromarray(arr, mode).save(b, "JPEG", **kw)
    return b.getvalue()


@pytest.fixture(scope="module")
def img():
    rng = np.random.default_rng(0)
    base = rng.random((8, 8, 3))
    return (np.kron(base, np.ones((28, 28, 1)))[:217, :203]
            * 255).astype(np.uint8)


@pytest.mark.parametrize("name,kw,mode", [
    ("gray_q90", dict(quality=90), "L"),
    ("rgb_444_q95", dict(quality=95, subsampling=0), "RGB"),
    ("rgb_422_q85", dict(quality=85, subsampling=1), "RGB"),
    ("rgb_420_q90", dict(quality=90, subsampling=2), "RGB"),
    ("rgb_420_dri", dict(quality=90, subsampling=2,
                         restart_marker_blocks=4), "RGB"),
])
def test_parse_matches_pil(img, name, kw, mode):
    data = _jpeg(img[..., 0] if mode == "L" else img, mode, **kw)
    info = _jpegc.parse(data)
    got = _ref_decode(info)
    ref = np.asarray(Image.open(io.BytesIO(data)).convert(mode), np.uint8)
    if ref.ndim == 2:
        ref = ref[..., None]
    d = np.abs(got.astype(int) - ref.astype(int))
    assert got.shape == ref.shape
    assert d.max() <= 2, (name, d.mean(), d.max())


def test_progressive_rejected(img):
    with pytest.raises(ValueError, match="progressive"):
        _jpegc.parse(_jpeg(img, "RGB", quality=90, progressive=True))


def test_garbage_rejected():
    with pytest.raises(ValueError):
        _jpegc.parse(b"definitely not a jpeg")

def _patch_marker(data: bytes, marker: int, offset: int, value: int) -> bytes:
    """Return a copy of `data` with byte `offset` (relative to the first
    `marker` segment's payload start, i.e. after the 2-byte length) set to
    `value`."""
    i = 2
    b = bytearray(data)
    while i + 4 <= len(b):
        assert b[i] == 0xFF
        m = b[i + 1]
        seglen = (b[i + 2] << 8) | b[i + 3]
        if m == marker:
            b[i + 4 + offset] = value
            return bytes(b)
        i += 2 + seglen
    raise AssertionError(f"marker {marker:#x} not found")


def test_bad_dqt_table_id_rejected(img):
    # ADVICE round-1 high: 4-bit table ids up to 15 indexed qt[4] —
    # OOB stack write. Must raise, not crash.
    data = _jpeg(img, "RGB", quality=90)
    bad = _patch_marker(data, 0xDB, 0, 0x0F)  # tq=15
    with pytest.raises(ValueError, match="DQT"):
        _jpegc.parse(bad)


def test_bad_dht_table_id_rejected(img):
    data = _jpeg(img, "RGB", quality=90)
    bad = _patch_marker(data, 0xC4, 0, 0x0F)  # tc=0, th=15
    with pytest.raises(ValueError, match="DHT"):
        _jpegc.parse(bad)


def test_bad_sof_quant_id_rejected(img):
    data = _jpeg(img, "RGB", quality=90)
    # SOF0 payload: precision(1) H(2) W(2) nc(1) then per-comp id,hv,tq
    bad = _patch_marker(data, 0xC0, 8, 0x0F)  # comp0 tq=15
    with pytest.raises(ValueError, match="quant"):
        _jpegc.parse(bad)


def test_bad_sos_huff_id_rejected(img):
    data = _jpeg(img, "RGB", quality=90)
    # SOS payload: ns(1) then per-comp id,tdta
    bad = _patch_marker(data, 0xDA, 2, 0xFF)  # comp0 td=15, ta=15
    with pytest.raises(ValueError, match="SOS"):
        _jpegc.parse(bad)


def test_corrupt_entropy_raises_runtime(img):
    # truncate/flip bits mid-scan until the huffman decoder objects; the
    # ingest worker must survive this (RuntimeError is caught there).
    data = bytearray(_jpeg(img, "RGB", quality=90))
    sos = data.find(b"\xff\xda")
    assert sos > 0
    raised = False
    for off in range(sos + 20, min(sos + 220, len(data) - 2)):
        trial = bytes(data[:off]) + bytes([data[off] ^ 0xFF]) + bytes(
            data[off + 1:])
        try:
            _jpegc.parse(trial)
        except (RuntimeError, ValueError):
            raised = True
            break
    assert raised
