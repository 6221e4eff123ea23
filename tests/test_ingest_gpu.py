"""GPU ingest pipeline vs the PIL correctness path (gpu-marked)."""
import io

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _jpeg_bytes(h, w):
    from PIL import Image
    arr = (np.random.default_rng(0).random((h, w, 3)) * 255).astype("uint8")
    buf = io.BytesIO()
    Image.fromarray(arr, "RGB").save(buf, "PNG")  # lossless for comparison
    return buf.getvalue()


def test_resize_norm_kernel_matches_pil():
    from tensorflow_web_deploy_amd.ingest.gpu_pipeline import GpuPreprocessor
    from tensorflow_web_deploy_amd.ingest.preprocess import decode_image
    blob = _jpeg_bytes(300, 400)
    pre = GpuPreprocessor(224, 224, 3, "unit")
    out = pre.process([blob])
    torch.cuda.synchronize()
    got = out[0].cpu().numpy()
    ref = decode_image(blob, (224, 224), 3, "unit")
    # PIL's bilinear and ours agree to within interpolation rounding
    assert got.shape == ref.shape == (224, 224, 3)
    err = np.abs(got - ref)
    assert np.median(err) < 0.01
    assert err.mean() < 0.02


def test_gpu_ingest_batch_and_sideband_stream():
    from tensorflow_web_deploy_amd.ingest.gpu_pipeline import GpuPreprocessor
    blobs = [_jpeg_bytes(100 + 7 * i, 150 + 11 * i) for i in range(5)]
    pre = GpuPreprocessor(64, 64, 3, "inception")
    out = pre.process(blobs)
    torch.cuda.synchronize()
    assert out.shape == (5, 64, 64, 3)
    v = out.cpu().numpy()
    assert v.min() >= -1.001 and v.max() <= 1.001
    assert np.abs(v).mean() > 0.05  # not all zeros


def test_gpu_ingest_feeds_engine():
    from tensorflow_web_deploy_amd.conf.config import Config
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    from tensorflow_web_deploy_amd.ingest.gpu_pipeline import GpuPreprocessor
    cfg = Config(model_name="lenet", device="cuda", input_height=28,
                 input_width=28, input_channels=1, num_classes=10)
    eng = LocalEngine(cfg)
    pre = GpuPreprocessor(28, 28, 1, "unit")
    blob = _jpeg_bytes(50, 50)
    batch = pre.process([blob])
    probs = eng.ex.run({"input": batch})[0]
    torch.cuda.synchronize()
    s = probs.sum(-1).float().cpu().numpy()
    np.testing.assert_allclose(s, 1.0, atol=1e-3)
