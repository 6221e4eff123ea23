"""End-to-end GPU tests: full models through the HIP-kernel executor vs
the CPU fp32 reference executor (gpu-marked)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _engines(model, h, w, c, nc):
    from tensorflow_web_deploy_amd.conf.config import Config
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    cfg = Config(model_name=model, input_height=h, input_width=w,
                 input_channels=c, num_classes=nc)
    gpu = LocalEngine(cfg.replace(device="cuda"))
    cpu = LocalEngine(cfg.replace(device="cpu"), graph=gpu.graph)
    return gpu, cpu


def test_lenet_gpu_vs_cpu():
    gpu, cpu = _engines("lenet", 28, 28, 1, 10)
    x = np.random.default_rng(0).random((8, 28, 28, 1), dtype=np.float32)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    assert pg.shape == pc.shape == (8, 10)
    # same argmax on almost all rows; prob close
    agree = (pg.argmax(1) == pc.argmax(1)).mean()
    assert agree >= 0.9, f"argmax agreement {agree}"
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    np.testing.assert_allclose(pg, pc, atol=0.05)


def test_resnet50_gpu_vs_cpu():
    gpu, cpu = _engines("resnet50", 224, 224, 3, 1000)
    x = (np.random.default_rng(1).random((2, 224, 224, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    assert pg.shape == (2, 1000)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    # bf16 through 53 random-init conv layers vs fp32: the GPU top-1 must
    # sit in the fp32 top-5, and the distributions must be close
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i], (pg[i].argmax(), top5[i])
    # distribution agreement (random-init nets have near-flat probs, so
    # absolute prob bounds are flaky; correlation is stable)
    corr = np.corrcoef(pg.ravel(), pc.ravel())[0, 1]
    assert corr > 0.95, corr


def test_resnet50_batch_consistency():
    """Same image in different batch slots gives the same probs."""
    gpu, _ = _engines("resnet50", 224, 224, 3, 1000)
    x = (np.random.rand(1, 224, 224, 3).astype(np.float32) - 0.5)
    xb = np.repeat(x, 4, axis=0)
    p = gpu.infer(xb)
    np.testing.assert_allclose(p[0], p[3], atol=1e-3)


def test_inception_v3_gpu_vs_cpu():
    gpu, cpu = _engines("inception_v3", 299, 299, 3, 1000)
    x = (np.random.default_rng(2).random((2, 299, 299, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i]
    assert np.abs(pg - pc).max() < 0.35


def test_mobilenet_v2_gpu_vs_cpu():
    gpu, cpu = _engines("mobilenet_v2", 224, 224, 3, 1000)
    x = (np.random.default_rng(5).random((2, 224, 224, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i]
    corr = np.corrcoef(pg.ravel(), pc.ravel())[0, 1]
    assert corr > 0.95, corr


def test_vit_b16_gpu_vs_cpu():
    gpu, cpu = _engines("vit_b16", 224, 224, 3, 1000)
    x = (np.random.default_rng(3).random((2, 224, 224, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i]
    assert np.abs(pg - pc).max() < 0.35


def test_vgg16_gpu_vs_cpu():
    gpu, cpu = _engines("vgg16", 224, 224, 3, 1000)
    x = (np.random.default_rng(7).random((2, 224, 224, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i]
    corr = np.corrcoef(pg.ravel(), pc.ravel())[0, 1]
    assert corr > 0.95, corr


def test_densenet121_gpu_vs_cpu():
    gpu, cpu = _engines("densenet121", 224, 224, 3, 1000)
    x = (np.random.default_rng(9).random((2, 224, 224, 3),
                                          dtype=np.float32) - 0.5)
    pg = gpu.infer(x)
    pc = cpu.infer(x)
    np.testing.assert_allclose(pg.sum(1), 1.0, atol=1e-3)
    top5 = np.argsort(pc, 1)[:, -5:]
    for i in range(pg.shape[0]):
        assert pg[i].argmax() in top5[i]
    corr = np.corrcoef(pg.ravel(), pc.ravel())[0, 1]
    assert corr > 0.95, corr
