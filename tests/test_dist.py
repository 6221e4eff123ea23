"""DP replica tests on CPU: gloo backend, world_size=2.

Validates the scatter/gather wire protocol and that a 2-replica
ReplicatedEngine matches the single-process LocalEngine bit-for-bit on
the fp32 CPU path.
"""
import numpy as np
import pytest
import torch

from tensorflow_web_deploy_amd.conf.config import Config
from tensorflow_web_deploy_amd.parallel.comm import shard_sizes


def test_shard_sizes():
    assert shard_sizes(10, 4) == [3, 3, 2, 2]
    assert shard_sizes(8, 8) == [1] * 8
    assert shard_sizes(3, 8) == [1, 1, 1, 0, 0, 0, 0, 0]
    assert shard_sizes(64, 2) == [32, 32]


@pytest.mark.timeout(300)
def test_replicated_engine_gloo_matches_local():
    cfg = Config(model_name="lenet", device="cpu", gpus=2)
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine

    local = LocalEngine(cfg.replace(gpus=1))
    rng = np.random.default_rng(5)
    x = rng.random((5, 28, 28, 1), dtype=np.float32)
    want = local.infer(x)

    eng = ReplicatedEngine(cfg, world=2, backend="gloo")
    try:
        got = eng.infer(x)
        np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)
        # odd batch (ragged shard) and batch smaller than world
        for n in (3, 1):
            got_n = eng.infer(x[:n])
            np.testing.assert_allclose(got_n, want[:n], rtol=1e-5, atol=1e-6)
        # second call reuses the group
        got2 = eng.infer(x)
        np.testing.assert_allclose(got2, want, rtol=1e-5, atol=1e-6)
    finally:
        eng.shutdown()


@pytest.mark.timeout(300)
def test_replica_death_detected():
    cfg = Config(model_name="lenet", device="cpu", gpus=2)
    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine
    eng = ReplicatedEngine(cfg, world=2, backend="gloo",
                           shrink_on_failure=False)
    try:
        eng.procs[1].terminate()
        eng.procs[1].join(timeout=10)
        with pytest.raises(RuntimeError):
            eng.infer(np.zeros((2, 28, 28, 1), np.float32))
    finally:
        eng.shutdown()


@pytest.mark.timeout(300)
def test_replica_shrink_serves_on_survivors():
    """Fault injection: kill a replica mid-service; the engine shrinks
    the group and the in-flight request is retried on the survivors."""
    cfg = Config(model_name="lenet", device="cpu", gpus=2)
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine
    local = LocalEngine(cfg.replace(gpus=1))
    x = np.random.default_rng(8).random((4, 28, 28, 1), dtype=np.float32)
    want = local.infer(x)
    eng = ReplicatedEngine(cfg, world=2, backend="gloo",
                           shrink_on_failure=True)
    try:
        np.testing.assert_allclose(eng.infer(x), want, rtol=1e-5, atol=1e-6)
        eng.procs[1].terminate()
        eng.procs[1].join(timeout=10)
        got = eng.infer(x)  # retried on the shrunken (world=1) group
        assert eng.world == 1
        np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)
    finally:
        eng.shutdown()


def test_blob_mode_world2_matches_world1():
    """DP serving fast path (mode-1 u8 slot scatter + per-replica
    resize): world=2 must agree exactly with world=1 on the same
    decoded slots (plumbing test — same resize math both sides)."""
    import io

    from PIL import Image

    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine

    cfg = Config(model_name="lenet", device="cpu", input_height=28,
                 input_width=28, input_channels=1, num_classes=10,
                 max_batch=16)

    def png(h, w):
        arr = (np.random.rand(h, w) * 255).astype("uint8")
        b = io.BytesIO()
        Image.fromarray(arr, "L").save(b, "PNG")
        return b.getvalue()

    blobs = [png(40, 40), png(28, 28), png(33, 17)]
    outs = []
    for world in (1, 2):
        eng = ReplicatedEngine(cfg, world=world, backend="gloo",
                               serve_blobs=True, start_timeout=120.0)
        try:
            d = eng.pre.decode_shm(blobs)
            assert d is not None
            p = eng.infer_decoded(d, len(blobs))
            assert p.shape == (3, 10)
            np.testing.assert_allclose(p.sum(1), 1.0, atol=1e-4)
            outs.append(p)
            # mode-0 path still works on the same group
            x = np.random.rand(4, 28, 28, 1).astype(np.float32)
            q = eng.infer(x)
            assert q.shape == (4, 10)
        finally:
            eng.shutdown()
    np.testing.assert_allclose(outs[0], outs[1], atol=1e-5)


@pytest.mark.gpu
def test_blob_mode_nccl_world1_matches_local():
    """mode-1 over nccl world=1: worker maps the server's decode arena,
    scatters u8 slots, resizes with the HIP kernel. Must match the
    in-process GpuServingEngine on the same JPEGs."""
    import io

    from PIL import Image

    from tensorflow_web_deploy_amd.engine.service import GpuServingEngine
    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    cfg = Config(model_name="lenet", device="cuda", input_height=28,
                 input_width=28, input_channels=1, num_classes=10,
                 max_batch=16)

    def png(h, w):
        arr = (np.random.rand(h, w) * 255).astype("uint8")
        b = io.BytesIO()
        Image.fromarray(arr, "L").save(b, "PNG")
        return b.getvalue()

    blobs = [png(40, 40), png(28, 28), png(33, 17)]
    local = GpuServingEngine(cfg)
    ref = local.infer_blobs(blobs)
    eng = ReplicatedEngine(cfg, world=1, backend="nccl",
                           serve_blobs=True, start_timeout=300.0)
    try:
        d = eng.pre.decode_shm(blobs)
        assert d is not None
        p = eng.infer_decoded(d, len(blobs))
        np.testing.assert_allclose(p, ref, atol=2e-3)
    finally:
        eng.shutdown()
