"""Dynamic-batcher unit tests (CPU, deterministic)."""
import threading
import time

import numpy as np
import pytest

from tensorflow_web_deploy_amd.sched.batcher import (BatchedEngine,
                                                     DynamicBatcher)


class RecordingEngine:
    def __init__(self, nc=10, delay=0.0):
        self.nc = nc
        self.delay = delay
        self.batches = []

    def infer(self, batch):
        self.batches.append(batch.shape[0])
        if self.delay:
            time.sleep(self.delay)
        out = np.zeros((batch.shape[0], self.nc), np.float32)
        # encode each row's input sum so demux order is checkable
        for i in range(batch.shape[0]):
            out[i, 0] = batch[i].sum()
        return out


def test_single_request_roundtrip():
    eng = RecordingEngine()
    b = DynamicBatcher(eng, max_batch=8, max_wait_ms=5.0)
    b.start()
    try:
        img = np.full((4, 4, 1), 2.0, np.float32)
        probs = b.infer_sync(img)
        assert probs[0] == pytest.approx(32.0)
    finally:
        b.stop()


def test_coalescing_and_demux():
    eng = RecordingEngine(delay=0.01)
    b = DynamicBatcher(eng, max_batch=16, max_wait_ms=20.0)
    b.start()
    try:
        imgs = [np.full((2, 2, 1), float(i), np.float32) for i in range(12)]
        futs = [b.submit(im) for im in imgs]
        res = [f.result(timeout=5.0) for f in futs]
        # each request got its own row back
        for i, r in enumerate(res):
            assert r[0] == pytest.approx(4.0 * i)
        # requests were coalesced into few batches (not 12 singles)
        assert max(eng.batches) > 1
        assert sum(eng.batches) == 12
    finally:
        b.stop()


def test_max_batch_respected():
    eng = RecordingEngine(delay=0.05)
    b = DynamicBatcher(eng, max_batch=4, max_wait_ms=50.0)
    b.start()
    try:
        futs = [b.submit(np.zeros((2, 2, 1), np.float32)) for _ in range(10)]
        [f.result(timeout=5.0) for f in futs]
        assert max(eng.batches) <= 4
        assert sum(eng.batches) == 10
    finally:
        b.stop()


def test_engine_error_propagates():
    class Boom:
        def infer(self, batch):
            raise ValueError("boom")

    b = DynamicBatcher(Boom(), max_batch=4, max_wait_ms=1.0)
    b.start()
    try:
        fut = b.submit(np.zeros((2, 2, 1), np.float32))
        with pytest.raises(ValueError):
            fut.result(timeout=5.0)
    finally:
        b.stop()


def test_batched_engine_adapter():
    eng = RecordingEngine()
    b = DynamicBatcher(eng, max_batch=8, max_wait_ms=5.0)
    b.start()
    try:
        be = BatchedEngine(b)
        batch = np.stack([np.full((2, 2, 1), float(i)) for i in range(5)])
        out = be.infer(batch.astype(np.float32))
        assert out.shape == (5, 10)
        for i in range(5):
            assert out[i, 0] == pytest.approx(4.0 * i)
    finally:
        b.stop()


def test_trace_recorded():
    eng = RecordingEngine()
    b = DynamicBatcher(eng, max_batch=4, max_wait_ms=1.0)
    b.start()
    try:
        b.infer_sync(np.zeros((2, 2, 1), np.float32))
        assert len(b.trace) >= 1
        t = b.trace[0]
        assert {"batch", "close", "done", "infer_ms"} <= set(t)
    finally:
        b.stop()


def test_concurrent_stress():
    """Race fuzz: many threads hammering submit while the
    batcher drains; every future must resolve exactly once, correctly."""
    import threading
    eng = RecordingEngine(delay=0.001)
    b = DynamicBatcher(eng, max_batch=32, max_wait_ms=1.0)
    b.start()
    results = {}
    lock = threading.Lock()

    def client(cid):
        for j in range(20):
            v = float(cid * 100 + j)
            r = b.infer_sync(np.full((2, 2, 1), v, np.float32), timeout=30)
            with lock:
                results[(cid, j)] = (r[0], 4.0 * v)

    threads = [threading.Thread(target=client, args=(i,)) for i in range(16)]
    try:
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
        assert len(results) == 16 * 20
        for (cid, j), (got, want) in results.items():
            assert got == pytest.approx(want), (cid, j)
        assert sum(eng.batches) == 16 * 20
    finally:
        b.stop()
