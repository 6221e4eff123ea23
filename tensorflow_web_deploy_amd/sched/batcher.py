"""Dynamic batcher: coalesces concurrent HTTP requests into super-batches.

(Reference runs one image per `sess.run`; the batcher is the serving-
performance layer the north star adds: window closes at max_batch or
max_wait_ms, whichever first; the super-batch goes to the engine —
LocalEngine on one GPU or ReplicatedEngine scatter/gather over DP
replicas — and per-request rows are demuxed back to their futures.)

Thread-based (Flask threaded server); `clock`/`sleeper` injectable for
deterministic tests.
"""
from __future__ import annotations

import threading
import time
from concurrent.futures import Future
from typing import Callable, List, Optional, Tuple

import numpy as np


class DynamicBatcher:
    def __init__(self, engine, max_batch: int = 64, max_wait_ms: float = 2.0,
                 clock: Callable[[], float] = time.monotonic):
        self.engine = engine
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1000.0
        self.clock = clock
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._queue: List[Tuple[np.ndarray, Future]] = []
        self._running = False
        self._thread: Optional[threading.Thread] = None
        # observability: ring buffer of batch spans
        self.trace: List[dict] = []
        self._trace_cap = 256

    # ------------------------------------------------------------ api

    def start(self):
        if self._running:
            return
        self._running = True
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def stop(self):
        with self._cv:
            self._running = False
            self._cv.notify_all()
        if self._thread:
            self._thread.join(timeout=5.0)

    def submit(self, img: np.ndarray) -> Future:
        """img: HWC float32. Returns a Future resolving to the prob row."""
        fut: Future = Future()
        with self._cv:
            self._queue.append((img, fut))
            self._cv.notify_all()
        return fut

    def infer_sync(self, img: np.ndarray, timeout: float = 30.0):
        return self.submit(img).result(timeout=timeout)

    # ----------------------------------------------------------- loop

    def _collect(self) -> List[Tuple[np.ndarray, Future]]:
        """Block until work; then close the window at max_batch or
        max_wait after the first request arrived."""
        with self._cv:
            while self._running and not self._queue:
                self._cv.wait(timeout=0.1)
            if not self._running:
                return []
            t0 = self.clock()
            while (len(self._queue) < self.max_batch
                   and self._running):
                remaining = self.max_wait - (self.clock() - t0)
                if remaining <= 0:
                    break
                self._cv.wait(timeout=remaining)
            items = self._queue[:self.max_batch]
            del self._queue[:self.max_batch]
            return items

    def _loop(self):
        while self._running:
            items = self._collect()
            if not items:
                continue
            t_close = self.clock()
            first = items[0][0]
            raw = isinstance(first, (bytes, bytearray))
            u8 = isinstance(first, np.ndarray) and first.dtype == np.uint8
            try:
                if raw:
                    probs = self.engine.infer_blobs(
                        [im for im, _ in items])
                elif u8:
                    probs = self.engine.infer_arrays(
                        [im for im, _ in items])
                else:
                    batch = np.stack([im for im, _ in items], axis=0)
                    probs = self.engine.infer(batch)
                t_done = self.clock()
                for i, (_, fut) in enumerate(items):
                    fut.set_result(probs[i])
            except Exception as e:
                t_done = self.clock()
                for _, fut in items:
                    if not fut.done():
                        fut.set_exception(e)
            self.trace.append({"batch": len(items), "close": t_close,
                               "done": t_done,
                               "infer_ms": (t_done - t_close) * 1e3})
            if len(self.trace) > self._trace_cap:
                del self.trace[:len(self.trace) - self._trace_cap]


class BatchedEngine:
    """Adapter giving the batcher the single-image `infer` the HTTP layer
    expects: enqueue one row, wait for the demuxed result."""

    def __init__(self, batcher: DynamicBatcher, raw: bool = False,
                 arrays: bool = False):
        self.batcher = batcher
        if raw:
            self.infer_blobs = self._infer_blobs
        if arrays:
            self.infer_arrays = self._infer_arrays
            self.decode = batcher.engine.decode

    def infer(self, batch: np.ndarray) -> np.ndarray:
        futs = [self.batcher.submit(batch[i]) for i in range(batch.shape[0])]
        return np.stack([f.result(timeout=60.0) for f in futs], axis=0)

    def _infer_blobs(self, blobs) -> np.ndarray:
        futs = [self.batcher.submit(b) for b in blobs]
        return np.stack([f.result(timeout=60.0) for f in futs], axis=0)

    def _infer_arrays(self, arrs) -> np.ndarray:
        futs = [self.batcher.submit(a) for a in arrs]
        return np.stack([f.result(timeout=60.0) for f in futs], axis=0)
