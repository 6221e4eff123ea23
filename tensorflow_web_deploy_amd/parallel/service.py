"""Shared inference service + multi-process HTTP serving.

The single-process Flask stack tops out near 600 req/s (werkzeug +
decode + GIL). Production shape (SURVEY §3b scaled out):

  service process: DynamicBatcher + engine (GPU; ReplicatedEngine for
      DP>1). Receives decoded uint8 images over an mp queue, batches,
      GPU-preprocesses (resize/normalize on the side stream), replies to
      per-worker response queues.
  N HTTP worker processes: each binds the SAME port with SO_REUSEPORT
      (kernel load-balances accepts), runs the Flask app threaded,
      decodes uploads in its own interpreter (parallel decode, no shared
      GIL), and round-trips the service.
"""
from __future__ import annotations

import itertools
import os
import queue
import socket
import threading
import time
import traceback
from multiprocessing import shared_memory
from typing import Optional

import numpy as np
import torch.multiprocessing as mp

from ..conf.config import Config

# Shared-memory request transport: an mp.Queue moves only ~1.3k
# 150-KB-image messages/s (measured; pickle + pipe); decoded images
# instead go through a per-worker SharedMemory slot ring and the queue
# carries ~100-byte descriptors.
SLOT_BYTES = 1 << 19   # 512 KiB per slot (room for ~400x400x3 u8)
SLOTS_PER_WORKER = 128


def _service_main(cfg: Config, in_q, resp_qs, err_q, ready_ev):
    try:
        import torch

        from ..engine.service import GpuServingEngine, LocalEngine
        from ..sched.batcher import DynamicBatcher
        if cfg.gpus > 1:
            from .replicas import ReplicatedEngine
            engine = ReplicatedEngine(cfg)
        elif torch.cuda.is_available():
            engine = GpuServingEngine(cfg)
        else:
            engine = LocalEngine(cfg)
        batcher = DynamicBatcher(engine, max_batch=cfg.max_batch,
                                 max_wait_ms=cfg.max_wait_ms)
        batcher.start()
        # warm common batch buckets so serving never hits a compile
        h, w, c = cfg.input_height, cfg.input_width, cfg.input_channels
        b = 1
        while True:
            engine.infer(np.zeros((b, h, w, c), np.float32))
            if b >= cfg.max_batch:
                break
            b = min(b * 2, cfg.max_batch)
        ready_ev.set()

        # responses ride the future's done-callback (runs in the batcher
        # thread): a per-request responder-thread pool caused a GIL
        # stampede after every batch (measured 25-45 ms per batch for
        # ~3 ms of GPU work)
        def respond(wid, rid, fut):
            try:
                resp_qs[wid].put((rid, fut.result(timeout=0), None))
            except Exception as e:  # noqa: BLE001
                resp_qs[wid].put((rid, None, repr(e)))

        shms: dict = {}
        stats = {"items": 0, "t0": time.time()}

        def _stat_loop():
            import sys
            while True:
                time.sleep(5.0)
                tr = batcher.trace[-200:]
                if tr:
                    bs = [t["batch"] for t in tr]
                    ms = [t["infer_ms"] for t in tr]
                    sg = stats.get("seg", {})
                    segs = " ".join(f"{k}={1000*v/max(stats['items'],1):.2f}ms"
                                    for k, v in sg.items())
                    print(f"[service-seg per-item] {segs}",
                          file=sys.stderr, flush=True)
                    print(f"[service] items={stats['items']} "
                          f"rate={stats['items']/(time.time()-stats['t0']):.0f}/s "
                          f"batches={len(batcher.trace)} "
                          f"avg_batch={sum(bs)/len(bs):.1f} "
                          f"avg_infer={sum(ms)/len(ms):.1f}ms "
                          f"max_infer={max(ms):.1f}ms",
                          file=sys.stderr, flush=True)
        if os.environ.get("TWD_SERVICE_STATS"):
            threading.Thread(target=_stat_loop, daemon=True).start()
        seg = {"get": 0.0, "copy": 0.0, "submit": 0.0}
        stats["seg"] = seg
        while True:
            tg0 = time.perf_counter()
            item = in_q.get()
            seg["get"] += time.perf_counter() - tg0
            if item is None:
                break
            stats["items"] += 1
            wid, rid, payload = item
            if isinstance(payload, tuple):  # shm descriptor
                shm_name, slot, shape = payload
                shm = shms.get(shm_name)
                if shm is None:
                    shm = shared_memory.SharedMemory(name=shm_name)
                    shms[shm_name] = shm
                nbytes = int(np.prod(shape))
                view = np.ndarray(shape, np.uint8,
                                  buffer=shm.buf,
                                  offset=slot * SLOT_BYTES)
                tc0 = time.perf_counter()
                arr = view.copy()  # slot is freed by the worker on reply
                seg["copy"] += time.perf_counter() - tc0
            else:
                arr = payload
            ts0 = time.perf_counter()
            fut = batcher.submit(arr)
            fut.add_done_callback(
                lambda f, wid=wid, rid=rid: respond(wid, rid, f))
            seg["submit"] += time.perf_counter() - ts0
        batcher.stop()
    except Exception:
        err_q.put(traceback.format_exc())
        raise


class ServiceClient:
    """Engine-interface adapter used inside an HTTP worker process."""

    def __init__(self, wid: int, in_q, resp_q, cfg: Config):
        self.wid = wid
        self.in_q = in_q
        self.resp_q = resp_q
        self.cfg = cfg
        self._rid = itertools.count()
        self._lock = threading.Lock()
        self._waiting: dict = {}
        self._shm = shared_memory.SharedMemory(
            create=True, size=SLOT_BYTES * SLOTS_PER_WORKER)
        self._free = list(range(SLOTS_PER_WORKER))
        self._stats = [] if os.environ.get("TWD_SERVICE_STATS") else None
        self._reader = threading.Thread(target=self._drain, daemon=True)
        self._reader.start()

    def _drain(self):
        while True:
            rid, row, err = self.resp_q.get()
            with self._lock:
                slot = self._waiting.pop(rid, None)
            if slot is not None:
                if slot.get("shm_slot") is not None:
                    with self._lock:
                        self._free.append(slot["shm_slot"])
                slot["row"] = row
                slot["err"] = err
                slot["ev"].set()

    def _roundtrip(self, arr: np.ndarray):
        t0 = time.perf_counter()
        rid = next(self._rid)
        slot = {"ev": threading.Event(), "shm_slot": None}
        payload = arr
        if arr.nbytes <= SLOT_BYTES:
            with self._lock:
                sid = self._free.pop() if self._free else None
            if sid is not None:
                dst = np.ndarray(arr.shape, np.uint8, buffer=self._shm.buf,
                                 offset=sid * SLOT_BYTES)
                dst[...] = arr
                slot["shm_slot"] = sid
                payload = (self._shm.name, sid, arr.shape)
        with self._lock:
            self._waiting[rid] = slot
        self.in_q.put((self.wid, rid, payload))
        if not slot["ev"].wait(timeout=60.0):
            raise TimeoutError("inference service timeout")
        if slot["err"]:
            raise RuntimeError(slot["err"])
        if self._stats is not None:
            self._stats.append(time.perf_counter() - t0)
            if len(self._stats) % 200 == 0:
                import sys
                ls = sorted(self._stats[-200:])
                print(f"[client {self.wid}] roundtrip p50="
                      f"{1000*ls[100]:.1f}ms p90={1000*ls[180]:.1f}ms "
                      f"n={len(self._stats)}", file=sys.stderr, flush=True)
        return slot["row"]

    # engine interface used by server/app.py
    def infer_arrays(self, arrs):
        return np.stack([self._roundtrip(a) for a in arrs], axis=0)

    def infer(self, batch: np.ndarray) -> np.ndarray:
        return np.stack([self._roundtrip(batch[i])
                         for i in range(batch.shape[0])], axis=0)

    def decode(self, blob: bytes) -> np.ndarray:
        """Entropy decode in THIS worker process (parallel across
        workers); resize/normalize happen on the GPU in the service."""
        import io

        from PIL import Image
        img = Image.open(io.BytesIO(blob))
        img = img.convert("L" if self.cfg.input_channels == 1 else "RGB")
        arr = np.asarray(img, dtype=np.uint8)
        if arr.ndim == 2:
            arr = arr[:, :, None]
        return arr


def _http_worker_main(wid: int, cfg: Config, in_q, resp_q):
    import sys
    # response-drain thread must not starve behind werkzeug request
    # threads: switch the GIL faster in HTTP workers
    sys.setswitchinterval(0.0005)
    from ..server.app import create_app
    client = ServiceClient(wid, in_q, resp_q, cfg)
    app = create_app(cfg, client)
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEPORT, 1)
    sock.bind((cfg.host, cfg.port))
    sock.listen(512)
    from werkzeug.serving import make_server
    srv = make_server(cfg.host, cfg.port, app, threaded=True,
                      fd=sock.fileno())
    srv.serve_forever()


class MultiprocServer:
    """N SO_REUSEPORT HTTP workers + 1 inference-service process."""

    def __init__(self, cfg: Config, http_workers: int = 8,
                 start_timeout: float = 900.0):
        self.cfg = cfg
        ctx = mp.get_context("spawn")
        self.in_q = ctx.Queue()
        self.err_q = ctx.Queue()
        self.resp_qs = [ctx.Queue() for _ in range(http_workers)]
        ready = ctx.Event()
        self.service = ctx.Process(
            target=_service_main,
            args=(cfg, self.in_q, self.resp_qs, self.err_q, ready),
            daemon=True)
        self.service.start()
        deadline = time.time() + start_timeout
        while not ready.wait(timeout=1.0):
            if not self.service.is_alive() or time.time() > deadline:
                try:
                    raise RuntimeError(self.err_q.get_nowait())
                except queue.Empty:
                    raise RuntimeError("inference service failed to start")
        self.workers = []
        for wid in range(http_workers):
            p = ctx.Process(target=_http_worker_main,
                            args=(wid, cfg, self.in_q, self.resp_qs[wid]),
                            daemon=True)
            p.start()
            self.workers.append(p)

    def shutdown(self):
        for p in self.workers:
            p.terminate()
        try:
            self.in_q.put(None)
        except Exception:
            pass
        self.service.join(timeout=10.0)
        if self.service.is_alive():
            self.service.terminate()


def main(argv=None):
    import argparse

    from ..conf.config import add_config_args, config_from_args
    p = argparse.ArgumentParser("multi-process serving")
    add_config_args(p)
    p.add_argument("--http-workers", type=int, default=8)
    args = p.parse_args(argv)
    cfg = config_from_args(args)
    srv = MultiprocServer(cfg, http_workers=args.http_workers)
    print(f"serving on {cfg.host}:{cfg.port} with {args.http_workers} "
          f"HTTP workers; Ctrl-C to stop")
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        srv.shutdown()


if __name__ == "__main__":
    main()
