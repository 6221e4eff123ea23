"""Batch-loop server over the C++ epoll frontend (`csrc/httpd.cpp`).

The Flask app (server/app.py) keeps the full reference-parity surface;
this module serves the same `POST /predict` contract through a C++
frontend that accepts, parses, batches and responds WITHOUT the GIL.
The Python process runs one loop: wait for a whole batch (GIL released
inside C++), decode+infer it, hand back responses. No per-request
Python ever runs, which removes the werkzeug/batcher GIL contention
measured in BASELINE.md (batch work 0.3-3 ms alone vs 73-91 ms in situ
under Flask).

  python -m tensorflow_web_deploy_amd.server.fast --model-name resnet50
"""
from __future__ import annotations

import json
import threading
from typing import Optional

import numpy as np

from ..conf.config import Config
from .labels import load_labels


class FastServer:
    def __init__(self, cfg: Config, engine=None, workers: int = 3,
                 port: Optional[int] = None):
        from .._httpd import Frontend
        self.cfg = cfg
        self.labels = load_labels(cfg.labels_path, cfg.num_classes)
        if engine is None:
            engine = _default_engine(cfg)
        self.engine = engine
        self.fe = Frontend(cfg.host or "0.0.0.0",
                           cfg.port if port is None else port,
                           workers, 8192)
        self.port = self.fe.port()
        self._stop = threading.Event()
        self.model_name = cfg.model_name or cfg.model_path
        # C-side /predict JSON: the frontend holds labels + model name
        # and formats whole batches from (ids, topk, probs) arrays
        self.fe.set_labels([str(x) for x in self.labels],
                           str(self.model_name))

    # -- one batch ----------------------------------------------------
    def _infer(self, bodies):
        """bodies -> list of (status, payload-json-str); isolates bad
        images so one corrupt upload cannot fail the whole batch."""
        eng = self.engine
        try:
            if hasattr(eng, "infer_blobs"):
                probs = eng.infer_blobs(bodies)
            else:
                arrs = [_decode(b, self.cfg) for b in bodies]
                probs = eng.infer(np.stack(arrs, 0))
            return [(200, self._payload(p)) for p in probs]
        except Exception:
            pass
        return self._infer_isolating(bodies)

    def _infer_isolating(self, bodies):
        """Per-item decode (NO shm ring — safe from any thread), bad
        images 400, rest served via the legacy pinned-arena path."""
        eng = self.engine
        out = [None] * len(bodies)
        good, idx = [], []
        for i, b in enumerate(bodies):
            try:
                good.append(_decode(b, self.cfg, raw=hasattr(
                    eng, "infer_blobs")))
                idx.append(i)
            except Exception as e:
                out[i] = (400, json.dumps(
                    {"error": f"cannot decode image: {e}"}))
        if good:
            try:
                if hasattr(eng, "infer_arrays"):
                    probs = eng.infer_arrays(good)
                else:
                    probs = eng.infer(np.stack(good, 0))
                for i, p in zip(idx, probs):
                    out[i] = (200, self._payload(p))
            except Exception as e:  # engine failure: honest 500s
                for i in idx:
                    out[i] = (500, json.dumps({"error": str(e)}))
        return out

    def _topk(self, probs):
        """Vectorized top-k for a whole (n, classes) batch ->
        (idx int32 [n,k], vals f32 [n,k])."""
        n, nc = probs.shape
        k = min(self.cfg.top_k, nc)
        if k < nc:
            part = np.argpartition(probs, -k, axis=1)[:, -k:]
            vals = np.take_along_axis(probs, part, axis=1)
            order = np.argsort(vals, axis=1)[:, ::-1]
            top = np.take_along_axis(part, order, axis=1)
        else:
            top = np.argsort(probs, axis=1)[:, ::-1][:, :k]
        vals = np.take_along_axis(probs, top, axis=1)
        return (np.ascontiguousarray(top, dtype=np.int32),
                np.ascontiguousarray(vals, dtype=np.float32))

    def _payload(self, probs) -> str:
        k = min(self.cfg.top_k, len(probs))
        if k < len(probs):
            part = np.argpartition(probs, -k)[-k:]
            top = part[np.argsort(probs[part])[::-1]]
        else:
            top = np.argsort(probs)[::-1]
        return json.dumps({
            "predictions": [
                {"label": self.labels[int(i)], "class_id": int(i),
                 "prob": float(probs[int(i)])} for i in top
            ],
            "model": self.model_name,
        })

    # -- loop ---------------------------------------------------------
    def step(self, timeout_ms: Optional[float] = None) -> int:
        """One wait_batch -> infer -> respond cycle; returns batch size."""
        batch = self.fe.wait_batch(self.cfg.max_batch,
                                   self.cfg.max_wait_ms)
        if not batch:
            return 0
        results = self._infer([b for (_, b) in batch])
        self.fe.respond_many([
            (rid, st, body.encode())
            for (rid, _), (st, body) in zip(batch, results)
        ])
        return len(batch)

    def serve_forever(self):
        if hasattr(self.engine, "infer_decoded"):
            return self._serve_pipelined()
        try:
            while not self._stop.is_set():
                self.step()
        finally:
            self.fe.stop()

    # -- two-stage pipeline: decode(N+1) overlaps GPU+respond(N) ------
    def _serve_pipelined(self):
        """Ingest thread: wait_batch + ring-decode into a slot half it
        OWNS (token from free_q; the GPU thread returns the token only
        after the H2D of that half has been issued and synced — without
        the token handoff the decode of batch N+2 could overwrite half
        A while batch N's DMA from it is still in flight). Main thread:
        upload + model + respond. Fallback paths never touch the ring
        (it is single-decoder by design)."""
        import os
        import queue
        import time as _t
        q = queue.Queue()
        free_q = queue.Queue()
        eng = self.engine
        # one token per decode window (N-phase pipeline: decode runs up
        # to N-1 batches ahead of the forward)
        nph = getattr(getattr(eng, "pre", None), "phases", 2)
        for _ph in range(nph):
            free_q.put(_ph)
        stats = ({"wait": 0.0, "decode": 0.0, "token": 0.0, "gpu": 0.0,
                  "json": 0.0, "respond": 0.0, "batches": 0, "imgs": 0}
                 if os.environ.get("TWD_FAST_STATS") else None)
        self.stats = stats

        # engines that can split upload (side stream, ingest thread)
        # from forward (default stream, this thread) overlap the H2D +
        # preprocess of batch N+1 with the forward of batch N
        can_upload = hasattr(eng, "upload_decoded")
        # dual GPU streams (opt-in TWD_GPU_STREAMS=2): a second executor
        # (own weights + HIP graphs + torch stream) runs alternate
        # batches so forward N+1 overlaps batch N's D2H/epilogue and
        # inter-launch Python. Off by default pending live-load
        # validation (the first rollout crashed on an unrelated NameError
        # and burned the round's A/B budget — see NOTES_ROUND2.md).
        n_gpu = 1
        if can_upload and hasattr(eng, "ex"):
            n_gpu = int(os.environ.get("TWD_GPU_STREAMS", "1"))
        self._fb_lock = threading.Lock()

        def ingest():
            while not self._stop.is_set():
                t0 = _t.perf_counter()
                batch = self.fe.wait_batch(self.cfg.max_batch,
                                           self.cfg.max_wait_ms)
                if not batch:
                    continue
                t1 = _t.perf_counter()
                bodies = [b for (_, b) in batch]
                ph = free_q.get()
                t2 = _t.perf_counter()
                handle = None
                try:
                    decoded = eng.pre.decode_shm(bodies, phase=ph)
                    if decoded is not None and can_upload:
                        handle = eng.upload_decoded(decoded,
                                                    len(bodies), ph)
                except Exception:
                    decoded = None  # bad image: isolate downstream
                if decoded is None:
                    free_q.put(ph)  # half unused
                    ph = None
                if stats is not None:
                    t3 = _t.perf_counter()
                    stats["wait"] += t1 - t0
                    stats["token"] += t2 - t1
                    stats["decode"] += t3 - t2
                    stats["batches"] += 1
                    stats["imgs"] += len(batch)
                q.put((batch, bodies, decoded, handle, ph))
            q.put(None)

        # stage 3: top-k + JSON + respond off the GPU thread (the
        # probs matrix is host numpy by then; JSON for a 256-batch is
        # 1-3 ms of pure-GIL work that would otherwise serialize with
        # the next batch's launch)
        respq = queue.Queue()

        def responder():
            while True:
                item = respq.get()
                if item is None:
                    return
                batch, kind, payload = item
                t0 = _t.perf_counter()
                if kind == "topk":
                    # top-k already computed on the GPU; JSON + enqueue
                    # entirely in C++
                    top, vals = payload
                    t1 = _t.perf_counter()
                    self.fe.respond_topk([rid for (rid, _) in batch],
                                         top, vals)
                elif kind == "probs":
                    top, vals = self._topk(payload)
                    t1 = _t.perf_counter()
                    self.fe.respond_topk([rid for (rid, _) in batch],
                                         top, vals)
                else:
                    t1 = _t.perf_counter()
                    self.fe.respond_many([
                        (rid, st, body.encode())
                        for (rid, _), (st, body) in zip(batch, payload)
                    ])
                if stats is not None:
                    stats["json"] += t1 - t0
                    stats["respond"] += _t.perf_counter() - t1

        k = min(self.cfg.top_k, self.cfg.num_classes)

        def gpu_loop(ex, stream):
            import torch
            while True:
                item = q.get()
                if item is None:
                    q.put(None)  # let sibling workers exit too
                    return
                batch, bodies, decoded, handle, ph = item
                if decoded is not None:
                    try:
                        t0 = _t.perf_counter()
                        if handle is not None and ex is not None:
                            bt, ev, n = handle
                            with torch.inference_mode(), \
                                    torch.cuda.stream(stream):
                                stream.wait_event(ev)
                                out = ex.run(
                                    {self.cfg.input_name: bt})[0]
                                # top-k ON the GPU: D2H shrinks from
                                # [n,classes] to 2x[n,k] and the numpy
                                # top-k leaves the responder stage
                                v, i = torch.topk(out.float()[:n], k, 1)
                                payload = ("topk",
                                           (i.to(torch.int32)
                                            .cpu().numpy(),
                                            v.cpu().numpy()))
                        elif (handle is not None
                              and hasattr(eng, "infer_uploaded_topk")):
                            payload = ("topk",
                                       eng.infer_uploaded_topk(handle, k))
                        elif handle is not None:
                            payload = ("probs",
                                       eng.infer_uploaded(handle))
                        else:
                            payload = ("probs",
                                       eng.infer_decoded(decoded,
                                                         len(bodies)))
                        if stats is not None:
                            stats["gpu"] += _t.perf_counter() - t0
                        respq.put((batch, payload[0], payload[1]))
                        continue
                    except Exception:
                        with self._fb_lock:
                            results = self._infer_isolating(bodies)
                    finally:
                        free_q.put(ph)  # forward synced: half reusable
                else:
                    with self._fb_lock:
                        results = self._infer_isolating(bodies)
                respq.put((batch, "results", results))

        t = threading.Thread(target=ingest, daemon=True)
        t.start()
        rt = threading.Thread(target=responder, daemon=True)
        rt.start()
        try:
            if n_gpu > 1:
                import torch

                from ..engine.gpu_executor import GpuExecutor
                workers = []
                for g in range(n_gpu):
                    if g == 0:
                        ex = eng.ex
                    else:
                        ex = GpuExecutor(eng.graph, device=eng.device,
                                         cfg=eng.cfg)
                        # pre-compile/capture the bucket plans so the
                        # first live batches don't pay graph capture
                        b = 1
                        cfgv = self.cfg
                        while True:
                            z = torch.zeros(
                                (b, cfgv.input_height, cfgv.input_width,
                                 cfgv.input_channels))
                            with torch.inference_mode():
                                for _ in range(2):  # warm + capture
                                    ex.run({cfgv.input_name: z})
                            if b >= cfgv.max_batch:
                                break
                            b = min(b * 2, cfgv.max_batch)
                    st = torch.cuda.Stream()
                    th = threading.Thread(target=gpu_loop,
                                          args=(ex, st), daemon=True)
                    th.start()
                    workers.append(th)
                while (not self._stop.is_set()
                       and any(w.is_alive() for w in workers)):
                    self._stop.wait(0.5)
            else:
                gpu_loop(None, None)
        finally:
            q.put(None)
            respq.put(None)
            if stats is not None and stats["batches"]:
                import sys
                b = stats["batches"]
                print("[fast stats] " + " ".join(
                    f"{k}={1000 * v / b:.2f}ms" for k, v in stats.items()
                    if isinstance(v, float)) +
                    f" batches={b} avg_bs={stats['imgs'] / b:.1f}",
                    file=sys.stderr, flush=True)
            self.fe.stop()

    def start_background(self) -> threading.Thread:
        t = threading.Thread(target=self.serve_forever, daemon=True)
        t.start()
        return t

    def stop(self):
        self._stop.set()
        self.fe.stop()


def _decode(data: bytes, cfg: Config, raw: bool = False):
    """bytes -> uint8 HWC (raw=True: original size for infer_arrays)
    or f32 model-input array."""
    if raw:
        import io

        from PIL import Image
        img = Image.open(io.BytesIO(data))
        img = img.convert("L" if cfg.input_channels == 1 else "RGB")
        a = np.asarray(img, dtype=np.uint8)
        return a[:, :, None] if a.ndim == 2 else a
    from ..ingest.preprocess import decode_image
    return decode_image(data, (cfg.input_width, cfg.input_height),
                        cfg.input_channels, cfg.normalize)


def _default_engine(cfg: Config):
    import torch
    if cfg.gpus > 1:
        from ..parallel.replicas import ReplicatedEngine
        return ReplicatedEngine(cfg, serve_blobs=True)
    if torch.cuda.is_available():
        from ..engine.service import GpuServingEngine
        return GpuServingEngine(cfg)
    from ..engine.service import LocalEngine
    return LocalEngine(cfg)


def main(argv=None):
    import argparse
    import gc

    from ..conf.config import add_config_args, config_from_args
    p = argparse.ArgumentParser("tensorflow_web_deploy_amd fast server")
    add_config_args(p)
    p.add_argument("--http-workers", type=int, default=3)
    args = p.parse_args(argv)
    cfg = config_from_args(args)
    srv = FastServer(cfg, workers=args.http_workers)
    gc.collect()
    gc.freeze()
    gc.set_threshold(200000, 100, 100)
    import signal
    try:
        signal.signal(signal.SIGTERM, lambda *_: srv.stop())
    except ValueError:
        pass  # not the main thread (embedded use)
    print(f"[fast] serving {srv.model_name} on :{srv.port}", flush=True)
    try:
        srv.serve_forever()
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
