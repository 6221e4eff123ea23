"""HTTP serving benchmark: images/sec (closed-loop) + p50/p99 latency
(open-loop) through the real Flask stack — the second half of the
BASELINE metric ("images/sec (whole node) + p50 HTTP latency").

Runs the server in-process on a real TCP port, drives it with
concurrent clients posting PNG images.

  python bench/serve_bench.py --model resnet50 --duration 10 \
      --clients 32 [--qps 500] [--gpus N]
"""
import argparse
import io
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402


def _png_bytes(h, w, gray=False):
    from PIL import Image
    arr = (np.random.rand(h, w, 3) * 255).astype("uint8")
    img = Image.fromarray(arr[..., 0] if gray else arr, "L" if gray else "RGB")
    buf = io.BytesIO()
    img.save(buf, "JPEG", quality=90)  # realistic upload payload
    return buf.getvalue()


def _client_proc(url, img, nthreads, per_thread_qps, duration, q):
    import threading
    import time
    import socket
    lat = []
    lock = threading.Lock()
    stop_at = time.perf_counter() + duration
    # pre-serialized request over a persistent keep-alive socket: the
    # urllib client cost ~3x this in client CPU (new TCP + header
    # parsing per request), and the load generator shares the box with
    # the decode workers — client cycles ARE server throughput.
    from urllib.parse import urlparse
    u = urlparse(url)
    host, port = u.hostname, u.port
    req_bytes = (f"POST {u.path} HTTP/1.1\r\nHost: {host}\r\n"
                 f"Content-Type: application/octet-stream\r\n"
                 f"Content-Length: {len(img)}\r\n\r\n"
                 ).encode() + img

    def run():
        sock = None
        buf = bytearray()

        def post_once():
            nonlocal sock, buf
            t0 = time.perf_counter()
            if sock is None:
                sock = socket.create_connection((host, port), timeout=30)
                sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                buf = bytearray()
            try:
                sock.sendall(req_bytes)
                # read headers
                while b"\r\n\r\n" not in buf:
                    d = sock.recv(65536)
                    if not d:
                        raise ConnectionError("closed")
                    buf += d
                head, _, rest = bytes(buf).partition(b"\r\n\r\n")
                hl = head.decode("latin-1", "replace").lower()
                clen = 0
                for line in hl.split("\r\n"):
                    if line.startswith("content-length:"):
                        clen = int(line.split(":", 1)[1])
                body = bytearray(rest)
                while len(body) < clen:
                    d = sock.recv(65536)
                    if not d:
                        raise ConnectionError("closed")
                    body += d
                buf = body[clen:]
                if "connection: close" in hl:
                    sock.close()
                    sock = None
            except Exception:
                try:
                    sock.close()
                except Exception:
                    pass
                sock = None
                raise
            dt = time.perf_counter() - t0
            with lock:
                lat.append(dt)

        if per_thread_qps > 0:
            interval = 1.0 / per_thread_qps
            nxt = time.perf_counter()
            while time.perf_counter() < stop_at:
                now = time.perf_counter()
                if now < nxt:
                    time.sleep(min(interval, nxt - now))
                    continue
                nxt += interval
                try:
                    post_once()
                except Exception:
                    pass
        else:
            while time.perf_counter() < stop_at:
                try:
                    post_once()
                except Exception:
                    pass

    ts = [threading.Thread(target=run, daemon=True) for _ in range(nthreads)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=duration + 30)
    q.put(lat)


def main():
    import sys as _sys
    _sys.setswitchinterval(0.002)  # fewer GIL handoffs under thread load
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--duration", type=float, default=10.0)
    p.add_argument("--clients", type=int, default=32)
    p.add_argument("--qps", type=float, default=0.0,
                   help=">0: open-loop at this rate (latency mode)")
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--port", type=int, default=5123)
    p.add_argument("--max-batch", type=int, default=64)
    p.add_argument("--max-wait-ms", type=float, default=2.0)
    p.add_argument("--http-workers", type=int, default=0,
                   help=">0: multi-process SO_REUSEPORT serving stack")
    p.add_argument("--fast", action="store_true", default=True,
                   help="C++ epoll frontend + batch loop (default)")
    p.add_argument("--flask", dest="fast", action="store_false",
                   help="measure the Flask/werkzeug stack instead")
    args = p.parse_args()

    import torch
    from tensorflow_web_deploy_amd.conf.config import Config
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    from tensorflow_web_deploy_amd.sched.batcher import (BatchedEngine,
                                                         DynamicBatcher)
    from tensorflow_web_deploy_amd.server.app import create_app
    from tensorflow_web_deploy_amd.parallel.service import MultiprocServer

    size = 299 if args.model == "inception_v3" else 224
    gray = args.model == "lenet"
    h = w = 28 if gray else size
    cfg = Config(model_name=args.model,
                 device="cuda" if torch.cuda.is_available() else "cpu",
                 input_height=h, input_width=w,
                 input_channels=1 if gray else 3,
                 num_classes=10 if gray else 1000,
                 max_batch=args.max_batch, max_wait_ms=args.max_wait_ms,
                 gpus=args.gpus, port=args.port)
    mserver = None
    batcher = None
    fast = None
    if args.fast:
        import dataclasses
        from tensorflow_web_deploy_amd.server.fast import FastServer
        cfg = dataclasses.replace(cfg, host="127.0.0.1")
        fast = FastServer(cfg, workers=3)
        backend = fast.engine
        b = 1
        while True:
            backend.infer(np.zeros((b, h, w, cfg.input_channels),
                                   np.float32))
            if b >= args.max_batch:
                break
            b = min(b * 2, args.max_batch)
        if hasattr(backend, "pre"):
            backend.pre.warm_pool()
        import gc
        gc.collect()
        gc.freeze()
        gc.set_threshold(200000, 100, 100)
        fast.start_background()
    elif args.http_workers > 0:
        mserver = MultiprocServer(cfg, http_workers=args.http_workers)
    elif args.gpus > 1:
        from tensorflow_web_deploy_amd.parallel.replicas import \
            ReplicatedEngine
        backend = ReplicatedEngine(cfg)
    elif torch.cuda.is_available():
        from tensorflow_web_deploy_amd.engine.service import GpuServingEngine
        backend = GpuServingEngine(cfg)
    else:
        backend = LocalEngine(cfg)
    if mserver is None and fast is None:
        # warm the engine (compile + HIP-graph capture for every bucket)
        b = 1
        while True:
            backend.infer(np.zeros((b, h, w, cfg.input_channels),
                                   np.float32))
            if b >= args.max_batch:
                break
            b = min(b * 2, args.max_batch)
        # GC hygiene: per-request ndarray churn triggers periodic gen-2
        # collections that showed up as 90-280 ms latency spikes
        import gc
        gc.collect()
        gc.freeze()
        gc.set_threshold(200000, 100, 100)
        batcher = DynamicBatcher(backend, max_batch=cfg.max_batch,
                                 max_wait_ms=cfg.max_wait_ms)
        batcher.start()
        engine = BatchedEngine(batcher, raw=hasattr(backend, "infer_blobs"))
        if hasattr(backend, "pre"):
            backend.pre.warm_pool()
        app = create_app(cfg, engine)

        th = threading.Thread(
            target=lambda: app.run(host="127.0.0.1", port=args.port,
                                   threaded=True, use_reloader=False),
            daemon=True)
        th.start()
    time.sleep(1.0)

    # Load generation runs in SEPARATE PROCESSES so the measurement is not
    # GIL-coupled to the server (each worker runs a few client threads).
    import multiprocessing as mp

    img = _png_bytes(h, w, gray)
    url = f"http://127.0.0.1:{args.port}/predict"

    # warmup through the full stack (retry while HTTP workers boot)
    import urllib.request
    deadline = time.time() + 120
    done = 0
    while done < 4:
        try:
            req = urllib.request.Request(url, data=img, headers={
                "Content-Type": "application/octet-stream"})
            urllib.request.urlopen(req, timeout=60).read()
            done += 1
        except Exception:
            if time.time() > deadline:
                raise
            time.sleep(0.5)

    nproc = min(args.clients, 16)
    per_proc_threads = max(1, args.clients // nproc)
    ctx = mp.get_context("fork")
    q = ctx.Queue()
    per_thread_qps = (args.qps / args.clients) if args.qps > 0 else 0.0
    procs = []
    for _ in range(nproc):
        pr = ctx.Process(target=_client_proc,
                         args=(url, img, per_proc_threads, per_thread_qps,
                               args.duration, q), daemon=True)
        pr.start()
        procs.append(pr)
    elapsed = args.duration
    lat = []
    for _ in range(nproc):
        lat.extend(q.get(timeout=args.duration + 60))
    for pr in procs:
        pr.join(timeout=10)
    ls = sorted(lat)
    n = len(ls)
    if batcher is not None:
        batcher.stop()
    out = {
        "metric": "serve_images_per_sec" if args.qps <= 0 else "serve_p50_ms",
        "mode": "closed" if args.qps <= 0 else f"open@{args.qps}qps",
        "model": args.model,
        "clients": args.clients,
        "images_per_sec": round(n / elapsed, 2),
        "p50_ms": round(1000 * ls[len(ls) // 2], 2) if ls else None,
        "p99_ms": round(1000 * ls[int(len(ls) * 0.99)], 2) if ls else None,
        "requests": n,
        "http_workers": args.http_workers,
        "fast": bool(args.fast),
        "batch_trace_tail": batcher.trace[-3:] if batcher else None,
    }
    print(json.dumps(out))
    if mserver is not None:
        mserver.shutdown()
    if fast is not None:
        fast.stop()


if __name__ == "__main__":
    main()
