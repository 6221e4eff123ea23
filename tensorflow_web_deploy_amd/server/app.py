"""Flask HTTP serving surface (reference-parity endpoints).

Reference class surface: `GET /` renders an upload form; `POST /predict`
takes a multipart image upload, runs preprocess -> model forward ->
top-k, returns predictions. We keep that surface and add operational
endpoints (/healthz, /metrics) and a JSON response schema:

  POST /predict  (multipart 'file', or raw image bytes)
    -> {"predictions": [{"label": str, "class_id": int, "prob": float}...],
        "model": str, "latency_ms": float}
"""
from __future__ import annotations

import time
from typing import Optional

import numpy as np
from flask import Flask, Response, jsonify, render_template, request

from ..conf.config import Config
from ..ingest.preprocess import decode_image
from .labels import load_labels

try:
    from prometheus_client import (CONTENT_TYPE_LATEST, Counter, Histogram,
                                   generate_latest)
    _PROM = True
except Exception:  # pragma: no cover
    _PROM = False


if _PROM:
    # module-level singletons: create_app may be called more than once
    # (tests, multi-app) and the default registry rejects duplicates
    req_counter = Counter("twd_requests_total", "predict requests", ["status"])
    lat_hist = Histogram("twd_request_latency_seconds",
                         "end-to-end predict latency")


def create_app(cfg: Config, engine, batcher=None) -> Flask:
    app = Flask(__name__)
    labels = load_labels(cfg.labels_path, cfg.num_classes)
    if batcher is None:
        batcher = getattr(engine, "batcher", None)

    @app.get("/")
    def index():
        return render_template("index.html", model=cfg.model_name or
                               cfg.model_path)

    @app.get("/healthz")
    def healthz():
        return jsonify({"status": "ok"})

    @app.post("/predict")
    def predict():
        t0 = time.perf_counter()
        data: Optional[bytes] = None
        if "file" in request.files:
            data = request.files["file"].read()
        elif request.data:
            data = request.data
        if not data:
            if _PROM:
                req_counter.labels("bad_request").inc()
            return jsonify({"error": "no image provided"}), 400
        if hasattr(engine, "infer_blobs"):
            # GPU-ingest path: decode in worker processes + side-stream
            # HIP preprocess inside the coalesced batch; request thread
            # only validates the header
            try:
                import io as _io

                from PIL import Image
                Image.open(_io.BytesIO(data)).verify()
            except Exception as e:
                if _PROM:
                    req_counter.labels("bad_image").inc()
                return jsonify({"error": f"cannot decode image: {e}"}), 400
            probs = engine.infer_blobs([data])[0]
        else:
            try:
                arr = decode_image(data, (cfg.input_width, cfg.input_height),
                                   cfg.input_channels, cfg.normalize)
            except Exception as e:
                if _PROM:
                    req_counter.labels("bad_image").inc()
                return jsonify({"error": f"cannot decode image: {e}"}), 400
            probs = engine.infer(arr[None])[0]
        k = min(cfg.top_k, len(probs))
        top = np.argsort(probs)[::-1][:k]
        dt = (time.perf_counter() - t0) * 1000.0
        if _PROM:
            req_counter.labels("ok").inc()
            lat_hist.observe(dt / 1000.0)
        return jsonify({
            "predictions": [
                {"label": labels[int(i)], "class_id": int(i),
                 "prob": float(probs[int(i)])}
                for i in top
            ],
            "model": cfg.model_name or cfg.model_path,
            "latency_ms": dt,
        })

    if _PROM:
        @app.get("/metrics")
        def metrics():
            return Response(generate_latest(), mimetype=CONTENT_TYPE_LATEST)

    @app.get("/debug/trace")
    def debug_trace():
        """Batching/latency spans ring buffer (SURVEY §5 tracing)."""
        spans = list(batcher.trace) if batcher is not None else []
        return jsonify({
            "batches": spans[-64:],
            "config": {"max_batch": cfg.max_batch,
                       "max_wait_ms": cfg.max_wait_ms,
                       "model": cfg.model_name or cfg.model_path,
                       "gpus": cfg.gpus},
        })

    return app


def main(argv=None):
    import sys as _sys
    _sys.setswitchinterval(0.002)  # fewer GIL handoffs under thread load
    import argparse
    from ..conf.config import add_config_args, config_from_args
    from ..engine.service import LocalEngine

    p = argparse.ArgumentParser("tensorflow_web_deploy_amd server")
    add_config_args(p)
    p.add_argument("--no-batcher", action="store_true",
                   help="serve without dynamic batching")
    args = p.parse_args(argv)
    cfg = config_from_args(args)
    import torch
    if cfg.gpus > 1:
        from ..parallel.replicas import ReplicatedEngine
        backend_engine = ReplicatedEngine(cfg)
    elif torch.cuda.is_available():
        from ..engine.service import GpuServingEngine
        backend_engine = GpuServingEngine(cfg)
    else:
        backend_engine = LocalEngine(cfg)
    if args.no_batcher:
        engine = backend_engine
    else:
        from ..sched.batcher import BatchedEngine, DynamicBatcher
        batcher = DynamicBatcher(backend_engine, max_batch=cfg.max_batch,
                                 max_wait_ms=cfg.max_wait_ms)
        batcher.start()
        engine = BatchedEngine(batcher,
                               raw=hasattr(backend_engine, "infer_blobs"))
    import gc
    gc.collect()
    gc.freeze()
    gc.set_threshold(200000, 100, 100)
    app = create_app(cfg, engine)
    app.run(host=cfg.host, port=cfg.port, threaded=True)


if __name__ == "__main__":
    main()
