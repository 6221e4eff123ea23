"""HTTP-layer tests: Flask routes + JSON schema, engine-agnostic."""
import io

import numpy as np
import pytest
from PIL import Image

from tensorflow_web_deploy_amd.conf.config import Config
from tensorflow_web_deploy_amd.engine.service import FakeEngine, LocalEngine
from tensorflow_web_deploy_amd.server.app import create_app


def _png_bytes(w=28, h=28, gray=True):
    arr = (np.random.rand(h, w) * 255).astype("uint8")
    img = Image.fromarray(arr, "L" if gray else "RGB")
    buf = io.BytesIO()
    img.save(buf, "PNG")
    return buf.getvalue()


@pytest.fixture
def client():
    cfg = Config(model_name="lenet", device="cpu")
    app = create_app(cfg, FakeEngine(10))
    app.config["TESTING"] = True
    return app.test_client()


def test_index(client):
    r = client.get("/")
    assert r.status_code == 200
    assert b"form" in r.data


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    assert r.get_json()["status"] == "ok"


def test_predict_multipart(client):
    r = client.post("/predict", data={
        "file": (io.BytesIO(_png_bytes()), "x.png"),
    }, content_type="multipart/form-data")
    assert r.status_code == 200
    j = r.get_json()
    assert "predictions" in j and len(j["predictions"]) == 5
    p0 = j["predictions"][0]
    assert set(p0) == {"label", "class_id", "prob"}
    assert j["latency_ms"] >= 0
    probs = [p["prob"] for p in j["predictions"]]
    assert probs == sorted(probs, reverse=True)


def test_predict_raw_body(client):
    r = client.post("/predict", data=_png_bytes(),
                    content_type="application/octet-stream")
    assert r.status_code == 200


def test_predict_no_image(client):
    r = client.post("/predict")
    assert r.status_code == 400


def test_predict_bad_image(client):
    r = client.post("/predict", data=b"not an image",
                    content_type="application/octet-stream")
    assert r.status_code == 400


def test_metrics(client):
    client.post("/predict", data=_png_bytes(),
                content_type="application/octet-stream")
    r = client.get("/metrics")
    assert r.status_code == 200
    assert b"twd_requests_total" in r.data


def test_e2e_lenet_from_pb(lenet_pb):
    """Full path: frozen .pb from disk -> engine -> HTTP predict."""
    cfg = Config(model_path=lenet_pb, device="cpu")
    engine = LocalEngine(cfg)
    app = create_app(cfg, engine)
    app.config["TESTING"] = True
    c = app.test_client()
    r = c.post("/predict", data=_png_bytes(),
               content_type="application/octet-stream")
    assert r.status_code == 200
    j = r.get_json()
    s = sum(p["prob"] for p in j["predictions"])
    assert 0 < s <= 1.0 + 1e-5


def test_debug_trace_endpoint():
    from tensorflow_web_deploy_amd.sched.batcher import (BatchedEngine,
                                                         DynamicBatcher)
    cfg = Config(model_name="lenet", device="cpu", max_batch=8)
    batcher = DynamicBatcher(FakeEngine(10), max_batch=8, max_wait_ms=1.0)
    batcher.start()
    try:
        engine = BatchedEngine(batcher)
        app = create_app(cfg, engine, batcher=batcher)
        app.config["TESTING"] = True
        c = app.test_client()
        c.post("/predict", data=_png_bytes(),
               content_type="application/octet-stream")
        r = c.get("/debug/trace")
        assert r.status_code == 200
        j = r.get_json()
        assert "batches" in j and len(j["batches"]) >= 1
        assert j["config"]["max_batch"] == 8
    finally:
        batcher.stop()


def test_conf_presets_load():
    """Every shipped YAML preset parses into a Config."""
    import glob
    import os

    from tensorflow_web_deploy_amd.conf.config import Config, load_config
    root = os.path.join(os.path.dirname(__file__), "..",
                        "tensorflow_web_deploy_amd", "conf", "presets")
    paths = sorted(glob.glob(os.path.join(root, "*.yaml")))
    assert len(paths) == 5
    for p in paths:
        cfg = load_config(p)
        assert isinstance(cfg, Config)
        assert cfg.model_name or cfg.model_path
