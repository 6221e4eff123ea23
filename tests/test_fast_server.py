"""C++ epoll frontend (_httpd) + FastServer loop, CPU-only.

Exercises the real TCP path: raw-body POST /predict, keep-alive reuse,
inline /healthz + 404, bad-image isolation within a batch, and a
concurrent multi-client burst (the GIL-contention scenario the frontend
exists to remove — BASELINE.md serving analysis)."""
import http.client
import io
import json
import threading
import urllib.error
import urllib.request

import numpy as np
import pytest
from PIL import Image

from tensorflow_web_deploy_amd.conf.config import Config
from tensorflow_web_deploy_amd.engine.service import FakeEngine


def _png_bytes(w=28, h=28):
    arr = (np.random.rand(h, w) * 255).astype("uint8")
    img = Image.fromarray(arr, "L")
    buf = io.BytesIO()
    img.save(buf, "PNG")
    return buf.getvalue()


@pytest.fixture
def server():
    from tensorflow_web_deploy_amd.server.fast import FastServer
    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 max_batch=8, max_wait_ms=5)
    srv = FastServer(cfg, engine=FakeEngine(10), workers=2, port=0)
    srv.start_background()
    yield srv
    srv.stop()


def _post(port, body, path="/predict", timeout=10):
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}{path}", data=body,
        headers={"Content-Type": "application/octet-stream"})
    with urllib.request.urlopen(req, timeout=timeout) as r:
        return r.status, json.loads(r.read())


def test_predict_roundtrip(server):
    status, j = _post(server.port, _png_bytes())
    assert status == 200
    assert len(j["predictions"]) == 5
    p0 = j["predictions"][0]
    assert set(p0) == {"label", "class_id", "prob"}
    probs = [p["prob"] for p in j["predictions"]]
    assert probs == sorted(probs, reverse=True)


def test_index_form_and_chunked_rejected(server):
    with urllib.request.urlopen(
            f"http://127.0.0.1:{server.port}/", timeout=10) as r:
        body = r.read()
        assert r.status == 200 and b"form" in body
    # chunked request bodies are answered with 411, not mis-parsed
    import socket
    s = socket.create_connection(("127.0.0.1", server.port), timeout=10)
    s.sendall(b"POST /predict HTTP/1.1\r\nHost: x\r\n"
              b"Transfer-Encoding: chunked\r\n\r\n"
              b"5\r\nhello\r\n0\r\n\r\n")
    resp = s.recv(4096)
    assert b"411" in resp.split(b"\r\n", 1)[0]
    s.close()


def test_healthz_inline_and_404(server):
    with urllib.request.urlopen(
            f"http://127.0.0.1:{server.port}/healthz", timeout=10) as r:
        assert json.loads(r.read())["status"] == "ok"
    with pytest.raises(urllib.error.HTTPError) as ei:
        urllib.request.urlopen(
            f"http://127.0.0.1:{server.port}/nope", timeout=10)
    assert ei.value.code == 404


def test_keep_alive_reuse(server):
    conn = http.client.HTTPConnection("127.0.0.1", server.port, timeout=10)
    try:
        for _ in range(3):
            conn.request("POST", "/predict", body=_png_bytes())
            r = conn.getresponse()
            assert r.status == 200
            json.loads(r.read())  # must fully drain to reuse
    finally:
        conn.close()


def test_bad_image_isolated(server):
    """A corrupt upload 400s; a good one in the same window still 200s."""
    results = {}

    def go(name, body):
        try:
            results[name] = _post(server.port, body)[0]
        except urllib.error.HTTPError as e:
            results[name] = e.code

    ts = [threading.Thread(target=go, args=("bad", b"not an image")),
          threading.Thread(target=go, args=("good", _png_bytes()))]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=15)
    assert results["bad"] == 400
    assert results["good"] == 200


def test_concurrent_burst(server):
    ok = []
    lock = threading.Lock()

    def client(n):
        for _ in range(n):
            status, j = _post(server.port, _png_bytes())
            assert status == 200 and len(j["predictions"]) == 5
            with lock:
                ok.append(1)

    ts = [threading.Thread(target=client, args=(5,)) for _ in range(8)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert len(ok) == 40
    st = server.fe.stats()
    assert st["responded"] >= 40


def test_stats_and_stop():
    from tensorflow_web_deploy_amd.server.fast import FastServer
    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 max_batch=4, max_wait_ms=2)
    srv = FastServer(cfg, engine=FakeEngine(10), workers=1, port=0)
    t = srv.start_background()
    _post(srv.port, _png_bytes())
    assert srv.fe.stats()["requests"] >= 1
    srv.stop()
    t.join(timeout=10)
    assert not t.is_alive()


@pytest.mark.gpu
def test_fast_server_gpu_end_to_end():
    """Real GpuServingEngine through the whole fast stack (C++ frontend,
    shm-ring decode incl the GPU-jpeg path, dual-stream pipeline):
    responses must match in-process infer_blobs on the same JPEGs."""
    import torch
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    from tensorflow_web_deploy_amd.engine.service import GpuServingEngine
    from tensorflow_web_deploy_amd.server.fast import FastServer

    from tensorflow_web_deploy_amd.conf.config import Config as Cfg
    cfg = Cfg(model_name="lenet", device="cuda", host="127.0.0.1",
              input_height=28, input_width=28, input_channels=1,
              num_classes=10, max_batch=16, max_wait_ms=3)
    eng = GpuServingEngine(cfg)
    srv = FastServer(cfg, engine=eng, workers=2, port=0)
    srv.start_background()
    try:
        imgs = [_png_bytes() for _ in range(12)]
        ref = eng.infer_blobs(imgs)  # in-process reference
        got = {}

        def post(i):
            status, j = _post(srv.port, imgs[i])
            assert status == 200
            got[i] = j["predictions"][0]["class_id"]

        ts = [threading.Thread(target=post, args=(i,))
              for i in range(len(imgs))]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=60)
        assert len(got) == len(imgs)
        for i in range(len(imgs)):
            assert got[i] == int(ref[i].argmax()), i
    finally:
        srv.stop()


def test_pipelined_loop_cpu_via_replicas():
    """_serve_pipelined on CPU: ReplicatedEngine(gloo, world=1,
    serve_blobs=True) has infer_decoded + pre.decode_shm, so the full
    three-stage pipeline (incl the n_gpu dispatch block) runs without
    a GPU — this is the test that would have caught the NameError that
    killed the first dual-stream rollout."""
    from tensorflow_web_deploy_amd.parallel.replicas import ReplicatedEngine
    from tensorflow_web_deploy_amd.server.fast import FastServer
    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 input_height=28, input_width=28, input_channels=1,
                 num_classes=10, max_batch=8, max_wait_ms=3)
    eng = ReplicatedEngine(cfg, world=1, backend="gloo",
                           serve_blobs=True, start_timeout=120.0)
    srv = FastServer(cfg, engine=eng, workers=2, port=0)
    srv.start_background()
    try:
        oks = []

        def post():
            status, j = _post(srv.port, _png_bytes(), timeout=30)
            assert status == 200 and len(j["predictions"]) == 5
            # this path responds through the C++ respond_topk JSON
            assert set(j) == {"predictions", "model"}
            for pr in j["predictions"]:
                assert set(pr) == {"label", "class_id", "prob"}
                assert isinstance(pr["class_id"], int)
            probs = [pr["prob"] for pr in j["predictions"]]
            assert probs == sorted(probs, reverse=True)
            oks.append(1)

        ts = [threading.Thread(target=post) for _ in range(10)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=60)
        assert len(oks) == 10
    finally:
        srv.stop()
        eng.shutdown()


def test_c_json_schema_matches_python(server):
    """respond_topk's C-formatted JSON must carry the same schema and
    ordering as the Flask/Python payloads (label escaping included)."""
    status, j = _post(server.port, _png_bytes())
    assert status == 200
    assert set(j) == {"predictions", "model"}
    for p in j["predictions"]:
        assert set(p) == {"label", "class_id", "prob"}
        assert isinstance(p["label"], str)
        assert isinstance(p["class_id"], int)
        assert 0.0 <= p["prob"] <= 1.0
    probs = [p["prob"] for p in j["predictions"]]
    assert probs == sorted(probs, reverse=True)


def test_idle_connection_reaped():
    """A connection that sends nothing is closed by the idle sweep; an
    active request is never reaped mid-flight."""
    import socket
    import time

    from tensorflow_web_deploy_amd.server.fast import FastServer
    from tensorflow_web_deploy_amd._httpd import Frontend
    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 max_batch=4, max_wait_ms=2)
    srv = FastServer(cfg, engine=FakeEngine(10), workers=1, port=0)
    # swap in a frontend with a 1 s idle timeout for the test
    srv.fe.stop()
    srv.fe = Frontend("127.0.0.1", 0, 1, 1024, idle_timeout_s=1)
    srv.fe.set_labels([str(i) for i in range(10)], "lenet")
    srv.port = srv.fe.port()
    srv.start_background()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=10)
        time.sleep(7)  # > timeout + sweep period
        # the server should have closed us: recv returns EOF
        s.settimeout(3)
        assert s.recv(64) == b""
        s.close()
        # normal requests still work afterwards
        status, j = _post(srv.port, _png_bytes())
        assert status == 200 and len(j["predictions"]) == 5
    finally:
        srv.stop()


def test_stress_mixed_traffic(server):
    """Race-class stress: concurrent keep-alive and one-shot clients,
    interleaved bad payloads and healthz probes; every request gets a
    correct, well-formed answer and the frontend counters reconcile."""
    import random

    errors = []
    ok = []
    lock = threading.Lock()
    rng = random.Random(7)

    def keepalive_client(n):
        conn = http.client.HTTPConnection("127.0.0.1", server.port,
                                          timeout=15)
        try:
            for _ in range(n):
                bad = rng.random() < 0.2
                body = b"garbage-bytes" if bad else _png_bytes()
                conn.request("POST", "/predict", body=body)
                r = conn.getresponse()
                data = r.read()
                j = json.loads(data)
                if bad:
                    assert r.status == 400 and "error" in j, j
                else:
                    assert r.status == 200, j
                    assert len(j["predictions"]) == 5
                with lock:
                    ok.append(1)
        except Exception as e:  # pragma: no cover
            with lock:
                errors.append(repr(e))
        finally:
            conn.close()

    def oneshot_client(n):
        for _ in range(n):
            try:
                if rng.random() < 0.3:
                    with urllib.request.urlopen(
                            f"http://127.0.0.1:{server.port}/healthz",
                            timeout=15) as r:
                        assert json.loads(r.read())["status"] == "ok"
                else:
                    status, j = _post(server.port, _png_bytes(),
                                      timeout=15)
                    assert status == 200
                with lock:
                    ok.append(1)
            except Exception as e:  # pragma: no cover
                with lock:
                    errors.append(repr(e))

    ts = ([threading.Thread(target=keepalive_client, args=(8,))
           for _ in range(6)] +
          [threading.Thread(target=oneshot_client, args=(8,))
           for _ in range(6)])
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=60)
    assert not errors, errors[:3]
    assert len(ok) == 12 * 8
    st = server.fe.stats()
    assert st["responded"] + st["rejected_503"] >= 50


def test_multipart_upload(server):
    """The HTML form posts multipart/form-data; the frontend extracts
    the file part in C++ (same contract as Flask request.files)."""
    img = _png_bytes()
    boundary = "----twdboundary123"
    body = (
        f"--{boundary}\r\n"
        "Content-Disposition: form-data; name=\"note\"\r\n\r\n"
        "hello\r\n"
        f"--{boundary}\r\n"
        "Content-Disposition: form-data; name=\"file\"; "
        "filename=\"x.png\"\r\n"
        "Content-Type: image/png\r\n\r\n"
    ).encode() + img + f"\r\n--{boundary}--\r\n".encode()
    req = urllib.request.Request(
        f"http://127.0.0.1:{server.port}/predict", data=body,
        headers={"Content-Type":
                 f"multipart/form-data; boundary={boundary}"})
    with urllib.request.urlopen(req, timeout=15) as r:
        j = json.loads(r.read())
        assert r.status == 200
        assert len(j["predictions"]) == 5


def test_metrics_prometheus(server):
    _post(server.port, _png_bytes())
    with urllib.request.urlopen(
            f"http://127.0.0.1:{server.port}/metrics", timeout=10) as r:
        text = r.read().decode()
        assert r.status == 200
        assert "twd_requests_total" in text
        assert "twd_responses_total" in text


def test_frontend_limit_edges(server):
    """C++ frontend hardening: runaway headers -> 400, Content-Length
    beyond 32 MiB (incl. negative wrap) -> 413, garbage request line ->
    connection closed, pipelined requests on one socket all answered."""
    import socket

    def raw(data, expect_close=False, nresp=1):
        s = socket.create_connection(("127.0.0.1", server.port), timeout=10)
        s.sendall(data)
        buf = b""
        try:
            while buf.count(b"HTTP/1.1") < nresp:
                d = s.recv(65536)
                if not d:
                    break
                buf += d
        except socket.timeout:
            pass
        s.close()
        return buf

    # runaway header block (>64 KiB without CRLFCRLF) -> 400
    r = raw(b"POST /predict HTTP/1.1\r\nX-Junk: " + b"a" * 70000)
    assert b"400" in r.split(b"\r\n")[0]

    # oversized Content-Length -> 413 before any body is read
    r = raw(b"POST /predict HTTP/1.1\r\nHost: x\r\n"
            b"Content-Length: 999999999999\r\n\r\n")
    assert b"413" in r.split(b"\r\n")[0]

    # negative Content-Length wraps to huge -> 413 (never a hang)
    r = raw(b"POST /predict HTTP/1.1\r\nHost: x\r\n"
            b"Content-Length: -1\r\n\r\n")
    assert b"413" in r.split(b"\r\n")[0]

    # garbage request line -> closed without a crash
    r = raw(b"NONSENSE\r\n\r\n")
    assert b"predictions" not in r

    # two pipelined healthz on one socket -> two responses
    r = raw(b"GET /healthz HTTP/1.1\r\nHost: x\r\n\r\n"
            b"GET /healthz HTTP/1.1\r\nHost: x\r\n\r\n", nresp=2)
    assert r.count(b"200 OK") == 2

    # server still healthy afterwards
    status, j = _post(server.port, _png_bytes())
    assert status == 200
