"""Smoke test for the multi-process SO_REUSEPORT serving stack
(parallel/service.py — kept as the documented alternative architecture;
off by default, see BASELINE.md serving history)."""
import io
import json
import urllib.request

import numpy as np
import pytest
from PIL import Image

from tensorflow_web_deploy_amd.conf.config import Config


@pytest.mark.timeout(180)
def test_multiproc_stack_serves():
    from tensorflow_web_deploy_amd.parallel.service import MultiprocServer
    cfg = Config(model_name="lenet", device="cpu", host="127.0.0.1",
                 port=0, input_height=28, input_width=28,
                 input_channels=1, num_classes=10, max_batch=4,
                 max_wait_ms=3)
    # port=0 is not supported by SO_REUSEPORT sharing: pick a free one
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    cfg = cfg.replace(port=port)
    srv = MultiprocServer(cfg, http_workers=2, start_timeout=300.0)
    try:
        arr = (np.random.rand(28, 28) * 255).astype("uint8")
        buf = io.BytesIO()
        Image.fromarray(arr, "L").save(buf, "PNG")
        img = buf.getvalue()
        import time
        deadline = time.time() + 120
        j = None
        while time.time() < deadline:
            try:
                req = urllib.request.Request(
                    f"http://127.0.0.1:{port}/predict", data=img,
                    headers={"Content-Type": "application/octet-stream"})
                with urllib.request.urlopen(req, timeout=20) as r:
                    j = json.loads(r.read())
                break
            except Exception:
                time.sleep(1.0)
        assert j is not None, "no response from multiproc stack"
        assert len(j["predictions"]) == 5
    finally:
        srv.shutdown()
