"""Flow inference service: TestClient round-trips for every response
format (json / .flo bytes / color PNG)."""

import io
import struct

import numpy as np
import pytest
import torch
from PIL import Image

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402


@pytest.fixture(scope="module")
def client():
    from deepof_amd.models import build_model
    from deepof_amd.serve import create_app

    torch.manual_seed(0)
    model, scales, _ = build_model("flownets")
    app = create_app(model, scales, (127.5,) * 3, torch.device("cpu"),
                     precision="fp32")
    return TestClient(app)


def _img_bytes(seed, h=64, w=96):
    rng = np.random.default_rng(seed)
    arr = rng.integers(0, 255, (h, w, 3), dtype=np.uint8)
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format="PNG")
    return buf.getvalue()


def _payload():
    import base64

    return {"img1": base64.b64encode(_img_bytes(0)).decode(),
            "img2": base64.b64encode(_img_bytes(1)).decode()}


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"
    assert r.json()["model"] == "FlowNetS"


def test_flow_json(client):
    r = client.post("/flow", json=_payload())
    assert r.status_code == 200
    d = r.json()
    assert d["shape"] == [64, 96, 2]
    assert np.isfinite(d["mean_magnitude"])
    assert d["max_magnitude"] >= d["mean_magnitude"] >= 0


def test_flow_flo_bytes(client):
    r = client.post("/flow?format=flo", json=_payload())
    assert r.status_code == 200
    data = r.content
    magic, = struct.unpack_from("<f", data, 0)
    w, h = struct.unpack_from("<ii", data, 4)
    assert abs(magic - 202021.25) < 1e-3
    assert (w, h) == (96, 64)
    flow = np.frombuffer(data[12:], dtype="<f4").reshape(h, w, 2)
    assert np.isfinite(flow).all()


def test_flow_png(client):
    r = client.post("/flow?format=png", json=_payload())
    assert r.status_code == 200
    img = Image.open(io.BytesIO(r.content))
    assert img.size == (96, 64)


def test_metrics_endpoint(client):
    client.post("/flow", json=_payload())
    r = client.get("/metrics")
    assert r.status_code == 200
    body = r.text
    assert "deepof_flow_requests_total" in body
    assert "deepof_flow_latency_seconds" in body


def test_flow_odd_image_size(client):
    """Arbitrary (non-power-of-two) image sizes must round-trip: the
    decoder crops 2x-upsampled skips to odd sizes."""
    import base64

    payload = {"img1": base64.b64encode(_img_bytes(5, h=70, w=90)).decode(),
               "img2": base64.b64encode(_img_bytes(6, h=70, w=90)).decode()}
    r = client.post("/flow", json=payload)
    assert r.status_code == 200
    assert r.json()["shape"] == [70, 90, 2]
