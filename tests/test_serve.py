"""Serving endpoint tests (CPU eager path via TestClient)."""

import io

import numpy as np
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from PIL import Image  # noqa: E402


@pytest.fixture(scope="module")
def client():
    from serve import create_app

    return TestClient(create_app())


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"


def test_enhance_roundtrip(client):
    rng = np.random.default_rng(0)
    img = rng.integers(0, 256, size=(32, 40, 3), dtype=np.uint8)
    buf = io.BytesIO()
    Image.fromarray(img).save(buf, format="PNG")
    r = client.post("/enhance", content=buf.getvalue())
    assert r.status_code == 200
    out = np.asarray(Image.open(io.BytesIO(r.content)).convert("RGB"))
    assert out.shape == img.shape and out.dtype == np.uint8


@pytest.mark.gpu
def test_serve_gpu_engine():
    """On a GPU the server routes /8-divisible frames through the hipGraph
    InferenceEngine and caches one engine per resolution."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    from serve import Server

    s = Server()
    rng = np.random.default_rng(1)
    img = rng.integers(0, 256, size=(64, 80, 3), dtype=np.uint8)
    out1 = s.enhance(img)
    out2 = s.enhance(img)
    assert out1.shape == img.shape
    assert (out1 == out2).all()
    assert (64, 80) in s._engines


def test_enhance_rejects_garbage_body(client):
    r = client.post("/enhance", content=b"this is not an image")
    assert r.status_code == 400
    assert "not a decodable image" in r.text
