"""Serving endpoint (runners/serve.py): health + micro-batched predict on a
tiny CPU model."""

import io

import numpy as np
import pytest
import torch
from PIL import Image


@pytest.fixture(scope="module")
def client():
    from starlette.testclient import TestClient

    from deepfake_detection_amd.runners.serve import InferenceEngine, create_app

    class TinyEngine(InferenceEngine):
        def __init__(self):
            # bypass the B7-scale constructor: tiny conv net with the same
            # (B, 12, 600, 600) -> (B, 2) contract
            self.device = "cpu"
            self.dtype = torch.float32
            self.model = torch.nn.Sequential(
                torch.nn.Conv2d(12, 4, 3, stride=8), torch.nn.AdaptiveAvgPool2d(1),
                torch.nn.Flatten(), torch.nn.Linear(4, 2), torch.nn.Softmax(-1)).eval()
            self.max_batch = 4
            self.batch_window_s = 0.005
            import asyncio

            self._queue = asyncio.Queue()
            self._worker = None

    engine = TinyEngine()
    with TestClient(create_app(engine)) as c:
        yield c


def _png_bytes():
    img = Image.fromarray((np.random.rand(64, 48, 3) * 255).astype(np.uint8))
    buf = io.BytesIO()
    img.save(buf, format="PNG")
    return buf.getvalue()


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"


def test_predict(client):
    r = client.post("/predict", content=_png_bytes(),
                    headers={"Content-Type": "application/octet-stream"})
    assert r.status_code == 200
    body = r.json()
    assert 0.0 <= body["fake_probability"] <= 1.0
    assert body["label"] in ("fake", "real")


def test_predict_concurrent_batched(client):
    import concurrent.futures as cf

    def one(_):
        return client.post("/predict", content=_png_bytes(),
                           headers={"Content-Type": "application/octet-stream"})

    with cf.ThreadPoolExecutor(8) as ex:
        rs = list(ex.map(one, range(8)))
    assert all(r.status_code == 200 for r in rs)
