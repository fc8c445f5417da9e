"""examples/serve.py endpoint test (the reference's Pluto inference demo,
/root/reference/bin/pluto.jl:373-382, as an HTTP service)."""

import io
import os
import sys

import numpy as np
import pytest
import torch

pytest.importorskip("fastapi")

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples"))


def _jpeg_bytes(w=64, h=48):
    from PIL import Image

    arr = (np.random.default_rng(0).random((h, w, 3)) * 255).astype("uint8")
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format="JPEG")
    return buf.getvalue()


def test_classify_endpoint_topk():
    from fastapi.testclient import TestClient
    from serve import build_app

    from fluxdistributed_amd.models import build_model

    model = build_model("resnet18", num_classes=10, small_input=True).eval()
    app = build_app(model, torch.device("cpu"), names=[f"c{i}" for i in range(10)])
    client = TestClient(app)

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"

    r = client.post("/classify", content=_jpeg_bytes(),
                    headers={"Content-Type": "image/jpeg"})
    assert r.status_code == 200
    preds = r.json()["predictions"]
    assert len(preds) == 3
    probs = [p["prob"] for p in preds]
    assert probs == sorted(probs, reverse=True)
    assert all(0.0 <= p <= 1.0 for p in probs)
    assert all(p["class"].startswith("c") for p in preds)
