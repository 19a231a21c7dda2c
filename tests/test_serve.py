"""Serving endpoint (CPU path; the GPU path reuses PackedInference which
is covered by the gpu-marked tests)."""

import numpy as np
import torch

from bdbnn_amd.serve import InferenceService, build_app


def test_inference_service_cpu():
    svc = InferenceService("resnet18", image=64)
    logits, latency = svc.predict(torch.randn(2, 3, 64, 64))
    assert logits.shape == (2, 1000)
    assert latency > 0


def test_fastapi_endpoint():
    from starlette.testclient import TestClient
    svc = InferenceService("resnet18", image=32)
    app = build_app(svc)
    client = TestClient(app)
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    x = np.random.randn(2, 3 * 32 * 32).astype("float32")
    r = client.post("/predict", json={"inputs": x.tolist()})
    assert r.status_code == 200
    body = r.json()
    assert len(body["top1"]) == 2
    assert body["logits_shape"] == [2, 1000]
