"""Minimal serving endpoint for binary inference (production-serving
path on MI355X: packed 1-bit weights + hipGraph replay).

    python -m bdbnn_amd.serve --arch resnet18 --port 8321
    # POST /predict  {"inputs": [[...CHW floats...], ...]}
    #   -> {"top1": [...], "logits_shape": [N, 1000], "latency_ms": ...}

Runs on CPU too (eval-mode model) so the API is testable without a GPU.
"""

import argparse
import time

import torch

from .models import imagenet as imagenet_models
from . import _C


class InferenceService:
    def __init__(self, arch="resnet18", image=224, use_graph=True,
                 batch_capture=None):
        self.image = image
        model = imagenet_models.__dict__[arch](False)
        self.gpu = torch.cuda.is_available() and _C.has_native()
        if self.gpu:
            from .engine import PackedInference
            self.engine = PackedInference(model)
            if use_graph and batch_capture:
                self.engine.capture((batch_capture, 3, image, image))
        else:
            self.model = model.eval()

    @torch.no_grad()
    def predict(self, x: torch.Tensor):
        t0 = time.perf_counter()
        if self.gpu:
            logits = self.engine(x).float().cpu()
        else:
            logits = self.model(x)
        latency = (time.perf_counter() - t0) * 1000
        return logits, latency


def build_app(service: InferenceService):
    from fastapi import FastAPI
    from pydantic import BaseModel

    class PredictRequest(BaseModel):
        inputs: list  # N x (3*H*W) flat or N x 3 x H x W nested

    app = FastAPI(title="bdbnn_amd binary inference")

    @app.get("/health")
    def health():
        return {"status": "ok", "gpu": service.gpu}

    @app.post("/predict")
    def predict(req: PredictRequest):
        x = torch.tensor(req.inputs, dtype=torch.float32)
        if x.dim() == 2:  # flat CHW
            n = x.shape[0]
            x = x.view(n, 3, service.image, service.image)
        logits, latency = service.predict(x)
        top1 = logits.argmax(dim=1).tolist()
        return {"top1": top1, "logits_shape": list(logits.shape),
                "latency_ms": round(latency, 3)}

    return app


def main():
    import uvicorn
    p = argparse.ArgumentParser()
    p.add_argument("--arch", default="resnet18")
    p.add_argument("--port", type=int, default=8321)
    p.add_argument("--image", type=int, default=224)
    p.add_argument("--capture-batch", type=int, default=None)
    args = p.parse_args()
    svc = InferenceService(args.arch, args.image,
                           batch_capture=args.capture_batch)
    uvicorn.run(build_app(svc), host="127.0.0.1", port=args.port)


if __name__ == "__main__":
    main()
