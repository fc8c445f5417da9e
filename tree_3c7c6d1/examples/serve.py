#!/usr/bin/env python3
"""Tiny inference server — the interactive half of the reference's Pluto
demo (webcam frame -> top-3 ImageNet classes) as a FastAPI endpoint.

    python examples/serve.py --checkpoint weights/resnet34_final.pt
    curl --data-binary @cat.jpg -H 'Content-Type: image/jpeg' \
        http://127.0.0.1:8000/classify
"""

import argparse
import io
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.utils.checkpoint import load_checkpoint
from fluxdistributed_amd.data.preprocess import preprocess


def build_app(model, device, names=None, topk=3):
    from fastapi import FastAPI, Request

    app = FastAPI(title="fluxdistributed_amd classifier")

    # raw request body, not multipart form: python-multipart is not in the
    # offline image, and a single image per request needs no form framing
    @app.post("/classify")
    async def classify(request: Request):
        import numpy as np
        from PIL import Image

        data = await request.body()
        with Image.open(io.BytesIO(data)) as im:
            arr = np.asarray(im.convert("RGB"), dtype="float32") / 255.0
        x = preprocess(torch.from_numpy(arr).permute(2, 0, 1)).unsqueeze(0).to(device)
        with torch.no_grad():
            probs = torch.softmax(model(x).float().cpu()[0], dim=-1)
        top = torch.topk(probs, min(topk, probs.shape[-1]))
        return {
            "predictions": [
                {"class": (names[int(j)] if names else int(j)), "prob": float(v)}
                for v, j in zip(top.values, top.indices)
            ]
        }

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(device)}

    return app


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--model", default="resnet34")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--labels", default=None)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()

    model = build_model(args.model, num_classes=args.num_classes)
    load_checkpoint(args.checkpoint, model)
    model.eval()
    device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
    model = model.to(device)
    names = None
    if args.labels:
        from fluxdistributed_amd.data.imagenet import labels

        names = [desc for _, desc in labels(args.labels)]

    import uvicorn

    uvicorn.run(build_app(model, device, names), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
