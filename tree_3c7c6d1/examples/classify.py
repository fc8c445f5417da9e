#!/usr/bin/env python3
"""Inference demo — the reference's Pluto notebook (bin/pluto.jl) re-expressed
as a CLI: load a trained checkpoint, classify images, print top-k
predictions. (The reference's webcam UI becomes --serve, a small HTTP
endpoint, in examples/serve.py.)

    python examples/classify.py --checkpoint weights/resnet34_final.pt \
        --model resnet34 --labels /data/imagenet img1.jpg img2.jpg
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.utils.checkpoint import load_checkpoint
from fluxdistributed_amd.utils.metrics import showpreds
from fluxdistributed_amd.data.preprocess import preprocess


def load_image(path: str) -> torch.Tensor:
    import numpy as np
    from PIL import Image

    with Image.open(path) as im:
        arr = np.asarray(im.convert("RGB"), dtype="float32") / 255.0
    return preprocess(torch.from_numpy(arr).permute(2, 0, 1))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("images", nargs="+")
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--model", default="resnet34")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--labels", default=None,
                   help="dataset root containing LOC_synset_mapping.txt")
    p.add_argument("--topk", type=int, default=3)
    args = p.parse_args()

    model = build_model(args.model, num_classes=args.num_classes)
    load_checkpoint(args.checkpoint, model)
    model.eval()
    device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
    model = model.to(device)
    if device.type == "cuda":
        # inference on the native kernel path: channels_last bf16
        from fluxdistributed_amd.utils.precision import to_mixed_bf16

        model = to_mixed_bf16(model.to(memory_format=torch.channels_last))

    names = None
    if args.labels:
        from fluxdistributed_amd.data.imagenet import labels

        names = [desc for _, desc in labels(args.labels)]

    x = torch.stack([load_image(pth) for pth in args.images]).to(device)
    if device.type == "cuda":
        x = x.bfloat16().contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        logits = model(x).float().cpu()
    probs = torch.softmax(logits, dim=-1)
    top = torch.topk(probs, min(args.topk, probs.shape[-1]), dim=-1)
    for i, pth in enumerate(args.images):
        preds = ", ".join(
            f"{(names[j] if names else str(int(j)))}: {float(v):.3f}"
            for v, j in zip(top.values[i], top.indices[i])
        )
        print(f"{pth}: {preds}")


if __name__ == "__main__":
    main()
