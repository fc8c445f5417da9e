"""Probe: is one graphed step the SAME MATH as one eager step?

Two identically-seeded models train 5 steps on the same fixed batch
sequence — one eager, one through GraphedTrainStep. Parameters must
match to bf16-reorder tolerance; a real semantic difference (stale
data, missed grad, double update) shows up as O(1) drift. Run on GPU."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.engine import make_train_step
from fluxdistributed_amd.utils.precision import to_mixed_bf16


import argparse

ARGS = argparse.Namespace(model="resnet18", batch=16, classes=16, size=32,
                          lr=0.05, steps=5, pool=5)


def mk_model():
    torch.manual_seed(7)
    m = build_model(ARGS.model, num_classes=ARGS.classes,
                    small_input=ARGS.size <= 64).cuda() \
        .to(memory_format=torch.channels_last)
    m = to_mixed_bf16(m)
    return m.train()


def batches(n):
    out = []
    for i in range(n):
        g = torch.Generator().manual_seed(100 + i)
        x = torch.randn(ARGS.batch, 3, ARGS.size, ARGS.size,
                        generator=g).bfloat16().cuda() \
            .contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, ARGS.classes, (ARGS.batch,), generator=g).cuda()
        out.append((x, y))
    return out


def main():
    p = argparse.ArgumentParser()
    for k, v in vars(ARGS).items():
        p.add_argument(f"--{k}", type=type(v), default=v)
    globals()["ARGS"] = p.parse_args()
    pool = batches(ARGS.pool)
    bs = [pool[i % ARGS.pool] for i in range(ARGS.steps)]
    losses = {}
    for mode in ("eager", "graph"):
        m = mk_model()
        opt = FusedSGDMomentum(m.parameters(), lr=ARGS.lr, momentum=0.9)
        step = make_train_step(m, opt, logit_cross_entropy,
                               example_batch=bs[0] if mode == "graph" else None,
                               use_graph=mode == "graph")
        if mode == "eager":
            # match GraphedTrainStep's 3 executed warmup steps on bs[0]
            # (capture itself records without executing)
            for _ in range(3):
                step(*bs[0])
        ls = [float(step(x, y)) for (x, y) in bs]
        torch.cuda.synchronize()
        losses[mode] = ls
        snap = {k: v.detach().float().clone()
                for k, v in m.named_parameters()}
        snap.update({k: v.detach().float().clone()
                     for k, v in m.named_buffers()
                     if "num_batches" not in k})
        if mode == "eager":
            pe = snap
        else:
            pg = snap
    print("eager losses:", [f"{v:.4f}" for v in losses["eager"]])
    print("graph losses:", [f"{v:.4f}" for v in losses["graph"]])
    drifts = []
    for k in pe:
        scale = max(float(pe[k].abs().max()), 1e-2)
        err = float((pe[k] - pg[k]).abs().max())
        drifts.append((err / scale, err, k))
    drifts.sort(reverse=True)
    for rel, ab, k in drifts[:8]:
        print(f"  drift rel={rel:.5f} abs={ab:.5f} {k}")
    print("parity:", "OK" if drifts[0][0] < 0.05 else "BROKEN")


if __name__ == "__main__":
    main()
