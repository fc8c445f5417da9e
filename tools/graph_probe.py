"""Probe: does GraphedTrainStep's replay actually consume new data?

Trains 30 replays on batch A, then one replay on unseen batch B, then A
again. If the static-input copy works, loss(B) must jump well above the
memorized loss(A); if replay is stuck on capture-time data, loss(B)
continues A's trajectory. Run on a GPU box."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.engine import make_train_step
from fluxdistributed_amd.utils.precision import to_mixed_bf16


def batch(seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(16, 3, 32, 32, generator=g).bfloat16().cuda() \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 16, (16,), generator=g).cuda()
    return x, y


def main():
    torch.manual_seed(0)
    m = build_model("resnet18", num_classes=16, small_input=True).cuda() \
        .to(memory_format=torch.channels_last)
    m = to_mixed_bf16(m)
    m.train()
    opt = FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9)
    A, B = batch(1), batch(2)
    step = make_train_step(m, opt, logit_cross_entropy, example_batch=A,
                           use_graph=True)
    la = None
    for _ in range(30):
        la = float(step(*A))
    lb = float(step(*B))
    la2 = float(step(*A))
    print(f"lossA(after 30)={la:.4f} lossB(unseen)={lb:.4f} lossA2={la2:.4f}")
    print("dataflow:", "OK" if lb > max(la, 0.5) * 3 else "BROKEN")


if __name__ == "__main__":
    main()
