"""Minimal repro driver for the threaded 2-replica abort (see
docs/notes-round3.md). Usage: python tools/fanout_repro.py [nreplicas]"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.parallel.task_ddp import prepare_training, train
from fluxdistributed_amd.utils.precision import to_mixed_bf16


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 2
    dtype = os.environ.get("REPRO_DTYPE", "bf16")
    torch.manual_seed(31)
    m = build_model("resnet18", num_classes=16, small_input=True)
    m = m.to(memory_format=torch.channels_last)
    if dtype == "bf16":
        m = to_mixed_bf16(m)
    m.train()

    def shards(step):
        out = []
        for r in range(n):
            g = torch.Generator().manual_seed(1000 + step * 10 + r)
            x = torch.randn(4, 3, 32, 32, generator=g)
            x = (x.bfloat16() if dtype == "bf16" else x).cuda() \
                .contiguous(memory_format=torch.channels_last)
            y = torch.randint(0, 16, (4,), generator=g).cuda()
            out.append((x, y))
        return out

    st = prepare_training(
        m, None, ["cuda:0"] * n,
        lambda mm: FusedSGDMomentum(mm.parameters(), lr=0.05, momentum=0.9),
    )
    train(logit_cross_entropy, st, steps=3, batches=shards, log_every=0)
    torch.cuda.synchronize()
    print("OK", n, dtype)


if __name__ == "__main__":
    main()
