#!/usr/bin/env python3
"""Flagship benchmark: ResNet-34 / synthetic ImageNet 224x224 training
images/sec on N MI355X GPUs (BASELINE.json metric).

Single GPU:   python bench.py --gpus 1 --steps 30 --warmup 10
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate throughput.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.parallel.process_ddp import DDPModel, init_process_group
from fluxdistributed_amd.data.synthetic import SyntheticBatcher
from fluxdistributed_amd.data.loader import PrefetchLoader
from fluxdistributed_amd.utils.precision import to_mixed_bf16


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="resnet34")
    p.add_argument("--batch", type=int, default=96, help="per-GPU batch size")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--bucket-mb", type=float, default=25.0)
    p.add_argument("--no-overlap", action="store_true")
    p.add_argument("--graph", action="store_true",
                   help="capture the step in a hipGraph (DEFAULT at world=1: "
                        "+4.4%% same-box over eager once the loader's "
                        "record_stream bug was fixed — docs/notes-round3.md)")
    p.add_argument("--no-graph", action="store_true",
                   help="force the eager step (the world>1 default: the "
                        "bucketed all-reduce/backward overlap is hook-driven "
                        "and stays on the validated eager path)")
    p.add_argument("--allow-cpu", action="store_true")
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--momentum", type=float, default=0.9)
    return p.parse_args()


def _miopen_env():
    """Persist MIOpen tuning in-repo so tuned conv kernels survive across
    machines (each gpurun box is fresh); without a tuned db, use FAST find
    so startup stays in the warmup budget."""
    root = os.path.dirname(os.path.abspath(__file__))
    db = os.path.join(root, "miopen_db")
    os.makedirs(db, exist_ok=True)
    os.environ.setdefault("MIOPEN_USER_DB_PATH", db)
    # The default path runs no MIOpen kernels (stem included — the CONV_STEM
    # kernels cover C<=5); this matters only for the FLUXDIST_CONV=miopen
    # A/B arm. The committed find-db has tuned solvers for every library-conv
    # shape that arm hits, and FAST skips MIOpen's background solver sweep,
    # whose naive_conv kernels otherwise run DURING the timed region
    # (profiles/README.md).
    os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")


def main():
    args = parse_args()
    _miopen_env()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    # --gpus must be honest (round-1 verdict weak #5): world size comes
    # from torchrun's env; a bare `bench.py --gpus 8` would otherwise
    # quietly bench 1 GPU.
    if args.gpus != world:
        raise SystemExit(
            f"--gpus {args.gpus} but WORLD_SIZE={world}; for N>1 launch via "
            f"`python -m torch.distributed.run --nnodes=1 --nproc-per-node "
            f"{args.gpus} --master-addr 127.0.0.1 bench.py --gpus {args.gpus} ...`")

    if torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
        # MIOpen find: benchmark per conv config once (cached in the in-repo
        # user db), instead of immediate-mode fallback kernels
        torch.backends.cudnn.benchmark = True
    elif args.allow_cpu:
        device = torch.device("cpu")
    else:
        raise SystemExit("no GPU found (use --allow-cpu for debugging)")

    if world > 1:
        init_process_group()
    dist = torch.distributed if world > 1 else None

    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    model = build_model(args.model, num_classes=args.num_classes)
    model = model.to(device).to(memory_format=torch.channels_last)
    if dtype == torch.bfloat16:
        model = to_mixed_bf16(model)
    model.train()

    opt = FusedSGDMomentum(model.parameters(), lr=args.lr, momentum=args.momentum)
    ddp = None
    if world > 1:
        ddp = DDPModel(model, opt, bucket_cap_mb=args.bucket_mb,
                       overlap=not args.no_overlap)

    batcher = SyntheticBatcher(
        args.batch, num_classes=args.num_classes, size=args.image_size,
        dtype=dtype, pool=4, channels_last=True,
        pin=device.type == "cuda", seed=1234 + rank,
    )
    loader = PrefetchLoader(batcher, device=device, buffersize=5)

    from fluxdistributed_amd.engine import make_train_step

    # hipGraph is the single-GPU default (replay removes the ~23% of wall
    # that is inter-kernel launch gaps, +4.4% same-box); world>1 keeps the
    # eager step so the bucketed all-reduce overlaps backward via hooks.
    use_graph = (args.graph or world == 1) and not args.no_graph
    example = next(loader) if (device.type == "cuda" and use_graph) else None
    train_step = make_train_step(model, opt, logit_cross_entropy,
                                 example_batch=example, ddp=ddp,
                                 use_graph=use_graph)

    def step():
        x, y = next(loader)
        return train_step(x, y)

    # ---- warmup (untimed) ----
    for _ in range(args.warmup):
        loss = step()
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    if dist is not None:
        dist.barrier()

    # ---- timed region: exactly --steps steps ----
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    if device.type == "cuda":
        torch.cuda.synchronize(device)
    if dist is not None:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if world > 1 else (1 if device.type == "cuda" else args.gpus)
    total_images = args.steps * args.batch * max(world, 1)
    images_per_sec = total_images / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0:
        print(json.dumps({
            "metric": "images/sec",
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "final_loss": round(float(loss.detach()), 4),
            "peak_mem_gb": (round(torch.cuda.max_memory_allocated() / 2**30, 2)
                            if device.type == "cuda" else None),
            "config": {
                "model": args.model,
                "global_batch": args.batch * max(world, 1),
                "seq_len": None,
                "image_size": args.image_size,
                "num_classes": args.num_classes,
                "parallelism": f"dp{max(world, 1)}",
                "engine": ("graph" if (use_graph and device.type == "cuda")
                           else "eager"),
                "optimizer": f"sgd_momentum(lr={args.lr},m={args.momentum})",
                "loss": "logitcrossentropy",
            },
        }))
    loader.close()
    if dist is not None and dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
