#!/usr/bin/env python3
"""Training driver — the reference's bin/driver.jl + bin/main.jl equivalent.

Task mode (one process, N devices — reference ddp_tasks.jl path):
    python train.py --mode task --devices 2 --model resnet34 --steps 100

Process mode (one process per GPU over RCCL — reference sync.jl path):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --mode process --model resnet50

Data: --data synthetic (default) or --data <name-or-path> for an ILSVRC
tree registered in Data.yaml (see fluxdistributed_amd/data/registry.py).
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from fluxdistributed_amd.models import build_model
from fluxdistributed_amd.ops import FusedSGDMomentum, FusedAdam, logit_cross_entropy
from fluxdistributed_amd.parallel.task_ddp import prepare_training, train
from fluxdistributed_amd.parallel.process_ddp import init_process_group, syncgrads_worker
from fluxdistributed_amd.data.synthetic import SyntheticBatcher
from fluxdistributed_amd.data.loader import PrefetchLoader
from fluxdistributed_amd.utils.device import resolve_devices
from fluxdistributed_amd.utils.precision import to_mixed_bf16
from fluxdistributed_amd.utils.checkpoint import save_checkpoint, load_checkpoint
from fluxdistributed_amd.utils.logging import get_logger

log = get_logger("train")


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--mode", choices=["task", "process"], default="task")
    p.add_argument("--model", default="resnet34")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--small-input", action="store_true")
    p.add_argument("--devices", type=int, default=None,
                   help="task mode: number of devices (default: all GPUs, or 1)")
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--batch", type=int, default=96)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--dtype", default="bf16" if torch.cuda.is_available() else "fp32",
                   choices=["bf16", "fp32"])
    p.add_argument("--optimizer", choices=["momentum", "adam"], default="momentum")
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--data", default="synthetic",
                   help="'synthetic' or a dataset name/path (ILSVRC layout)")
    p.add_argument("--nsamples", type=int, default=None,
                   help="imagenet: images per minibatch (default: --batch)")
    p.add_argument("--classes", type=int, nargs="*", default=None,
                   help="imagenet: restrict to these class indices")
    p.add_argument("--buffersize", type=int, default=5)
    p.add_argument("--val-every", type=int, default=50)
    p.add_argument("--log-every", type=int, default=10)
    p.add_argument("--checkpoint-every", type=int, default=0)
    p.add_argument("--checkpoint-dir", default="weights")
    p.add_argument("--resume", default=None)
    p.add_argument("--seed", type=int, default=42,
                   help="base seed for data sharding / per-rank RNG")
    return p.parse_args()


def make_opt_factory(args):
    if args.optimizer == "adam":
        return lambda m: FusedAdam(m.parameters(), lr=args.lr)
    return lambda m: FusedSGDMomentum(m.parameters(), lr=args.lr,
                                      momentum=args.momentum)


def make_batch_fn(args, rank=0, world=1):
    """Returns data(nsamples) -> (x, y) host batches for one rank/replica.

    ImageNet keys are sharded rank-disjoint with a deterministic per-rank
    RNG (the reference's per-device sharding, ddp_tasks.jl:257-258);
    synthetic data gets a per-rank seed.
    """
    import random

    if args.data == "synthetic":
        dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
        b = SyntheticBatcher(args.batch, num_classes=args.num_classes,
                             size=args.image_size, dtype=dtype,
                             pin=torch.cuda.is_available(), seed=1000 + rank)
        return lambda n=None: b()
    from fluxdistributed_amd.data.registry import dataset
    from fluxdistributed_amd.data.imagenet import (
        minibatch, shard_key, train_solutions,
    )

    root = dataset(args.data)
    key = train_solutions(root, classes=args.classes)
    shard = shard_key(key, rank, world, seed=args.seed)
    rng = random.Random(args.seed * 7_919 + rank)
    ns = args.nsamples or args.batch
    return lambda n=None: minibatch(root, shard, nsamples=n or ns, rng=rng)


def run_task(args):
    devices = resolve_devices(args.devices)
    log.info("task-DDP on %s", devices)
    model = build_model(args.model, num_classes=args.num_classes,
                        small_input=args.small_input)
    if torch.cuda.is_available():
        model = model.to(memory_format=torch.channels_last)
        if args.dtype == "bf16":
            model = to_mixed_bf16(model)
    st = prepare_training(
        model,
        [make_batch_fn(args, rank=i, world=len(devices))
         for i in range(len(devices))],
        devices, make_opt_factory(args),
        nsamples=args.batch, buffersize=args.buffersize,
    )
    if args.resume:
        for r in st.replicas:
            load_checkpoint(args.resume, r.model, r.optimizer)

    def on_cycle_end(cycle, state):
        if args.checkpoint_every and cycle % args.checkpoint_every == 0:
            r = state.replicas[0]
            save_checkpoint(
                os.path.join(args.checkpoint_dir, f"{args.model}_cycle{cycle}.pt"),
                r.model, r.optimizer, step=cycle,
            )

    results = train(logit_cross_entropy, st, steps=args.steps,
                    log_every=args.log_every, val_every=args.val_every,
                    on_cycle_end=on_cycle_end)
    log.info("done: %d cycles, %d missed; timers=%s", st.cycles, st.num_missed,
             st.timers.summary())
    r = st.replicas[0]
    save_checkpoint(os.path.join(args.checkpoint_dir, f"{args.model}_final.pt"),
                    r.model, r.optimizer, step=st.cycles)
    return results


def run_process(args):
    rank, world = init_process_group()
    device = (torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}")
              if torch.cuda.is_available() else torch.device("cpu"))
    log.info("process-DDP rank %d/%d on %s", rank, world, device)
    model = build_model(args.model, num_classes=args.num_classes,
                        small_input=args.small_input).to(device)
    if device.type == "cuda":
        model = model.to(memory_format=torch.channels_last)
        if args.dtype == "bf16":
            model = to_mixed_bf16(model)
    opt = make_opt_factory(args)(model)
    if args.resume:
        load_checkpoint(args.resume, model, opt)

    batch_fn = make_batch_fn(args, rank, world)
    loader = PrefetchLoader(batch_fn, device=device, buffersize=args.buffersize)
    model_, opt_, stats = syncgrads_worker(
        model, opt, logit_cross_entropy, loader, steps=args.steps,
        checkpoint_every=args.checkpoint_every,
        checkpoint_dir=args.checkpoint_dir,
    )
    if rank == 0:
        log.info("done: %s", stats)
        save_checkpoint(os.path.join(args.checkpoint_dir, f"{args.model}_final.pt"),
                        model_, opt_, step=args.steps)
    loader.close()
    torch.distributed.destroy_process_group()


def main():
    args = parse_args()
    if args.mode == "process" or int(os.environ.get("WORLD_SIZE", "1")) > 1:
        run_process(args)
    else:
        run_task(args)


if __name__ == "__main__":
    main()
