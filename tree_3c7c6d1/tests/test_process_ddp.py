"""Process-DDP over gloo, world_size 2 (the multi-process path that runs
RCCL on GPU boxes — reference src/sync.jl semantics, fixed world size bug)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy


def _mlp(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                               torch.nn.Linear(16, 4))


def _run_ddp_worker(rank, world, port, overlap, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from fluxdistributed_amd.parallel.process_ddp import DDPModel

        model = _mlp(seed=42 + rank)  # ranks start DIFFERENT on purpose
        opt = FusedSGDMomentum(model.parameters(), lr=0.05, momentum=0.9)
        ddp = DDPModel(model, opt, bucket_cap_mb=0.001, overlap=overlap)

        torch.manual_seed(7)  # same data on both ranks -> same behavior as solo
        xs = [torch.randn(4, 8) for _ in range(3)]
        ys = [torch.randint(0, 4, (4,)) for _ in range(3)]
        for x, y in zip(xs, ys):
            out = ddp(x)
            loss = logit_cross_entropy(out, y)
            opt.zero_grad()
            loss.backward()
            ddp.finalize_backward()
            opt.step()
        params = [p.detach().clone() for p in model.parameters()]
        q.put((rank, [p.numpy() for p in params]))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {type(e).__name__}: {e}"))


@pytest.mark.parametrize("overlap", [True, False])
def test_ddp_ranks_converge_identically(overlap):
    """Broadcast at init + averaged grads => both ranks end bit-identical,
    and identical to a solo run on the same data from rank0's init."""
    world = 2
    port = 29600 + (os.getpid() + int(overlap)) % 500
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_run_ddp_worker, args=(r, world, port, overlap, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, payload = q.get(timeout=120)
        assert not isinstance(payload, str), payload
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)

    # ranks identical
    for a, b in zip(results[0], results[1]):
        assert (a == b).all()

    # identical data on every rank => averaged grad == each rank's grad =>
    # equivalent to solo training from rank0's initialization
    solo = _mlp(seed=42)  # rank0's init was broadcast
    opt = FusedSGDMomentum(solo.parameters(), lr=0.05, momentum=0.9)
    torch.manual_seed(7)
    xs = [torch.randn(4, 8) for _ in range(3)]
    ys = [torch.randint(0, 4, (4,)) for _ in range(3)]
    for x, y in zip(xs, ys):
        opt.zero_grad()
        logit_cross_entropy(solo(x), y).backward()
        opt.step()
    for p_solo, arr in zip(solo.parameters(), results[0]):
        assert torch.allclose(p_solo.detach(), torch.from_numpy(arr),
                              rtol=1e-5, atol=1e-6)


def _run_sentinel_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from fluxdistributed_amd.parallel.process_ddp import syncgrads_worker

        model = _mlp(seed=1)
        opt = FusedSGDMomentum(model.parameters(), lr=0.01)
        torch.manual_seed(3)
        batches = [(torch.randn(2, 8), torch.randint(0, 4, (2,))) for _ in range(10)]
        # rank 0 wants to stop at step 2, rank 1 at step 4 ->
        # cooperative stop happens at step 4 (ALL ranks must vote)
        stop_at = 2 if rank == 0 else 4
        steps_done = {"n": 0}

        def should_stop(step):
            steps_done["n"] = step + 1
            return step >= stop_at

        syncgrads_worker(model, opt, logit_cross_entropy, iter(batches),
                         steps=10, should_stop=should_stop)
        q.put((rank, steps_done["n"]))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {type(e).__name__}: {e}"))


def test_cooperative_sentinel_shutdown():
    """All-`nothing` sentinel semantics (sync.jl:49-53): stop only when
    every rank votes stop."""
    world = 2
    port = 29700 + os.getpid() % 500
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_run_sentinel_worker, args=(r, world, port, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    counts = {}
    for _ in range(world):
        rank, n = q.get(timeout=120)
        assert not isinstance(n, str), n
        counts[rank] = n
    for p in procs:
        p.join(timeout=30)
    assert counts[0] == counts[1] == 5  # stopped after step index 4
