#!/usr/bin/env python3
"""2-rank data-parallel gradient diagnostic: runs one synchronous step on
two ranks (gloo over one GPU) and compares the post-all-reduce flat
gradient buffers element-wise across ranks, mapping any divergence back to
the owning parameter. This is the tool that caught the double-counted
bucket-readiness bug (profiles/README.md r1.10).

    python tools/dbg2rank.py 1    # overlapped buckets
    python tools/dbg2rank.py 0    # one-shot all-reduce
"""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def worker(rank, world, port, results, overlap):
    import torch.distributed as dist
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK="0")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    torch.manual_seed(1000 + rank)
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
    from fluxdistributed_amd.parallel.process_ddp import DDPModel
    from fluxdistributed_amd.utils.precision import to_mixed_bf16
    model = build_model("resnet18", num_classes=32, small_input=True)
    model = to_mixed_bf16(model.to("cuda:0").to(memory_format=torch.channels_last))
    model.train()
    opt = FusedSGDMomentum(model.parameters(), lr=0.05, momentum=0.9)
    ddp = DDPModel(model, opt, bucket_cap_mb=1.0, overlap=overlap)
    g = torch.Generator().manual_seed(123 + rank)
    x = torch.randn(4, 3, 32, 32, generator=g).bfloat16().cuda() \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 32, (4,), generator=g).cuda()
    out = ddp(x)
    loss = logit_cross_entropy(out, y)
    opt.zero_grad()
    loss.backward()
    ddp.finalize_backward()
    torch.cuda.synchronize()
    results[f"G{rank}"] = [grp.G.float().cpu().clone() for grp in opt.groups]
    if rank == 0:
        names = []
        for gi, grp in enumerate(opt.groups):
            for p, off in zip(grp.params, grp.offsets):
                nm = [n for n, q in model.named_parameters() if q is p]
                names.append((gi, off, p.numel(), nm[0] if nm else "?"))
        results["names"] = names
    dist.destroy_process_group()


def main(overlap):
    import torch.multiprocessing as mp
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(worker, args=(2, 29671, results, overlap), nprocs=2, join=True)
    bad = 0
    for gi in range(len(results["G0"])):
        a, b = results["G0"][gi], results["G1"][gi]
        diff = (a - b).abs()
        print(f"group {gi}: maxdiff {diff.max().item():.3e} "
              f"nnz-diff {int((diff > 0).sum())}")
        if diff.max() > 0:
            bad += 1
            idx = int(diff.argmax())
            for (g2, off, n, nm) in results["names"]:
                if g2 == gi and off <= idx < off + n:
                    print("   diverging param:", nm, "off", off, "n", n)
    print("RESULT:", "DIVERGED" if bad else "OK", "overlap =", overlap)


if __name__ == "__main__":
    main(overlap=len(sys.argv) > 1 and sys.argv[1] == "1")
