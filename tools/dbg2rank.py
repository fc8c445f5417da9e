import os, sys, torch
sys.path.insert(0, "/root/repo")

def worker(rank, world, port, results, overlap):
    import torch.distributed as dist
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), LOCAL_RANK="0")
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
    x = torch.randn(4, 3, 32, 32, generator=g).bfloat16().cuda().contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 32, (4,), generator=g).cuda()
    # instrument bucket launches
    import fluxdistributed_amd.parallel.bucketing as bk
    seq = []
    pname = {}
    for n_, p_ in model.named_parameters():
        pname[id(p_)] = n_
    events = []
    borig = bk.GradBucketer._on_grad
    def on_grad_patched(self, p):
        bi = self._param2bucket.get(id(p), -1)
        pend = self.buckets[bi]["pending"] if bi >= 0 else -1
        events.append((pname.get(id(p), "??"), bi, pend))
        borig(self, p)
    bk.GradBucketer._on_grad = on_grad_patched
    orig = bk.GradBucketer._launch_ready_locked
    def patched(self):
        nl0 = self._next_launch
        while self._next_launch < len(self.buckets) and self._ready[self._next_launch]:
            b = self.buckets[self._next_launch]
            seg = b["flat"][b["lo"]:b["hi"]]
            torch.cuda.current_stream(seg.device).synchronize()
            pre = float(seg.float().sum())
            import torch.distributed as dist2
            dist2.all_reduce(seg, op=dist2.ReduceOp.SUM, group=self.pg)
            post = float(seg.float().sum())
            seq.append((self._next_launch, int(seg.numel()), b["lo"], pre, post))
            b["postcopy"] = seg.clone()
            self._next_launch += 1
    bk.GradBucketer._launch_ready_locked = patched
    out = ddp(x); loss = logit_cross_entropy(out, y)
    opt.zero_grad(); loss.backward()
    torch.cuda.synchronize()
    # BEFORE finalize: which buckets changed since their reduce?
    changed = []
    for bi, b in enumerate(ddp.bucketer.buckets):
        if "postcopy" in b:
            seg = b["flat"][b["lo"]:b["hi"]]
            d = (seg.float() - b["postcopy"].float()).abs()
            if d.max() > 0:
                changed.append((bi, float(d.max()), int(d.argmax()) + b["lo"],
                                int((d > 0).sum())))
    results[f"changed{rank}"] = changed
    ddp.finalize_backward()
    torch.cuda.synchronize()
    results[f"seq{rank}"] = seq
    if rank == 0:
        results["events"] = events
        results["n_notify"] = len(bk._NOTIFY)
    results[f"G{rank}"] = [grp.G.float().cpu().clone() for grp in opt.groups]
    # map offsets to names
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
    mgr = mp.Manager(); results = mgr.dict()
    mp.spawn(worker, args=(2, 29671, results, overlap), nprocs=2, join=True)
    for gi in range(len(results["G0"])):
        a, b = results["G0"][gi], results["G1"][gi]
        diff = (a - b).abs()
        print(f"group {gi}: maxdiff {diff.max().item():.3e} at {int(diff.argmax())} nnz-diff {int((diff>0).sum())}")
        if diff.max() > 0:
            idx = int(diff.argmax())
            for (g2, off, n, nm) in results["names"]:
                if g2 == gi and off <= idx < off + n:
                    print("   diverging param:", nm, "off", off, "n", n)
    print("changed after reduce, rank0:", results["changed0"])
    print("changed after reduce, rank1:", results["changed1"])
    print("n_notify subscribers:", results.get("n_notify"))
    from collections import Counter
    cnt = Counter(e[0] for e in results["events"])
    dups = {k: v for k, v in cnt.items() if v > 1}
    print("params with >1 grad event:", dups)
    print("first 20 events:", results["events"][:20])
    s0, s1 = results["seq0"], results["seq1"]
    print("seq lens", len(s0), len(s1))
    for a, b in zip(s0, s1):
        mark = "" if abs(a[4]-b[4]) < 1e-3 else "   <-- POST MISMATCH"
        print(f"b{a[0]:02d} n={a[1]:8d} lo={a[2]:8d} pre0={a[3]:+.4f} pre1={b[3]:+.4f} post0={a[4]:+.4f} post1={b[4]:+.4f}{mark}")
    print("done overlap=", overlap)

if __name__ == "__main__":
    main(overlap=sys.argv[1] == "1")
