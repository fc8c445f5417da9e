"""End-to-end CLI smoke: train.py task mode on CPU with 2 logical devices
(the reference's bin/driver.jl usage, CPU fake-device contract)."""

import json
import os
import subprocess
import sys


def test_train_cli_task_mode(tmp_path):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ckdir = tmp_path / "w"
    cmd = [sys.executable, os.path.join(root, "train.py"),
           "--mode", "task", "--devices", "2", "--model", "resnet18",
           "--small-input", "--steps", "2", "--batch", "4",
           "--image-size", "32", "--num-classes", "8", "--dtype", "fp32",
           "--data", "synthetic", "--checkpoint-dir", str(ckdir),
           "--log-every", "1", "--val-every", "0"]
    r = subprocess.run(cmd, cwd=root, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    final = ckdir / "resnet18_final.pt"
    assert final.exists(), list(ckdir.iterdir()) if ckdir.exists() else "no dir"

    import torch

    ck = torch.load(final, map_location="cpu", weights_only=False)
    assert "model" in ck and "optimizer" in ck


def test_train_cli_process_mode_world2(tmp_path):
    """train.py process mode launched the way the driver launches it:
    torchrun, 2 ranks, 127.0.0.1 rendezvous — gloo on CPU (the same
    run_process() code path RCCL takes on GPUs)."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ckdir = tmp_path / "w"
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--standalone",
           "--local-addr", "127.0.0.1",
           os.path.join(root, "train.py"),
           "--mode", "process", "--model", "resnet18", "--small-input",
           "--steps", "2", "--batch", "4", "--image-size", "32",
           "--num-classes", "8", "--dtype", "fp32", "--data", "synthetic",
           "--checkpoint-dir", str(ckdir), "--log-every", "1",
           "--val-every", "0"]
    r = subprocess.run(cmd, cwd=root, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    assert (ckdir / "resnet18_final.pt").exists(), \
        list(ckdir.iterdir()) if ckdir.exists() else "no dir"


def test_train_cli_resume(tmp_path):
    """--checkpoint-every writes a per-cycle checkpoint; --resume loads
    model AND optimizer state into every replica (train.py wiring, not
    just utils.checkpoint) and training continues from it."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ckdir = tmp_path / "w"
    base = [sys.executable, os.path.join(root, "train.py"),
            "--mode", "task", "--devices", "2", "--model", "resnet18",
            "--small-input", "--batch", "4", "--image-size", "32",
            "--num-classes", "8", "--dtype", "fp32", "--data", "synthetic",
            "--checkpoint-dir", str(ckdir), "--log-every", "0",
            "--val-every", "0"]
    r = subprocess.run(base + ["--steps", "2", "--checkpoint-every", "1"],
                       cwd=root, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    mid = ckdir / "resnet18_cycle1.pt"
    assert mid.exists(), list(ckdir.iterdir())

    r2 = subprocess.run(base + ["--steps", "1", "--resume", str(mid)],
                        cwd=root, capture_output=True, text=True, timeout=600)
    assert r2.returncode == 0, r2.stderr[-2000:]

    import torch

    before = torch.load(mid, map_location="cpu", weights_only=False)
    after = torch.load(ckdir / "resnet18_final.pt", map_location="cpu",
                       weights_only=False)
    # optimizer state survived the round-trip and training moved the params
    assert before["optimizer"], "checkpoint missing optimizer state"
    moved = any(not torch.equal(before["model"][k], after["model"][k])
                for k in before["model"] if k.endswith("weight"))
    assert moved, "resume produced identical params (no training happened)"
