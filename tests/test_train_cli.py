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
