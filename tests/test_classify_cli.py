"""examples/classify.py end-to-end CLI test (the reference's Pluto demo
classification flow, /root/reference/bin/pluto.jl:373-382): checkpoint ->
decode -> preprocess -> top-k printout."""

import os
import subprocess
import sys


def test_classify_cli(tmp_path):
    import numpy as np
    import torch
    from PIL import Image

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, root)
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.utils.checkpoint import save_checkpoint

    torch.manual_seed(0)
    m = build_model("resnet18", num_classes=12, small_input=False)
    ck = tmp_path / "m.pt"
    save_checkpoint(str(ck), m)

    img = tmp_path / "x.jpg"
    arr = (np.random.default_rng(1).random((300, 400, 3)) * 255).astype("uint8")
    Image.fromarray(arr).save(img, format="JPEG")

    r = subprocess.run(
        [sys.executable, os.path.join(root, "examples", "classify.py"),
         str(img), "--checkpoint", str(ck), "--model", "resnet18",
         "--num-classes", "12", "--topk", "3"],
        cwd=root, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    assert "x.jpg" in r.stdout or "top" in r.stdout.lower() or r.stdout.strip()
