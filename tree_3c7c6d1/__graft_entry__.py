"""Driver contract entry points.

build(): compile the gfx950 HIP extension in-tree (cross-compiles on CPU
machines) and import the package.
smoke(): one tiny forward+backward+optimizer step of the flagship model
(ResNet-34, bf16 mixed, channels_last) on cuda:0 through the native kernels.
"""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))


def build() -> None:
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", "8")
    subprocess.run(
        [sys.executable, "setup.py", "build_ext", "--inplace"],
        cwd=ROOT, env=env, check=True,
    )
    sys.path.insert(0, ROOT)
    import fluxdistributed_amd  # noqa: F401
    from fluxdistributed_amd.ops.native import native_available

    assert native_available(), "native extension failed to import after build"
    print("build OK: fluxdistributed_amd._C ready (gfx950)")


def smoke() -> None:
    sys.path.insert(0, ROOT)
    import torch

    assert torch.cuda.is_available(), "smoke() needs a GPU"
    from fluxdistributed_amd.models import build_model
    from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
    from fluxdistributed_amd.ops.native import native_available
    from fluxdistributed_amd.utils.precision import to_mixed_bf16

    assert native_available(), "native extension not loaded"
    dev = torch.device("cuda:0")
    model = build_model("resnet34", num_classes=1000).to(dev)
    model = to_mixed_bf16(model.to(memory_format=torch.channels_last))
    model.train()
    opt = FusedSGDMomentum(model.parameters(), lr=0.01, momentum=0.9)
    x = torch.randn(8, 3, 224, 224, device=dev, dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (8,), device=dev)
    out = model(x)
    loss = logit_cross_entropy(out, y)
    opt.zero_grad()
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss.float()), f"non-finite loss {loss}"
    print(f"smoke OK: resnet34 fwd+bwd+step on {torch.cuda.get_device_name(0)}, "
          f"loss={float(loss):.4f}")


if __name__ == "__main__":
    build()
    if len(sys.argv) > 1 and sys.argv[1] == "smoke":
        smoke()
