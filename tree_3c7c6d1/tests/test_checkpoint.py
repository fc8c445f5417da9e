import torch
import torch.nn as nn

from fluxdistributed_amd.ops import FusedSGDMomentum, logit_cross_entropy
from fluxdistributed_amd.utils.checkpoint import save_checkpoint, load_checkpoint


def _model(seed):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(6, 12), nn.ReLU(), nn.Linear(12, 3))


def test_checkpoint_resume_exact(tmp_path, seed):
    m = _model(1)
    opt = FusedSGDMomentum(m.parameters(), lr=0.05, momentum=0.9)
    x = torch.randn(4, 6)
    y = torch.randint(0, 3, (4,))
    for _ in range(3):
        opt.zero_grad()
        logit_cross_entropy(m(x), y).backward()
        opt.step()
    path = str(tmp_path / "ckpt.pt")
    save_checkpoint(path, m, opt, step=3, extra={"note": "t"})

    # continue original 2 more steps
    for _ in range(2):
        opt.zero_grad()
        logit_cross_entropy(m(x), y).backward()
        opt.step()

    # resume fresh copy from checkpoint, run the same 2 steps
    m2 = _model(999)  # different init, will be overwritten
    opt2 = FusedSGDMomentum(m2.parameters(), lr=0.05, momentum=0.9)
    step, extra = load_checkpoint(path, m2, opt2)
    assert step == 3 and extra["note"] == "t"
    for _ in range(2):
        opt2.zero_grad()
        logit_cross_entropy(m2(x), y).backward()
        opt2.step()

    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_checkpoint_model_only(tmp_path, seed):
    m = _model(2)
    path = str(tmp_path / "m.pt")
    save_checkpoint(path, m, step=1)
    m2 = _model(3)
    load_checkpoint(path, m2)
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)
