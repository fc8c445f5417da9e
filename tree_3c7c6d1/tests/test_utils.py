"""Utility-layer tests: timers, schedules, metrics pretty-printer, loggers."""

import json

import pytest
import torch

from fluxdistributed_amd.utils.timers import StageTimers, Throughput
from fluxdistributed_amd.utils.schedule import step_decay, cosine
from fluxdistributed_amd.utils.metrics import topkaccuracy, showpreds
from fluxdistributed_amd.utils.logging import JSONLLogger


def test_stage_timers_accumulate():
    t = StageTimers()
    with t.stage("fwd"):
        pass
    with t.stage("fwd"):
        pass
    s = t.summary()
    assert "fwd" in s


def test_throughput_counts():
    th = Throughput()
    th.add(10)
    th.add(10)
    assert th.rate() >= 0.0


def test_schedules_move_lr():
    m = torch.nn.Linear(2, 2)
    opt = torch.optim.SGD(m.parameters(), lr=1.0)
    step_decay(opt, base_lr=1.0, factor=0.5, every=1)(2)
    lr_after_decay = opt.param_groups[0]["lr"]
    assert lr_after_decay < 1.0
    cosine(opt, base_lr=1.0, total_cycles=10)(10)
    assert opt.param_groups[0]["lr"] <= lr_after_decay


def test_topk_and_showpreds():
    logits = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1]])
    target = torch.tensor([1, 0])
    assert topkaccuracy(logits, target, k=1) == 1.0
    out = showpreds(logits, target, class_names=["a", "b", "c"], k=2)
    assert "b" in out and "a" in out


def test_jsonl_logger(tmp_path):
    p = tmp_path / "m.jsonl"
    lg = JSONLLogger(str(p), config={"run": "t"})
    lg.log({"loss": 0.5}, step=1)
    lg.finish()
    lines = p.read_text().splitlines()
    assert json.loads(lines[0])["config"]["run"] == "t"
    rec = json.loads(lines[1])
    assert rec["step"] == 1 and rec["loss"] == 0.5


def test_wandb_logger_optional_import():
    """WandbLogger is the one-file adapter (/root/reference/src/loggers/
    wandb.jl:1); wandb is absent offline, so construction must raise
    ImportError and nothing else in the package may depend on it."""
    from fluxdistributed_amd.utils.logging import WandbLogger

    try:
        import wandb  # noqa: F401
        pytest.skip("wandb installed; adapter exercised in wandb envs")
    except ImportError:
        pass
    with pytest.raises(ImportError):
        WandbLogger(project="x")
