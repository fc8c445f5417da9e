"""Pluggable logging — the reference routes all metrics through Julia's
Logging stdlib so any backend (incl. Wandb) can swap in
(/root/reference/src/loggers/wandb.jl, README.md:72-92). Here: stdlib
`logging` plus a MetricsLogger protocol with JSONL and in-memory backends
(wandb itself is not installed in this image; the protocol matches)."""

import json
import logging
import sys
import time
from typing import Dict, Optional


def get_logger(name: str = "fluxdistributed_amd") -> logging.Logger:
    log = logging.getLogger(name)
    if not logging.getLogger("fluxdistributed_amd").handlers:
        root = logging.getLogger("fluxdistributed_amd")
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter("[%(asctime)s %(name)s] %(message)s", "%H:%M:%S"))
        root.addHandler(h)
        root.setLevel(logging.INFO)
    return log


class MetricsLogger:
    """Backend-agnostic metrics sink. `log(dict)` per step; `config` for run
    metadata (the reference's Wandb config dict, README.md:84-87)."""

    def __init__(self, config: Optional[Dict] = None):
        self.config = dict(config or {})
        self.records = []

    def log(self, metrics: Dict, step: Optional[int] = None):
        rec = {"_t": time.time(), **({"step": step} if step is not None else {}), **metrics}
        self.records.append(rec)

    def finish(self):
        pass


class WandbLogger(MetricsLogger):
    """One-file wandb adapter behind an optional import — the reference's
    src/loggers/wandb.jl:1 shim loaded via @require
    (/root/reference/src/FluxDistributed.jl:22-24). Raises ImportError at
    construction when wandb is absent (it is not in this offline image);
    everything else in the framework works without it.
    """

    def __init__(self, project: str = "fluxdistributed-amd",
                 config: Optional[Dict] = None, **init_kw):
        super().__init__(config)
        import wandb  # optional dependency; absent offline

        self._wandb = wandb
        self._run = wandb.init(project=project, config=self.config, **init_kw)

    def get_config(self, key: str):
        """The reference adapter's single method: Wandb.get_config(lg, str)."""
        return self._run.config[key]

    def log(self, metrics, step=None):
        super().log(metrics, step)
        self._wandb.log(metrics, step=step)

    def finish(self):
        self._run.finish()


class JSONLLogger(MetricsLogger):
    def __init__(self, path: str, config: Optional[Dict] = None):
        super().__init__(config)
        self.path = path
        with open(path, "w") as f:
            f.write(json.dumps({"config": self.config}) + "\n")

    def log(self, metrics, step=None):
        super().log(metrics, step)
        with open(self.path, "a") as f:
            f.write(json.dumps(self.records[-1]) + "\n")
