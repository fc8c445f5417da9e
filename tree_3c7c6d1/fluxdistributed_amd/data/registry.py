"""Dataset registry — the reference's DataSets.jl Data.toml equivalent
(/root/reference/Data.toml:1-27): named datasets resolve to storage roots,
so dataset names are configuration, not hard-coded call-site literals
(fixing the reference's "imagenet_local" literals).

Registry file: Data.yaml at the repo root (or $FLUXDIST_DATA_CONFIG), e.g.

    datasets:
      imagenet_local:
        driver: filesystem
        path: /data/imagenet
      cifar10:
        driver: filesystem
        path: /data/cifar10
"""

import os
from typing import Dict, Optional

import yaml

_DEFAULT_PATHS = ("Data.yaml", "data.yaml")


def _config_path() -> Optional[str]:
    env = os.environ.get("FLUXDIST_DATA_CONFIG")
    if env and os.path.exists(env):
        return env
    for base in (os.getcwd(), os.path.dirname(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))):
        for name in _DEFAULT_PATHS:
            p = os.path.join(base, name)
            if os.path.exists(p):
                return p
    return None


def load_registry() -> Dict[str, dict]:
    path = _config_path()
    if path is None:
        return {}
    with open(path) as f:
        cfg = yaml.safe_load(f) or {}
    return cfg.get("datasets", {})


def dataset(name: str) -> str:
    """Resolve a dataset name to its filesystem root. A literal path that
    exists is accepted directly (convenience)."""
    reg = load_registry()
    if name in reg:
        entry = reg[name]
        if entry.get("driver", "filesystem") != "filesystem":
            raise ValueError(f"dataset '{name}': unsupported driver "
                             f"{entry.get('driver')!r} (offline image)")
        return entry["path"]
    if os.path.exists(name):
        return name
    raise KeyError(
        f"dataset '{name}' not in registry ({sorted(reg) or 'empty'}) and not a path"
    )
