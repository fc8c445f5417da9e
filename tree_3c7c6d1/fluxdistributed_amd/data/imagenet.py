"""ImageNet (ILSVRC) data pipeline — reference /root/reference/src/imagenet.jl.

`labels` parses LOC_synset_mapping.txt, `train_solutions` parses
LOC_train_solution.csv, `minibatch` samples rows and decodes JPEGs with a
thread pool (the reference's Threads.@spawn fproc fan-out,
imagenet.jl:37-48). JPEG decode uses PIL (available offline in this image).

The dataset root replaces the reference's DataSets.jl Data.toml registry
(SURVEY.md §5.6): pass a filesystem path; names are not hard-coded at call
sites (fixing the reference's "imagenet_local" literals).
"""

import csv
import os
import random
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .preprocess import preprocess


def labels(root: str) -> List[Tuple[str, str]]:
    """Parse LOC_synset_mapping.txt -> [(synset_id, description)] in file
    order; row index = class index (reference imagenet.jl:8-21)."""
    path = os.path.join(root, "LOC_synset_mapping.txt")
    out = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            syn, _, desc = line.partition(" ")
            out.append((syn, desc))
    return out


def train_solutions(root: str, classes: Optional[Sequence[int]] = None
                    ) -> List[Tuple[str, int]]:
    """Parse LOC_train_solution.csv -> [(image_id, class_index)], optionally
    filtered to `classes` (reference imagenet.jl:58-75)."""
    syn2idx: Dict[str, int] = {s: i for i, (s, _) in enumerate(labels(root))}
    path = os.path.join(root, "LOC_train_solution.csv")
    rows = []
    with open(path) as f:
        for rec in csv.DictReader(f):
            img_id = rec["ImageId"]
            synset = rec["PredictionString"].split()[0]
            cls = syn2idx[synset]
            if classes is None or cls in classes:
                rows.append((img_id, cls))
    return rows


def shard_key(key: List[Tuple[str, int]], shard: int, nshards: int,
              seed: int = 0) -> List[Tuple[str, int]]:
    """Disjoint contiguous partition of the key, shuffled within the shard —
    the reference's per-device sharding (/root/reference/src/ddp_tasks.jl:
    257-258: partition row range into Ndev chunks, shuffle within chunk).

    Deterministic for a given (shard, nshards, seed); the N shards are
    pairwise disjoint and together cover the whole key.
    """
    if not 0 <= shard < nshards:
        raise ValueError(f"shard {shard} out of range for nshards={nshards}")
    n = len(key)
    lo = shard * n // nshards
    hi = (shard + 1) * n // nshards
    rows = list(key[lo:hi])
    random.Random(seed * 1_000_003 + shard).shuffle(rows)
    return rows


def makepaths(image_id: str, root: str, split: str = "train") -> str:
    """ILSVRC layout (reference imagenet.jl:50-56)."""
    if split == "train":
        synset = image_id.split("_")[0]
        return os.path.join(root, "ILSVRC", "Data", "CLS-LOC", "train", synset,
                            image_id + ".JPEG")
    return os.path.join(root, "ILSVRC", "Data", "CLS-LOC", split, image_id + ".JPEG")


def _fproc(path: str, out_view: torch.Tensor):
    """Decode one JPEG into a (3,224,224) view (reference fproc,
    imagenet.jl:28-35: decode -> preprocess -> per-image standardize)."""
    from PIL import Image

    with Image.open(path) as im:
        arr = np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0
    img = torch.from_numpy(arr).permute(2, 0, 1)  # HWC -> CHW
    x = preprocess(img)
    # reference additionally applies Flux.normalise per-image (imagenet.jl:34)
    x = (x - x.mean()) / (x.std() + 1e-5)
    out_view.copy_(x)


def minibatch(root: str, key: List[Tuple[str, int]], nsamples: int = 32,
              ids: Optional[Sequence[int]] = None, num_workers: int = 8,
              rng: Optional[random.Random] = None
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Random minibatch of decoded, preprocessed images
    (reference imagenet.jl:23-48). Returns (x[N,3,224,224] f32, y[N] i64).

    Decode fan-out uses a bounded thread pool with a hard join before
    return (the reference's @sync barrier; @async deadlocked there —
    imagenet.jl:42-43 comment)."""
    rng = rng or random
    if ids is None:
        ids = [rng.randrange(len(key)) for _ in range(nsamples)]
    rows = [key[i] for i in ids]
    x = torch.zeros(len(rows), 3, 224, 224, dtype=torch.float32)
    y = torch.tensor([cls for _, cls in rows], dtype=torch.long)
    with ThreadPoolExecutor(max_workers=num_workers) as pool:
        futs = [
            pool.submit(_fproc, makepaths(img_id, root), x[i])
            for i, (img_id, _) in enumerate(rows)
        ]
        for f in futs:
            f.result()  # re-raise decode errors; barrier
    return x, y
