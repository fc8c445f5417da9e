"""Image preprocessing (reference /root/reference/src/preprocess.jl:26-81):
resize smallest dimension to 256 with Gaussian lowpass when downscaling,
center-crop 224, ImageNet mean/std normalize. Implemented on torch CPU
tensors (CHW float in [0,1])."""

import math
from typing import Tuple

import torch
import torch.nn.functional as F

IMAGENET_MEAN = torch.tensor([0.485, 0.456, 0.406]).view(3, 1, 1)
IMAGENET_STD = torch.tensor([0.229, 0.224, 0.225]).view(3, 1, 1)


def _gaussian_kernel1d(sigma: float) -> torch.Tensor:
    radius = max(1, int(math.ceil(3.0 * sigma)))
    x = torch.arange(-radius, radius + 1, dtype=torch.float32)
    k = torch.exp(-(x ** 2) / (2 * sigma * sigma))
    return k / k.sum()


def resize_smallest_dimension(img: torch.Tensor, target: int = 256) -> torch.Tensor:
    """Resize so min(H, W) == target; Gaussian lowpass before downscaling
    (reference preprocess.jl:30-42 uses σ = 0.75 * inv_scale)."""
    c, h, w = img.shape
    scale = target / min(h, w)
    nh, nw = max(target, round(h * scale)), max(target, round(w * scale))
    x = img.unsqueeze(0)
    if scale < 1.0:
        sigma = 0.75 / scale
        k = _gaussian_kernel1d(sigma).to(img.dtype)
        r = (len(k) - 1) // 2
        kx = k.view(1, 1, 1, -1).expand(c, 1, 1, -1)
        ky = k.view(1, 1, -1, 1).expand(c, 1, -1, 1)
        x = F.pad(x, (r, r, r, r), mode="reflect")
        x = F.conv2d(x, ky, groups=c)
        x = F.conv2d(x, kx, groups=c)
    x = F.interpolate(x, size=(nh, nw), mode="bilinear", align_corners=False,
                      antialias=False)
    return x.squeeze(0)


def center_crop(img: torch.Tensor, size: int = 224) -> torch.Tensor:
    """(reference preprocess.jl:45-49)"""
    c, h, w = img.shape
    top = (h - size) // 2
    left = (w - size) // 2
    return img[:, top : top + size, left : left + size]


def preprocess(img: torch.Tensor, crop: int = 224, resize: int = 256,
               normalize: bool = True) -> torch.Tensor:
    """Full pipeline (reference preprocess.jl:51-70). Input CHW float [0,1];
    output CHW float32, ImageNet-normalized."""
    x = resize_smallest_dimension(img, resize)
    x = center_crop(x, crop)
    if normalize:
        x = (x - IMAGENET_MEAN.to(x.dtype)) / IMAGENET_STD.to(x.dtype)
    return x.float()


def topk_probs(logits: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """(reference preprocess.jl:74-75)"""
    probs = torch.softmax(logits.float(), dim=-1)
    return torch.topk(probs, k, dim=-1)
