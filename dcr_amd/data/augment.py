"""Batch-level augmentations (CutMix / Mixup family).

Capability parity: the MAE/DeiT-derived augmentation utilities carried in
/root/reference/utils_ret.py:23-297 (CutMix/SegMix family — vestigial for
the DCR pipelines but part of the library surface)."""
from __future__ import annotations

from typing import Tuple

import numpy as np
import torch


def rand_bbox(size: Tuple[int, ...], lam: float):
    """Random box covering (1-lam) of the image area (CutMix paper)."""
    H, W = size[2], size[3]
    cut_rat = np.sqrt(1.0 - lam)
    cut_w, cut_h = int(W * cut_rat), int(H * cut_rat)
    cx, cy = np.random.randint(W), np.random.randint(H)
    bbx1 = int(np.clip(cx - cut_w // 2, 0, W))
    bby1 = int(np.clip(cy - cut_h // 2, 0, H))
    bbx2 = int(np.clip(cx + cut_w // 2, 0, W))
    bby2 = int(np.clip(cy + cut_h // 2, 0, H))
    return bbx1, bby1, bbx2, bby2


def cutmix_data(x: torch.Tensor, y: torch.Tensor, alpha: float = 1.0):
    """Returns (mixed_x, y_a, y_b, lam)."""
    lam = float(np.random.beta(alpha, alpha)) if alpha > 0 else 1.0
    index = torch.randperm(x.size(0), device=x.device)
    bbx1, bby1, bbx2, bby2 = rand_bbox(x.size(), lam)
    mixed = x.clone()
    mixed[:, :, bby1:bby2, bbx1:bbx2] = x[index, :, bby1:bby2, bbx1:bbx2]
    lam_adj = 1 - ((bbx2 - bbx1) * (bby2 - bby1) / (x.size(-1) * x.size(-2)))
    return mixed, y, y[index], lam_adj


def mixup_data(x: torch.Tensor, y: torch.Tensor, alpha: float = 1.0):
    """Returns (mixed_x, y_a, y_b, lam)."""
    lam = float(np.random.beta(alpha, alpha)) if alpha > 0 else 1.0
    index = torch.randperm(x.size(0), device=x.device)
    mixed = lam * x + (1 - lam) * x[index]
    return mixed, y, y[index], lam


def mixup_criterion(criterion, pred, y_a, y_b, lam: float):
    return lam * criterion(pred, y_a) + (1 - lam) * criterion(pred, y_b)
