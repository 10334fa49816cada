"""Image transforms (PIL + torch; torchvision is not available here).

Reference behavior: torchvision Resize(bilinear, shorter-side) +
Center/RandomCrop + RandomHorizontalFlip + ToTensor + Normalize(0.5, 0.5)
(/root/reference/datasets.py:59-67).
"""
from __future__ import annotations

import random

import numpy as np
import torch
from PIL import Image


def resize_shorter(img: Image.Image, size: int, resample=Image.BILINEAR) -> Image.Image:
    w, h = img.size
    if w <= h:
        nw, nh = size, max(size, round(h * size / w))
    else:
        nw, nh = max(size, round(w * size / h)), size
    return img.resize((nw, nh), resample)


def center_crop(img: Image.Image, size: int) -> Image.Image:
    w, h = img.size
    left = (w - size) // 2
    top = (h - size) // 2
    return img.crop((left, top, left + size, top + size))


def random_crop(img: Image.Image, size: int, rng: random.Random | None = None) -> Image.Image:
    rng = rng or random
    w, h = img.size
    left = rng.randint(0, max(0, w - size))
    top = rng.randint(0, max(0, h - size))
    return img.crop((left, top, left + size, top + size))


def to_tensor(img: Image.Image) -> torch.Tensor:
    arr = np.asarray(img, dtype=np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(arr.copy()).permute(2, 0, 1).float().div_(255.0)
    return t


def normalize(t: torch.Tensor, mean: float = 0.5, std: float = 0.5) -> torch.Tensor:
    return (t - mean) / std


class TrainTransform:
    """resize(shorter=size) -> crop(size) -> maybe-hflip -> [-1,1] tensor."""

    def __init__(self, size: int, center_crop: bool = False, random_flip: bool = False):
        self.size = size
        self.center_crop = center_crop
        self.random_flip = random_flip

    def __call__(self, img: Image.Image) -> torch.Tensor:
        img = resize_shorter(img, self.size)
        img = center_crop(img, self.size) if self.center_crop else random_crop(img, self.size)
        if self.random_flip and random.random() < 0.5:
            img = img.transpose(Image.FLIP_LEFT_RIGHT)
        return normalize(to_tensor(img))


class EvalTransform:
    """resize(shorter=size) -> center crop -> normalize (retrieval/metrics)."""

    def __init__(self, size: int, mean=(0.485, 0.456, 0.406), std=(0.229, 0.224, 0.225)):
        self.size = size
        self.mean = torch.tensor(mean).view(3, 1, 1)
        self.std = torch.tensor(std).view(3, 1, 1)

    def __call__(self, img: Image.Image) -> torch.Tensor:
        img = resize_shorter(img, self.size, Image.BICUBIC)
        img = center_crop(img, self.size)
        t = to_tensor(img)
        if t.shape[0] == 1:
            t = t.expand(3, -1, -1)
        return (t - self.mean) / self.std
