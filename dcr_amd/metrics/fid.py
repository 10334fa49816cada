"""FID — Frechet Inception Distance.

Capability parity: /root/reference/metrics/fid.py — pool3 (2048-d)
activations at 299x299, Gaussian statistics, Frechet distance via
scipy.linalg.sqrtm (host-side), recursive path glob, .npz stat caching,
`calculate_fid_given_paths` CLI surface (batch 50, dims 2048 defaults
matching diff_retrieval.py:597-600).
"""
from __future__ import annotations

from pathlib import Path
from typing import Iterable, List, Union

import numpy as np
import torch
from PIL import Image
from scipy import linalg
from torch.utils.data import DataLoader, Dataset

from .inception import InceptionV3

IMG_EXTS = {".bmp", ".jpg", ".jpeg", ".pgm", ".png", ".ppm", ".tif", ".tiff", ".webp"}


class _ImagePathDataset(Dataset):
    def __init__(self, files: List[Path]):
        self.files = files

    def __len__(self):
        return len(self.files)

    def __getitem__(self, i):
        img = Image.open(self.files[i]).convert("RGB")
        arr = np.asarray(img, dtype=np.uint8)
        return torch.from_numpy(arr.copy()).permute(2, 0, 1).float() / 255.0


def _list_images(path: Union[str, Path]) -> List[Path]:
    path = Path(path)
    files = sorted(f for f in path.rglob("*") if f.suffix.lower() in IMG_EXTS)
    if not files:
        raise FileNotFoundError(f"no images under {path}")
    return files


@torch.no_grad()
def get_activations(source, model: InceptionV3, batch_size: int = 50,
                    dims: int = 2048, device: str = "cpu",
                    num_workers: int = 2) -> np.ndarray:
    """source: directory path, list of files, or a [N,3,H,W] float tensor
    in [0,1]."""
    if isinstance(source, torch.Tensor):
        ds = [source[i] for i in range(source.shape[0])]
        loader = DataLoader(ds, batch_size=batch_size)
    else:
        files = _list_images(source) if isinstance(source, (str, Path)) else \
            [Path(f) for f in source]
        loader = DataLoader(_ImagePathDataset(files), batch_size=batch_size,
                            num_workers=num_workers)
    model = model.to(device).eval()
    acts = []
    for batch in loader:
        # variable sizes: resize happens inside the model (299 bilinear)
        pred = model(batch.to(device))[0]
        if pred.dim() == 4 and (pred.shape[2] != 1 or pred.shape[3] != 1):
            pred = torch.nn.functional.adaptive_avg_pool2d(pred, 1)
        acts.append(pred.squeeze(-1).squeeze(-1).cpu().numpy())
    return np.concatenate(acts, axis=0)


def calculate_activation_statistics(source, model, batch_size=50, dims=2048,
                                    device="cpu", num_workers=2):
    act = get_activations(source, model, batch_size, dims, device, num_workers)
    mu = np.mean(act, axis=0)
    sigma = np.cov(act, rowvar=False)
    return mu, sigma


def calculate_frechet_distance(mu1, sigma1, mu2, sigma2, eps: float = 1e-6) -> float:
    """||mu1-mu2||^2 + Tr(s1 + s2 - 2 sqrt(s1 s2)) (reference fid.py:142-196)."""
    mu1, mu2 = np.atleast_1d(mu1), np.atleast_1d(mu2)
    sigma1, sigma2 = np.atleast_2d(sigma1), np.atleast_2d(sigma2)
    diff = mu1 - mu2
    covmean, _ = linalg.sqrtm(sigma1.dot(sigma2), disp=False)
    if not np.isfinite(covmean).all():
        offset = np.eye(sigma1.shape[0]) * eps
        covmean = linalg.sqrtm((sigma1 + offset).dot(sigma2 + offset))
    if np.iscomplexobj(covmean):
        if not np.allclose(np.diagonal(covmean).imag, 0, atol=1e-3):
            raise ValueError(f"Imaginary component {np.max(np.abs(covmean.imag))}")
        covmean = covmean.real
    return float(diff.dot(diff) + np.trace(sigma1) + np.trace(sigma2)
                 - 2 * np.trace(covmean))


def _stats_for_path(path, model, batch_size, dims, device, num_workers):
    path = Path(path) if isinstance(path, (str, Path)) else path
    if isinstance(path, Path) and path.suffix == ".npz":
        with np.load(str(path)) as f:
            return f["mu"][:], f["sigma"][:]
    return calculate_activation_statistics(path, model, batch_size, dims,
                                           device, num_workers)


def calculate_fid_given_paths(paths: Iterable, batch_size: int = 50,
                              device: str = "cpu", dims: int = 2048,
                              num_workers: int = 2,
                              weights_path: str | None = None) -> float:
    """Reference surface (fid.py:239-255): paths = [dir_or_npz, dir_or_npz]."""
    block = InceptionV3.BLOCK_INDEX_BY_DIM[dims]
    model = InceptionV3([block], weights_path=weights_path)
    p1, p2 = list(paths)
    mu1, s1 = _stats_for_path(p1, model, batch_size, dims, device, num_workers)
    mu2, s2 = _stats_for_path(p2, model, batch_size, dims, device, num_workers)
    return calculate_frechet_distance(mu1, s1, mu2, s2)


def save_fid_stats(path, out_npz, batch_size: int = 50, device: str = "cpu",
                   dims: int = 2048, num_workers: int = 2):
    """Cacheable stats (reference fid.py:258-275)."""
    block = InceptionV3.BLOCK_INDEX_BY_DIM[dims]
    model = InceptionV3([block])
    mu, sigma = calculate_activation_statistics(path, model, batch_size, dims,
                                                device, num_workers)
    np.savez(out_npz, mu=mu, sigma=sigma)
