"""Improved Precision & Recall (manifold metric).

Capability parity: /root/reference/metrics/ipr.py — VGG16-fc2 (4096-d)
features, pairwise L2 distance matrix in float64 (||x||²-2xy+||y||²),
k-NN radii (k=3 default), manifold membership counting, realism score,
.npz manifold caching.
"""
from __future__ import annotations

from collections import namedtuple
from pathlib import Path
from typing import Union

import numpy as np
import torch
from PIL import Image
from torch.utils.data import DataLoader, Dataset

from ..data.transforms import EvalTransform
from ..retrieval.backbones import VGG16

Manifold = namedtuple("Manifold", ["features", "radii"])
PrecisionAndRecall = namedtuple("PrecisionAndRecall", ["precision", "recall"])


class _Paths(Dataset):
    def __init__(self, files, size=224):
        self.files = list(files)
        self.tf = EvalTransform(size)

    def __len__(self):
        return len(self.files)

    def __getitem__(self, i):
        return self.tf(Image.open(self.files[i]).convert("RGB"))


def compute_pairwise_distances(X: np.ndarray, Y: np.ndarray | None = None) -> np.ndarray:
    """Squared-expansion trick in float64 (reference ipr.py:184-219)."""
    X = X.astype(np.float64)
    Y = X if Y is None else Y.astype(np.float64)
    X_norm = (X ** 2).sum(axis=1, keepdims=True)
    Y_norm = (Y ** 2).sum(axis=1, keepdims=True)
    d2 = X_norm - 2 * X.dot(Y.T) + Y_norm.T
    np.maximum(d2, 0, out=d2)
    return np.sqrt(d2)


def distances2radii(distances: np.ndarray, k: int = 3) -> np.ndarray:
    """radius = distance to k-th nearest neighbor (self excluded)."""
    return np.sort(distances, axis=1)[:, k]


class IPR:
    def __init__(self, batch_size: int = 50, k: int = 3, num_samples: int = 10000,
                 model=None, device: str = "cpu"):
        self.batch_size = batch_size
        self.k = k
        self.num_samples = num_samples
        self.device = device
        self.vgg16 = (model or VGG16()).to(device).eval()
        self.manifold_ref = None

    # -- features ----------------------------------------------------------
    @torch.no_grad()
    def extract_features(self, source: Union[str, Path, torch.Tensor]) -> np.ndarray:
        if isinstance(source, torch.Tensor):
            loader = DataLoader(list(source), batch_size=self.batch_size)
        else:
            files = sorted(p for p in Path(source).rglob("*")
                           if p.suffix.lower() in {".png", ".jpg", ".jpeg"})
            files = files[: self.num_samples]
            loader = DataLoader(_Paths(files), batch_size=self.batch_size)
        feats = []
        for batch in loader:
            f = self.vgg16.fc2_features(batch.to(self.device))
            feats.append(f.cpu().numpy())
        return np.concatenate(feats, axis=0)

    def compute_manifold(self, source) -> Manifold:
        if isinstance(source, (str, Path)) and str(source).endswith(".npz"):
            with np.load(str(source)) as f:
                return Manifold(f["features"][:], f["radii"][:])
        feats = self.extract_features(source)
        distances = compute_pairwise_distances(feats)
        return Manifold(feats, distances2radii(distances, self.k))

    def compute_manifold_ref(self, source):
        self.manifold_ref = self.compute_manifold(source)

    def save_ref(self, out_npz):
        np.savez(out_npz, features=self.manifold_ref.features,
                 radii=self.manifold_ref.radii)

    # -- metrics -----------------------------------------------------------
    @staticmethod
    def _membership_fraction(subject_feats, manifold: Manifold) -> float:
        d = compute_pairwise_distances(subject_feats, manifold.features)
        inside = (d < manifold.radii[None, :]).any(axis=1)
        return float(inside.mean())

    def precision_and_recall(self, subject_source) -> PrecisionAndRecall:
        assert self.manifold_ref is not None, "call compute_manifold_ref first"
        manifold_subject = self.compute_manifold(subject_source)
        precision = self._membership_fraction(manifold_subject.features,
                                              self.manifold_ref)
        recall = self._membership_fraction(self.manifold_ref.features,
                                           manifold_subject)
        return PrecisionAndRecall(precision, recall)

    def realism(self, image_feats: np.ndarray) -> float:
        """max over ref of radius/distance (reference ipr.py:255-263),
        computed against the half of the reference manifold with the
        smallest radii (their convention uses the median cut)."""
        ref = self.manifold_ref
        cut = np.median(ref.radii)
        mask = ref.radii < cut
        d = compute_pairwise_distances(image_feats[None] if image_feats.ndim == 1
                                       else image_feats, ref.features[mask])
        ratios = ref.radii[mask][None, :] / np.maximum(d, 1e-10)
        return float(ratios.max())
