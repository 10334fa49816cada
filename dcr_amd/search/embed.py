"""LAION-scale embedding generation (embedding_search L5, part 1).

Capability parity: /root/reference/embedding_search/
download_and_generate_embedding.py + utils.py — embed images with the
SSCD descriptor and dump `embedding.pkl` with the reference's format:
{'features': np.float32 [N, D], 'indexes': list[str]}.

This environment has no network (no img2dataset/webdataset), so sources
are: a local image folder, a list of files, or synthetic random images
("laion-shaped" index for BASELINE config 5). The reference's
webdataset-tar path is represented by the folder path; its CLI arg bugs
(utils.py:58 args.url vs args.tars, 5-vs-4 arg call — SURVEY.md §2.6.5)
are not reproduced.
"""
from __future__ import annotations

import pickle
from pathlib import Path
from typing import List, Optional, Sequence

import numpy as np
import torch
from PIL import Image
from torch.utils.data import DataLoader, Dataset

from ..data.transforms import EvalTransform
from ..retrieval.backbones import load_sscd


class _FolderDataset(Dataset):
    def __init__(self, files: Sequence[Path], size: int = 224):
        self.files = list(files)
        self.tf = EvalTransform(size, mean=(0.485, 0.456, 0.406),
                                std=(0.229, 0.224, 0.225))

    def __len__(self):
        return len(self.files)

    def __getitem__(self, i):
        img = Image.open(self.files[i]).convert("RGB")
        return self.tf(img), i


class SyntheticLAIONDataset(Dataset):
    """Random images with LAION-like keys (no-network BASELINE config 5)."""

    def __init__(self, n: int, size: int = 224, seed: int = 0):
        self.n = n
        self.size = size
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 10_000_019 + i)
        return torch.randn(3, self.size, self.size, generator=g), i

    def key(self, i):
        return f"{self.seed:05d}{i:09d}"


@torch.no_grad()
def extract_features_custom(model, loader: DataLoader, device,
                            use_fp16: bool = False) -> np.ndarray:
    """Forward loop appending CPU features (reference utils.py:78-113)."""
    feats = []
    for batch, _ in loader:
        batch = batch.to(device, non_blocking=True)
        if use_fp16:
            batch = batch.half()
        out = model(batch)
        if out.dim() > 2:
            out = out.flatten(1)
        feats.append(out.float().cpu())
    return torch.cat(feats).numpy().astype(np.float32)


def generate_embeddings(
    source,
    out_pickle: str | Path,
    pt_model: str = "sscd",
    batch_size: int = 128,
    device: Optional[str] = None,
    num_workers: int = 4,
    synthetic_n: Optional[int] = None,
    seed: int = 0,
) -> dict:
    """Embed a source and dump the reference pickle format
    (download_and_generate_embedding.py:89-99)."""
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    model = load_sscd(pt_model, device=device)

    if synthetic_n is not None:
        ds = SyntheticLAIONDataset(synthetic_n, seed=seed)
        indexes: List[str] = [ds.key(i) for i in range(synthetic_n)]
    else:
        src = Path(source)
        files = sorted(f for f in src.rglob("*")
                       if f.suffix.lower() in {".png", ".jpg", ".jpeg", ".webp"})
        ds = _FolderDataset(files)
        indexes = [str(f) for f in files]

    loader = DataLoader(ds, batch_size=batch_size, num_workers=num_workers)
    feats = extract_features_custom(model, loader, device)
    blob = {"features": feats, "indexes": indexes}
    out_pickle = Path(out_pickle)
    out_pickle.parent.mkdir(parents=True, exist_ok=True)
    with open(out_pickle, "wb") as fh:
        pickle.dump(blob, fh, protocol=pickle.HIGHEST_PROTOCOL)
    return blob
