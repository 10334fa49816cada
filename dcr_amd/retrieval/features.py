"""Distributed feature extraction (the reference's hot retrieval loop).

Capability parity: /root/reference/utils_ret.py:704-787 extract_features —
batched forward over a (Distributed)Sampler'd dataset, per-batch
all_gather of (index, features) and rank-0 index_copy_ reassembly.

MI355X design: the all-gathers are RCCL over xGMI; small [B, D] messages
are latency-bound so both gathers are launched back-to-back (single
synchronization point per batch). On one process it is a plain loop.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch.utils.data import DataLoader

from ..parallel import dist as dist_utils


@torch.no_grad()
def extract_features(model, data_loader: DataLoader, device,
                     multiscale: bool = False, dims: Optional[int] = None,
                     dtype: torch.dtype = torch.float32) -> Optional[torch.Tensor]:
    """Returns the full [N, D] feature matrix on every rank (rank0-authoritative).

    data_loader yields (images, index) — SynthDataset convention.
    """
    n_total = len(data_loader.dataset)
    features = None
    world = dist_utils.get_world_size()

    for batch in data_loader:
        samples, index = batch[0], batch[1]
        samples = samples.to(device, non_blocking=True)
        index = index.to(device, non_blocking=True)
        if multiscale:
            feats = _multiscale(samples, model)
        else:
            feats = model(samples)
        if feats.dim() > 2:
            feats = feats.flatten(1)
        feats = feats.to(dtype)

        if features is None:
            D = dims or feats.shape[-1]
            features = torch.zeros(n_total, D, device=device, dtype=dtype)

        if world > 1 and dist.is_initialized():
            idx_all = [torch.zeros_like(index) for _ in range(world)]
            f_all = [torch.zeros_like(feats) for _ in range(world)]
            h1 = dist.all_gather(idx_all, index, async_op=True)
            h2 = dist.all_gather(f_all, feats, async_op=True)
            h1.wait()
            h2.wait()
            idx_cat = torch.cat(idx_all)
            f_cat = torch.cat(f_all)
        else:
            idx_cat, f_cat = index, feats
        features.index_copy_(0, idx_cat, f_cat)
    return features


@torch.no_grad()
def _multiscale(samples: torch.Tensor, model) -> torch.Tensor:
    """Multi-scale eval (reference utils_ret.py:676-698): 1x, 1/sqrt(2), 1/2,
    summed then renormalized."""
    v = None
    for s in (1.0, 1.0 / (2 ** 0.5), 0.5):
        if s == 1.0:
            inp = samples.clone()
        else:
            inp = F.interpolate(samples, scale_factor=s, mode="bilinear",
                                align_corners=False)
        feats = model(inp)
        if feats.dim() > 2:
            feats = feats.flatten(1)
        v = feats if v is None else v + feats
    v /= 3
    return v / (v.norm(dim=-1, keepdim=True) + 1e-8)
