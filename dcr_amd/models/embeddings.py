"""Timestep embeddings (sinusoidal + MLP), SD-2.1 semantics.

Reference behavior: diffusers `Timesteps(flip_sin_to_cos=True,
downscale_freq_shift=0)` + `TimestepEmbedding` as used by the UNet the
reference finetunes (/root/reference/diff_train.py:644 via diffusers).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


def timestep_embedding(
    timesteps: torch.Tensor,
    dim: int,
    flip_sin_to_cos: bool = True,
    downscale_freq_shift: float = 0.0,
    max_period: int = 10000,
) -> torch.Tensor:
    """[N] int/float timesteps -> [N, dim] sinusoidal embedding (fp32)."""
    half = dim // 2
    exponent = -math.log(max_period) * torch.arange(
        half, dtype=torch.float32, device=timesteps.device
    )
    exponent = exponent / (half - downscale_freq_shift)
    emb = timesteps.float()[:, None] * exponent.exp()[None, :]
    sin, cos = emb.sin(), emb.cos()
    if flip_sin_to_cos:
        emb = torch.cat([cos, sin], dim=-1)
    else:
        emb = torch.cat([sin, cos], dim=-1)
    if dim % 2 == 1:
        emb = F.pad(emb, (0, 1))
    return emb


class TimestepEmbedding(nn.Module):
    """2-layer SiLU MLP: time_embed_dim = 4 * block_out_channels[0]."""

    def __init__(self, in_channels: int, time_embed_dim: int):
        super().__init__()
        self.linear_1 = nn.Linear(in_channels, time_embed_dim)
        self.linear_2 = nn.Linear(time_embed_dim, time_embed_dim)

    def forward(self, sample: torch.Tensor) -> torch.Tensor:
        return self.linear_2(F.silu(self.linear_1(sample)))
