"""Shared nn.Module wrappers around dcr_amd.ops kernels.

Parameter names match diffusers (`weight`/`bias` on modules named
`norm*`) so state dicts interop with diffusers-format checkpoints
(SURVEY.md §5.4 checkpoint-layout contract).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class GroupNormOp(nn.Module):
    """GroupNorm with optional fused SiLU, fp32 accumulation (HIP kernel)."""

    def __init__(self, num_groups: int, num_channels: int, eps: float = 1e-6,
                 fused_silu: bool = False):
        super().__init__()
        self.num_groups = num_groups
        self.num_channels = num_channels
        self.eps = eps
        self.fused_silu = fused_silu
        self.weight = nn.Parameter(torch.ones(num_channels))
        self.bias = nn.Parameter(torch.zeros(num_channels))

    def forward(self, x):
        return ops.group_norm_silu(x, self.weight, self.bias, self.num_groups,
                                   self.eps, self.fused_silu)

    def extra_repr(self):
        return f"{self.num_groups}, {self.num_channels}, eps={self.eps}, fused_silu={self.fused_silu}"


class LayerNormOp(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, elementwise_affine: bool = True):
        super().__init__()
        self.dim = dim
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        return ops.layer_norm(x, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.dim}, eps={self.eps}"
