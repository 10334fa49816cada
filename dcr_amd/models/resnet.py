"""UNet/VAE ResNet blocks and up/down-samplers (SD-2.1 shapes).

Reference behavior: diffusers ResnetBlock2D / Downsample2D / Upsample2D as
exercised by the finetune loop (/root/reference/diff_train.py:644) —
rebuilt on dcr_amd ops: GroupNorm+SiLU is one fused HIP kernel
(SURVEY.md §2.4.A), convs go through the native implicit-GEMM kernel (ops/conv.py)
with the time-embedding and residual adds fused into its epilogue.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.conv import Conv2d
from .layers import GroupNormOp


class ResnetBlock2D(nn.Module):
    def __init__(
        self,
        in_channels: int,
        out_channels: Optional[int] = None,
        temb_channels: Optional[int] = 1280,
        groups: int = 32,
        eps: float = 1e-6,
        dropout: float = 0.0,
        output_scale_factor: float = 1.0,
    ):
        super().__init__()
        out_channels = out_channels or in_channels
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.output_scale_factor = output_scale_factor

        self.norm1 = GroupNormOp(groups, in_channels, eps=eps, fused_silu=True)
        self.conv1 = Conv2d(in_channels, out_channels, 3, padding=1)
        if temb_channels is not None:
            self.time_emb_proj = nn.Linear(temb_channels, out_channels)
        else:
            self.time_emb_proj = None
        self.norm2 = GroupNormOp(groups, out_channels, eps=eps, fused_silu=True)
        self.dropout = nn.Dropout(dropout)
        self.conv2 = Conv2d(out_channels, out_channels, 3, padding=1)
        if in_channels != out_channels:
            self.conv_shortcut = Conv2d(in_channels, out_channels, 1)
        else:
            self.conv_shortcut = None

    def forward(self, x: torch.Tensor, temb: Optional[torch.Tensor] = None) -> torch.Tensor:
        h = self.norm1(x)          # fused GN+SiLU
        # time-embedding projection rides conv1's epilogue (one fused
        # add instead of a broadcast elementwise pass); the residual add
        # rides conv2's epilogue the same way (ops/conv.py)
        tv = None
        if self.time_emb_proj is not None and temb is not None:
            tv = self.time_emb_proj(F.silu(temb))
        h = self.conv1(h, temb=tv)
        h = self.norm2(h)          # fused GN+SiLU
        h = self.dropout(h)
        sc = self.conv_shortcut(x) if self.conv_shortcut is not None else x
        h = self.conv2(h, res=sc)
        if self.output_scale_factor != 1.0:
            h = h / self.output_scale_factor
        return h


class Downsample2D(nn.Module):
    def __init__(self, channels: int, out_channels: Optional[int] = None):
        super().__init__()
        out_channels = out_channels or channels
        self.conv = Conv2d(channels, out_channels, 3, stride=2, padding=1)

    def forward(self, x):
        return self.conv(x)


class Upsample2D(nn.Module):
    def __init__(self, channels: int, out_channels: Optional[int] = None):
        super().__init__()
        out_channels = out_channels or channels
        self.conv = Conv2d(channels, out_channels, 3, padding=1)

    def forward(self, x):
        x = F.interpolate(x, scale_factor=2.0, mode="nearest")
        return self.conv(x)
