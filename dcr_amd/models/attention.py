"""Transformer blocks for the UNet (SD-2.1: linear projections, GEGLU FF).

Reference behavior: diffusers Transformer2DModel / BasicTransformerBlock /
Attention as run by the finetune hot loop (/root/reference/diff_train.py:644)
and sampling (diff_inference.py:190). MI355X design: LayerNorm and the
GEGLU gate are single fused HIP kernels; attention routes through
dcr_amd.ops.attention (flash-style CDNA4 kernel on GPU); the QKV/out
projections route through the in-tree bf16 MFMA GEMM
(dcr_amd/ops/hip/gemm.hip) on eligible shapes (DcrLinear).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from .layers import GroupNormOp, LayerNormOp
from ..ops.linear import DcrLinear


class Attention(nn.Module):
    """Multi-head attention with diffusers param naming (to_q/to_k/to_v/to_out.0)."""

    def __init__(
        self,
        query_dim: int,
        cross_attention_dim: Optional[int] = None,
        heads: int = 8,
        dim_head: int = 64,
        dropout: float = 0.0,
        bias: bool = False,
        out_bias: bool = True,
    ):
        super().__init__()
        inner_dim = heads * dim_head
        kv_dim = cross_attention_dim or query_dim
        self.heads = heads
        self.dim_head = dim_head
        self.to_q = DcrLinear(query_dim, inner_dim, bias=bias)
        self.to_k = DcrLinear(kv_dim, inner_dim, bias=bias)
        self.to_v = DcrLinear(kv_dim, inner_dim, bias=bias)
        self.to_out = nn.ModuleList([DcrLinear(inner_dim, query_dim, bias=out_bias),
                                     nn.Dropout(dropout)])

    def forward(self, x: torch.Tensor, context: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, L, _ = x.shape
        ctx = context if context is not None else x
        # transpose-free [B, L, H, D] layout: the flash kernel reads it
        # with a strided row pitch, so no permute copies are materialized
        q = self.to_q(x).view(B, L, self.heads, self.dim_head)
        k = self.to_k(ctx).view(B, ctx.shape[1], self.heads, self.dim_head)
        v = self.to_v(ctx).view(B, ctx.shape[1], self.heads, self.dim_head)
        out = ops.attention(q, k, v, layout="blhd")
        out = out.reshape(B, L, self.heads * self.dim_head)
        out = self.to_out[0](out)
        return self.to_out[1](out)


class FeedForward(nn.Module):
    """GEGLU FF: proj to 2*inner, fused a*gelu(g) kernel, proj back."""

    def __init__(self, dim: int, mult: int = 4, dropout: float = 0.0):
        super().__init__()
        inner = dim * mult
        # diffusers naming: ff.net.0.proj (GEGLU), ff.net.1 (Dropout), ff.net.2 (Linear)
        geglu_proj = nn.Module()
        geglu_proj.proj = DcrLinear(dim, inner * 2)
        self.net = nn.ModuleList([geglu_proj, nn.Dropout(dropout), DcrLinear(inner, dim)])

    def forward(self, x):
        x = self.net[0].proj(x)
        x = ops.geglu(x)
        x = self.net[1](x)
        return self.net[2](x)


class BasicTransformerBlock(nn.Module):
    def __init__(
        self,
        dim: int,
        heads: int,
        dim_head: int,
        cross_attention_dim: Optional[int] = None,
        dropout: float = 0.0,
    ):
        super().__init__()
        self.norm1 = LayerNormOp(dim)
        self.attn1 = Attention(dim, heads=heads, dim_head=dim_head, dropout=dropout)
        self.norm2 = LayerNormOp(dim)
        self.attn2 = Attention(dim, cross_attention_dim=cross_attention_dim,
                               heads=heads, dim_head=dim_head, dropout=dropout)
        self.norm3 = LayerNormOp(dim)
        self.ff = FeedForward(dim, dropout=dropout)

    def forward(self, x, context=None):
        x = x + self.attn1(self.norm1(x))
        x = x + self.attn2(self.norm2(x), context=context)
        x = x + self.ff(self.norm3(x))
        return x


class Transformer2DModel(nn.Module):
    """Spatial transformer: GN -> (linear) proj_in -> blocks -> proj_out + residual.

    SD-2.1 uses use_linear_projection=True (proj_in/out are Linear on the
    [B, HW, C] layout, not 1x1 convs).
    """

    def __init__(
        self,
        in_channels: int,
        heads: int,
        dim_head: int,
        depth: int = 1,
        cross_attention_dim: Optional[int] = None,
        norm_num_groups: int = 32,
        use_linear_projection: bool = True,
        dropout: float = 0.0,
    ):
        super().__init__()
        inner_dim = heads * dim_head
        self.use_linear_projection = use_linear_projection
        self.norm = GroupNormOp(norm_num_groups, in_channels, eps=1e-6, fused_silu=False)
        if use_linear_projection:
            self.proj_in = DcrLinear(in_channels, inner_dim)
            self.proj_out = DcrLinear(inner_dim, in_channels)
        else:
            self.proj_in = nn.Conv2d(in_channels, inner_dim, 1)
            self.proj_out = nn.Conv2d(inner_dim, in_channels, 1)
        self.transformer_blocks = nn.ModuleList([
            BasicTransformerBlock(inner_dim, heads, dim_head,
                                  cross_attention_dim=cross_attention_dim, dropout=dropout)
            for _ in range(depth)
        ])

    def forward(self, x: torch.Tensor, context: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, C, H, W = x.shape
        residual = x
        h = self.norm(x)
        if self.use_linear_projection:
            h = h.permute(0, 2, 3, 1).reshape(B, H * W, C)
            h = self.proj_in(h)
        else:
            h = self.proj_in(h).permute(0, 2, 3, 1).reshape(B, H * W, -1)
        for block in self.transformer_blocks:
            h = block(h, context=context)
        if self.use_linear_projection:
            h = self.proj_out(h)
            h = h.reshape(B, H, W, C).permute(0, 3, 1, 2)
        else:
            h = h.reshape(B, H, W, -1).permute(0, 3, 1, 2)
            h = self.proj_out(h)
        return h + residual
