"""Hand-written MFMA GEMM dispatch for the transformer linear layers.

The reference runs every UNet linear (QKV/out projections, GEGLU
FeedForward matmuls, proj_in/out — /root/reference/diff_train.py:644 via
diffusers) through cuBLAS; round 1 left them on rocBLAS/Tensile (~20-25%
of GPU busy time). ``dcr_linear`` routes eligible shapes through the
in-tree bf16 MFMA kernel (dcr_amd/ops/hip/gemm.hip): forward
``x @ W^T + b``, backward ``dy @ W`` (dgrad) and ``dy^T @ x`` (wgrad,
with the bias-grad column-sum fused into the staging pass so the aten
reduce disappears).

Dispatch: bf16 CUDA tensors with K % 8 == 0 and M large enough to fill
the 128x128 tile grid. Everything else (tiny time-embed MLPs, fp32,
CPU) falls back to F.linear. DCR_NATIVE_GEMM=0 opts out entirely.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch import nn

from . import use_hip, require_hip, count_dispatch


def _native_gemm_on() -> bool:
    return os.environ.get("DCR_NATIVE_GEMM", "1") != "0"


def _eligible(x2d: torch.Tensor, weight: torch.Tensor) -> bool:
    M, K = x2d.shape
    N = weight.shape[0]
    return (x2d.dtype == torch.bfloat16 and weight.dtype == torch.bfloat16
            and K % 8 == 0 and N % 8 == 0 and M >= 256)


class _NativeLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, weight, bias):
        m = require_hip("gemm_bf16")
        count_dispatch("gemm")
        (y,) = m.gemm_bf16(x2d, weight, bias, False, False, False)
        ctx.save_for_backward(x2d, weight)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, weight = ctx.saved_tensors
        m = require_hip("gemm_bf16")
        dy = dy.contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            (dx,) = m.gemm_bf16(dy, weight, None, False, True, False)
        dw = db = None
        if ctx.needs_input_grad[1]:
            if ctx.has_bias and ctx.needs_input_grad[2]:
                dw, db = m.gemm_bf16(dy, x2d, None, True, True, True)
                db = db.to(dy.dtype)
            else:
                (dw,) = m.gemm_bf16(dy, x2d, None, True, True, False)
        return dx, dw, db


def dcr_linear(x: torch.Tensor, weight: torch.Tensor,
               bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear with the hand-written MFMA GEMM on eligible GPU shapes."""
    if use_hip(x) and _native_gemm_on():
        x2d = x.reshape(-1, x.shape[-1])
        if _eligible(x2d, weight):
            y = _NativeLinear.apply(x2d.contiguous(), weight.contiguous(),
                                    bias)
            return y.reshape(*x.shape[:-1], weight.shape[0])
    return F.linear(x, weight, bias)


class DcrLinear(nn.Linear):
    """nn.Linear whose forward routes through the MFMA GEMM when eligible.

    State-dict compatible with nn.Linear (and so with the diffusers
    checkpoint naming the models use)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # type: ignore[override]
        return dcr_linear(x, self.weight, self.bias)
