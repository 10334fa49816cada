"""Linear-layer dispatch: rocBLAS forward/dgrad + hand-written MFMA wgrad.

The reference runs every UNet linear (QKV/out projections, GEGLU
FeedForward matmuls, proj_in/out — /root/reference/diff_train.py:644 via
diffusers) through cuBLAS. Round-2 measurement on MI355X
(scripts/bench_gemm.py, gpurun_out/r02c2_bench_gemm.log):

* fwd / dgrad: the in-tree 128x128 2-barrier MFMA kernel reaches
  77-373 TF = 0.42-0.59x rocBLAS/Tensile on the SD-2.1 shapes — Tensile
  keeps those passes.
* wgrad + fused bias-grad measured 0.98-1.13x in the ISOLATED microbench
  at contraction >= 4096, but the whole-model profile
  (profiles/r02_prof_head.md) shows it costing 40 ms/6-steps vs the
  ~25 ms of Tensile wgrad + aten bias-reduce it replaced — the
  microbench's rocBLAS side was inflated by an fp32 cast in the
  reference timing. Net in-model loss, so native linear dispatch is
  DEFAULT OFF.

DCR_NATIVE_GEMM=1 enables the hybrid (native wgrad) path;
DCR_NATIVE_GEMM=full forces all three passes native (benching only).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch import nn

from . import use_hip, require_hip, count_dispatch


def _mode() -> str:
    return os.environ.get("DCR_NATIVE_GEMM", "0")


def _wgrad_eligible(x2d: torch.Tensor, weight: torch.Tensor) -> bool:
    M, K = x2d.shape
    N = weight.shape[0]
    return (x2d.dtype == torch.bfloat16 and weight.dtype == torch.bfloat16
            and K % 8 == 0 and N % 8 == 0 and M >= 4096)


class _HybridLinear(torch.autograd.Function):
    """rocBLAS fwd/dgrad, native MFMA wgrad (+fused dbias)."""

    @staticmethod
    def forward(ctx, x2d, weight, bias):
        y = F.linear(x2d, weight, bias)
        ctx.save_for_backward(x2d, weight)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dy @ weight if ctx.needs_input_grad[0] else None
        dw = db = None
        if ctx.needs_input_grad[1]:
            m = require_hip("gemm_bf16")
            count_dispatch("gemm_wgrad")
            want_db = ctx.has_bias and ctx.needs_input_grad[2]
            if want_db:
                dw, db = m.gemm_bf16(dy, x2d, None, True, True, True)
                db = db.to(dy.dtype)
            else:
                (dw,) = m.gemm_bf16(dy, x2d, None, True, True, False)
        elif ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.sum(0)
        return dx, dw, db


class _FullNativeLinear(torch.autograd.Function):
    """All three passes on gemm.hip (DCR_NATIVE_GEMM=full — benching)."""

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
    """F.linear with the native MFMA wgrad in its measured-win regime."""
    mode = _mode()
    if mode != "0" and use_hip(x) and torch.is_grad_enabled() \
            and weight.requires_grad:
        x2d = x.reshape(-1, x.shape[-1])
        if _wgrad_eligible(x2d, weight):
            fn = _FullNativeLinear if mode == "full" else _HybridLinear
            y = fn.apply(x2d.contiguous(), weight.contiguous(), bias)
            return y.reshape(*x.shape[:-1], weight.shape[0])
    return F.linear(x, weight, bias)


class DcrLinear(nn.Linear):
    """nn.Linear routing through dcr_linear (state-dict compatible with
    nn.Linear and so with the diffusers checkpoint naming)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # type: ignore[override]
        return dcr_linear(x, self.weight, self.bias)
