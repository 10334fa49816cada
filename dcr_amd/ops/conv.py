"""Native implicit-GEMM conv dispatch (forward HIP kernel, MIOpen backward).

Measured on MI355X (scripts/bench_conv.py, bs16 SD-2.1 shapes): the
hand-written NHWC implicit-GEMM forward (dcr_amd/ops/hip/conv_nhwc.hip,
BK=64 + split-K) runs 0.96-1.25x MIOpen — so eligible convs dispatch to
it; backward goes through aten::convolution_backward (MIOpen), which is
mathematically independent of which forward produced the output.

Eligibility (checked per call): CUDA bf16 channels_last input+weight,
C % 32 == 0, K % 64 == 0, 3x3(pad 1, dil 1) or 1x1(pad 0), stride 1|2,
no groups. Anything else falls through to torch's conv. Opt out with
DCR_NATIVE_CONV=0.
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import count_dispatch, ext


def _enabled() -> bool:
    return os.environ.get("DCR_NATIVE_CONV", "1") != "0"


def _eligible(x: torch.Tensor, m: nn.Conv2d) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4):
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    w = m.weight
    if w.dtype != torch.bfloat16 or \
            not w.is_contiguous(memory_format=torch.channels_last):
        return False
    C, K = w.shape[1], w.shape[0]
    R, S = w.shape[2], w.shape[3]
    if C % 32 != 0 or K % 64 != 0 or m.groups != 1:
        return False
    if m.dilation != (1, 1):
        return False
    if (R, S) == (3, 3):
        if m.padding != (1, 1):
            return False
    elif (R, S) == (1, 1):
        if m.padding != (0, 0):
            return False
    else:
        return False
    return m.stride[0] == m.stride[1] and m.stride[0] in (1, 2)


class _NativeConvFn(torch.autograd.Function):
    """Native conv fwd with optional fused epilogue adds:
    res — ResnetBlock2D's residual (d_res = dy, no extra kernel);
    temb — the [N, K] time-embedding projection broadcast
    (d_temb = dy.sum over H, W — what autograd's broadcast-add backward
    did anyway). Each fusion deletes one full elementwise pass."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding, res, temb):
        m = ext()
        count_dispatch('conv_nhwc')
        y = m.conv2d_nhwc_fwd_v2(x, weight, bias, stride, padding,
                                 res, temb)
        ctx.save_for_backward(x, weight)
        ctx.conf = (stride, padding, bias is not None,
                    res is not None, temb is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        stride, padding, has_bias, has_res, has_temb = ctx.conf
        dy = dy.contiguous(memory_format=torch.channels_last)
        d_res = dy if has_res else None
        d_temb = dy.sum(dim=(2, 3)) if has_temb else None
        # native backward (conv_nhwc_bwd.hip: dgrad + wgrad + fused
        # bias-grad) — numerics GPU-validated round 2; C % 64 shapes only
        if os.environ.get("DCR_NATIVE_CONV_BWD", "0") == "1" \
                and weight.shape[1] % 64 == 0:
            m = ext()
            count_dispatch('conv_nhwc_bwd')
            dx, dw, db = m.conv2d_nhwc_bwd(dy, x, weight, stride, padding)
            if has_bias and ctx.needs_input_grad[2]:
                db = db.to(dy.dtype)
            else:
                db = None
            return dx, dw, db, None, None, d_res, d_temb
        dx, dw, db = torch.ops.aten.convolution_backward(
            dy, x, weight,
            [weight.shape[0]] if has_bias else None,
            [stride, stride], [padding, padding], [1, 1], False, [0, 0], 1,
            [ctx.needs_input_grad[0], ctx.needs_input_grad[1],
             has_bias and ctx.needs_input_grad[2]])
        return dx, dw, db, None, None, d_res, d_temb


class Conv2d(nn.Conv2d):
    """nn.Conv2d with the native MI355X forward when eligible (identical
    parameters/state-dict; backward via MIOpen either way). `res`/`temb`
    optionally fuse a residual / per-(n,k) time-embedding add into the
    native epilogue; ineligible shapes fall back to the unfused ops."""

    def forward(self, x, res=None, temb=None):
        if _enabled() and _eligible(x, self):
            fuse_res = res is None or (
                res.dtype == torch.bfloat16
                and res.is_contiguous(memory_format=torch.channels_last))
            fuse_temb = temb is None or (temb.dtype == torch.bfloat16
                                         and temb.is_contiguous())
            if fuse_res and fuse_temb:
                return _NativeConvFn.apply(x, self.weight, self.bias,
                                           self.stride[0], self.padding[0],
                                           res, temb)
            y = _NativeConvFn.apply(x, self.weight, self.bias,
                                    self.stride[0], self.padding[0],
                                    None, None)
            if temb is not None:
                y = y + temb[:, :, None, None]
            if res is not None:
                y = y + res
            return y
        y = super().forward(x)
        if temb is not None:
            y = y + temb[:, :, None, None]
        if res is not None:
            y = y + res
        return y
