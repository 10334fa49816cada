"""Functional op wrappers: HIP kernels on GPU, PyTorch reference on CPU.

Each op corresponds to a fused hot spot of the SD-2.1 finetune / sampling
loop the reference runs through libraries:

* ``group_norm_silu``  — every ResNet-block norm in UNet/VAE
  (reference: diffusers ResnetBlock2D, /root/reference/diff_train.py:644).
* ``layer_norm``       — transformer blocks in UNet + CLIP text encoder.
* ``geglu``            — UNet transformer FeedForward gate.
* ``attention``        — self/cross attention (flash-style HIP kernel).
* ``add_noise`` / ``get_velocity`` — DDPM scheduler math
  (reference: diff_train.py:632,650).
* ``cfg_combine``      — classifier-free guidance in sampling.
"""
from __future__ import annotations

import math
import os
from typing import Optional

import torch
import torch.nn.functional as F

from . import use_hip, require_hip, count_dispatch


# --------------------------------------------------------------------------
# GroupNorm (+ optional fused SiLU)
# --------------------------------------------------------------------------
class _GroupNormSiLUNHWC(torch.autograd.Function):
    """channels_last path: stats/apply kernels over the native NHWC walk
    (dcr_amd/ops/hip/norms_nhwc.hip) — no transpose round-trips."""

    @staticmethod
    def forward(ctx, x, weight, bias, num_groups, eps, apply_silu):
        m = require_hip("group_norm_silu_nhwc")
        count_dispatch('groupnorm_nhwc')
        y, mean, rstd = m.groupnorm_silu_nhwc_fwd(x, weight, bias, num_groups,
                                                  eps, apply_silu)
        ctx.save_for_backward(x, weight, bias, mean, rstd)
        ctx.num_groups = num_groups
        ctx.apply_silu = apply_silu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, bias, mean, rstd = ctx.saved_tensors
        m = require_hip("group_norm_silu_nhwc")
        dx, dw, db = m.groupnorm_silu_nhwc_bwd(
            dy.contiguous(memory_format=torch.channels_last), x, weight, bias,
            mean, rstd, ctx.num_groups, ctx.apply_silu)
        return dx, dw, db, None, None, None


class _GroupNormSiLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, num_groups, eps, apply_silu):
        m = require_hip("group_norm_silu")
        if m is None:  # debug fallback on GPU
            return _gn_silu_ref(x, weight, bias, num_groups, eps, apply_silu)
        count_dispatch('groupnorm')
        y, mean, rstd = m.groupnorm_silu_fwd(x, weight, bias, num_groups, eps, apply_silu)
        ctx.save_for_backward(x, weight, bias, mean, rstd)
        ctx.num_groups = num_groups
        ctx.apply_silu = apply_silu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, bias, mean, rstd = ctx.saved_tensors
        m = require_hip("group_norm_silu")
        dx, dw, db = m.groupnorm_silu_bwd(
            dy.contiguous(), x, weight, bias, mean, rstd, ctx.num_groups, ctx.apply_silu
        )
        return dx, dw, db, None, None, None


def _gn_silu_ref(x, weight, bias, num_groups, eps, apply_silu):
    # fp32 accumulation regardless of input dtype (matches HIP kernel numerics)
    y = F.group_norm(x.float(), num_groups, weight.float(), bias.float(), eps)
    if apply_silu:
        y = F.silu(y)
    return y.to(x.dtype)


def group_norm_silu(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    num_groups: int = 32,
    eps: float = 1e-5,
    apply_silu: bool = True,
) -> torch.Tensor:
    if use_hip(x):
        if x.dim() == 4 and x.is_contiguous(memory_format=torch.channels_last) \
                and not x.is_contiguous() and x.shape[1] % 4 == 0:
            return _GroupNormSiLUNHWC.apply(x, weight, bias, num_groups, eps,
                                            apply_silu)
        return _GroupNormSiLU.apply(x.contiguous(), weight, bias, num_groups, eps, apply_silu)
    return _gn_silu_ref(x, weight, bias, num_groups, eps, apply_silu)


# --------------------------------------------------------------------------
# LayerNorm
# --------------------------------------------------------------------------
class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        m = require_hip("layer_norm")
        if m is None:
            return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)
        count_dispatch('layernorm')
        y, mean, rstd = m.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        m = require_hip("layer_norm")
        dx, dw, db = m.layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layer_norm(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    eps: float = 1e-5,
) -> torch.Tensor:
    if use_hip(x):
        return _LayerNorm.apply(x.contiguous(), weight, bias, eps)
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


# --------------------------------------------------------------------------
# GEGLU: out = a * gelu(g) with [a, g] = split(x, 2, dim=-1)
# --------------------------------------------------------------------------
class _GEGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        m = require_hip("geglu")
        if m is None:
            a, g = x.chunk(2, dim=-1)
            return a * F.gelu(g)
        count_dispatch('geglu')
        y = m.geglu_fwd(x)
        ctx.save_for_backward(x)
        return y

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        m = require_hip("geglu")
        return m.geglu_bwd(dy.contiguous(), x)


def geglu(x: torch.Tensor) -> torch.Tensor:
    """x[..., :N] * gelu(x[..., N:]) — exact (erf) GELU."""
    if use_hip(x):
        return _GEGLU.apply(x.contiguous())
    a, g = x.chunk(2, dim=-1)
    return a * F.gelu(g)


# --------------------------------------------------------------------------
# Attention (hand-written CDNA4 flash kernel). q,k,v: [B, H, L, D].
# --------------------------------------------------------------------------
class _FlashAttention(torch.autograd.Function):
    """bf16 flash attention, head_dim 64 (dcr_amd/ops/hip/attention.hip)."""

    @staticmethod
    def forward(ctx, q, k, v, scale, causal):
        m = require_hip("attn")
        count_dispatch('attention')
        # schedule dispatch (measured, gpurun_out/r02c7): the swapped-QK^T
        # in-register-softmax v4 wins at Lq >= 256 (1.17x @256, 1.27x
        # @1024, 1.73x @4096 vs v1); the 64x64-tile v1 stays for short
        # sequences and CLIP's causal 77. v4's LSE is bit-identical so
        # the shared FlashAttention-2 backward applies to both.
        # DCR_ATTN_V2/V3: env-gated draft schedules (A/B only).
        if os.environ.get("DCR_ATTN_V3") == "1":
            fwd = m.attn_fwd_v3
        elif os.environ.get("DCR_ATTN_V2") == "1":
            fwd = m.attn_fwd_v2
        elif q.shape[1] >= 256 and os.environ.get("DCR_ATTN_V4", "1") != "0":
            fwd = m.attn_fwd_v4
        else:
            fwd = m.attn_fwd
        o, lse = fwd(q, k, v, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, dO):
        q, k, v, o, lse = ctx.saved_tensors
        m = require_hip("attn")
        # measured dispatch (gpurun_out/r02c16): the swapped-operand v4
        # backward wins for self-attention (1.27x @256, 1.13x @1024,
        # 1.56x @4096 vs v1) and loses at tiny L / 77-token cross-attn
        # where its 256-row blocks starve. DCR_ATTN_BWD_V4=0/1 overrides.
        env = os.environ.get("DCR_ATTN_BWD_V4", "")
        use_v4 = (env == "1") if env in ("0", "1") \
            else (q.shape[1] >= 256 and k.shape[1] >= 256)
        bwd = m.attn_bwd_v4 if use_v4 else m.attn_bwd
        dQ, dK, dV = bwd(q, k, v, o, dO.contiguous(), lse,
                         ctx.scale, ctx.causal)
        return dQ, dK, dV, None, None


def _attention_math(q, k, v, scale, causal):
    """Composite path (rocBLAS GEMMs + native softmax) for shapes no HIP
    kernel covers (now only the VAE's single 512-d head and training-mode
    non-64 dims; SD-1.4's 40/80/160 run the gen kernel). No Triton."""
    count_dispatch('attention_math')
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    if causal:
        Lq, Lk = q.shape[-2], k.shape[-2]
        mask = torch.ones(Lq, Lk, dtype=torch.bool, device=q.device).tril_()
        s = s.masked_fill(~mask, float("-inf"))
    p = s.softmax(dim=-1)
    return (p @ v.float()).to(q.dtype)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = False,
    scale: Optional[float] = None,
    layout: str = "bhld",
) -> torch.Tensor:
    """layout "bhld": q,k,v are [B, H, L, D]. layout "blhd": [B, L, H, D] —
    the transpose-free layout the models use (the flash kernel walks it
    with a strided row pitch, so no head-split permute copies happen)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if use_hip(q):
        if q.dtype == torch.bfloat16 and q.shape[-1] == 64 and k.shape[-1] == 64:
            if layout == "blhd":
                return _FlashAttention.apply(q.contiguous(), k.contiguous(),
                                             v.contiguous(), scale, causal)
            shp = q.shape
            out = _FlashAttention.apply(
                q.reshape(-1, shp[-2], 64).contiguous(),
                k.reshape(-1, k.shape[-2], 64).contiguous(),
                v.reshape(-1, v.shape[-2], 64).contiguous(),
                scale, causal)
            return out.reshape(shp)
        if q.dtype == torch.bfloat16 and q.shape[-1] in (40, 80, 160) \
                and not (torch.is_grad_enabled() and
                         (q.requires_grad or k.requires_grad or v.requires_grad)):
            # SD-1.4 head dims (sd_mitigation / diff_inference sampling,
            # reference sd_mitigation.py:46) — forward-only HIP kernel
            m = require_hip("attn_gen")
            count_dispatch('attention_gen')
            if layout == "blhd":
                return m.attn_fwd_gen(q.contiguous(), k.contiguous(),
                                      v.contiguous(), scale, causal)
            d = q.shape[-1]
            shp = q.shape
            out = m.attn_fwd_gen(
                q.reshape(-1, shp[-2], d).contiguous(),
                k.reshape(-1, k.shape[-2], d).contiguous(),
                v.reshape(-1, v.shape[-2], d).contiguous(), scale, causal)
            return out.reshape(shp)
        if layout == "blhd":
            out = _attention_math(q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3),
                                  v.permute(0, 2, 1, 3), scale, causal)
            return out.permute(0, 2, 1, 3)
        return _attention_math(q, k, v, scale, causal)
    if layout == "blhd":
        out = F.scaled_dot_product_attention(
            q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3), v.permute(0, 2, 1, 3),
            is_causal=causal, scale=scale)
        return out.permute(0, 2, 1, 3)
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal, scale=scale)


# --------------------------------------------------------------------------
# Diffusion scheduler math (no autograd needed: inputs are leaves w/o grad)
# --------------------------------------------------------------------------
def _gather_sqrt(alphas_cumprod: torch.Tensor, t: torch.Tensor, ndim: int):
    ac = alphas_cumprod.to(torch.float32).gather(0, t)
    shape = [t.shape[0]] + [1] * (ndim - 1)
    sqrt_ac = ac.sqrt().view(shape)
    sqrt_1mac = (1.0 - ac).sqrt().view(shape)
    return sqrt_ac, sqrt_1mac


def add_noise(
    x0: torch.Tensor, noise: torch.Tensor, alphas_cumprod: torch.Tensor, t: torch.Tensor
) -> torch.Tensor:
    """x_t = sqrt(ac_t) x0 + sqrt(1-ac_t) noise, per-sample t."""
    from . import ext

    m = ext()
    if use_hip(x0) and m is not None:
        count_dispatch('add_noise')
        ac = alphas_cumprod if (alphas_cumprod.device == x0.device and
                                alphas_cumprod.dtype == torch.float32) \
            else alphas_cumprod.to(x0.device, torch.float32)
        return m.add_noise(x0.contiguous(), noise.contiguous(), ac,
                           t.contiguous())
    sa, sb = _gather_sqrt(alphas_cumprod.to(x0.device), t, x0.dim())
    return (sa * x0.float() + sb * noise.float()).to(x0.dtype)


def get_velocity(
    x0: torch.Tensor, noise: torch.Tensor, alphas_cumprod: torch.Tensor, t: torch.Tensor
) -> torch.Tensor:
    """v = sqrt(ac_t) noise - sqrt(1-ac_t) x0 (v-prediction target)."""
    from . import ext

    m = ext()
    if use_hip(x0) and m is not None:
        ac = alphas_cumprod if (alphas_cumprod.device == x0.device and
                                alphas_cumprod.dtype == torch.float32) \
            else alphas_cumprod.to(x0.device, torch.float32)
        return m.get_velocity(x0.contiguous(), noise.contiguous(), ac,
                              t.contiguous())
    sa, sb = _gather_sqrt(alphas_cumprod.to(x0.device), t, x0.dim())
    return (sa * noise.float() - sb * x0.float()).to(x0.dtype)


def _dense(t: torch.Tensor) -> bool:
    """dense (no holes) in ANY layout — elementwise kernels walk raw memory."""
    if t.is_contiguous():
        return True
    return t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last)


def lincomb(x: torch.Tensor, y: torch.Tensor, a: float, b: float,
            z: Optional[torch.Tensor] = None, c: float = 0.0) -> torch.Tensor:
    """out = a*x + b*y (+ c*z) — one fused HIP kernel; the DDIM and
    DPM-Solver++ sampler updates are expressed through this."""
    from . import ext

    m = ext()
    if use_hip(x) and m is not None and x.numel() % 4 == 0 \
            and _dense(x) and x.stride() == y.stride() \
            and (z is None or z.stride() == x.stride()):
        count_dispatch('lincomb')
        return m.lincomb(x, y, z, float(a), float(b), float(c))
    out = a * x.float() + b * y.float()
    if z is not None:
        out = out + c * z.float()
    return out.to(x.dtype)


def cfg_combine(eps_uncond: torch.Tensor, eps_text: torch.Tensor, scale: float) -> torch.Tensor:
    """Classifier-free guidance: eps_u + s * (eps_t - eps_u)."""
    from . import ext

    m = ext()
    if use_hip(eps_uncond) and m is not None \
            and eps_uncond.stride() == eps_text.stride() \
            and _dense(eps_uncond):
        return m.cfg_combine(eps_uncond, eps_text, float(scale))
    return eps_uncond + scale * (eps_text - eps_uncond)
