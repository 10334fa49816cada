"""Flat-buffer fused AdamW for MI355X.

Reference behavior: torch.optim.AdamW over ~865M UNet params
(/root/reference/diff_train.py:435-446; betas 0.9/0.999, eps 1e-8,
weight_decay 1e-2). Instead of ~700 per-tensor kernel launches per step,
this optimizer re-parameterizes the model so that

* every parameter tensor is a VIEW into one contiguous flat buffer,
* gradients land in one contiguous flat arena via batched
  ``gather_grads()`` (autograd owns per-param grads; no per-param adds),
* Adam state (m, v) are two more flat buffers,

so one step is ONE elementwise HIP kernel over four flat arrays —
HBM-bound by design (~8 TB/s on MI355X), and the flat grad buffer doubles
as the bucket arena for RCCL all-reduce (dcr_amd/parallel/ddp.py).

Flatten AFTER moving the model to its device; ``p.data`` is replaced by
buffer views.
"""
from __future__ import annotations

from typing import Iterable, List

import torch

from . import use_hip, require_hip, count_dispatch


class FusedAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 5e-6,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 1e-2,
        device_state: bool = False,
        max_grad_norm: float = 1.0,
    ):
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.max_grad_norm = max_grad_norm

        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("FusedAdamW: no trainable parameters")
        device = self.params[0].device
        dtype = self.params[0].dtype
        for p in self.params:
            if p.device != device or p.dtype != dtype:
                raise ValueError("FusedAdamW requires uniform param dtype/device")

        total = sum(p.numel() for p in self.params)
        self.flat_param = torch.empty(total, device=device, dtype=dtype)
        self.flat_grad = torch.zeros(total, device=device, dtype=dtype)
        self.exp_avg = torch.zeros(total, device=device, dtype=torch.float32)
        self.exp_avg_sq = torch.zeros(total, device=device, dtype=torch.float32)
        # pure-bf16 training: bf16 model params/grads + fp32 master copy
        self.master = None
        if dtype == torch.bfloat16:
            self.master = torch.empty(total, device=device, dtype=torch.float32)

        # Re-parameterize: params become views of flat_param. Grads are NOT
        # pre-assigned as arena views: with p.grad set, autograd's
        # AccumulateGrad issues one small `add_` kernel per parameter
        # (~700 launch-bound kernels, ~3.5 ms/step measured in the round-1
        # profile). Leaving p.grad unset lets autograd hand over the
        # computed tensor for free; gather_grads() then moves them into
        # the arena in a handful of batched _foreach kernels.
        offset = 0
        self.offsets: List[int] = []
        self._grad_views: List[torch.Tensor] = []
        for p in self.params:
            n = p.numel()
            sl = self.flat_param[offset : offset + n]
            gl = self.flat_grad[offset : offset + n]
            if (p.data.dim() == 4 and not p.data.is_contiguous()
                    and p.data.is_contiguous(
                        memory_format=torch.channels_last)):
                # channels_last conv weights: store in NHWC memory order
                # and view back with channels_last strides — a plain
                # view_as() would silently revert them to NCHW, making
                # every conv pay a per-call weight relayout (measured
                # ~2 ms/step of aten::copy_, profile_aten r02c8) and
                # de-eligibilizing the native conv kernel for every
                # TRAINED module (the frozen VAE was the only one
                # actually running it)
                K, C, Hh, Ww = p.data.shape
                sl.copy_(p.data.permute(0, 2, 3, 1).reshape(-1))
                p.data = sl.view(K, Hh, Ww, C).permute(0, 3, 1, 2)
                self._grad_views.append(
                    gl.view(K, Hh, Ww, C).permute(0, 3, 1, 2))
            else:
                sl.copy_(p.data.reshape(-1))
                p.data = sl.view_as(p.data)
                self._grad_views.append(gl.view_as(p.data))
            self.offsets.append(offset)
            offset += n
        self.numel = total
        self._view_of = {id(p): v for p, v in zip(self.params, self._grad_views)}
        self._gathered: set = set()
        if self.master is not None:
            self.master.copy_(self.flat_param.float())

        # device-state mode (round-2 draft, DCR_DEV_ADAMW=1): all per-step
        # scalars live in hyper[8] on device so the whole optimizer tail is
        # hipGraph-capturable — see elementwise.hip adamw_dev section.
        # hyper = {lr, b1^t, b2^t, inv_bc1, inv_bc2, clip_coef, gnorm_sq, t}
        self.hyper = None
        if device_state and device.type == "cuda":
            self.hyper = torch.tensor([lr, 1, 1, 1, 1, 1, 0, 0],
                                      dtype=torch.float32, device=device)

    # -- torch.optim-ish surface ------------------------------------------
    def zero_grad(self, set_to_none: bool = False):
        # the arena-wide zero exists so slices whose params produce no
        # grad read 0 — when EVERY param was gathered last cycle (the
        # normal training case) the next cycle's first gather per param
        # is a copy, so the zero is skippable (1.7 GB write saved).
        # _ensure_cold_slices() covers the rare dynamic-graph case where
        # a later cycle leaves some params ungathered.
        if len(self._gathered) != len(self.params):
            self.flat_grad.zero_()
            self._skipped_zero = False
        else:
            self._skipped_zero = True
        self._gathered.clear()
        for p in self.params:
            p.grad = None

    def _ensure_cold_slices(self):
        """After the final gather of a cycle whose zero was skipped, any
        param that produced no grad this cycle still holds LAST cycle's
        grad in its arena slice — zero exactly those."""
        if not getattr(self, "_skipped_zero", False):
            return
        if len(self._gathered) == len(self.params):
            return
        stale = [self._view_of[id(p)] for p in self.params
                 if id(p) not in self._gathered]
        if stale:
            torch._foreach_zero_(stale)
        self._skipped_zero = False

    @torch.no_grad()
    def gather_grads(self, params=None):
        """Move autograd-produced p.grad tensors into the flat arena with
        batched _foreach kernels (first gather per param after zero_grad
        is a copy over the zeroed slice; later gathers — gradient
        accumulation micro-steps — add). Called per-bucket by the DDP
        engine (before each bucket's all-reduce) and by finalize()."""
        copy_d, copy_s, add_d, add_s = [], [], [], []
        for p in (params if params is not None else self.params):
            g = p.grad
            if g is None:
                continue
            view = self._view_of[id(p)]
            if g.dtype != view.dtype:
                g = g.to(view.dtype)
            if id(p) in self._gathered:
                add_d.append(view)
                add_s.append(g)
            else:
                copy_d.append(view)
                copy_s.append(g)
                self._gathered.add(id(p))
            p.grad = None
        if copy_d:
            torch._foreach_copy_(copy_d, copy_s)
        if add_d:
            torch._foreach_add_(add_d, add_s)

    @torch.no_grad()
    def step(self, lr: float | None = None):
        self.gather_grads()  # no-op when finalize() already ran
        self._ensure_cold_slices()
        if lr is not None:
            self.lr = lr
        self.step_count += 1
        t = self.step_count
        if use_hip(self.flat_param):
            m = require_hip("adamw")
            if m is not None:
                count_dispatch('adamw')
                if self.master is not None:
                    m.adamw_step_bf16(
                        self.flat_param, self.flat_grad, self.master,
                        self.exp_avg, self.exp_avg_sq, self.lr, self.beta1,
                        self.beta2, self.eps, self.weight_decay, t)
                else:
                    m.adamw_step(
                        self.flat_param, self.flat_grad, self.exp_avg, self.exp_avg_sq,
                        self.lr, self.beta1, self.beta2, self.eps, self.weight_decay, t,
                    )
                return
        # reference implementation (CPU tests / debug fallback)
        bc1 = 1.0 - self.beta1 ** t
        bc2 = 1.0 - self.beta2 ** t
        g = self.flat_grad.float()
        self.exp_avg.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
        self.exp_avg_sq.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(self.eps)
        upd = (self.exp_avg / bc1) / denom
        if self.master is not None:
            self.master.add_(upd + self.weight_decay * self.master, alpha=-self.lr)
            self.flat_param.copy_(self.master.to(self.flat_param.dtype))
        else:
            self.flat_param.add_(
                (upd + self.weight_decay * self.flat_param.float()).to(self.flat_param.dtype),
                alpha=-self.lr,
            )

    @torch.no_grad()
    def step_dev(self, lr: float | None = None):
        """Device-state step (clip INCLUDED — do not call clip_grad_norm_
        separately): 3 stream-ordered kernels reading every scalar from the
        hyper buffer; capturable into a hipGraph. The only host work is the
        4-byte lr write when the schedule moves."""
        assert self.hyper is not None, "built without device_state=True"
        self.gather_grads()  # no-op when finalize() already ran
        self._ensure_cold_slices()
        if lr is not None and lr != self.lr:
            self.lr = lr
            self.hyper[0].fill_(lr)  # outside any captured graph
        self.step_count += 1
        m = require_hip("adamw")
        count_dispatch('adamw_dev')
        m.adamw_step_dev(self.flat_param, self.flat_grad, self.master,
                         self.exp_avg, self.exp_avg_sq, self.hyper,
                         self.beta1, self.beta2, self.eps, self.weight_decay,
                         self.max_grad_norm)

    @torch.no_grad()
    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        """Global L2 grad clip (reference: diff_train.py:657-663, max 1.0).
        fp32 accumulation via the dtype arg — `.float()` would
        materialize a full fp32 copy of the 1.7 GB bf16 arena first
        (~1 ms/step, profile_aten r02c8)."""
        self.gather_grads()  # no-op when finalize() already ran
        self._ensure_cold_slices()
        norm = torch.linalg.vector_norm(self.flat_grad, dtype=torch.float32)
        scale = max_norm / (norm + 1e-6)
        if float(norm) > max_norm:
            self.flat_grad.mul_(scale.to(self.flat_grad.dtype))
        return norm

    # -- checkpointing -----------------------------------------------------
    def state_dict(self):
        d = {
            "step": self.step_count,
            "lr": self.lr,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "flat_param": self.flat_param,
        }
        if self.master is not None:
            d["master"] = self.master
        if self.hyper is not None:
            d["hyper"] = self.hyper
        return d

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.lr = sd["lr"]
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.flat_param.copy_(sd["flat_param"])
        if self.hyper is not None and "hyper" in sd:
            self.hyper.copy_(sd["hyper"])
        if self.master is not None:
            if "master" in sd:
                self.master.copy_(sd["master"])
            else:
                self.master.copy_(self.flat_param.float())
