"""DDIM sampler (50-step default inference path).

Capability parity: diffusers DDIMScheduler inside StableDiffusionPipeline
(/root/reference/diff_inference.py:190-193, 50 steps). eta=0
deterministic update; epsilon and v-prediction supported.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path
from typing import Optional

import torch

from .. import ops
from .ddpm import make_betas


@dataclass
class DDIMConfig:
    num_train_timesteps: int = 1000
    beta_start: float = 0.00085
    beta_end: float = 0.012
    beta_schedule: str = "scaled_linear"
    prediction_type: str = "epsilon"
    steps_offset: int = 1
    clip_sample: bool = False
    set_alpha_to_one: bool = False


class _StepOut:
    def __init__(self, prev_sample, pred_original_sample=None):
        self.prev_sample = prev_sample
        self.pred_original_sample = pred_original_sample


class DDIMScheduler:
    def __init__(self, **kwargs):
        self.config = DDIMConfig(**kwargs)
        c = self.config
        self.betas = make_betas(c.num_train_timesteps, c.beta_start, c.beta_end,
                                c.beta_schedule).to(torch.float32)
        self.alphas_cumprod = torch.cumprod(1.0 - self.betas, dim=0)
        self.final_alpha_cumprod = (torch.tensor(1.0) if c.set_alpha_to_one
                                    else self.alphas_cumprod[0])
        self.prediction_type = c.prediction_type
        self.init_noise_sigma = 1.0
        self.timesteps: Optional[torch.Tensor] = None
        self.num_inference_steps: Optional[int] = None

    def set_timesteps(self, num_inference_steps: int, device=None):
        c = self.config
        self.num_inference_steps = num_inference_steps
        step = c.num_train_timesteps // num_inference_steps
        ts = (torch.arange(0, num_inference_steps) * step).round().flip(0).long()
        ts = ts + c.steps_offset
        ts = ts.clamp(max=c.num_train_timesteps - 1)
        self.timesteps = ts.to(device) if device is not None else ts

    def scale_model_input(self, sample, timestep=None):
        return sample

    def step(self, model_output: torch.Tensor, timestep: int, sample: torch.Tensor,
             eta: float = 0.0, generator=None) -> _StepOut:
        t = int(timestep)
        prev_t = t - self.config.num_train_timesteps // self.num_inference_steps
        ac_t = self.alphas_cumprod[t]
        ac_prev = self.alphas_cumprod[prev_t] if prev_t >= 0 else self.final_alpha_cumprod

        # fast path (GPU): the eta=0 update is linear in (sample,
        # model_output) -> one fused HIP kernel (ops.lincomb)
        if eta == 0.0 and not self.config.clip_sample and sample.is_cuda:
            a_t = float(ac_t.sqrt())
            s_t = float((1 - ac_t).sqrt())
            a_p = float(ac_prev.sqrt())
            s_p = float((1 - ac_prev).sqrt())     # dir coefficient, sigma = 0
            if self.prediction_type == "epsilon":
                A = a_p / a_t
                B = s_p - a_p * s_t / a_t
            elif self.prediction_type == "v_prediction":
                A = a_p * a_t + s_p * s_t
                B = s_p * a_t - a_p * s_t
            else:
                raise ValueError(self.prediction_type)
            return _StepOut(ops.lincomb(sample, model_output, A, B))

        ac_t = ac_t.to(sample.device)
        ac_prev = ac_prev.to(sample.device)

        mo = model_output.float()
        s = sample.float()
        if self.prediction_type == "epsilon":
            x0 = (s - (1 - ac_t).sqrt() * mo) / ac_t.sqrt()
            eps = mo
        elif self.prediction_type == "v_prediction":
            x0 = ac_t.sqrt() * s - (1 - ac_t).sqrt() * mo
            eps = ac_t.sqrt() * mo + (1 - ac_t).sqrt() * s
        else:
            raise ValueError(self.prediction_type)
        if self.config.clip_sample:
            x0 = x0.clamp(-1, 1)

        var = (1 - ac_prev) / (1 - ac_t) * (1 - ac_t / ac_prev)
        sigma = eta * var.sqrt()
        dir_xt = (1 - ac_prev - sigma ** 2).clamp(min=0).sqrt() * eps
        prev = ac_prev.sqrt() * x0 + dir_xt
        if eta > 0:
            noise = torch.randn(sample.shape, generator=generator,
                                device=sample.device, dtype=torch.float32)
            prev = prev + sigma * noise
        return _StepOut(prev.to(sample.dtype), x0.to(sample.dtype))

    def save_pretrained(self, path):
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        d = asdict(self.config)
        d["_class_name"] = "DDIMScheduler"
        (path / "scheduler_config.json").write_text(json.dumps(d, indent=2))

    @classmethod
    def from_pretrained(cls, path):
        d = json.loads((Path(path) / "scheduler_config.json").read_text())
        d.pop("_class_name", None)
        known = {k: v for k, v in d.items() if k in DDIMConfig.__dataclass_fields__}
        return cls(**known)
