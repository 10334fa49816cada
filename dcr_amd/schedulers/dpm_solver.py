"""DPM-Solver++ (2M) multistep sampler.

Capability parity: diffusers DPMSolverMultistepScheduler as used for
stock-SD sampling and mitigation runs
(/root/reference/diff_inference.py:93, sd_mitigation.py:58; 50 steps).
Data-prediction (dpmsolver++) variant, order 2, no thresholding
(latent-space SD).
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path
from typing import List, Optional

import torch

from .. import ops
from .ddpm import make_betas


@dataclass
class DPMConfig:
    num_train_timesteps: int = 1000
    beta_start: float = 0.00085
    beta_end: float = 0.012
    beta_schedule: str = "scaled_linear"
    prediction_type: str = "epsilon"
    solver_order: int = 2
    steps_offset: int = 1


class _StepOut:
    def __init__(self, prev_sample):
        self.prev_sample = prev_sample


class DPMSolverMultistepScheduler:
    def __init__(self, **kwargs):
        self.config = DPMConfig(**kwargs)
        c = self.config
        self.betas = make_betas(c.num_train_timesteps, c.beta_start, c.beta_end,
                                c.beta_schedule).to(torch.float32)
        self.alphas_cumprod = torch.cumprod(1.0 - self.betas, dim=0)
        self.alpha_t = self.alphas_cumprod.sqrt()
        self.sigma_t = (1 - self.alphas_cumprod).sqrt()
        self.lambda_t = self.alpha_t.log() - self.sigma_t.log()
        self.prediction_type = c.prediction_type
        self.init_noise_sigma = 1.0
        self.timesteps: Optional[torch.Tensor] = None
        self.model_outputs: List[Optional[torch.Tensor]] = []
        self.lower_order_nums = 0
        self._step_index = 0

    def set_timesteps(self, num_inference_steps: int, device=None):
        c = self.config
        # linspace in timestep space, matching diffusers' default
        ts = torch.linspace(0, c.num_train_timesteps - 1, num_inference_steps + 1) \
            .round().long().flip(0)[:-1]
        self.timesteps = ts.to(device) if device is not None else ts
        self.num_inference_steps = num_inference_steps
        self.model_outputs = [None] * c.solver_order
        self.lower_order_nums = 0
        self._step_index = 0

    def scale_model_input(self, sample, timestep=None):
        return sample

    def _to_x0(self, model_output: torch.Tensor, t: int, sample: torch.Tensor):
        a, s = float(self.alpha_t[t]), float(self.sigma_t[t])
        if self.prediction_type == "epsilon":
            cx, cm = 1.0 / a, -s / a
        elif self.prediction_type == "v_prediction":
            cx, cm = a, -s
        else:
            raise ValueError(self.prediction_type)
        if sample.is_cuda:
            return ops.lincomb(sample, model_output, cx, cm)
        return cx * sample.float() + cm * model_output.float()

    def step(self, model_output: torch.Tensor, timestep: int, sample: torch.Tensor,
             generator=None) -> _StepOut:
        i = self._step_index
        t = int(timestep)
        prev_t = int(self.timesteps[i + 1]) if i + 1 < len(self.timesteps) else 0

        x0 = self._to_x0(model_output, t, sample)
        self.model_outputs = self.model_outputs[1:] + [x0]

        lam_t = self.lambda_t[t]
        lam_prev = self.lambda_t[prev_t]
        h = (lam_prev - lam_t).to(sample.device)
        a_prev = self.alpha_t[prev_t].to(sample.device)
        s_prev = self.sigma_t[prev_t].to(sample.device)
        s_t = self.sigma_t[t].to(sample.device)

        K = float(-a_prev * torch.expm1(-h))
        c_x = float(s_prev / s_t)
        # diffusers parity: lower_order_final only drops to first order on
        # the last step when num_inference_steps < 15
        final_first_order = (i + 1 >= len(self.timesteps)
                             and len(self.timesteps) < 15)
        if self.lower_order_nums < 1 or self.model_outputs[-2] is None or final_first_order:
            # first-order (DDIM-like) update in x0-parameterization
            if sample.is_cuda:
                prev = ops.lincomb(sample, x0, c_x, K)
            else:
                prev = c_x * sample.float() + K * x0
        else:
            t_prev2 = int(self.timesteps[i - 1])
            h_last = lam_t - self.lambda_t[t_prev2]
            r = float(h_last / h)
            x0_prev = self.model_outputs[-2]
            g = 1.0 / (2 * r)
            if sample.is_cuda:
                prev = ops.lincomb(sample, x0, c_x, K * (1 + g), x0_prev, -K * g)
            else:
                prev = c_x * sample.float() + K * ((1 + g) * x0 - g * x0_prev)

        self.lower_order_nums = min(self.lower_order_nums + 1, self.config.solver_order)
        self._step_index += 1
        return _StepOut(prev.to(sample.dtype))

    def save_pretrained(self, path):
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        d = asdict(self.config)
        d["_class_name"] = "DPMSolverMultistepScheduler"
        (path / "scheduler_config.json").write_text(json.dumps(d, indent=2))

    @classmethod
    def from_pretrained(cls, path):
        d = json.loads((Path(path) / "scheduler_config.json").read_text())
        d.pop("_class_name", None)
        known = {k: v for k, v in d.items() if k in DPMConfig.__dataclass_fields__}
        return cls(**known)

    @classmethod
    def from_config(cls, scheduler_or_config):
        """Mirror diffusers' `DPMSolverMultistepScheduler.from_config(pipe.scheduler.config)`."""
        cfg = getattr(scheduler_or_config, "config", scheduler_or_config)
        if not isinstance(cfg, dict):
            cfg = asdict(cfg)
        known = {k: v for k, v in cfg.items() if k in DPMConfig.__dataclass_fields__}
        return cls(**known)
