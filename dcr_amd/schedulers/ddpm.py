"""DDPM noise scheduler (training-side math).

Capability parity: diffusers DDPMScheduler as used by the reference
finetune loop (/root/reference/diff_train.py:448,624-632,650):
`add_noise` and `get_velocity` with per-sample timesteps, scaled_linear
betas (SD defaults: beta 0.00085..0.012, 1000 steps). The elementwise
math runs as one HIP kernel on GPU (dcr_amd.ops.add_noise/get_velocity).
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path

import torch

from .. import ops


def make_betas(num_train_timesteps: int, beta_start: float, beta_end: float,
               beta_schedule: str) -> torch.Tensor:
    if beta_schedule == "linear":
        return torch.linspace(beta_start, beta_end, num_train_timesteps, dtype=torch.float64)
    if beta_schedule == "scaled_linear":
        return torch.linspace(beta_start ** 0.5, beta_end ** 0.5, num_train_timesteps,
                              dtype=torch.float64) ** 2
    if beta_schedule == "squaredcos_cap_v2":
        import math
        def alpha_bar(t):
            return math.cos((t + 0.008) / 1.008 * math.pi / 2) ** 2
        betas = []
        for i in range(num_train_timesteps):
            t1, t2 = i / num_train_timesteps, (i + 1) / num_train_timesteps
            betas.append(min(1 - alpha_bar(t2) / alpha_bar(t1), 0.999))
        return torch.tensor(betas, dtype=torch.float64)
    raise ValueError(f"unknown beta schedule {beta_schedule}")


@dataclass
class DDPMConfig:
    num_train_timesteps: int = 1000
    beta_start: float = 0.00085
    beta_end: float = 0.012
    beta_schedule: str = "scaled_linear"
    prediction_type: str = "epsilon"
    clip_sample: bool = False
    steps_offset: int = 1


class DDPMScheduler:
    def __init__(self, **kwargs):
        self.config = DDPMConfig(**kwargs)
        c = self.config
        self.betas = make_betas(c.num_train_timesteps, c.beta_start, c.beta_end,
                                c.beta_schedule).to(torch.float32)
        self.alphas = 1.0 - self.betas
        self.alphas_cumprod = torch.cumprod(self.alphas, dim=0)
        self.num_train_timesteps = c.num_train_timesteps
        self.prediction_type = c.prediction_type
        self.init_noise_sigma = 1.0

    def _ac_on(self, device) -> torch.Tensor:
        # device-cached alphas_cumprod: a per-call pageable H2D copy is
        # both wasteful and hipGraph-capture-illegal (r02c11)
        cache = getattr(self, "_ac_cache", None)
        if cache is None or cache.device != device:
            self._ac_cache = self.alphas_cumprod.to(device, torch.float32)
        return self._ac_cache

    def add_noise(self, original_samples: torch.Tensor, noise: torch.Tensor,
                  timesteps: torch.Tensor) -> torch.Tensor:
        return ops.add_noise(original_samples, noise,
                             self._ac_on(original_samples.device), timesteps)

    def get_velocity(self, sample: torch.Tensor, noise: torch.Tensor,
                     timesteps: torch.Tensor) -> torch.Tensor:
        return ops.get_velocity(sample, noise,
                                self._ac_on(sample.device), timesteps)

    # ancestral DDPM sampling step (used mainly by tests; inference uses DDIM/DPM)
    def step(self, model_output: torch.Tensor, timestep: int, sample: torch.Tensor,
             generator=None):
        t = int(timestep)
        ac_t = self.alphas_cumprod[t]
        ac_prev = self.alphas_cumprod[t - 1] if t > 0 else torch.tensor(1.0)
        beta_t = self.betas[t]
        if self.prediction_type == "epsilon":
            x0 = (sample - (1 - ac_t).sqrt() * model_output) / ac_t.sqrt()
        elif self.prediction_type == "v_prediction":
            x0 = ac_t.sqrt() * sample - (1 - ac_t).sqrt() * model_output
        else:
            raise ValueError(self.prediction_type)
        if self.config.clip_sample:
            x0 = x0.clamp(-1, 1)
        coef_x0 = (ac_prev.sqrt() * beta_t) / (1 - ac_t)
        coef_xt = (self.alphas[t].sqrt() * (1 - ac_prev)) / (1 - ac_t)
        mean = coef_x0 * x0 + coef_xt * sample
        if t > 0:
            var = beta_t * (1 - ac_prev) / (1 - ac_t)
            noise = torch.randn(sample.shape, generator=generator, device=sample.device,
                                dtype=sample.dtype)
            mean = mean + var.clamp(min=1e-20).sqrt() * noise
        return mean

    def save_pretrained(self, path):
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        d = asdict(self.config)
        d["_class_name"] = "DDPMScheduler"
        (path / "scheduler_config.json").write_text(json.dumps(d, indent=2))

    @classmethod
    def from_pretrained(cls, path):
        path = Path(path)
        f = path / "scheduler_config.json"
        d = json.loads(f.read_text())
        d.pop("_class_name", None)
        known = {k: v for k, v in d.items() if k in DDPMConfig.__dataclass_fields__}
        return cls(**known)
