"""StableDiffusionPipeline equivalent (text→image sampling), MI355X-native.

Capability parity: diffusers StableDiffusionPipeline as used by
/root/reference/diff_inference.py:85-106,190-193 and sd_mitigation.py —
50-step sampling with classifier-free guidance, plus the `Newpipe`
Gaussian-embedding-noise mitigation (diff_inference.py:3-6: the prompt
embedding gets `lam * randn_like(e)` added before the denoise loop).

CFG batching doubles the UNet batch (uncond ++ cond) like the reference
stack; the combine is one HIP kernel (ops.cfg_combine).
"""
from __future__ import annotations

from pathlib import Path
from typing import List, Optional, Union

import torch

from .. import ops
from ..data.tokenizer import load_tokenizer
from ..models import AutoencoderKL, CLIPTextModel, UNet2DConditionModel
from ..schedulers import DDIMScheduler, DPMSolverMultistepScheduler
from ..utils.image import tensor_to_pil


class _PipeOut:
    def __init__(self, images):
        self.images = images


class StableDiffusionPipeline:
    def __init__(self, unet, vae, text_encoder, tokenizer, scheduler,
                 embed_noise_lam: float = 0.0):
        self.unet = unet
        self.vae = vae
        self.text_encoder = text_encoder
        self.tokenizer = tokenizer
        self.scheduler = scheduler
        self.embed_noise_lam = embed_noise_lam
        self._device = next(unet.parameters()).device

    # ------------------------------------------------------------------
    @classmethod
    def from_pretrained(cls, path, scheduler: Optional[object] = None,
                        torch_dtype: Optional[torch.dtype] = None,
                        embed_noise_lam: float = 0.0):
        path = Path(path)
        unet = UNet2DConditionModel.from_pretrained(path / "unet")
        vae = AutoencoderKL.from_pretrained(path / "vae")
        te = CLIPTextModel.from_pretrained(path / "text_encoder")
        tok = load_tokenizer(path / "tokenizer")
        if scheduler is None:
            sched_dir = path / "scheduler"
            if sched_dir.exists():
                import json
                cls_name = json.loads(
                    (sched_dir / "scheduler_config.json").read_text()).get("_class_name")
                sched_cls = {"DDIMScheduler": DDIMScheduler,
                             "DPMSolverMultistepScheduler": DPMSolverMultistepScheduler,
                             "DDPMScheduler": DDIMScheduler}.get(cls_name, DDIMScheduler)
                scheduler = sched_cls.from_pretrained(sched_dir)
            else:
                scheduler = DDIMScheduler()
        if torch_dtype is not None:
            unet.to(torch_dtype)
            vae.to(torch_dtype)
            te.to(torch_dtype)
        pipe = cls(unet, vae, te, tok, scheduler, embed_noise_lam)
        return pipe

    def to(self, device, channels_last: bool = True):
        device = torch.device(device)
        self.unet.to(device)
        self.vae.to(device)
        self.text_encoder.to(device)
        if device.type == "cuda":
            torch.backends.cudnn.benchmark = True
            if channels_last:
                # NHWC convs + NHWC GroupNorm kernels
                self.unet.to(memory_format=torch.channels_last)
                self.vae.to(memory_format=torch.channels_last)
        self._device = device
        return self

    @property
    def device(self):
        return self._device

    # ------------------------------------------------------------------
    def _encode_prompt(self, prompt: List[str], num_images_per_prompt: int,
                       do_cfg: bool, generator=None):
        tok = self.tokenizer(prompt, truncation=True, padding="max_length",
                             max_length=self.tokenizer.model_max_length,
                             return_tensors="pt")
        with torch.no_grad():
            emb = self.text_encoder(tok.input_ids.to(self.device))[0]
        if self.embed_noise_lam > 0:
            # Newpipe mitigation (diff_inference.py:3-6)
            noise = torch.randn(emb.shape, generator=generator,
                                device=emb.device, dtype=torch.float32).to(emb.dtype)
            emb = emb + self.embed_noise_lam * noise
        emb = emb.repeat_interleave(num_images_per_prompt, dim=0)
        if not do_cfg:
            return emb
        uncond = self.tokenizer([""] * len(prompt), truncation=True,
                                padding="max_length",
                                max_length=self.tokenizer.model_max_length,
                                return_tensors="pt")
        with torch.no_grad():
            uemb = self.text_encoder(uncond.input_ids.to(self.device))[0]
        uemb = uemb.repeat_interleave(num_images_per_prompt, dim=0)
        return torch.cat([uemb, emb], dim=0)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def __call__(
        self,
        prompt: Union[str, List[str]],
        height: int = 256,
        width: int = 256,
        num_inference_steps: int = 50,
        guidance_scale: float = 7.5,
        num_images_per_prompt: int = 1,
        generator: Optional[torch.Generator] = None,
        latents: Optional[torch.Tensor] = None,
        output_type: str = "pil",
    ) -> _PipeOut:
        prompts = [prompt] if isinstance(prompt, str) else list(prompt)
        do_cfg = guidance_scale > 1.0
        emb = self._encode_prompt(prompts, num_images_per_prompt, do_cfg, generator)

        n = len(prompts) * num_images_per_prompt
        lc = self.vae.config.latent_channels
        lh, lw = height // 8, width // 8
        dtype = self.unet.dtype
        if latents is None:
            latents = torch.randn((n, lc, lh, lw), generator=generator,
                                  device=self.device, dtype=torch.float32).to(dtype)
        if self.device.type == "cuda" and \
                self.unet.conv_in.weight.is_contiguous(memory_format=torch.channels_last):
            latents = latents.to(memory_format=torch.channels_last)
        latents = latents * self.scheduler.init_noise_sigma

        self.scheduler.set_timesteps(num_inference_steps, device=self.device)
        for t in self.scheduler.timesteps:
            model_in = torch.cat([latents] * 2) if do_cfg else latents
            model_in = self.scheduler.scale_model_input(model_in, t)
            eps = self.unet(model_in, t.expand(model_in.shape[0]).to(self.device),
                            emb.to(dtype))
            if do_cfg:
                eps_u, eps_t = eps.chunk(2)
                eps = ops.cfg_combine(eps_u, eps_t, guidance_scale)
            latents = self.scheduler.step(eps, int(t), latents).prev_sample.to(dtype)

        images = self.vae.decode(latents / self.vae.config.scaling_factor).sample
        images = (images.float() / 2 + 0.5).clamp(0, 1)
        if output_type == "pil":
            return _PipeOut([tensor_to_pil(im * 2 - 1) for im in images])
        return _PipeOut(images)

    # ------------------------------------------------------------------
    def save_pretrained(self, path):
        from ..models.model_io import save_pipeline_index
        path = Path(path)
        self.unet.save_pretrained(path / "unet")
        self.vae.save_pretrained(path / "vae")
        self.text_encoder.save_pretrained(path / "text_encoder")
        self.tokenizer.save_pretrained(path / "tokenizer")
        self.scheduler.save_pretrained(path / "scheduler")
        save_pipeline_index(path)


def Newpipe(path, lam: float = 0.1, **kw):
    """Reference Newpipe: StableDiffusionPipeline with embedding noise."""
    return StableDiffusionPipeline.from_pretrained(path, embed_noise_lam=lam, **kw)
