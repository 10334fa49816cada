"""Diffusers-layout checkpoint I/O (safetensors), no diffusers dependency.

Layout contract (/root/reference/diff_train.py:709-728):

    output_dir/checkpoint_{step}/
        model_index.json
        unet/{config.json, diffusion_pytorch_model.safetensors}
        vae/{config.json, diffusion_pytorch_model.safetensors}
        text_encoder/{config.json, model.safetensors}
        tokenizer/...
        scheduler/scheduler_config.json
        state.pt            # NEW vs reference: optimizer/sampler/RNG resume state

The reference cannot resume training (no optimizer state saved); we add
state.pt next to the diffusers layout without breaking it.
"""
from __future__ import annotations

import json
from pathlib import Path

import torch
from safetensors.torch import load_file, save_file


_WEIGHT_NAMES = {
    "UNet2DConditionModel": "diffusion_pytorch_model.safetensors",
    "AutoencoderKL": "diffusion_pytorch_model.safetensors",
    "CLIPTextModel": "model.safetensors",
}


def _weight_file(cfg_json: str) -> str:
    try:
        cls = json.loads(cfg_json).get("_class_name", "")
    except Exception:
        cls = ""
    return _WEIGHT_NAMES.get(cls, "diffusion_pytorch_model.safetensors")


def save_module(module: torch.nn.Module, path: Path, cfg_json: str):
    path = Path(path)
    path.mkdir(parents=True, exist_ok=True)
    (path / "config.json").write_text(cfg_json)
    sd = {k: v.detach().contiguous().cpu() for k, v in module.state_dict().items()}
    save_file(sd, str(path / _weight_file(cfg_json)))


def load_module(module: torch.nn.Module, path: Path, strict: bool = True):
    path = Path(path)
    cfg_json = (path / "config.json").read_text() if (path / "config.json").exists() else "{}"
    wf = path / _weight_file(cfg_json)
    if not wf.exists():
        # tolerate either name, but deterministically: prefer known
        # diffusers/transformers weight filenames, and refuse to guess
        # between multiple unknown candidates (glob order is fs-dependent)
        known = ("diffusion_pytorch_model.safetensors", "model.safetensors")
        cands = sorted(path.glob("*.safetensors"))
        preferred = [c for c in cands if c.name in known]
        if preferred:
            wf = preferred[0]
        elif len(cands) == 1:
            wf = cands[0]
        elif not cands:
            raise FileNotFoundError(f"no safetensors weights under {path}")
        else:
            raise FileNotFoundError(
                f"ambiguous safetensors weights under {path}: "
                f"{[c.name for c in cands]}")
    sd = load_file(str(wf))
    module.load_state_dict(sd, strict=strict)
    return module


def save_pipeline_index(path: Path, scheduler: str = "DDPMScheduler"):
    path = Path(path)
    path.mkdir(parents=True, exist_ok=True)
    index = {
        "_class_name": "StableDiffusionPipeline",
        "_diffusers_version": "dcr_amd-0.1.0",
        "scheduler": ["diffusers", scheduler],
        "text_encoder": ["transformers", "CLIPTextModel"],
        "tokenizer": ["transformers", "CLIPTokenizer"],
        "unet": ["diffusers", "UNet2DConditionModel"],
        "vae": ["diffusers", "AutoencoderKL"],
    }
    (path / "model_index.json").write_text(json.dumps(index, indent=2))
