#!/usr/bin/env python3
"""Convert a local diffusers StableDiffusionPipeline checkpoint directory
into this framework's layout (and back).

Because dcr_amd's state-dict key names match diffusers', conversion is a
config translation + weight copy with key validation — no renaming. This
is the "switch from the reference stack" path: a checkpoint trained with
the reference (diffusers save_pretrained) loads here and vice versa.

Usage:
    python scripts/convert_diffusers_checkpoint.py SRC_DIR DST_DIR [--check]
"""
from __future__ import annotations

import argparse
import json
import shutil
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
from safetensors.torch import load_file

from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                            UNet2DConditionModel, UNetConfig, VAEConfig)


def _load_weights(subdir: Path):
    for name in ("diffusion_pytorch_model.safetensors", "model.safetensors"):
        f = subdir / name
        if f.exists():
            return load_file(str(f))
    for f in subdir.glob("*.bin"):
        return torch.load(str(f), map_location="cpu", weights_only=True)
    raise FileNotFoundError(f"no weights in {subdir}")


def convert(src: Path, dst: Path, check: bool = True):
    dst.mkdir(parents=True, exist_ok=True)

    # UNet
    ucfg = UNetConfig.from_json((src / "unet" / "config.json").read_text())
    unet = UNet2DConditionModel(ucfg)
    sd = _load_weights(src / "unet")
    missing, unexpected = unet.load_state_dict(sd, strict=False)
    print(f"unet: {len(sd)} keys, missing={len(missing)} unexpected={len(unexpected)}")
    if check and (missing or unexpected):
        print("  missing:", missing[:10])
        print("  unexpected:", unexpected[:10])
    unet.save_pretrained(dst / "unet")

    # VAE
    vcfg = VAEConfig.from_json((src / "vae" / "config.json").read_text())
    vae = AutoencoderKL(vcfg)
    sd = _load_weights(src / "vae")
    missing, unexpected = vae.load_state_dict(sd, strict=False)
    print(f"vae: {len(sd)} keys, missing={len(missing)} unexpected={len(unexpected)}")
    vae.save_pretrained(dst / "vae")

    # text encoder
    tc_file = src / "text_encoder" / "config.json"
    tcfg_raw = json.loads(tc_file.read_text())
    tcfg = CLIPTextConfig(**{k: tcfg_raw[k] for k in tcfg_raw
                             if k in CLIPTextConfig.__dataclass_fields__})
    te = CLIPTextModel(tcfg)
    sd = _load_weights(src / "text_encoder")
    sd = {k: v for k, v in sd.items()
          if not k.endswith("position_ids")}  # transformers buffer, not a param
    missing, unexpected = te.load_state_dict(sd, strict=False)
    print(f"text_encoder: {len(sd)} keys, missing={len(missing)} "
          f"unexpected={len(unexpected)}")
    te.save_pretrained(dst / "text_encoder")

    # tokenizer + scheduler + index: copy through
    for sub in ("tokenizer", "scheduler"):
        if (src / sub).exists():
            shutil.copytree(src / sub, dst / sub, dirs_exist_ok=True)
    from dcr_amd.models.model_io import save_pipeline_index
    save_pipeline_index(dst)
    print(f"converted -> {dst}")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("src", type=Path)
    ap.add_argument("dst", type=Path)
    ap.add_argument("--check", action="store_true")
    a = ap.parse_args()
    convert(a.src, a.dst, a.check)
