#!/usr/bin/env python3
"""Inference-time mitigation CLI — parity with /root/reference/sd_mitigation.py.

Reproduces known-replicating generations on the SD-1.4 architecture with
the 12 hardcoded LAION prompts (reference :81), optionally with the
Gaussian-embedding-noise mitigation (Newpipe) or prompt augmentation,
DPM-Solver++ 50 steps. No network: weights are random-init SD-1.4 unless
a local checkpoint directory is passed via --modelpath.
"""
from __future__ import annotations

import argparse
import os

import torch

from diff_inference import prompt_augmentation, resize  # shared helpers (dedup of
# the reference's copy at sd_mitigation.py:3-30)

PROMPT_LIST = [
    "Wall View 002", "Wall View 003", "Chamberly - Alloy 5 Piece Sectional",
    "Hopped-Up Gaming: East", "Pantomine - Driftwood 4 Piece Sectional",
    "Cresson - Pewter 4 Piece Sectional",
    "Jinllingsly - Chocolate 3 Piece Sectional",
    "Maier - Charcoal 2 Piece Sectional", "Classic Cars for Sale",
    "Mothers influence on her young hippo",
    "Living in the Light with Ann Graham Lotz",
    "The No Limits Business Woman Podcast",
]


def main(args):
    from dcr_amd.pipelines import StableDiffusionPipeline
    from dcr_amd.schedulers import DPMSolverMultistepScheduler
    from dcr_amd.data.tokenizer import HashTokenizer

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    if args.modelpath and os.path.isdir(args.modelpath):
        pipe = StableDiffusionPipeline.from_pretrained(
            args.modelpath, scheduler=DPMSolverMultistepScheduler(),
            embed_noise_lam=args.rand_noise_lam or 0.0)
        tokenizer = pipe.tokenizer
    else:
        from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                                    UNet2DConditionModel, UNetConfig, VAEConfig)
        if args.model_size == "tiny":
            ucfg, vcfg, tcfg = UNetConfig.tiny(), VAEConfig.tiny(), CLIPTextConfig.tiny()
        else:  # SD-1.4 (reference checkpath CompVis/stable-diffusion-v1-4)
            ucfg, vcfg, tcfg = UNetConfig.sd14(), VAEConfig.sd(), CLIPTextConfig.sd14()
        tokenizer = HashTokenizer()
        pipe = StableDiffusionPipeline(
            UNet2DConditionModel(ucfg), AutoencoderKL(vcfg), CLIPTextModel(tcfg),
            tokenizer, DPMSolverMultistepScheduler(),
            embed_noise_lam=args.rand_noise_lam or 0.0)
    if device.type == "cuda":
        pipe.to(device)
        for m in (pipe.unet, pipe.vae, pipe.text_encoder):
            m.to(torch.bfloat16)
    generator = torch.Generator(device.type).manual_seed(args.seed)

    savepath = f"./mitigationSD/inf_{args.seed}/gen"
    if args.rand_noise_lam is not None:
        savepath = f"{savepath}_ginfer{args.rand_noise_lam}"
    elif args.rand_augs is not None:
        savepath = f"{savepath}_auginfer_{args.rand_augs}_{args.rand_aug_repeats}"
    else:
        savepath = f"{savepath}_nomit"
    os.makedirs(f"{savepath}/generations", exist_ok=True)

    prompt_list = list(PROMPT_LIST)
    if args.rand_augs is not None:
        prompt_list = [prompt_augmentation(p, args.rand_augs, tokenizer,
                                           args.rand_aug_repeats)
                       for p in prompt_list]
    with open(f"{savepath}/prompts.txt", "w") as f:
        for line in prompt_list:
            f.write(f"{line}\n")

    count = 0
    for prompt in prompt_list:
        for _ in range(args.nbatches):
            images = pipe(prompt, height=args.resolution, width=args.resolution,
                          num_inference_steps=50,
                          num_images_per_prompt=args.im_batch,
                          generator=generator).images
            for image in images:
                if image.size[0] > args.resolution:
                    image = resize(args.resolution, args.resolution, image)
                image.save(f"{savepath}/generations/{count}.png")
                count += 1
    print(f"wrote {count} generations to {savepath}")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--synset_map", type=str, default=None,
                        help="(reference flag; unused)")
    parser.add_argument("-nb", "--nbatches", type=int, default=1,
                        help="generation rounds per prompt")
    parser.add_argument("-imb", "--im_batch", type=int, default=1)
    parser.add_argument("--resolution", type=int, default=512)
    parser.add_argument("--iternum", default=None, type=int,
                        help="(reference flag; unused)")
    parser.add_argument("--rand_noise_lam", type=float, default=None)
    parser.add_argument("--rand_augs", type=str, default=None)
    parser.add_argument("--rand_aug_repeats", type=int, default=4)
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--modelpath", type=str, default=None)
    parser.add_argument("--model_size", type=str, default="sd14",
                        choices=["sd14", "tiny"])
    main(parser.parse_args())
