#!/usr/bin/env python3
"""Inference CLI — flag/behavior parity with /root/reference/diff_inference.py.

Generates N batches of images from a finetuned checkpoint (or the stock
SD-2.1 architecture), deriving dataset/caption-style/save-path from the
model path (reference :44-81, :227-239), building the prompt list per
caption mode (:121-170), applying inference-time prompt augmentations
(:171-176) and the Newpipe Gaussian-embedding-noise mitigation, writing
`prompts.txt` (:179-181) and numbered PNGs (:183-201).

MI355X addition (BASELINE config 4): generation is embarrassingly
parallel — under torchrun each rank renders its slice of the batch list
(global PNG numbering preserved), one pipeline per GPU.
"""
from __future__ import annotations

import argparse
import ast
import json
import os

import numpy as np
import torch
from PIL import Image

from dcr_amd.data import get_classnames, insert_rand_word
from dcr_amd.data.tokenizer import load_tokenizer


def resize(w_val, l_val, img):
    return img.resize((w_val, l_val), Image.Resampling.LANCZOS)


def prompt_augmentation(prompt, aug_style, tokenizer=None, repeat_num=2):
    """Inference-time prompt perturbation (semantics of reference
    diff_inference.py:14-30): insert `repeat_num` extra tokens at random
    word boundaries — random integers, random vocabulary words, or words
    already present in the prompt."""
    original_words = prompt.split(" ")

    def pick_word() -> str:
        if aug_style == "rand_numb_add":
            return str(np.random.choice(100000))
        if aug_style == "rand_word_add":
            token_id = int(np.random.randint(49400))
            return tokenizer.decode([token_id])
        if aug_style == "rand_word_repeat":
            return str(np.random.choice(original_words))
        raise Exception(f"unknown prompt augmentation style: {aug_style!r}")

    for _ in range(repeat_num):
        prompt = insert_rand_word(prompt, pick_word())
    return prompt


def derive_savepath(args) -> str:
    """Reference :44-81 savepath derivation (the path IS the config)."""
    if args.modelpath is None:
        savepath = f"./inferences/defaultsd/{args.dataset}/{args.capstyle}"
    else:
        mp = os.path.basename(os.path.normpath(args.modelpath))
        if "traintext" not in args.modelpath:
            if "imagenette" in args.modelpath:
                args.dataset = "imagenette10"
                savepath = f"./inferences/imagenette10_frozentext/{mp}"
            elif "aesthetics" in args.modelpath:
                args.dataset = "laionaesthetics"
                savepath = f"./inferences/laionaesthetics_ft/{mp}"
            elif "laion" in args.modelpath:
                args.dataset = "laion"
                savepath = f"./inferences/laion_frozentext/{mp}"
            elif "l100kaion" in args.modelpath:
                args.dataset = "l100kaion"
                savepath = f"./inferences/l100kaion_frozentext/{mp}"
            else:
                args.dataset = args.dataset or "custom"
                savepath = f"./inferences/{args.dataset}_frozentext/{mp}"
        else:
            if "imagenette" in args.modelpath:
                args.dataset = "imagenette10"
                savepath = f"./inferences/imagenette10_traintext/{mp}"
            elif "laion" in args.modelpath:
                args.dataset = "laion"
                savepath = f"./inferences/laion_traintext/{mp}"
            else:
                args.dataset = args.dataset or "custom"
                savepath = f"./inferences/{args.dataset}_traintext/{mp}"
    if args.iternum is not None:
        savepath = f"{savepath}_{args.iternum}"
    savepath = f"{savepath}/{args.modelstyle}"
    if args.rand_noise_lam is not None:
        savepath = f"{savepath}_ginfer{args.rand_noise_lam}"
    if args.rand_augs is not None:
        savepath = f"{savepath}_auginfer_{args.rand_augs}_{args.rand_aug_repeats}"
    return savepath


def build_prompt_list(args, tokenizer):
    """Reference :121-170."""
    nb = args.nbatches
    if args.modelstyle == "nolevel":
        return ["An image"] * nb
    if args.modelstyle == "classlevel":
        objects = get_classnames("imagenette")
        np.random.seed(args.seed)
        return [f"An image of {x}" for x in np.random.choice(objects, nb)]
    if args.modelstyle in ("instancelevel_blip", "instancelevel_random"):
        json_map = {
            ("imagenette10", "instancelevel_blip"):
                "./data/imagenette2-320/blip_captions.json",
            ("imagenette10", "instancelevel_random"):
                "./data/imagenette2-320/random_captions_4.json",
            ("laionaesthetics", "instancelevel_blip"):
                "./data/laion_10k_random_aesthetics_5plus/laion_aesthetics_combined_captions.json",
            ("laion", "instancelevel_blip"):
                "./data/laion_10k_random/laion_combined_captions.json",
            ("l100kaion", "instancelevel_blip"):
                "./data/laion_100k_random_sdv2p1/l100kaion_combined_captions.json",
        }
        pj = args.prompt_json or json_map.get((args.dataset, args.modelstyle))
        if pj and os.path.exists(pj):
            with open(pj) as f:
                all_prompts = json.load(f)
            okprompts = [v[0] for v in all_prompts.values()]
            if args.dataset == "l100kaion":
                okprompts = okprompts[:10000]
        else:
            # no caption json on disk (no network): synthetic prompt pool
            okprompts = [f"An image of {c} variant {i}"
                         for c in get_classnames("imagenette") for i in range(100)]
        np.random.seed(args.seed)
        prompt_list = list(np.random.choice(okprompts, nb))
        if args.modelstyle == "instancelevel_random":
            prompt_list = [tokenizer.decode(ast.literal_eval(p)) for p in prompt_list]
        return prompt_list
    raise ValueError(f"unknown modelstyle {args.modelstyle}")


def main(args):
    from dcr_amd.parallel import init_distributed_mode
    from dcr_amd.pipelines import StableDiffusionPipeline
    from dcr_amd.schedulers import DPMSolverMultistepScheduler

    rank, world, local = init_distributed_mode(gate_print=False)
    device = torch.device("cuda", local) if torch.cuda.is_available() \
        else torch.device("cpu")

    savepath = derive_savepath(args)
    os.makedirs(savepath, exist_ok=True)
    os.makedirs(f"{savepath}/generations", exist_ok=True)

    if args.modelpath is None:
        checkpath = "stabilityai/stable-diffusion-2-1"
    elif args.iternum is not None:
        checkpath = f"{args.modelpath}/checkpoint_{args.iternum}/"
    else:
        checkpath = f"{args.modelpath}/checkpoint/"

    if os.path.isdir(checkpath):
        pipe = StableDiffusionPipeline.from_pretrained(
            checkpath, embed_noise_lam=args.rand_noise_lam or 0.0)
    else:
        # stock model path: no weights on disk -> random-init SD-2.1 arch
        from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                                    UNet2DConditionModel, UNetConfig, VAEConfig)
        from dcr_amd.data.tokenizer import HashTokenizer
        size = args.model_size
        ucfg = UNetConfig.tiny() if size == "tiny" else UNetConfig.sd21()
        vcfg = VAEConfig.tiny() if size == "tiny" else VAEConfig.sd()
        tcfg = CLIPTextConfig.tiny() if size == "tiny" else CLIPTextConfig.sd21()
        pipe = StableDiffusionPipeline(
            UNet2DConditionModel(ucfg), AutoencoderKL(vcfg), CLIPTextModel(tcfg),
            HashTokenizer(), DPMSolverMultistepScheduler(),
            embed_noise_lam=args.rand_noise_lam or 0.0)
    pipe.scheduler = DPMSolverMultistepScheduler.from_config(pipe.scheduler) \
        if args.modelpath is None else pipe.scheduler
    if torch.cuda.is_available():
        pipe.to(device)
        for m in (pipe.unet, pipe.vae, pipe.text_encoder):
            m.to(torch.bfloat16)
    tokenizer = pipe.tokenizer if args.modelpath is not None else load_tokenizer()

    prompt_list = build_prompt_list(args, tokenizer)
    if args.rand_augs is not None:
        prompt_list = [prompt_augmentation(p, args.rand_augs, tokenizer,
                                           args.rand_aug_repeats)
                       for p in prompt_list]

    if rank == 0:
        with open(f"{savepath}/prompts.txt", "w") as f:
            for line in prompt_list:
                f.write(f"{line}\n")

    generator = torch.Generator(device.type).manual_seed(42 + rank)
    for i in range(args.nbatches):
        if world > 1 and i % world != rank:
            continue  # rank-sharded generation (global numbering kept)
        prompt = str(prompt_list[i])
        images = pipe(prompt=prompt, height=args.resolution, width=args.resolution,
                      num_inference_steps=50, num_images_per_prompt=args.im_batch,
                      generator=generator if args.modelpath is None else None).images
        for j, image in enumerate(images):
            if image.size[0] > args.resolution:
                image = resize(args.resolution, args.resolution, image)
            image.save(f"{savepath}/generations/{i * args.im_batch + j}.png")
    print(f"rank {rank}: wrote generations to {savepath}")


if __name__ == "__main__":
    parser = argparse.ArgumentParser(description="Preprocess images")
    parser.add_argument("--modelpath", type=str, default=None)
    parser.add_argument("--dataset", type=str, default=None)
    parser.add_argument("--capstyle", type=str, default=None)
    parser.add_argument("--captoken", type=str, default=None)
    parser.add_argument("-nb", "--nbatches", type=int, required=True)
    parser.add_argument("-imb", "--im_batch", type=int, default=1)
    parser.add_argument("--resolution", type=int, default=256)
    parser.add_argument("--iternum", default=None, type=int)
    parser.add_argument("--rand_noise_lam", type=float, default=None)
    parser.add_argument("--rand_augs", type=str, default=None)
    parser.add_argument("--rand_aug_repeats", type=int, default=2)
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--prompt_json", type=str, default=None)
    parser.add_argument("--model_size", type=str, default="sd21",
                        choices=["sd21", "tiny"])
    args = parser.parse_args()

    assert not (args.modelpath is None and args.capstyle is None), \
        "Modelpath and caption style cant be None at the same time"
    assert not (args.modelpath is None and args.dataset is None), \
        "Modelpath and Dataset name cant be None at the same time"

    if args.capstyle is not None and args.capstyle in \
            ["nolevel", "classlevel", "instancelevel_blip", "instancelevel_random"]:
        args.modelstyle = args.capstyle
    elif "nolevel" in args.modelpath:
        args.modelstyle = "nolevel"
    elif "classlevel" in args.modelpath:
        args.modelstyle = "classlevel"
    elif "instancelevel_blip" in args.modelpath:
        args.modelstyle = "instancelevel_blip"
    elif "instancelevel_random" in args.modelpath:
        args.modelstyle = "instancelevel_random"
    else:
        args.modelstyle = "nolevel"

    if args.rand_augs:
        assert args.modelstyle == "instancelevel_blip", \
            "Random caption augmentations can only be applied if model is trained on blip captions"
    main(args)
