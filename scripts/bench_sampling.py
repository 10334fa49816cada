#!/usr/bin/env python3
"""Sampling benchmark (BASELINE config 4): 50-step DDIM/DPM generation
throughput at 256px/512px on random-init SD-2.1, bf16, CFG on.

Prints one JSON line per configuration: images/sec (whole-process).
Multi-GPU generation is embarrassingly parallel (diff_inference.py shards
batches across ranks), so 1-GPU numbers scale ~linearly.
"""
from __future__ import annotations

import argparse
import json
import time

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--resolutions", type=int, nargs="+", default=[256, 512])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--repeat", type=int, default=2)
    ap.add_argument("--scheduler", choices=["ddim", "dpm"], default="ddim")
    ap.add_argument("--model", choices=["sd21", "tiny"], default="sd21")
    args = ap.parse_args()

    from dcr_amd.data.tokenizer import HashTokenizer
    from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                                UNet2DConditionModel, UNetConfig, VAEConfig)
    from dcr_amd.pipelines import StableDiffusionPipeline
    from dcr_amd.schedulers import DDIMScheduler, DPMSolverMultistepScheduler

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda" if use_cuda else "cpu")
    if use_cuda:
        torch.backends.cudnn.benchmark = True

    torch.manual_seed(0)
    if args.model == "tiny":
        ucfg, vcfg, tcfg = UNetConfig.tiny(), VAEConfig.tiny(), CLIPTextConfig.tiny()
    else:
        ucfg, vcfg, tcfg = UNetConfig.sd21(), VAEConfig.sd(), CLIPTextConfig.sd21()
    sched = DDIMScheduler() if args.scheduler == "ddim" else DPMSolverMultistepScheduler()
    pipe = StableDiffusionPipeline(
        UNet2DConditionModel(ucfg), AutoencoderKL(vcfg), CLIPTextModel(tcfg),
        HashTokenizer(), sched).to(device)
    if use_cuda:
        for m in (pipe.unet, pipe.vae, pipe.text_encoder):
            m.to(torch.bfloat16)
    pipe.unet.eval(); pipe.vae.eval(); pipe.text_encoder.eval()

    for res in args.resolutions:
        # warmup (MIOpen find for this resolution's conv shapes)
        pipe("warmup prompt", height=res, width=res,
             num_inference_steps=4, num_images_per_prompt=args.batch,
             output_type="pt")
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for r in range(args.repeat):
            pipe(f"bench prompt {r}", height=res, width=res,
                 num_inference_steps=args.steps,
                 num_images_per_prompt=args.batch, output_type="pt")
        if use_cuda:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        n = args.batch * args.repeat
        print(json.dumps({
            "metric": "ddim50_imgs_per_sec" if args.scheduler == "ddim"
                      else "dpm50_imgs_per_sec",
            "value": round(n / dt, 4),
            "resolution": res,
            "batch": args.batch,
            "steps": args.steps,
            "sec_per_image": round(dt / n, 4),
            "dtype": "bf16" if use_cuda else "fp32",
            "model": args.model,
            "cfg": True,
        }))


if __name__ == "__main__":
    main()
