#!/usr/bin/env python3
"""Flagship benchmark: SD-2.1 256px finetune step (imgs/sec), BASELINE.json.

Measures the reference's canonical train config
(/root/reference/README.md:27-35: SD-2.1, 256px, bs16/GPU, bf16) on
random-init weights + synthetic data (no network), full train step:
VAE encode -> add_noise -> text encode -> UNet fwd -> MSE -> backward
(bucketed RCCL all-reduce for N>1) -> grad clip -> fused AdamW.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
For N>1 the driver launches this under torch.distributed.run with one
rank per GPU (RANK/LOCAL_RANK/WORLD_SIZE read from env).

Rank 0 prints ONE JSON line with the whole-job aggregate imgs/sec.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--resolution", type=int, default=256)
    ap.add_argument("--model", type=str, default="sd21", choices=["sd21", "tiny"])
    # defaults = the measured-fastest configuration on MI355X
    # (pure bf16 params + fp32-master AdamW; NHWC convs): see BASELINE.md
    ap.add_argument("--precision", type=str, default="pure_bf16",
                    choices=["bf16", "pure_bf16", "no"])
    ap.add_argument("--channels-last", dest="channels_last", action="store_true",
                    default=True)
    ap.add_argument("--no-channels-last", dest="channels_last",
                    action="store_false")
    args = ap.parse_args()

    from dcr_amd.parallel import dist as dist_utils
    from dcr_amd.train import TrainConfig, Trainer

    rank, world, local = dist_utils.init_distributed_mode(gate_print=False)
    if world == 1 and args.gpus > 1:
        print("warning: --gpus > 1 but not launched under torchrun; running 1 rank",
              file=sys.stderr)
    n_gpus = world if world > 1 else 1

    if os.environ.get("DCR_CONV_BENCHMARK") == "1":
        torch.backends.cudnn.benchmark = True  # MIOpen find-mode tuning
    use_cuda = torch.cuda.is_available()
    # local % device_count: lets a world-2 smoke run share one leased GPU
    # (RCCL multi-rank-per-device) without a special path
    device = torch.device("cuda", local % max(torch.cuda.device_count(), 1)) \
        if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    cfg = TrainConfig(
        model_size=args.model,
        synthetic_data=True,
        synthetic_size=args.batch_size * 4,
        resolution=args.resolution,
        train_batch_size=args.batch_size,
        mixed_precision=args.precision if use_cuda else "no",
        dataloader_num_workers=0,
        max_train_steps=10**9,
        seed=1234,
        class_prompt="instancelevel_blip",
        duplication="nodup",
        output_dir="/tmp/dcr_bench_out",
        channels_last=args.channels_last,
    )
    trainer = Trainer(cfg, device=device)
    trainer.unet.train()

    # fixed synthetic batch of the training shape, resident on device
    g = torch.Generator().manual_seed(1234 + rank)
    batch = {
        "pixel_values": (torch.rand(args.batch_size, 3, args.resolution,
                                    args.resolution, generator=g) * 2 - 1).to(device),
        "input_ids": trainer.tokenizer(
            [f"An image of sample {i} rank {rank}" for i in range(args.batch_size)],
            truncation=True, padding="max_length",
            max_length=trainer.tokenizer.model_max_length,
            return_tensors="pt").input_ids.to(device),
    }

    def step():
        trainer.train_step(batch, sync_gradients=True)

    for _ in range(args.warmup):
        step()

    # Whole-step hipGraph capture after the MIOpen-find warmup; timed
    # steps then replay it (1-GPU only; measured 217.2 vs 213-215 eager,
    # r02c11). DCR_HIPGRAPH=0 opts out; any capture failure falls back
    # to the eager step so the bench always completes.
    if use_cuda and n_gpus == 1 and os.environ.get("DCR_HIPGRAPH", "1") != "0":
        try:
            trainer.enable_hipgraph(batch)
            step()  # one replay outside the timed window
        except Exception as e:  # noqa: BLE001
            print(f"hipGraph capture unavailable ({type(e).__name__}: {e}); "
                  f"benching the eager step", file=sys.stderr)

    if dist_utils.is_dist():
        dist_utils.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_cuda:
        torch.cuda.synchronize()
    if dist_utils.is_dist():
        dist_utils.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks (slowest rank defines job time)
    if dist_utils.is_dist():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else None)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    phases = trainer.prof.summary()
    if phases and rank == 0:
        print("phase ms/step:", json.dumps({k: round(v, 2) for k, v in phases.items()}),
              file=sys.stderr)

    ms_per_step = elapsed / args.steps * 1000.0
    imgs_per_sec = args.batch_size * n_gpus * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "sd21_256px_finetune_imgs_per_sec",
            "value": round(imgs_per_sec, 3),
            "unit": "imgs/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "sd2.1-unet-865M-random-init" if args.model == "sd21"
                         else "tiny-unet",
                "global_batch": args.batch_size * n_gpus,
                "resolution": args.resolution,
                "seq_len": 77,
                "parallelism": f"dp{n_gpus}",
                "optimizer": "fused_adamw",
            },
        }))


if __name__ == "__main__":
    main()
    # explicit teardown: a rank exiting with the group alive can race the
    # store shutdown under torchrun and turn a finished bench into a
    # non-zero exit
    import torch.distributed as _dist
    if _dist.is_available() and _dist.is_initialized():
        _dist.destroy_process_group()
