#!/usr/bin/env python3
"""Training CLI — flag-compatible with /root/reference/diff_train.py:54-280.

Finetunes SD (default SD-2.1 architecture; random-init when weights are
not on disk — no network here) on an ImageFolder dataset with
caption-conditioning modes, duplication schemes and train-time copying
mitigations. Multi-GPU: one process per GPU under
`python -m torch.distributed.run --nproc-per-node N diff_train.py ...`
(replaces the reference's `accelerate launch`); gradients sync through
dcr_amd's bucketed RCCL all-reduce.
"""
from __future__ import annotations

import argparse
import os


def parse_args():
    p = argparse.ArgumentParser(description="DCR-AMD Stable Diffusion finetune")
    p.add_argument("--pretrained_model_name_or_path", type=str,
                   default="stabilityai/stable-diffusion-2-1")
    p.add_argument("--revision", type=str, default=None)
    p.add_argument("--tokenizer_name", type=str, default=None)
    p.add_argument("--instance_data_dir", type=str, default=None)
    p.add_argument("--instance_prompt_loc", type=str, default=None,
                   help="path to the caption json (BLIP/random captions)")
    p.add_argument("--class_prompt", type=str, default="nolevel",
                   choices=["nolevel", "classlevel", "instancelevel_blip",
                            "instancelevel_random", "instancelevel_ogcap"])
    p.add_argument("--class_data_dir", type=str, default=None)
    p.add_argument("--num_class_images", type=int, default=100)
    p.add_argument("--seed", type=int, default=None)
    p.add_argument("--generation_seed", type=int, default=1024)
    p.add_argument("--resolution", type=int, default=256)
    p.add_argument("--center_crop", action="store_true")
    p.add_argument("--random_flip", action="store_true")
    p.add_argument("--train_text_encoder", action="store_true")
    p.add_argument("--train_batch_size", type=int, default=4)
    p.add_argument("--sample_batch_size", type=int, default=4)
    p.add_argument("--num_train_epochs", type=int, default=1)
    p.add_argument("--max_train_steps", type=int, default=None)
    p.add_argument("--save_steps", type=int, default=500)
    p.add_argument("--gradient_accumulation_steps", type=int, default=1)
    p.add_argument("--gradient_checkpointing", action="store_true")
    p.add_argument("--learning_rate", type=float, default=5e-6)
    p.add_argument("--scale_lr", action="store_true")
    p.add_argument("--lr_scheduler", type=str, default="constant",
                   choices=["linear", "cosine", "cosine_with_restarts",
                            "polynomial", "constant", "constant_with_warmup"])
    p.add_argument("--lr_warmup_steps", type=int, default=500)
    p.add_argument("--adam_beta1", type=float, default=0.9)
    p.add_argument("--adam_beta2", type=float, default=0.999)
    p.add_argument("--adam_weight_decay", type=float, default=1e-2)
    p.add_argument("--adam_epsilon", type=float, default=1e-08)
    p.add_argument("--max_grad_norm", type=float, default=1.0)
    p.add_argument("--push_to_hub", action="store_true")
    p.add_argument("--hub_token", type=str, default=None)
    p.add_argument("--hub_model_id", type=str, default=None)
    p.add_argument("--logging_dir", type=str, default="logs")
    p.add_argument("--mixed_precision", type=str, default="bf16",
                   choices=["no", "fp16", "bf16", "pure_bf16"])
    p.add_argument("--channels_last", action="store_true",
                   help="NHWC convs + NHWC GroupNorm kernels (fastest on MI355X)")
    p.add_argument("--local_rank", type=int, default=-1)
    p.add_argument("-j", "--num_workers", type=int, default=4)
    p.add_argument("--modelsavesteps", type=int, default=1000)
    p.add_argument("--output_dir", type=str, default="model_out")
    p.add_argument("--resume", type=str, default=None,
                   help="'auto' = latest checkpoint_*/ (or checkpoint/) in "
                        "the mangled output dir; or an explicit checkpoint "
                        "directory. Restores weights, optimizer state, step "
                        "count and RNG streams (state.pt).")
    p.add_argument("--duplication", type=str, default="nodup",
                   choices=["nodup", "dup_both", "dup_image"])
    p.add_argument("--weight_pc", type=float, default=0.05)
    p.add_argument("--dup_weight", type=float, default=5.0)
    p.add_argument("--rand_noise_lam", type=float, default=0.0)
    p.add_argument("--mixup_noise_lam", type=float, default=0.0)
    p.add_argument("--trainspecial", type=str, default=None,
                   choices=["allcaps", "randrepl", "randwordadd", "wordrepeat"])
    p.add_argument("--trainspecial_prob", type=float, default=0.5)
    p.add_argument("--trainsubset", type=float, default=None)
    p.add_argument("--unet_from_scratch", type=str, default="no",
                   choices=["no", "yes"])
    p.add_argument("--unet_config", type=str, default="./unet_config.json")
    p.add_argument("--model_size", type=str, default="sd21",
                   choices=["sd21", "tiny"],
                   help="architecture family when weights are not local")
    p.add_argument("--synthetic_data", action="store_true",
                   help="random images/captions instead of instance_data_dir")

    args = p.parse_args()
    env_local_rank = int(os.environ.get("LOCAL_RANK", -1))
    if env_local_rank not in (-1, args.local_rank):
        args.local_rank = env_local_rank
    return args


def main():
    args = parse_args()
    from dcr_amd.parallel import init_distributed_mode, is_main_process
    from dcr_amd.train import TrainConfig, Trainer, mangle_output_dir, validate
    from dcr_amd.utils.image import concat_h

    init_distributed_mode()

    cfg = TrainConfig(
        pretrained_model_name_or_path=args.pretrained_model_name_or_path,
        tokenizer_name=args.tokenizer_name,
        revision=args.revision,
        unet_from_scratch=args.unet_from_scratch,
        unet_config=args.unet_config if os.path.exists(args.unet_config) else None,
        model_size=args.model_size,
        train_text_encoder=args.train_text_encoder,
        instance_data_dir=args.instance_data_dir,
        synthetic_data=args.synthetic_data or args.instance_data_dir is None,
        class_prompt=args.class_prompt,
        prompt_json=args.instance_prompt_loc,
        duplication=args.duplication,
        weight_pc=args.weight_pc,
        dup_weight=args.dup_weight,
        trainspecial=args.trainspecial,
        trainspecial_prob=args.trainspecial_prob,
        trainsubset=args.trainsubset,
        resolution=args.resolution,
        center_crop=args.center_crop,
        random_flip=args.random_flip,
        dataloader_num_workers=args.num_workers,
        train_batch_size=args.train_batch_size,
        num_train_epochs=args.num_train_epochs,
        max_train_steps=args.max_train_steps or 100000,
        gradient_accumulation_steps=args.gradient_accumulation_steps,
        gradient_checkpointing=args.gradient_checkpointing,
        learning_rate=args.learning_rate,
        scale_lr=args.scale_lr,
        lr_scheduler=args.lr_scheduler,
        lr_warmup_steps=args.lr_warmup_steps,
        adam_beta1=args.adam_beta1,
        adam_beta2=args.adam_beta2,
        adam_weight_decay=args.adam_weight_decay,
        adam_epsilon=args.adam_epsilon,
        max_grad_norm=args.max_grad_norm,
        mixed_precision=args.mixed_precision,
        channels_last=args.channels_last,
        seed=args.seed,
        rand_noise_lam=args.rand_noise_lam,
        mixup_noise_lam=args.mixup_noise_lam,
        output_dir=args.output_dir,
        save_steps=args.save_steps,
        modelsavesteps=args.modelsavesteps,
        generation_seed=args.generation_seed,
    )
    validate(cfg)
    cfg.output_dir = mangle_output_dir(cfg)
    if is_main_process():
        print(f"output_dir: {cfg.output_dir}")

    trainer = Trainer(cfg)

    if args.resume:
        from pathlib import Path as _P
        if args.resume == "auto":
            root = _P(cfg.output_dir)
            cands = sorted(root.glob("checkpoint_*"),
                           key=lambda p: int(p.name.split("_")[-1]))
            ck = cands[-1] if cands else root / "checkpoint"
        else:
            ck = _P(args.resume)
        if (ck / "state.pt").exists() or (ck / "unet").exists():
            trainer.load_checkpoint(ck)
            if is_main_process():
                print(f"resumed from {ck} at step {trainer.global_step}")
        elif args.resume != "auto":
            raise SystemExit(f"--resume: no checkpoint at {ck}")
        elif is_main_process():
            print("--resume auto: no checkpoint found, starting fresh")

    def sample_fn(tr: Trainer, out_dir):
        """Periodic sample grids (reference diff_train.py:571-611,673-701)."""
        import torch
        from dcr_amd.pipelines import StableDiffusionPipeline
        from dcr_amd.schedulers import DDIMScheduler
        pipe = StableDiffusionPipeline(tr.unet, tr.vae, tr.text_encoder,
                                       tr.tokenizer, DDIMScheduler())
        was_training = tr.unet.training
        tr.unet.eval()
        import numpy as np
        objects = tr.dataset.objects if hasattr(tr.dataset, "objects") else \
            ["church", "garbage truck", "tench"]
        # instancelevel modes sample prompts from the caption pool
        # (reference diff_train.py:582-591)
        rand_prompts = None
        prompts_map = getattr(tr.dataset, "prompts", None)
        if cfg.class_prompt.startswith("instancelevel") and prompts_map:
            choicelist = [v[0] for v in prompts_map.values()]
            rand_prompts = list(np.random.choice(choicelist, 3))
        genseed = cfg.generation_seed
        for count, obj in enumerate(objects):
            if count > 2:
                break
            if cfg.class_prompt == "nolevel":
                genseed += 1
                prompt = "An image"
            elif cfg.class_prompt == "classlevel":
                prompt = f"An image of {obj}"
            elif rand_prompts is not None:
                prompt = str(rand_prompts[count])
            else:
                prompt = f"An image of {obj}"
            gen = torch.Generator(tr.device.type).manual_seed(genseed)
            images = pipe(prompt=prompt, height=cfg.resolution, width=cfg.resolution,
                          num_inference_steps=50, num_images_per_prompt=4,
                          generator=gen).images
            concat_h(images).save(
                os.path.join(out_dir, "generations", f"{tr.global_step:04d}_{obj}.png"))
        if was_training:
            tr.unet.train()

    trainer.fit(sample_fn=sample_fn)


if __name__ == "__main__":
    main()
