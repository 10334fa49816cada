"""Typed training config + CLI-compatible output-dir mangling.

Flag surface parity: /root/reference/diff_train.py:54-280 (argparse) and
:736-764 (output-dir name mangling — downstream tools parse the config
back out of the path, so the format is a compatibility contract).
"""
from __future__ import annotations

import math
from dataclasses import dataclass, asdict
from typing import Optional


@dataclass
class TrainConfig:
    # model
    pretrained_model_name_or_path: str = "stabilityai/stable-diffusion-2-1"
    revision: Optional[str] = None
    tokenizer_name: Optional[str] = None    # explicit tokenizer dir (diff_train.py:371)
    unet_from_scratch: str = "no"          # 'yes' => random-init UNet (BASELINE path)
    unet_config: Optional[str] = None       # ./unet_config.json when from scratch
    model_size: str = "sd21"                # sd21 | tiny (tiny = CPU tests)
    train_text_encoder: bool = False
    prediction_type: Optional[str] = None   # override scheduler (epsilon/v_prediction)

    # data
    instance_data_dir: Optional[str] = None
    synthetic_data: bool = False            # random images (no-network benches)
    synthetic_size: int = 1024
    class_prompt: str = "instancelevel_blip"
    prompt_json: Optional[str] = None
    duplication: str = "nodup"              # nodup | dup_both | dup_image
    weight_pc: float = 0.05
    dup_weight: float = 5.0
    trainspecial: Optional[str] = None      # allcaps|randrepl|randwordadd|wordrepeat
    trainspecial_prob: float = 0.5
    trainsubset: Optional[int] = None
    resolution: int = 256
    center_crop: bool = True
    random_flip: bool = True
    dataloader_num_workers: int = 4

    # optimization
    train_batch_size: int = 16
    num_train_epochs: int = 1
    max_train_steps: Optional[int] = 100000
    gradient_accumulation_steps: int = 1
    gradient_checkpointing: bool = False
    learning_rate: float = 5e-6
    scale_lr: bool = False
    lr_scheduler: str = "constant_with_warmup"
    lr_warmup_steps: int = 5000
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_weight_decay: float = 1e-2
    adam_epsilon: float = 1e-8
    max_grad_norm: float = 1.0
    mixed_precision: str = "bf16"           # no | fp16 | bf16 | pure_bf16
    channels_last: bool = False             # NHWC convs (MI355X igemm path)
    seed: Optional[int] = None

    # mitigations (train-time)
    rand_noise_lam: float = 0.0             # gaussian embedding noise
    mixup_noise_lam: float = 0.0            # embedding mixup, lam ~ Beta(a, 1)

    # io / logging
    output_dir: str = "model_out"
    save_steps: int = 500                   # sample-grid cadence
    modelsavesteps: int = 2000              # checkpoint cadence
    generation_seed: int = 1000
    log_every: int = 10
    project: str = "diffrep_ft"

    # distributed
    ddp_bucket_mb: float = 64.0

    def to_dict(self):
        return asdict(self)


def mangle_output_dir(cfg: TrainConfig) -> str:
    """Reference: diff_train.py:745-760 — exact naming contract."""
    out = cfg.output_dir
    if cfg.trainsubset is not None:
        out = f"{out}_{cfg.trainsubset}subset"
    if cfg.unet_from_scratch == "no":
        out = f"{out}_{cfg.class_prompt}_{cfg.duplication}"
    else:
        out = f"{out}_{cfg.class_prompt}_{cfg.duplication}_unetfromscr"
    if cfg.duplication in ("dup_both", "dup_image"):
        out = f"{out}_{cfg.weight_pc}_{cfg.dup_weight}"
    if cfg.rand_noise_lam > 0:
        out = f"{out}_glam{cfg.rand_noise_lam}"
    if cfg.mixup_noise_lam > 0:
        out = f"{out}_mixlam{cfg.mixup_noise_lam}"
    if cfg.trainspecial is not None:
        out = f"{out}_special_{cfg.trainspecial}_{cfg.trainspecial_prob}"
    return out


def validate(cfg: TrainConfig):
    """Reference arg asserts: diff_train.py:739-743."""
    if cfg.duplication == "dup_image" and cfg.class_prompt == "instancelevel_ogcap":
        raise AssertionError(
            "Duplicating just the image in original captions scenario is not acceptable")
    if cfg.trainspecial and cfg.class_prompt != "instancelevel_blip":
        raise Exception("Cant train special without blip captions")


def get_lr(cfg: TrainConfig, step: int, world_size: int = 1) -> float:
    """LR schedules: all six reference choices (diff_train.py:178-189 via
    diffusers get_scheduler, called with only warmup/total steps at
    diff_train.py:504-509 — so cosine_with_restarts uses 1 cycle and
    polynomial uses power=1, lr_end=1e-7, the library defaults).
    scale_lr multiplies the base lr by accum*batch*world
    (diff_train.py:419-422)."""
    base = cfg.learning_rate
    if cfg.scale_lr:
        base *= cfg.gradient_accumulation_steps * cfg.train_batch_size * world_size
    total = cfg.max_train_steps or 1
    warm = cfg.lr_warmup_steps
    s = cfg.lr_scheduler
    if s == "constant":
        return base
    if s == "constant_with_warmup":
        return base * min(1.0, (step + 1) / max(1, warm))
    if s == "linear":
        if step < warm:
            return base * (step + 1) / max(1, warm)
        return base * max(0.0, (total - step) / max(1, total - warm))
    if s == "cosine":
        if step < warm:
            return base * (step + 1) / max(1, warm)
        prog = (step - warm) / max(1, total - warm)
        return base * 0.5 * (1 + math.cos(math.pi * min(1.0, prog)))
    if s == "cosine_with_restarts":
        if step < warm:
            return base * (step + 1) / max(1, warm)
        prog = (step - warm) / max(1, total - warm)
        if prog >= 1.0:
            return 0.0
        num_cycles = 1
        return base * max(0.0, 0.5 * (1 + math.cos(math.pi * ((num_cycles * prog) % 1.0))))
    if s == "polynomial":
        lr_end, power = 1e-7, 1.0
        if step < warm:
            return base * (step + 1) / max(1, warm)
        if step > total:
            return lr_end
        remaining = 1 - (step - warm) / max(1, total - warm)
        return (base - lr_end) * remaining ** power + lr_end
    raise ValueError(f"unknown lr_scheduler {s}")
