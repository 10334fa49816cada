"""Training/retrieval datasets with the reference's caption + duplication
semantics, torchvision-free.

Capability parity (/root/reference/datasets.py, diff_retrieval.py:61-111):

* ``ObjectAttributeDataset`` — ImageFolder with caption modes
  {nolevel, classlevel, instancelevel_blip, instancelevel_random,
  instancelevel_ogcap}, duplication schemes {nodup, dup_both, dup_image}
  with pickle-cached sampling weights
  (weights_{pc}_{w}_seed{s}.pickle, datasets.py:76-90), and train-time
  caption interventions {allcaps, randrepl, randwordadd, wordrepeat}.
* ``SynthDataset`` — flat image dir + prompts.txt (generation outputs).
* ``SyntheticImageDataset`` — random images/captions for benchmarks
  (no network ⇒ synthetic data, BASELINE.json configs).
"""
from __future__ import annotations

import ast
import json
import pickle
import random
from pathlib import Path
from typing import List, Optional

import numpy as np
import torch
from PIL import Image
from torch.utils.data import Dataset

from .transforms import TrainTransform, EvalTransform

IMG_EXTS = {".jpg", ".jpeg", ".png", ".bmp", ".webp", ".tiff"}

IMAGENETTE_CLASSES = [
    "tench", "English springer", "cassette player", "chain saw", "church",
    "French horn", "garbage truck", "gas pump", "golf ball", "parachute",
]


def get_classnames(datasetpath: str) -> List[str]:
    """Reference: datasets.py:25-29 — hardcoded Imagenette class lists."""
    if "imagenette_2class" in str(datasetpath):
        return ["church", "garbage truck"]
    return list(IMAGENETTE_CLASSES)


def scan_image_folder(root) -> tuple[list[tuple[str, int]], list[str]]:
    """torchvision.ImageFolder semantics: class-per-subdir, sorted."""
    root = Path(root)
    classes = sorted(d.name for d in root.iterdir() if d.is_dir())
    samples = []
    for idx, cls in enumerate(classes):
        for f in sorted((root / cls).rglob("*")):
            if f.suffix.lower() in IMG_EXTS:
                samples.append((str(f), idx))
    if not samples:
        raise FileNotFoundError(f"no class-subdir images under {root}")
    return samples, classes


def insert_rand_word(sentence: str, word: str, rng: Optional[random.Random] = None) -> str:
    rng = rng or random
    sent_list = sentence.split(" ")
    sent_list.insert(rng.randint(0, len(sent_list)), word)
    return " ".join(sent_list)


class ObjectAttributeDataset(Dataset):
    def __init__(
        self,
        instance_data_root: str,
        tokenizer,
        class_prompt: Optional[str] = None,
        size: int = 320,
        center_crop: bool = False,
        random_flip: bool = False,
        prompt_json: Optional[str] = None,
        duplication: str = "nodup",
        trainspecial: Optional[str] = None,
        trainspecial_prob: float = 0.5,
        weight_pc: float = 0.05,
        dup_weight: float = 5.0,
        seed: Optional[int] = None,
    ):
        self.root = str(instance_data_root)
        self.samples, self.classes = scan_image_folder(instance_data_root)
        self.tokenizer = tokenizer
        self.class_prompt = class_prompt
        self.size = size
        self.duplication = duplication
        self.trainspecial = trainspecial
        self.trainspecial_prob = trainspecial_prob
        self.objects = get_classnames(self.root)
        self.image_transforms = TrainTransform(size, center_crop, random_flip)

        self.prompts = None
        if class_prompt in ("instancelevel_blip", "instancelevel_ogcap", "instancelevel_random") \
                or trainspecial is not None:
            if prompt_json is None:
                raise ValueError(f"class_prompt={class_prompt} needs prompt_json")
            with open(prompt_json) as f:
                self.prompts = json.load(f)

        self.samplingweights = None
        if duplication in ("dup_both", "dup_image"):
            sw_path = Path(self.root) / f"weights_{weight_pc}_{dup_weight}_seed{seed}.pickle"
            if sw_path.exists():
                with open(sw_path, "rb") as fh:
                    self.samplingweights = pickle.load(fh)
            else:
                w = [1] * len(self.samples)
                n_dup = int(weight_pc * len(self.samples))
                chosen = np.random.choice(len(self.samples), n_dup, replace=False)
                for i in chosen:
                    w[i] = w[i] * dup_weight // 1
                with open(sw_path, "wb") as fh:
                    pickle.dump(w, fh, protocol=pickle.HIGHEST_PROTOCOL)
                self.samplingweights = w

    def __len__(self):
        return len(self.samples)

    def _prompt_for(self, path_img: str, label: int, index: int) -> str:
        if self.trainspecial is not None:
            caps = self.prompts[path_img]
            if self.trainspecial == "allcaps":
                return str(np.random.choice(caps, 1)[0])
            prompt = caps[0]
            if self.trainspecial == "randrepl":
                if np.random.uniform() <= self.trainspecial_prob:
                    ids = list(np.random.randint(49400, size=4))
                    prompt = self.tokenizer.decode(ids)
            elif self.trainspecial == "randwordadd":
                if np.random.uniform() <= self.trainspecial_prob:
                    for _ in range(2):
                        rw = self.tokenizer.decode(list(np.random.randint(49400, size=1)))
                        prompt = insert_rand_word(prompt, rw)
            elif self.trainspecial == "wordrepeat":
                if np.random.uniform() <= self.trainspecial_prob:
                    wl = prompt.split(" ")
                    for _ in range(2):
                        prompt = insert_rand_word(prompt, str(np.random.choice(wl)))
            return prompt

        if self.class_prompt == "nolevel":
            return "An image"
        if self.class_prompt == "classlevel":
            return f"An image of {self.objects[label]}"
        if self.class_prompt in ("instancelevel_blip", "instancelevel_random",
                                 "instancelevel_ogcap"):
            if self.duplication == "dup_image" and self.samplingweights is not None \
                    and self.samplingweights[index] > 1:
                prompt = str(np.random.choice(self.prompts[path_img], 1)[0])
            else:
                prompt = self.prompts[path_img][0]
            if self.class_prompt == "instancelevel_random":
                prompt = self.tokenizer.decode(ast.literal_eval(prompt))
            return prompt
        return "An image"

    def __getitem__(self, index):
        path_img, label = self.samples[index]
        img = Image.open(path_img)
        if img.mode != "RGB":
            img = img.convert("RGB")
        example = {"instance_images": self.image_transforms(img)}
        prompt = self._prompt_for(path_img, label, index)
        example["instance_prompt_ids"] = self.tokenizer(
            prompt, truncation=True, padding="max_length",
            max_length=self.tokenizer.model_max_length, return_tensors="pt",
        ).input_ids
        return example


def collate_fn(examples):
    """Reference: diff_train.py:283-297."""
    pixel_values = torch.stack([e["instance_images"] for e in examples]) \
        .to(memory_format=torch.contiguous_format).float()
    input_ids = torch.cat([e["instance_prompt_ids"] for e in examples], dim=0)
    return {"pixel_values": pixel_values, "input_ids": input_ids}


class SynthDataset(Dataset):
    """Flat directory of generated images + optional prompts.txt
    (reference: diff_retrieval.py:61-111)."""

    def __init__(self, main_dir: str, transform=None, size: int = 224,
                 with_prompts: bool = False):
        self.main_dir = Path(main_dir)
        self.transform = transform or EvalTransform(size)
        files = [f for f in sorted(self.main_dir.iterdir())
                 if f.suffix.lower() in IMG_EXTS]
        # natural sort by numeric stem when possible (generations are 0.png, 1.png, ...)
        def key(f):
            try:
                return (0, int(f.stem))
            except ValueError:
                return (1, f.stem)
        self.files = sorted(files, key=key)
        self.prompts = None
        pf = self.main_dir / "prompts.txt"
        if with_prompts and pf.exists():
            self.prompts = pf.read_text().splitlines()

    def __len__(self):
        return len(self.files)

    def __getitem__(self, idx):
        img = Image.open(self.files[idx]).convert("RGB")
        t = self.transform(img)
        if self.prompts is not None:
            p = self.prompts[idx] if idx < len(self.prompts) else ""
            return t, idx, p
        return t, idx


class SyntheticImageDataset(Dataset):
    """Random images + captions of the training shape (benchmarks; no network)."""

    def __init__(self, n: int, size: int = 256, tokenizer=None, seed: int = 0,
                 n_classes: int = 10):
        self.n = n
        self.size = size
        self.tokenizer = tokenizer
        self.seed = seed
        self.n_classes = n_classes

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + idx)
        img = torch.rand(3, self.size, self.size, generator=g) * 2 - 1
        example = {"instance_images": img}
        if self.tokenizer is not None:
            prompt = f"An image of {IMAGENETTE_CLASSES[idx % self.n_classes]} sample {idx}"
            example["instance_prompt_ids"] = self.tokenizer(
                prompt, truncation=True, padding="max_length",
                max_length=self.tokenizer.model_max_length, return_tensors="pt",
            ).input_ids
        return example
