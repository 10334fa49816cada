"""Minimal CLIP (image + text) for CLIP-score and `--pt_style clip`.

Capability parity: the openai `clip` package usage at
/root/reference/utils_ret.py:1046-1066 (gen_clipscore, ViT-B/16) and
diff_retrieval.py:268-275. Implemented from scratch on dcr_amd ops;
random init unless local weights exist (no network).
"""
from __future__ import annotations

from pathlib import Path
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.tokenizer import load_tokenizer
from ..models.clip_text import CLIPTextConfig, CLIPTextModel
from .dino import VisionTransformer


class CLIPModel(nn.Module):
    """ViT image tower + causal text tower with joint embedding space."""

    def __init__(self, embed_dim: int = 512, image_size: int = 224,
                 patch: int = 16, vision_width: int = 768, vision_depth: int = 12,
                 text_cfg: CLIPTextConfig | None = None):
        super().__init__()
        self.visual = VisionTransformer(img_size=image_size, patch_size=patch,
                                        embed_dim=vision_width, depth=vision_depth,
                                        num_heads=vision_width // 64)
        self.visual_proj = nn.Parameter(torch.randn(vision_width, embed_dim) * 0.02)
        tc = text_cfg or CLIPTextConfig(hidden_size=512, intermediate_size=2048,
                                        num_hidden_layers=12, num_attention_heads=8)
        self.text = CLIPTextModel(tc)
        self.text_proj = nn.Parameter(torch.randn(tc.hidden_size, embed_dim) * 0.02)
        self.logit_scale = nn.Parameter(torch.tensor(4.6052))  # ln(100)

    def encode_image(self, images: torch.Tensor) -> torch.Tensor:
        feats = self.visual(images)
        return F.normalize(feats @ self.visual_proj, dim=-1)

    def encode_text(self, input_ids: torch.Tensor) -> torch.Tensor:
        out = self.text(input_ids)
        return F.normalize(out.pooler_output @ self.text_proj, dim=-1)

    def forward(self, images, input_ids):
        im = self.encode_image(images)
        tx = self.encode_text(input_ids)
        return im @ tx.t() * self.logit_scale.exp()


def load_clip(name: str = "ViT-B/16", device="cpu"):
    torch.manual_seed(0)
    model = CLIPModel()
    p = Path("./pretrainedmodels") / "clip_vitb16.pt"
    if p.exists():
        model.load_state_dict(torch.load(str(p), map_location="cpu",
                                         weights_only=True), strict=False)
    return model.to(device).eval(), load_tokenizer()


@torch.no_grad()
def gen_clipscore(model: CLIPModel, tokenizer, images: torch.Tensor,
                  prompts: List[str], device="cpu", batch_size: int = 64):
    """Image-prompt CLIP alignment (reference utils_ret.py:1046-1066):
    mean cosine similarity between each image and its own prompt."""
    scores = []
    for i in range(0, images.shape[0], batch_size):
        im = images[i:i + batch_size].to(device)
        ids = tokenizer(prompts[i:i + batch_size], truncation=True,
                        padding="max_length", max_length=77,
                        return_tensors="pt").input_ids.to(device)
        ie = model.encode_image(im)
        te = model.encode_text(ids)
        scores.append((ie * te).sum(-1).cpu())
    return torch.cat(scores)
