"""DINO-style Vision Transformer backbone (self-contained).

Capability parity: /root/reference/dino_vits.py — ViT with patch embed,
pre-norm blocks, positional-embedding interpolation and
`get_intermediate_layers`, plus the vit_small/vit_base //8//16 factories
used by `--pt_style dino` (diff_retrieval.py:249-267). No network here:
weights are random unless a local state dict is provided.
"""
from __future__ import annotations

import math
from pathlib import Path
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..models.layers import LayerNormOp
from .. import ops


class PatchEmbed(nn.Module):
    def __init__(self, img_size=224, patch_size=16, in_chans=3, embed_dim=768):
        super().__init__()
        self.num_patches = (img_size // patch_size) ** 2
        self.img_size = img_size
        self.patch_size = patch_size
        self.proj = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size,
                              stride=patch_size)

    def forward(self, x):
        return self.proj(x).flatten(2).transpose(1, 2)  # [B, N, C]


class ViTAttention(nn.Module):
    def __init__(self, dim, num_heads, qkv_bias=True):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.proj = nn.Linear(dim, dim)

    def forward(self, x):
        B, N, C = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.num_heads, self.head_dim)
        q = qkv[:, :, 0]
        k = qkv[:, :, 1]
        v = qkv[:, :, 2]
        out = ops.attention(q, k, v, layout="blhd")
        return self.proj(out.reshape(B, N, C))


class ViTMlp(nn.Module):
    def __init__(self, dim, hidden):
        super().__init__()
        self.fc1 = nn.Linear(dim, hidden)
        self.fc2 = nn.Linear(hidden, dim)

    def forward(self, x):
        return self.fc2(F.gelu(self.fc1(x)))


class ViTBlock(nn.Module):
    def __init__(self, dim, num_heads, mlp_ratio=4.0):
        super().__init__()
        self.norm1 = LayerNormOp(dim, eps=1e-6)
        self.attn = ViTAttention(dim, num_heads)
        self.norm2 = LayerNormOp(dim, eps=1e-6)
        self.mlp = ViTMlp(dim, int(dim * mlp_ratio))

    def forward(self, x):
        x = x + self.attn(self.norm1(x))
        x = x + self.mlp(self.norm2(x))
        return x


class VisionTransformer(nn.Module):
    def __init__(self, img_size=224, patch_size=16, embed_dim=768, depth=12,
                 num_heads=12, mlp_ratio=4.0):
        super().__init__()
        self.embed_dim = embed_dim
        self.patch_embed = PatchEmbed(img_size, patch_size, 3, embed_dim)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(
            torch.zeros(1, self.patch_embed.num_patches + 1, embed_dim))
        self.blocks = nn.ModuleList([
            ViTBlock(embed_dim, num_heads, mlp_ratio) for _ in range(depth)])
        self.norm = LayerNormOp(embed_dim, eps=1e-6)
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)

    def interpolate_pos_encoding(self, x, w, h):
        """Bicubic pos-embed interpolation (reference dino_vits.py:213-233)."""
        npatch = x.shape[1] - 1
        N = self.pos_embed.shape[1] - 1
        if npatch == N and w == h:
            return self.pos_embed
        cls_pos = self.pos_embed[:, :1]
        patch_pos = self.pos_embed[:, 1:]
        dim = x.shape[-1]
        ps = self.patch_embed.patch_size
        w0, h0 = w // ps, h // ps
        patch_pos = F.interpolate(
            patch_pos.reshape(1, int(math.sqrt(N)), int(math.sqrt(N)), dim)
            .permute(0, 3, 1, 2),
            size=(w0, h0), mode="bicubic", align_corners=False)
        patch_pos = patch_pos.permute(0, 2, 3, 1).view(1, -1, dim)
        return torch.cat((cls_pos, patch_pos), dim=1)

    def prepare_tokens(self, x):
        B, C, W, H = x.shape
        tok = self.patch_embed(x)
        cls = self.cls_token.expand(B, -1, -1)
        tok = torch.cat((cls, tok), dim=1)
        return tok + self.interpolate_pos_encoding(tok, W, H)

    def forward(self, x):
        x = self.prepare_tokens(x)
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        return x[:, 0]  # CLS token

    def get_intermediate_layers(self, x, n=1):
        x = self.prepare_tokens(x)
        out = []
        for i, blk in enumerate(self.blocks):
            x = blk(x)
            if len(self.blocks) - i <= n:
                out.append(self.norm(x))
        return out


def vit_small(patch_size=16, **kw):
    return VisionTransformer(patch_size=patch_size, embed_dim=384, depth=12,
                             num_heads=6, **kw)


def vit_base(patch_size=16, **kw):
    return VisionTransformer(patch_size=patch_size, embed_dim=768, depth=12,
                             num_heads=12, **kw)


_DINO_FACTORY = {
    "dino_vits16": (vit_small, 16), "dino_vits8": (vit_small, 8),
    "dino_vitb16": (vit_base, 16), "dino_vitb8": (vit_base, 8),
}


class ResNetDINO(nn.Module):
    """DINO ResNet-50 backbone (reference dino_vits.py:416-431: torchvision
    resnet50 with fc=Identity): 2048-d global-average-pooled features.
    Reuses this framework's ResNet50 trunk."""

    def __init__(self):
        super().__init__()
        from .backbones import ResNet50
        self.backbone = ResNet50()

    def forward(self, x):
        return self.backbone.forward_features(x).mean(dim=(2, 3))


def load_dino(arch: str = "dino_vitb16", weights: Optional[str] = None,
              device="cpu"):
    """Factory mirroring dino_vits.py:340-487 hub loaders; local weights
    (state dict .pth) are loaded when provided/found, else random init."""
    if arch == "dino_resnet50":
        model = ResNetDINO()
        cand = Path(weights) if weights else Path("./pretrainedmodels") / f"{arch}.pth"
        if cand.exists():
            sd = torch.load(str(cand), map_location="cpu", weights_only=True)
            model.backbone.load_state_dict(sd, strict=False)
        return model.to(device).eval()
    fac, ps = _DINO_FACTORY.get(arch, _DINO_FACTORY["dino_vitb16"])
    model = fac(patch_size=ps)
    cand = Path(weights) if weights else Path("./pretrainedmodels") / f"{arch}.pth"
    if cand.exists():
        sd = torch.load(str(cand), map_location="cpu", weights_only=True)
        model.load_state_dict(sd, strict=False)
    else:
        torch.manual_seed(0)
        for m in model.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02)
    return model.to(device).eval()
