"""AutoencoderKL — SD VAE, MI355X-native implementation.

Capability parity: diffusers AutoencoderKL as used for latent encode in
training (/root/reference/diff_train.py:620-621, scale 0.18215) and decode
in sampling. Diffusers state-dict naming; GroupNorm+SiLU fused HIP kernel.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path
from typing import Optional, Tuple

import torch
import torch.nn as nn

from ..ops.conv import Conv2d
from .attention import Attention
from .layers import GroupNormOp
from .resnet import Downsample2D, ResnetBlock2D, Upsample2D


@dataclass
class VAEConfig:
    in_channels: int = 3
    out_channels: int = 3
    latent_channels: int = 4
    block_out_channels: Tuple[int, ...] = (128, 256, 512, 512)
    layers_per_block: int = 2
    norm_num_groups: int = 32
    sample_size: int = 768
    scaling_factor: float = 0.18215

    @classmethod
    def sd(cls) -> "VAEConfig":
        return cls()

    @classmethod
    def tiny(cls) -> "VAEConfig":
        return cls(block_out_channels=(16, 32, 32, 32), norm_num_groups=8,
                   layers_per_block=1, sample_size=32)

    def to_json(self) -> str:
        d = asdict(self)
        d["_class_name"] = "AutoencoderKL"
        return json.dumps(d, indent=2)

    @classmethod
    def from_json(cls, text: str) -> "VAEConfig":
        d = json.loads(text)
        d.pop("_class_name", None)
        known = {f: d[f] for f in d if f in cls.__dataclass_fields__}
        if isinstance(known.get("block_out_channels"), list):
            known["block_out_channels"] = tuple(known["block_out_channels"])
        return cls(**known)


class VAEAttention(nn.Module):
    """Single-head self-attention on [B,C,H,W] (VAE mid block)."""

    def __init__(self, channels: int, groups: int):
        super().__init__()
        self.group_norm = GroupNormOp(groups, channels, eps=1e-6, fused_silu=False)
        self.attn = Attention(channels, heads=1, dim_head=channels, bias=True)

    def forward(self, x):
        B, C, H, W = x.shape
        h = self.group_norm(x)
        h = h.permute(0, 2, 3, 1).reshape(B, H * W, C)
        h = self.attn(h)
        return x + h.reshape(B, H, W, C).permute(0, 3, 1, 2)

    # flatten attn.* keys to match diffusers AutoencoderKL mid_block.attentions.0.*
    def _named_members_remap(self):  # used by model_io for key remap
        return {"attn.": ""}


class DownEncoderBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, num_layers, groups, add_downsample):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(in_ch if i == 0 else out_ch, out_ch, temb_channels=None,
                          groups=groups, eps=1e-6)
            for i in range(num_layers)
        ])
        self.downsamplers = nn.ModuleList([Downsample2D(out_ch)]) if add_downsample else None

    def forward(self, x):
        for r in self.resnets:
            x = r(x, None)
        if self.downsamplers is not None:
            x = self.downsamplers[0](x)
        return x


class UpDecoderBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, num_layers, groups, add_upsample):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(in_ch if i == 0 else out_ch, out_ch, temb_channels=None,
                          groups=groups, eps=1e-6)
            for i in range(num_layers)
        ])
        self.upsamplers = nn.ModuleList([Upsample2D(out_ch)]) if add_upsample else None

    def forward(self, x):
        for r in self.resnets:
            x = r(x, None)
        if self.upsamplers is not None:
            x = self.upsamplers[0](x)
        return x


class MidBlock(nn.Module):
    def __init__(self, ch, groups):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(ch, ch, temb_channels=None, groups=groups, eps=1e-6),
            ResnetBlock2D(ch, ch, temb_channels=None, groups=groups, eps=1e-6),
        ])
        self.attentions = nn.ModuleList([VAEAttention(ch, groups)])

    def forward(self, x):
        x = self.resnets[0](x, None)
        x = self.attentions[0](x)
        x = self.resnets[1](x, None)
        return x


class Encoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = cfg.block_out_channels
        self.conv_in = Conv2d(cfg.in_channels, ch[0], 3, padding=1)
        self.down_blocks = nn.ModuleList()
        out_c = ch[0]
        for i in range(len(ch)):
            in_c, out_c = out_c, ch[i]
            self.down_blocks.append(DownEncoderBlock2D(
                in_c, out_c, cfg.layers_per_block, cfg.norm_num_groups,
                add_downsample=i < len(ch) - 1))
        self.mid_block = MidBlock(ch[-1], cfg.norm_num_groups)
        self.conv_norm_out = GroupNormOp(cfg.norm_num_groups, ch[-1], eps=1e-6,
                                         fused_silu=True)
        self.conv_out = Conv2d(ch[-1], 2 * cfg.latent_channels, 3, padding=1)

    def forward(self, x):
        x = self.conv_in(x)
        for blk in self.down_blocks:
            x = blk(x)
        x = self.mid_block(x)
        x = self.conv_norm_out(x)
        return self.conv_out(x)


class Decoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = list(reversed(cfg.block_out_channels))
        self.conv_in = Conv2d(cfg.latent_channels, ch[0], 3, padding=1)
        self.mid_block = MidBlock(ch[0], cfg.norm_num_groups)
        self.up_blocks = nn.ModuleList()
        out_c = ch[0]
        for i in range(len(ch)):
            in_c, out_c = out_c, ch[i]
            self.up_blocks.append(UpDecoderBlock2D(
                in_c, out_c, cfg.layers_per_block + 1, cfg.norm_num_groups,
                add_upsample=i < len(ch) - 1))
        self.conv_norm_out = GroupNormOp(cfg.norm_num_groups, ch[-1], eps=1e-6,
                                         fused_silu=True)
        self.conv_out = Conv2d(ch[-1], cfg.out_channels, 3, padding=1)

    def forward(self, z):
        z = self.conv_in(z)
        z = self.mid_block(z)
        for blk in self.up_blocks:
            z = blk(z)
        z = self.conv_norm_out(z)
        return self.conv_out(z)


class DiagonalGaussianDistribution:
    def __init__(self, parameters: torch.Tensor):
        self.parameters = parameters
        self.mean, self.logvar = torch.chunk(parameters, 2, dim=1)
        self.logvar = torch.clamp(self.logvar, -30.0, 20.0)
        self.std = torch.exp(0.5 * self.logvar)

    def sample(self, generator: Optional[torch.Generator] = None) -> torch.Tensor:
        noise = torch.randn(self.mean.shape, generator=generator,
                            device=self.mean.device, dtype=self.mean.dtype)
        return self.mean + self.std * noise

    def mode(self) -> torch.Tensor:
        return self.mean

    def kl(self) -> torch.Tensor:
        return 0.5 * torch.sum(
            self.mean.pow(2) + self.logvar.exp() - 1.0 - self.logvar, dim=[1, 2, 3])


class _EncodeOut:
    def __init__(self, dist):
        self.latent_dist = dist


class _DecodeOut:
    def __init__(self, sample):
        self.sample = sample


class AutoencoderKL(nn.Module):
    def __init__(self, config: Optional[VAEConfig] = None):
        super().__init__()
        cfg = config or VAEConfig.sd()
        self.config = cfg
        self.encoder = Encoder(cfg)
        self.decoder = Decoder(cfg)
        self.quant_conv = Conv2d(2 * cfg.latent_channels, 2 * cfg.latent_channels, 1)
        self.post_quant_conv = Conv2d(cfg.latent_channels, cfg.latent_channels, 1)

    @property
    def dtype(self):
        return self.quant_conv.weight.dtype

    def encode(self, x: torch.Tensor) -> _EncodeOut:
        moments = self.quant_conv(self.encoder(x))
        return _EncodeOut(DiagonalGaussianDistribution(moments))

    def decode(self, z: torch.Tensor) -> _DecodeOut:
        return _DecodeOut(self.decoder(self.post_quant_conv(z)))

    def forward(self, x, sample_posterior: bool = True):
        dist = self.encode(x).latent_dist
        z = dist.sample() if sample_posterior else dist.mode()
        return self.decode(z).sample

    def save_pretrained(self, path):
        from .model_io import save_module
        save_module(self, Path(path), self.config.to_json())

    @classmethod
    def from_pretrained(cls, path):
        from .model_io import load_module
        path = Path(path)
        cfg = VAEConfig.from_json((path / "config.json").read_text())
        model = cls(cfg)
        load_module(model, path)
        return model
