"""UNet2DConditionModel — SD-2.1 architecture, MI355X-native implementation.

Capability parity target: the diffusers UNet the reference finetunes and
samples (/root/reference/diff_train.py:18,399-408, diff_inference.py:190).
State-dict key names match diffusers so checkpoints interoperate
(SURVEY.md §5.4). All hot ops route through dcr_amd.ops HIP kernels.

Default config = stabilityai/stable-diffusion-2-1 unet/config.json:
block_out_channels [320,640,1280,1280], attention heads [5,10,20,20]
(dim_head 64), cross_attention_dim 1024, layers_per_block 2,
use_linear_projection, eps 1e-5.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path
from typing import Optional, Tuple

import torch
import torch.nn as nn

from .attention import Transformer2DModel
from .embeddings import TimestepEmbedding, timestep_embedding
from .layers import GroupNormOp
from ..ops.conv import Conv2d
from .resnet import Downsample2D, ResnetBlock2D, Upsample2D


@dataclass
class UNetConfig:
    sample_size: int = 96
    in_channels: int = 4
    out_channels: int = 4
    block_out_channels: Tuple[int, ...] = (320, 640, 1280, 1280)
    down_block_types: Tuple[str, ...] = (
        "CrossAttnDownBlock2D", "CrossAttnDownBlock2D", "CrossAttnDownBlock2D", "DownBlock2D",
    )
    up_block_types: Tuple[str, ...] = (
        "UpBlock2D", "CrossAttnUpBlock2D", "CrossAttnUpBlock2D", "CrossAttnUpBlock2D",
    )
    layers_per_block: int = 2
    attention_head_dim: Tuple[int, ...] = (5, 10, 20, 20)  # = num heads (legacy name)
    cross_attention_dim: int = 1024
    norm_num_groups: int = 32
    norm_eps: float = 1e-5
    use_linear_projection: bool = True
    flip_sin_to_cos: bool = True
    freq_shift: float = 0.0
    dropout: float = 0.0

    @classmethod
    def sd21(cls) -> "UNetConfig":
        return cls()

    @classmethod
    def sd14(cls) -> "UNetConfig":
        """CompVis/stable-diffusion-v1-4 unet: 8 heads per block, 768-d
        cross-attention (CLIP ViT-L), conv proj_in/out."""
        return cls(sample_size=64, attention_head_dim=(8, 8, 8, 8),
                   cross_attention_dim=768, use_linear_projection=False)

    @classmethod
    def tiny(cls) -> "UNetConfig":
        """A tiny config for CPU tests — same topology, small widths."""
        return cls(
            sample_size=8,
            block_out_channels=(32, 64, 64, 64),
            attention_head_dim=(1, 2, 2, 2),
            cross_attention_dim=32,
            norm_num_groups=8,
            layers_per_block=1,
        )

    def to_json(self) -> str:
        d = asdict(self)
        d["_class_name"] = "UNet2DConditionModel"
        return json.dumps(d, indent=2)

    @classmethod
    def from_json(cls, text: str) -> "UNetConfig":
        d = json.loads(text)
        d.pop("_class_name", None)
        known = {f: d[f] for f in d if f in cls.__dataclass_fields__}
        for k in ("block_out_channels", "down_block_types", "up_block_types",
                  "attention_head_dim"):
            if k in known and isinstance(known[k], list):
                known[k] = tuple(known[k])
        if "attention_head_dim" in known and isinstance(known["attention_head_dim"], int):
            known["attention_head_dim"] = tuple(
                [known["attention_head_dim"]] * len(known.get("block_out_channels", (1,) * 4)))
        return cls(**known)


class CrossAttnDownBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, temb_ch, num_layers, heads, cross_dim,
                 groups, eps, use_linear, add_downsample, dropout):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(in_ch if i == 0 else out_ch, out_ch, temb_ch,
                          groups=groups, eps=eps, dropout=dropout)
            for i in range(num_layers)
        ])
        self.attentions = nn.ModuleList([
            Transformer2DModel(out_ch, heads, out_ch // heads, depth=1,
                               cross_attention_dim=cross_dim, norm_num_groups=groups,
                               use_linear_projection=use_linear, dropout=dropout)
            for _ in range(num_layers)
        ])
        self.downsamplers = nn.ModuleList([Downsample2D(out_ch)]) if add_downsample else None

    def forward(self, x, temb, context):
        out_states = []
        for resnet, attn in zip(self.resnets, self.attentions):
            x = resnet(x, temb)
            x = attn(x, context)
            out_states.append(x)
        if self.downsamplers is not None:
            x = self.downsamplers[0](x)
            out_states.append(x)
        return x, out_states


class DownBlock2D(nn.Module):
    def __init__(self, in_ch, out_ch, temb_ch, num_layers, groups, eps,
                 add_downsample, dropout):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(in_ch if i == 0 else out_ch, out_ch, temb_ch,
                          groups=groups, eps=eps, dropout=dropout)
            for i in range(num_layers)
        ])
        self.downsamplers = nn.ModuleList([Downsample2D(out_ch)]) if add_downsample else None

    def forward(self, x, temb, context=None):
        out_states = []
        for resnet in self.resnets:
            x = resnet(x, temb)
            out_states.append(x)
        if self.downsamplers is not None:
            x = self.downsamplers[0](x)
            out_states.append(x)
        return x, out_states


class UNetMidBlock2DCrossAttn(nn.Module):
    def __init__(self, ch, temb_ch, heads, cross_dim, groups, eps, use_linear, dropout):
        super().__init__()
        self.resnets = nn.ModuleList([
            ResnetBlock2D(ch, ch, temb_ch, groups=groups, eps=eps, dropout=dropout),
            ResnetBlock2D(ch, ch, temb_ch, groups=groups, eps=eps, dropout=dropout),
        ])
        self.attentions = nn.ModuleList([
            Transformer2DModel(ch, heads, ch // heads, depth=1,
                               cross_attention_dim=cross_dim, norm_num_groups=groups,
                               use_linear_projection=use_linear, dropout=dropout)
        ])

    def forward(self, x, temb, context):
        x = self.resnets[0](x, temb)
        x = self.attentions[0](x, context)
        x = self.resnets[1](x, temb)
        return x


class CrossAttnUpBlock2D(nn.Module):
    def __init__(self, in_ch, prev_ch, out_ch, temb_ch, num_layers, heads, cross_dim,
                 groups, eps, use_linear, add_upsample, dropout):
        super().__init__()
        resnets = []
        for i in range(num_layers):
            res_skip = in_ch if i == num_layers - 1 else out_ch
            res_in = prev_ch if i == 0 else out_ch
            resnets.append(ResnetBlock2D(res_in + res_skip, out_ch, temb_ch,
                                         groups=groups, eps=eps, dropout=dropout))
        self.resnets = nn.ModuleList(resnets)
        self.attentions = nn.ModuleList([
            Transformer2DModel(out_ch, heads, out_ch // heads, depth=1,
                               cross_attention_dim=cross_dim, norm_num_groups=groups,
                               use_linear_projection=use_linear, dropout=dropout)
            for _ in range(num_layers)
        ])
        self.upsamplers = nn.ModuleList([Upsample2D(out_ch)]) if add_upsample else None

    def forward(self, x, res_states, temb, context):
        for resnet, attn in zip(self.resnets, self.attentions):
            res = res_states.pop()
            x = torch.cat([x, res], dim=1)
            x = resnet(x, temb)
            x = attn(x, context)
        if self.upsamplers is not None:
            x = self.upsamplers[0](x)
        return x


class UpBlock2D(nn.Module):
    def __init__(self, in_ch, prev_ch, out_ch, temb_ch, num_layers, groups, eps,
                 add_upsample, dropout):
        super().__init__()
        resnets = []
        for i in range(num_layers):
            res_skip = in_ch if i == num_layers - 1 else out_ch
            res_in = prev_ch if i == 0 else out_ch
            resnets.append(ResnetBlock2D(res_in + res_skip, out_ch, temb_ch,
                                         groups=groups, eps=eps, dropout=dropout))
        self.resnets = nn.ModuleList(resnets)
        self.upsamplers = nn.ModuleList([Upsample2D(out_ch)]) if add_upsample else None

    def forward(self, x, res_states, temb, context=None):
        for resnet in self.resnets:
            res = res_states.pop()
            x = torch.cat([x, res], dim=1)
            x = resnet(x, temb)
        if self.upsamplers is not None:
            x = self.upsamplers[0](x)
        return x


class UNet2DConditionModel(nn.Module):
    def __init__(self, config: Optional[UNetConfig] = None):
        super().__init__()
        cfg = config or UNetConfig.sd21()
        self.config = cfg
        ch = cfg.block_out_channels
        temb_ch = ch[0] * 4
        self.conv_in = Conv2d(cfg.in_channels, ch[0], 3, padding=1)
        self.time_embedding = TimestepEmbedding(ch[0], temb_ch)

        self.down_blocks = nn.ModuleList()
        out_c = ch[0]
        for i, btype in enumerate(cfg.down_block_types):
            in_c, out_c = out_c, ch[i]
            is_final = i == len(ch) - 1
            if btype == "CrossAttnDownBlock2D":
                blk = CrossAttnDownBlock2D(
                    in_c, out_c, temb_ch, cfg.layers_per_block, cfg.attention_head_dim[i],
                    cfg.cross_attention_dim, cfg.norm_num_groups, cfg.norm_eps,
                    cfg.use_linear_projection, not is_final, cfg.dropout)
            elif btype == "DownBlock2D":
                blk = DownBlock2D(in_c, out_c, temb_ch, cfg.layers_per_block,
                                  cfg.norm_num_groups, cfg.norm_eps, not is_final, cfg.dropout)
            else:
                raise ValueError(f"unknown down block {btype}")
            self.down_blocks.append(blk)

        self.mid_block = UNetMidBlock2DCrossAttn(
            ch[-1], temb_ch, cfg.attention_head_dim[-1], cfg.cross_attention_dim,
            cfg.norm_num_groups, cfg.norm_eps, cfg.use_linear_projection, cfg.dropout)

        self.up_blocks = nn.ModuleList()
        rev_ch = list(reversed(ch))
        rev_heads = list(reversed(cfg.attention_head_dim))
        out_c = rev_ch[0]
        for i, btype in enumerate(cfg.up_block_types):
            prev_c, out_c = out_c, rev_ch[i]
            in_c = rev_ch[min(i + 1, len(ch) - 1)]
            is_final = i == len(ch) - 1
            n_layers = cfg.layers_per_block + 1
            if btype == "CrossAttnUpBlock2D":
                blk = CrossAttnUpBlock2D(
                    in_c, prev_c, out_c, temb_ch, n_layers, rev_heads[i],
                    cfg.cross_attention_dim, cfg.norm_num_groups, cfg.norm_eps,
                    cfg.use_linear_projection, not is_final, cfg.dropout)
            elif btype == "UpBlock2D":
                blk = UpBlock2D(in_c, prev_c, out_c, temb_ch, n_layers,
                                cfg.norm_num_groups, cfg.norm_eps, not is_final, cfg.dropout)
            else:
                raise ValueError(f"unknown up block {btype}")
            self.up_blocks.append(blk)

        self.conv_norm_out = GroupNormOp(cfg.norm_num_groups, ch[0], eps=cfg.norm_eps,
                                         fused_silu=True)
        self.conv_out = Conv2d(ch[0], cfg.out_channels, 3, padding=1)

    @property
    def dtype(self):
        return self.conv_in.weight.dtype

    def forward(
        self,
        sample: torch.Tensor,
        timestep: torch.Tensor,
        encoder_hidden_states: torch.Tensor,
    ) -> torch.Tensor:
        if timestep.dim() == 0:
            timestep = timestep[None].expand(sample.shape[0])
        t_emb = timestep_embedding(
            timestep, self.config.block_out_channels[0],
            flip_sin_to_cos=self.config.flip_sin_to_cos,
            downscale_freq_shift=self.config.freq_shift,
        ).to(sample.dtype)
        temb = self.time_embedding(t_emb)

        x = self.conv_in(sample)
        res_states = [x]
        for blk in self.down_blocks:
            x, states = blk(x, temb, encoder_hidden_states)
            res_states.extend(states)
        x = self.mid_block(x, temb, encoder_hidden_states)
        for blk in self.up_blocks:
            x = blk(x, res_states, temb, encoder_hidden_states)
        x = self.conv_norm_out(x)  # fused GN+SiLU
        return self.conv_out(x)

    # -- checkpoint I/O (diffusers directory layout) -----------------------
    def save_pretrained(self, path):
        from .model_io import save_module
        save_module(self, Path(path), self.config.to_json())

    @classmethod
    def from_pretrained(cls, path):
        from .model_io import load_module
        path = Path(path)
        cfg = UNetConfig.from_json((path / "config.json").read_text())
        model = cls(cfg)
        load_module(model, path)
        return model
