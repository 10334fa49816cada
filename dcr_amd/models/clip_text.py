"""CLIP text encoder (SD-2.1: OpenCLIP ViT-H text tower), MI355X-native.

Capability parity: transformers CLIPTextModel as used for conditioning
(/root/reference/diff_train.py:386-393,636). transformers-compatible
state-dict naming (text_model.encoder.layers.N....). LayerNorm routes
through the fused HIP kernel; attention through dcr_amd.ops.attention
with a causal mask.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, asdict
from pathlib import Path
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .layers import LayerNormOp


@dataclass
class CLIPTextConfig:
    vocab_size: int = 49408
    hidden_size: int = 1024
    intermediate_size: int = 4096
    num_hidden_layers: int = 23
    num_attention_heads: int = 16
    max_position_embeddings: int = 77
    layer_norm_eps: float = 1e-5
    hidden_act: str = "gelu"

    @classmethod
    def sd21(cls) -> "CLIPTextConfig":
        return cls()

    @classmethod
    def sd14(cls) -> "CLIPTextConfig":
        """CLIP ViT-L/14 text tower (SD-1.x conditioning)."""
        return cls(hidden_size=768, intermediate_size=3072,
                   num_hidden_layers=12, num_attention_heads=12,
                   hidden_act="quick_gelu")

    @classmethod
    def tiny(cls) -> "CLIPTextConfig":
        # full CLIP vocab so real tokenizer ids (bos 49406/eos 49407) stay valid
        return cls(vocab_size=49408, hidden_size=32, intermediate_size=64,
                   num_hidden_layers=2, num_attention_heads=2)

    def to_json(self) -> str:
        d = asdict(self)
        d["_class_name"] = "CLIPTextModel"
        return json.dumps(d, indent=2)

    @classmethod
    def from_json(cls, text: str) -> "CLIPTextConfig":
        d = json.loads(text)
        d.pop("_class_name", None)
        return cls(**{f: d[f] for f in d if f in cls.__dataclass_fields__})


class CLIPAttention(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.heads = cfg.num_attention_heads
        self.dim_head = cfg.hidden_size // cfg.num_attention_heads
        self.q_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.k_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.v_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.out_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, x):
        B, L, C = x.shape
        q = self.q_proj(x).view(B, L, self.heads, self.dim_head)
        k = self.k_proj(x).view(B, L, self.heads, self.dim_head)
        v = self.v_proj(x).view(B, L, self.heads, self.dim_head)
        out = ops.attention(q, k, v, causal=True, layout="blhd")
        return self.out_proj(out.reshape(B, L, C))


class CLIPMLP(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size)
        self.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        self.act = cfg.hidden_act

    def forward(self, x):
        x = self.fc1(x)
        if self.act == "quick_gelu":
            x = x * torch.sigmoid(1.702 * x)
        else:
            x = F.gelu(x)
        return self.fc2(x)


class CLIPEncoderLayer(nn.Module):
    def __init__(self, cfg: CLIPTextConfig):
        super().__init__()
        self.layer_norm1 = LayerNormOp(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.self_attn = CLIPAttention(cfg)
        self.layer_norm2 = LayerNormOp(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.mlp = CLIPMLP(cfg)

    def forward(self, x):
        x = x + self.self_attn(self.layer_norm1(x))
        x = x + self.mlp(self.layer_norm2(x))
        return x


class _TextModelOut:
    def __init__(self, last_hidden_state, pooler_output=None):
        self.last_hidden_state = last_hidden_state
        self.pooler_output = pooler_output

    def __getitem__(self, i):
        return (self.last_hidden_state, self.pooler_output)[i]


class CLIPTextModel(nn.Module):
    def __init__(self, config: Optional[CLIPTextConfig] = None):
        super().__init__()
        cfg = config or CLIPTextConfig.sd21()
        self.config = cfg
        tm = nn.Module()
        emb = nn.Module()
        emb.token_embedding = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        emb.position_embedding = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        tm.embeddings = emb
        enc = nn.Module()
        enc.layers = nn.ModuleList([CLIPEncoderLayer(cfg) for _ in range(cfg.num_hidden_layers)])
        tm.encoder = enc
        tm.final_layer_norm = LayerNormOp(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.text_model = tm

    @property
    def dtype(self):
        return self.text_model.embeddings.token_embedding.weight.dtype

    def forward(self, input_ids: torch.Tensor, attention_mask=None):
        tm = self.text_model
        L = input_ids.shape[1]
        pos = torch.arange(L, device=input_ids.device)
        x = tm.embeddings.token_embedding(input_ids) + tm.embeddings.position_embedding(pos)
        for layer in tm.encoder.layers:
            x = layer(x)
        x = tm.final_layer_norm(x)
        # pooled = hidden state at the argmax (EOS) token, CLIP convention
        eos_idx = input_ids.argmax(dim=-1)
        pooled = x[torch.arange(x.shape[0], device=x.device), eos_idx]
        return _TextModelOut(x, pooled)

    def save_pretrained(self, path):
        from .model_io import save_module
        save_module(self, Path(path), self.config.to_json())

    @classmethod
    def from_pretrained(cls, path):
        from .model_io import load_module
        path = Path(path)
        cfg = CLIPTextConfig.from_json((path / "config.json").read_text())
        model = cls(cfg)
        load_module(model, path)
        return model
