"""Tokenizer layer: CLIP BPE when vocab files exist, hash fallback otherwise.

The reference uses transformers' CLIPTokenizer
(/root/reference/diff_train.py:371-383). This container has no network to
fetch vocab/merges, so we provide:

* ``load_tokenizer(path)`` — a real ``transformers.CLIPTokenizer`` when a
  tokenizer directory (vocab.json + merges.txt) is available, else
* ``HashTokenizer`` — deterministic word-hash tokenizer with the same
  call surface (padding="max_length", model_max_length=77, bos/eos ids
  matching CLIP: 49406/49407) so every caption-pipeline code path runs
  identically on synthetic data.
"""
from __future__ import annotations

import json
import re
from pathlib import Path
from typing import List, Union

import torch


class _TokOut:
    def __init__(self, input_ids):
        self.input_ids = input_ids


class HashTokenizer:
    """Deterministic, vocab-free tokenizer with CLIP-compatible ids."""

    BOS = 49406
    EOS = 49407

    def __init__(self, model_max_length: int = 77, vocab_size: int = 49408):
        self.model_max_length = model_max_length
        self.vocab_size = vocab_size

    def _word_id(self, word: str) -> int:
        # stable non-crypto hash into [0, 49406)
        h = 2166136261
        for ch in word.encode("utf-8"):
            h = ((h ^ ch) * 16777619) & 0xFFFFFFFF
        return h % (self.BOS - 1) + 1  # avoid 0 (pad-ish) and specials

    def encode_words(self, text: str) -> List[int]:
        words = re.findall(r"[\w']+|[^\w\s]", str(text).lower())
        return [self._word_id(w) for w in words]

    def __call__(
        self,
        text: Union[str, List[str]],
        truncation: bool = True,
        padding: str = "max_length",
        max_length: int | None = None,
        return_tensors: str = "pt",
    ) -> _TokOut:
        max_length = max_length or self.model_max_length
        texts = [text] if isinstance(text, str) else list(text)
        rows = []
        for t in texts:
            ids = [self.BOS] + self.encode_words(t)
            if truncation:
                ids = ids[: max_length - 1]
            ids.append(self.EOS)
            if padding == "max_length":
                # CLIP pads with EOS token id (pad_token == eos for SD tokenizers)
                ids = ids + [self.EOS] * (max_length - len(ids))
            rows.append(ids)
        out = torch.tensor(rows, dtype=torch.long)
        return _TokOut(out)

    def decode(self, ids) -> str:
        # ids are hashes: decode to stable pseudo-words (round-trip not possible)
        toks = []
        for i in ids:
            i = int(i)
            if i in (self.BOS, self.EOS):
                continue
            toks.append(f"w{i % 9973}")
        return " ".join(toks)

    def save_pretrained(self, path):
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        (path / "tokenizer_config.json").write_text(json.dumps({
            "tokenizer_class": "HashTokenizer",
            "model_max_length": self.model_max_length,
            "vocab_size": self.vocab_size,
        }, indent=2))

    @classmethod
    def from_pretrained(cls, path):
        cfgf = Path(path) / "tokenizer_config.json"
        if cfgf.exists():
            d = json.loads(cfgf.read_text())
            if d.get("tokenizer_class") == "HashTokenizer":
                return cls(d.get("model_max_length", 77), d.get("vocab_size", 49408))
        return load_tokenizer(path)


def load_tokenizer(path=None, model_max_length: int = 77):
    """Real CLIPTokenizer if vocab files exist at `path`, else HashTokenizer."""
    if path is not None:
        p = Path(path)
        if (p / "vocab.json").exists() and (p / "merges.txt").exists():
            try:
                from transformers import CLIPTokenizer
                tok = CLIPTokenizer.from_pretrained(str(p))
                if tok.model_max_length > 10**6:  # no tokenizer_config.json
                    tok.model_max_length = model_max_length
                return tok
            except Exception:
                pass
        cfgf = p / "tokenizer_config.json"
        if cfgf.exists():
            d = json.loads(cfgf.read_text())
            if d.get("tokenizer_class") == "HashTokenizer":
                return HashTokenizer(d.get("model_max_length", 77),
                                     d.get("vocab_size", 49408))
    return HashTokenizer(model_max_length=model_max_length)
