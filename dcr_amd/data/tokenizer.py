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


def _bytes_to_unicode():
    """OpenAI CLIP's reversible byte<->unicode map (printable chars for
    every byte value so BPE merges operate on strings)."""
    bs = (list(range(ord("!"), ord("~") + 1))
          + list(range(ord("\xa1"), ord("\xac") + 1))
          + list(range(ord("\xae"), ord("\xff") + 1)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


class CLIPBPETokenizer:
    """In-house CLIP byte-level BPE (reference uses transformers'
    CLIPTokenizer, /root/reference/diff_train.py:371-383). Loads the
    standard vocab.json + merges.txt pair; matches transformers'
    CLIPTokenizer ids token-for-token (tested against it in
    tests/test_data.py on a constructed vocab — no network needed)."""

    def __init__(self, vocab_file, merges_file, model_max_length: int = 77):
        self.model_max_length = model_max_length
        self.encoder = json.loads(Path(vocab_file).read_text())
        self.decoder = {v: k for k, v in self.encoder.items()}
        merges = Path(merges_file).read_text().split("\n")
        if merges and merges[0].startswith("#version"):
            merges = merges[1:]
        merges = [tuple(m.split()) for m in merges if m.strip()]
        self.bpe_ranks = {m: i for i, m in enumerate(merges)}
        self.byte_encoder = _bytes_to_unicode()
        self.byte_decoder = {v: k for k, v in self.byte_encoder.items()}
        self.bos_token_id = self.encoder.get("<|startoftext|>", 49406)
        self.eos_token_id = self.encoder.get("<|endoftext|>", 49407)
        self.vocab_size = len(self.encoder)
        self._cache = {"<|startoftext|>": "<|startoftext|>",
                       "<|endoftext|>": "<|endoftext|>"}
        import regex
        self._pat = regex.compile(
            r"""<\|startoftext\|>|<\|endoftext\|>|'s|'t|'re|'ve|'m|'ll|'d|"""
            r"""[\p{L}]+|[\p{N}]|[^\s\p{L}\p{N}]+""", regex.IGNORECASE)

    def _bpe(self, token: str) -> str:
        if token in self._cache:
            return self._cache[token]
        word = tuple(token[:-1]) + (token[-1] + "</w>",)
        while len(word) > 1:
            pairs = {(word[i], word[i + 1]) for i in range(len(word) - 1)}
            best = min(pairs, key=lambda p: self.bpe_ranks.get(p, 1 << 30))
            if best not in self.bpe_ranks:
                break
            a, b = best
            out, i = [], 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == a and word[i + 1] == b:
                    out.append(a + b)
                    i += 2
                else:
                    out.append(word[i])
                    i += 1
            word = tuple(out)
        res = " ".join(word)
        self._cache[token] = res
        return res

    def encode_words(self, text: str) -> List[int]:
        text = re.sub(r"\s+", " ", str(text)).strip().lower()
        ids: List[int] = []
        for tok in self._pat.findall(text):
            tok = "".join(self.byte_encoder[b] for b in tok.encode("utf-8"))
            for piece in self._bpe(tok).split(" "):
                ids.append(self.encoder.get(
                    piece, self.encoder.get("<|endoftext|>", 0)))
        return ids

    def __call__(self, text, truncation=True, padding="max_length",
                 max_length=None, return_tensors="pt"):
        max_length = max_length or self.model_max_length
        texts = [text] if isinstance(text, str) else list(text)
        rows = []
        for t in texts:
            ids = [self.bos_token_id] + self.encode_words(t)
            if truncation:
                ids = ids[: max_length - 1]
            ids.append(self.eos_token_id)
            if padding == "max_length":
                ids = ids + [self.eos_token_id] * (max_length - len(ids))
            rows.append(ids)
        return _TokOut(torch.tensor(rows, dtype=torch.long))

    def decode(self, ids) -> str:
        toks = []
        for i in ids:
            i = int(i)
            if i in (self.bos_token_id, self.eos_token_id):
                continue
            toks.append(self.decoder.get(i, ""))
        text = "".join(toks)
        raw = bytearray(self.byte_decoder[c] for c in text
                        if c in self.byte_decoder)
        return raw.decode("utf-8", errors="replace").replace("</w>", " ").strip()

    def save_pretrained(self, path):
        path = Path(path)
        path.mkdir(parents=True, exist_ok=True)
        (path / "vocab.json").write_text(json.dumps(self.encoder))
        (path / "merges.txt").write_text(
            "#version: 0.2\n" + "\n".join(
                " ".join(m) for m, _ in
                sorted(self.bpe_ranks.items(), key=lambda kv: kv[1])) + "\n")
        (path / "tokenizer_config.json").write_text(json.dumps({
            "tokenizer_class": "CLIPTokenizer",
            "model_max_length": self.model_max_length}, indent=2))


def load_tokenizer(path=None, model_max_length: int = 77):
    """In-house CLIP BPE if vocab files exist at `path`, else HashTokenizer."""
    if path is not None:
        p = Path(path)
        if (p / "vocab.json").exists() and (p / "merges.txt").exists():
            try:
                cfgf = p / "tokenizer_config.json"
                mml = model_max_length
                if cfgf.exists():
                    mml = json.loads(cfgf.read_text()).get(
                        "model_max_length", model_max_length)
                return CLIPBPETokenizer(p / "vocab.json", p / "merges.txt",
                                        model_max_length=mml)
            except Exception:
                pass
        cfgf = p / "tokenizer_config.json"
        if cfgf.exists():
            d = json.loads(cfgf.read_text())
            if d.get("tokenizer_class") == "HashTokenizer":
                return HashTokenizer(d.get("model_max_length", 77),
                                     d.get("vocab_size", 49408))
    return HashTokenizer(model_max_length=model_max_length)
