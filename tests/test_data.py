"""Data-layer tests: ObjectAttributeDataset caption/duplication semantics
(/root/reference/datasets.py), collate, tokenizer."""
import json
import pickle

import numpy as np
import pytest
import torch
from PIL import Image

from dcr_amd.data import (HashTokenizer, ObjectAttributeDataset, SynthDataset,
                          SyntheticImageDataset, collate_fn, get_classnames,
                          insert_rand_word)


@pytest.fixture
def image_folder(tmp_path):
    root = tmp_path / "imagenette"
    prompts = {}
    for cls in ["church", "tench"]:
        d = root / cls
        d.mkdir(parents=True)
        for i in range(4):
            f = d / f"{i}.png"
            Image.new("RGB", (80, 70), color=(i * 30, 100, 50)).save(f)
            prompts[str(f)] = [f"a photo of a {cls} number {i}",
                               f"another caption for {cls} {i}"]
    pj = tmp_path / "blip.json"
    pj.write_text(json.dumps(prompts))
    return root, pj


def test_classnames():
    assert get_classnames("/data/imagenette_2class/train") == ["church", "garbage truck"]
    assert len(get_classnames("/data/imagenette/train")) == 10


def test_tokenizer_deterministic_and_shape():
    tok = HashTokenizer()
    out = tok(["hello world", "hello world"], return_tensors="pt").input_ids
    assert out.shape == (2, 77)
    assert torch.equal(out[0], out[1])
    assert out[0, 0] == 49406  # BOS
    assert (out[0] == 49407).any()  # EOS + padding


def test_dataset_classlevel(image_folder):
    root, pj = image_folder
    tok = HashTokenizer()
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="classlevel", size=64)
    ex = ds[0]
    assert ex["instance_images"].shape == (3, 64, 64)
    assert ex["instance_images"].min() >= -1.001 and ex["instance_images"].max() <= 1.001
    assert ex["instance_prompt_ids"].shape == (1, 77)


def test_dataset_instancelevel_blip(image_folder):
    root, pj = image_folder
    tok = HashTokenizer()
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="instancelevel_blip",
                                size=64, prompt_json=str(pj))
    ex = ds[0]
    path0 = ds.samples[0][0]
    expected = tok(ds.prompts[path0][0], truncation=True, padding="max_length",
                   max_length=77, return_tensors="pt").input_ids
    assert torch.equal(ex["instance_prompt_ids"], expected)


def test_dup_weights_pickled_and_cached(image_folder):
    root, pj = image_folder
    tok = HashTokenizer()
    np.random.seed(0)
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="classlevel", size=64,
                                duplication="dup_both", weight_pc=0.25,
                                dup_weight=5.0, seed=7)
    w = ds.samplingweights
    assert len(w) == len(ds.samples) == 8
    assert sum(1 for x in w if x > 1) == 2  # 25% of 8
    # cache file exists with reference naming (datasets.py:77)
    cache = root / "weights_0.25_5.0_seed7.pickle"
    assert cache.exists()
    with open(cache, "rb") as fh:
        assert pickle.load(fh) == w
    # second construction loads the cache (even with different rng state)
    np.random.seed(99)
    ds2 = ObjectAttributeDataset(str(root), tok, class_prompt="classlevel", size=64,
                                 duplication="dup_both", weight_pc=0.25,
                                 dup_weight=5.0, seed=7)
    assert ds2.samplingweights == w


def test_trainspecial_randrepl(image_folder):
    root, pj = image_folder
    tok = HashTokenizer()
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="instancelevel_blip",
                                size=64, prompt_json=str(pj),
                                trainspecial="randrepl", trainspecial_prob=1.0)
    ex = ds[0]  # always replaced with 4 random-token caption
    assert ex["instance_prompt_ids"].shape == (1, 77)


def test_insert_rand_word():
    s = insert_rand_word("a b c", "X")
    assert sorted(s.split(" ")) == ["X", "a", "b", "c"]


def test_collate(image_folder):
    root, pj = image_folder
    tok = HashTokenizer()
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="nolevel", size=64)
    batch = collate_fn([ds[0], ds[1]])
    assert batch["pixel_values"].shape == (2, 3, 64, 64)
    assert batch["pixel_values"].dtype == torch.float32
    assert batch["input_ids"].shape == (2, 77)


def test_synth_dataset_natural_order(tmp_path):
    d = tmp_path / "gens"
    d.mkdir()
    for i in [0, 1, 2, 10, 11]:
        Image.new("RGB", (32, 32)).save(d / f"{i}.png")
    (d / "prompts.txt").write_text("\n".join(f"p{i}" for i in range(5)))
    ds = SynthDataset(str(d), size=32, with_prompts=True)
    names = [f.name for f in ds.files]
    assert names == ["0.png", "1.png", "2.png", "10.png", "11.png"]
    t, idx, p = ds[3]
    assert t.shape == (3, 32, 32) and idx == 3 and p == "p3"


def test_synthetic_dataset_deterministic():
    tok = HashTokenizer()
    ds = SyntheticImageDataset(4, size=32, tokenizer=tok, seed=5)
    a, b = ds[2], ds[2]
    assert torch.equal(a["instance_images"], b["instance_images"])


def test_cutmix_mixup():
    from dcr_amd.data import cutmix_data, mixup_data, rand_bbox
    torch.manual_seed(0)
    np.random.seed(0)
    x = torch.rand(4, 3, 32, 32)
    y = torch.arange(4)
    mx, ya, yb, lam = cutmix_data(x, y, alpha=1.0)
    assert mx.shape == x.shape and 0.0 <= lam <= 1.0
    mx2, _, _, lam2 = mixup_data(x, y, alpha=1.0)
    assert torch.isfinite(mx2).all() and 0.0 <= lam2 <= 1.0
    b = rand_bbox(x.shape, 0.5)
    assert b[0] <= b[2] and b[1] <= b[3]


def test_bool_flag():
    import pytest as _pytest
    from dcr_amd.utils import bool_flag
    assert bool_flag("true") and bool_flag("on") and bool_flag("1")
    assert not bool_flag("false") and not bool_flag("off")
    with _pytest.raises(Exception):
        bool_flag("maybe")


def test_load_real_clip_tokenizer(tmp_path):
    """load_tokenizer uses the in-house CLIP BPE when vocab files exist
    (reference tokenizer path, diff_train.py:371-383)."""
    vocab = {"<|startoftext|>": 0, "<|endoftext|>": 1}
    # minimal BPE vocab: byte-level symbols + a merged token
    for i, ch in enumerate("abcdefghijklmnopqrstuvwxyz"):
        vocab[ch] = 2 + i
        vocab[ch + "</w>"] = 28 + i
    vocab["ab</w>"] = 60
    (tmp_path / "vocab.json").write_text(json.dumps(vocab))
    (tmp_path / "merges.txt").write_text("#version: 0.2\na b</w>\n")
    from dcr_amd.data.tokenizer import CLIPBPETokenizer, load_tokenizer
    tok = load_tokenizer(tmp_path, model_max_length=77)
    assert isinstance(tok, CLIPBPETokenizer)  # in-house BPE since round 2
    out = tok("ab", truncation=True, padding="max_length", max_length=16,
              return_tensors="pt")
    assert out.input_ids.shape == (1, 16)
    assert out.input_ids[0, 0].item() == 0  # bos
    assert out.input_ids[0, 1].item() == vocab["ab</w>"]


def test_transforms_shapes_and_range():
    from dcr_amd.data.transforms import (TrainTransform, EvalTransform,
                                         resize_shorter, center_crop)
    img = Image.new("RGB", (100, 60), color=(200, 30, 80))
    r = resize_shorter(img, 48)
    assert min(r.size) == 48 and r.size[0] == 80
    c = center_crop(r, 48)
    assert c.size == (48, 48)
    t = TrainTransform(32, center_crop=True)(img)
    assert t.shape == (3, 32, 32) and -1.001 <= t.min() and t.max() <= 1.001
    e = EvalTransform(32)(img)
    assert e.shape == (3, 32, 32)


def test_image_grid_utils():
    from dcr_amd.utils import concat_h, image_grid, tensor_to_pil
    imgs = [Image.new("RGB", (8, 8), color=(i * 40, 0, 0)) for i in range(4)]
    row = concat_h(imgs)
    assert row.size == (32, 8)
    grid = image_grid(imgs, rows=2, cols=2)
    assert grid.size == (16, 16)
    t = torch.rand(3, 8, 8) * 2 - 1
    p = tensor_to_pil(t)
    assert p.size == (8, 8)


def test_dataset_with_real_clip_tokenizer(tmp_path, image_folder):
    """ObjectAttributeDataset works with a real transformers CLIPTokenizer
    (same __call__/decode surface as HashTokenizer)."""
    root, pj = image_folder
    vocab = {"<|startoftext|>": 0, "<|endoftext|>": 1}
    for i, ch in enumerate("abcdefghijklmnopqrstuvwxyz"):
        vocab[ch] = 2 + i
        vocab[ch + "</w>"] = 28 + i
    (tmp_path / "tok" ).mkdir(exist_ok=True)
    (tmp_path / "tok" / "vocab.json").write_text(json.dumps(vocab))
    (tmp_path / "tok" / "merges.txt").write_text("#version: 0.2\n")
    from dcr_amd.data.tokenizer import load_tokenizer
    tok = load_tokenizer(tmp_path / "tok", model_max_length=77)
    ds = ObjectAttributeDataset(str(root), tok, class_prompt="classlevel",
                                size=64)
    ex = ds[0]
    assert ex["instance_prompt_ids"].shape == (1, 77)


def test_clip_bpe_matches_transformers(tmp_path):
    """In-house CLIP BPE (dcr_amd.data.tokenizer.CLIPBPETokenizer) produces
    token-for-token the ids transformers' CLIPTokenizer produces on the
    same vocab/merges files (constructed here — no network)."""
    import json
    from dcr_amd.data.tokenizer import CLIPBPETokenizer, _bytes_to_unicode

    b2u = _bytes_to_unicode()
    chars = list(b2u.values())
    vocab = {}
    for c in chars:
        vocab[c] = len(vocab)
    for c in chars:
        vocab[c + "</w>"] = len(vocab)
    merges = [("t", "h"), ("th", "e</w>"), ("a", "n"), ("an", "d</w>"),
              ("i", "n"), ("in", "g</w>"), ("r", "e"), ("o", "n</w>"),
              ("c", "a"), ("ca", "t</w>"), ("th", "e"), ("the", "re</w>")]
    for a, b in merges:
        for tok in (a, b, a + b):
            if tok not in vocab:
                vocab[tok] = len(vocab)
    vocab["<|startoftext|>"] = len(vocab)
    vocab["<|endoftext|>"] = len(vocab)
    (tmp_path / "vocab.json").write_text(json.dumps(vocab))
    (tmp_path / "merges.txt").write_text(
        "#version: 0.2\n" + "\n".join(f"{a} {b}" for a, b in merges) + "\n")

    ours = CLIPBPETokenizer(tmp_path / "vocab.json", tmp_path / "merges.txt")
    from transformers import CLIPTokenizer
    theirs = CLIPTokenizer(str(tmp_path / "vocab.json"),
                           str(tmp_path / "merges.txt"))

    samples = [
        "the cat sat on the mat",
        "A photograph of THE re-opening, and nothing else!",
        "there and then: 123 cats running, in-the rain...",
        "captain's log 42; the ending?",
        "",
    ]
    for s in samples:
        ref = theirs(s)["input_ids"]
        got = ours.encode_words(s)
        assert [vocab["<|startoftext|>"]] + got + [vocab["<|endoftext|>"]] \
            == ref, (s, got, ref)

    # padded batch surface (what the training pipeline calls)
    out = ours(samples[:2], max_length=16).input_ids
    assert out.shape == (2, 16)
    assert (out[:, 0] == vocab["<|startoftext|>"]).all()

    # save/load round-trip keeps ids
    ours.save_pretrained(tmp_path / "rt")
    from dcr_amd.data.tokenizer import load_tokenizer
    re_tok = load_tokenizer(tmp_path / "rt")
    assert type(re_tok).__name__ == "CLIPBPETokenizer"
    assert re_tok.encode_words(samples[0]) == ours.encode_words(samples[0])


def test_clip_bpe_decode_roundtrip(tmp_path):
    """decode() inverts encode for in-vocab text (strips specials and
    end-of-word markers)."""
    import json
    from dcr_amd.data.tokenizer import CLIPBPETokenizer, _bytes_to_unicode

    b2u = _bytes_to_unicode()
    vocab = {}
    for c in b2u.values():
        vocab[c] = len(vocab)
    for c in b2u.values():
        vocab[c + "</w>"] = len(vocab)
    vocab["<|startoftext|>"] = len(vocab)
    vocab["<|endoftext|>"] = len(vocab)
    (tmp_path / "vocab.json").write_text(json.dumps(vocab))
    (tmp_path / "merges.txt").write_text("#version: 0.2\n")
    tok = CLIPBPETokenizer(tmp_path / "vocab.json", tmp_path / "merges.txt")
    text = "a photo of a cat"
    ids = tok(text, max_length=32).input_ids[0].tolist()
    assert tok.decode(ids) == text
