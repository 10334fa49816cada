"""Property-based tests (hypothesis) for the pure-math helpers: these
hold for ARBITRARY shapes/values, not just the fixture shapes."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from dcr_amd import ops
from dcr_amd.data.tokenizer import HashTokenizer
from dcr_amd.retrieval.similarity import einsum_in_chunks
from dcr_amd.search import sharded_topk

SET = dict(max_examples=25, deadline=None)


@settings(**SET)
@given(n=st.integers(1, 7), m=st.integers(1, 6), c=st.integers(1, 9),
       p=st.integers(1, 5), chunk=st.integers(1, 8),
       stype=st.sampled_from(["cross", ""]))
def test_einsum_in_chunks_equals_dense(n, m, c, p, chunk, stype):
    g = torch.Generator().manual_seed(n * 1000 + m * 100 + c * 10 + p)
    a = torch.randn(n, c, p, generator=g)
    b = torch.randn(m, c, p, generator=g)
    out = einsum_in_chunks(a, b, chunk=chunk, stype=stype)
    if stype == "cross":
        ref = torch.einsum("ncp,mcq->nmpq", a, b).amax(dim=(2, 3))
    else:
        ref = torch.einsum("ncp,mcp->nmp", a, b).amax(dim=2)
    assert torch.allclose(out, ref, atol=1e-5)


@settings(**SET)
@given(q=st.integers(1, 6), ns=st.integers(1, 40), d=st.integers(2, 16),
       k=st.integers(1, 5), chunk=st.integers(1, 16))
def test_sharded_topk_equals_full(q, ns, d, k, chunk):
    k = min(k, ns)
    g = torch.Generator().manual_seed(q * 100 + ns)
    query = torch.randn(q, d, generator=g)
    shard = torch.randn(ns, d, generator=g)
    v, i = sharded_topk(query, shard, k=k, chunk=chunk, global_offset=7)
    ref = (query @ shard.t()).topk(k, dim=1)
    assert torch.allclose(v, ref.values, atol=1e-5)
    assert torch.equal(i - 7, ref.indices)


@settings(**SET)
@given(n=st.integers(1, 64),
       a=st.floats(-3, 3), b=st.floats(-3, 3), c=st.floats(-3, 3))
def test_lincomb_formula(n, a, b, c):
    g = torch.Generator().manual_seed(n)
    x = torch.randn(n, generator=g)
    y = torch.randn(n, generator=g)
    z = torch.randn(n, generator=g)
    out = ops.lincomb(x, y, a, b, z, c)
    assert torch.allclose(out, a * x + b * y + c * z, atol=1e-5)
    out2 = ops.lincomb(x, y, a, b)
    assert torch.allclose(out2, a * x + b * y, atol=1e-5)


@settings(**SET)
@given(text=st.text(max_size=200), max_length=st.integers(4, 77))
def test_hash_tokenizer_invariants(text, max_length):
    tok = HashTokenizer()
    ids = tok([text], max_length=max_length).input_ids[0].tolist()
    assert len(ids) == max_length
    assert ids[0] == tok.BOS
    assert tok.EOS in ids
    assert all(0 <= i < tok.vocab_size for i in ids)
    # deterministic
    again = tok([text], max_length=max_length).input_ids[0].tolist()
    assert ids == again
    # body ids never collide with specials
    body = ids[1:ids.index(tok.EOS)]
    assert all(i not in (tok.BOS, tok.EOS, 0) for i in body)


@settings(**SET)
@given(n=st.integers(1, 4), t=st.integers(0, 999))
def test_add_noise_velocity_identity(n, t):
    """sqrt(ac)*x_t - sqrt(1-ac)*v == x0 exactly (closed-form identity
    the v-prediction target relies on, diff_train.py:647-652)."""
    from dcr_amd.schedulers import DDPMScheduler
    s = DDPMScheduler()
    g = torch.Generator().manual_seed(n * 7 + t)
    x0 = torch.randn(n, 2, 4, 4, generator=g)
    eps = torch.randn(n, 2, 4, 4, generator=g)
    tt = torch.full((n,), t, dtype=torch.long)
    xt = s.add_noise(x0, eps, tt)
    v = s.get_velocity(x0, eps, tt)
    ac = s.alphas_cumprod[tt].view(-1, 1, 1, 1)
    rec = ac.sqrt() * xt - (1 - ac).sqrt() * v
    assert torch.allclose(rec, x0, atol=1e-5)


@settings(**SET)
@given(text=st.text(alphabet=st.characters(min_codepoint=32,
                                           max_codepoint=126), max_size=60),
       max_length=st.integers(8, 64))
def test_clip_bpe_tokenizer_properties(tmp_path_factory, text, max_length):
    """CLIPBPETokenizer: deterministic, BOS/EOS framing, fixed padded
    length, ids in-vocab — for arbitrary printable input."""
    import json
    from dcr_amd.data.tokenizer import CLIPBPETokenizer, _bytes_to_unicode
    global _BPE_TOK
    try:
        tok = _BPE_TOK
    except NameError:
        d = tmp_path_factory.mktemp("bpe")
        b2u = _bytes_to_unicode()
        vocab = {}
        for c in b2u.values():
            vocab[c] = len(vocab)
        for c in b2u.values():
            vocab[c + "</w>"] = len(vocab)
        vocab["<|startoftext|>"] = len(vocab)
        vocab["<|endoftext|>"] = len(vocab)
        (d / "vocab.json").write_text(json.dumps(vocab))
        (d / "merges.txt").write_text("#version: 0.2\n")
        tok = _BPE_TOK = CLIPBPETokenizer(d / "vocab.json", d / "merges.txt")
    ids = tok([text], max_length=max_length).input_ids[0].tolist()
    assert len(ids) == max_length
    assert ids[0] == tok.bos_token_id
    assert tok.eos_token_id in ids[1:]
    assert all(0 <= i < tok.vocab_size for i in ids)
    assert ids == tok([text], max_length=max_length).input_ids[0].tolist()


@settings(**SET)
@given(sizes=st.lists(st.integers(1, 40), min_size=1, max_size=6),
       accum=st.integers(1, 3))
def test_gather_grads_equals_autograd_sum(sizes, accum):
    """gather_grads over `accum` backward passes reproduces exactly what
    plain autograd accumulation would (copy-then-add invariant of the
    flat-arena design)."""
    torch.manual_seed(0)
    params = [torch.nn.Parameter(torch.randn(n)) for n in sizes]
    from dcr_amd.ops.adamw import FusedAdamW
    opt = FusedAdamW(params, lr=0.0, weight_decay=0.0)
    opt.zero_grad()
    expect = [torch.zeros_like(p) for p in params]
    for a in range(accum):
        loss = sum(((i + 1) * p * (a + 1)).sum() for i, p in enumerate(params))
        loss.backward()
        for i, p in enumerate(params):
            expect[i] += torch.full_like(p, float((i + 1) * (a + 1)))
        opt.gather_grads()
    for i, p in enumerate(params):
        assert torch.allclose(opt._view_of[id(p)], expect[i], atol=1e-5)
