"""Retrieval + metrics + search tests (CPU).

BASELINE config 1: SSCD dot-product retrieval on 128 random 64x64
images, CPU only — the full plumbing must run without a GPU."""
import pickle

import numpy as np
import pytest
import torch
from PIL import Image
from torch.utils.data import DataLoader

from dcr_amd.retrieval import (SSCDModel, extract_features, l2_normalize,
                               sim_matrix, topk_stats, glcm_entropy, jpeg_size,
                               tv_loss, load_sscd)
from dcr_amd.retrieval.similarity import einsum_in_chunks


def test_sscd_descriptor_shape_and_norm():
    torch.manual_seed(0)
    m = SSCDModel().eval()
    with torch.no_grad():
        f = m(torch.randn(2, 3, 64, 64))
    assert f.shape == (2, 512)
    assert torch.allclose(f.norm(dim=-1), torch.ones(2), atol=1e-5)


def test_baseline_config1_cpu_retrieval():
    """128 random 64x64 images -> SSCD features -> dot-product sim -> stats."""
    torch.manual_seed(1)
    model = load_sscd("sscd", device="cpu")
    imgs = torch.randn(128, 3, 64, 64)
    ds = [(imgs[i], i) for i in range(128)]
    loader = DataLoader(ds, batch_size=32)
    feats = extract_features(model, loader, torch.device("cpu"))
    assert feats.shape == (128, 512)
    feats = l2_normalize(feats)
    sim = sim_matrix(feats[:64], feats[64:])
    assert sim.shape == (64, 64)
    stats = topk_stats(sim, sim_matrix(feats[64:], feats[64:]))
    for k in ("sim_mean", "sim_std", "sim_95pc", "sim_gt_05pc",
              "bg_mean", "bg_95pc"):
        assert k in stats and np.isfinite(stats[k])
    # identical sets => self-similarity 1.0 on the diagonal
    sim_self = sim_matrix(feats, feats)
    assert torch.allclose(sim_self.diag(), torch.ones(128), atol=1e-4)


def test_extract_features_index_order():
    """features land at their dataset indices regardless of batch order."""
    model = torch.nn.Flatten()
    data = torch.arange(8, dtype=torch.float32).view(8, 1).repeat(1, 4)
    ds = [(data[i].view(1, 2, 2), i) for i in reversed(range(8))]
    loader = DataLoader(ds, batch_size=3)
    feats = extract_features(model, loader, torch.device("cpu"))
    assert torch.equal(feats[5], data[5])


def test_einsum_in_chunks_matches_dense():
    torch.manual_seed(0)
    a = torch.randn(6, 8, 3)
    b = torch.randn(5, 8, 3)
    out = einsum_in_chunks(a, b, chunk=2)
    ref = torch.einsum("ncp,mcq->nmpq", a, b).amax(dim=(2, 3))
    assert torch.allclose(out, ref, atol=1e-5)
    # same-patch variant (reference's live splitloss, diff_retrieval.py:397-400)
    out2 = einsum_in_chunks(a, b, chunk=2, stype="")
    ref2 = torch.einsum("ncp,mcp->nmp", a, b).amax(dim=2)
    assert torch.allclose(out2, ref2, atol=1e-5)
    # cross >= same-patch everywhere (max over a superset of pairs)
    assert (out >= out2 - 1e-5).all()


def test_backbone_arch_mapping():
    """Reference (pt_style, arch) pairs select the right backbone
    (diff_retrieval.py:249-285)."""
    import argparse
    from diff_retrieval import build_backbone
    def mk(**kw):
        d = dict(pt_style="sscd", arch="resnet50", pretrained="")
        d.update(kw)
        return argparse.Namespace(**d)
    m = build_backbone(mk(arch="resnet50_disc"), "cpu")  # sscd_disc_large: 1024-d
    x = torch.randn(2, 3, 224, 224)
    assert m(x).shape[-1] == 1024
    assert build_backbone(mk(), "cpu")(x).shape[-1] == 512
    d = build_backbone(mk(pt_style="dino", arch="vit_small"), "cpu")
    assert d.patch_embed.proj.kernel_size == (16, 16)
    d8 = build_backbone(mk(pt_style="dino", arch="vit_base8"), "cpu")
    assert d8.patch_embed.proj.kernel_size == (8, 8)
    rn = build_backbone(mk(pt_style="dino", arch="resnet50"), "cpu")
    assert rn(torch.randn(1, 3, 64, 64)).shape == (1, 2048)
    with pytest.raises(NotImplementedError):
        build_backbone(mk(pt_style="dino", arch="vit_base_cifar10"), "cpu")


def test_sharded_topk_bf16_rerank():
    """bf16-GEMM candidate pass + fp32 re-rank returns exact fp32 scores
    and recovers orderings that bf16 alone cannot resolve."""
    from dcr_amd.search import sharded_topk
    torch.manual_seed(0)
    q = torch.nn.functional.normalize(torch.randn(10, 64), dim=1)
    shard = torch.nn.functional.normalize(torch.randn(500, 64), dim=1)
    v32, i32 = sharded_topk(q, shard, k=3, chunk=128)
    vbf, ibf = sharded_topk(q, shard, k=3, chunk=128,
                            compute_dtype=torch.bfloat16)
    # re-ranked scores are exact fp32 values of the selected rows
    sel = torch.einsum("qd,qkd->qk", q, shard[ibf])
    assert torch.allclose(vbf, sel, atol=1e-6)
    # random cosine scores are well separated at k=3: same result as fp32
    assert torch.equal(i32, ibf)
    assert torch.allclose(v32, vbf, atol=1e-6)

    # crafted near-tie below bf16 resolution: two candidates score
    # 0.900000 vs 0.900150 — bf16 rounds both to the same value; the fp32
    # re-rank inside the margin must still order them correctly
    base = torch.nn.functional.normalize(torch.randn(1, 64), dim=1)
    a = torch.nn.functional.normalize(
        0.9 * base + (1 - 0.9**2) ** 0.5 *
        torch.nn.functional.normalize(torch.randn(1, 64) - base * (torch.randn(1, 64) @ base.t()), dim=1), dim=1)
    # build candidates with controlled exact scores against `base`
    d = 64
    e1 = torch.zeros(1, d); e1[0, 0] = 1.0
    e2 = torch.zeros(1, d); e2[0, 1] = 1.0
    qq = e1  # query along axis 0
    c_lo = 0.900000 * e1 + (1 - 0.900000**2) ** 0.5 * e2
    c_hi = 0.900150 * e1 + (1 - 0.900150**2) ** 0.5 * e2
    filler = torch.nn.functional.normalize(torch.randn(50, d), dim=1) * 0.5
    shard2 = torch.cat([c_lo, filler, c_hi], dim=0)
    v, i = sharded_topk(qq, shard2, k=1, chunk=16,
                        compute_dtype=torch.bfloat16)
    assert int(i[0, 0]) == shard2.shape[0] - 1, "re-rank missed the true top-1"
    assert abs(float(v[0, 0]) - 0.900150) < 1e-5


def test_complexity_metrics():
    rng = np.random.default_rng(0)
    noise = rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)
    flat = np.full((64, 64, 3), 128, np.uint8)
    assert glcm_entropy(noise) > glcm_entropy(flat)
    assert jpeg_size(Image.fromarray(noise)) > jpeg_size(Image.fromarray(flat))
    t_noise = torch.rand(3, 32, 32)
    t_flat = torch.full((3, 32, 32), 0.5)
    assert tv_loss(t_noise) > tv_loss(t_flat)


def test_fid_identical_sets_near_zero(tmp_path):
    from dcr_amd.metrics import calculate_fid_given_paths
    rng = np.random.default_rng(0)
    d1, d2 = tmp_path / "a", tmp_path / "b"
    d1.mkdir(); d2.mkdir()
    for i in range(8):
        arr = rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)
        Image.fromarray(arr).save(d1 / f"{i}.png")
        Image.fromarray(arr).save(d2 / f"{i}.png")
    fid = calculate_fid_given_paths([str(d1), str(d2)], batch_size=4,
                                    device="cpu", dims=64)
    assert fid < 1e-3, fid


def test_fid_different_sets_positive(tmp_path):
    from dcr_amd.metrics import calculate_fid_given_paths
    rng = np.random.default_rng(0)
    d1, d2 = tmp_path / "a", tmp_path / "b"
    d1.mkdir(); d2.mkdir()
    for i in range(8):
        Image.fromarray(rng.integers(0, 128, (64, 64, 3)).astype(np.uint8)) \
            .save(d1 / f"{i}.png")
        Image.fromarray(rng.integers(128, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(d2 / f"{i}.png")
    fid = calculate_fid_given_paths([str(d1), str(d2)], batch_size=4,
                                    device="cpu", dims=64)
    assert fid > 0.01


def test_ipr_precision_recall():
    from dcr_amd.metrics.ipr import IPR
    torch.manual_seed(0)
    ipr = IPR(batch_size=8, k=3)
    ref = torch.rand(16, 3, 32, 32)
    ipr.compute_manifold_ref(ref)
    pr = ipr.precision_and_recall(ref + 0.01 * torch.randn_like(ref))
    assert 0.0 <= pr.precision <= 1.0 and 0.0 <= pr.recall <= 1.0
    assert pr.precision > 0.5  # nearly identical distributions overlap


def test_embedding_pickle_contract(tmp_path):
    """embedding.pkl = {'features': [N,D] f32, 'indexes': list}."""
    from dcr_amd.search import generate_embeddings
    blob = generate_embeddings(None, tmp_path / "embedding.pkl",
                               batch_size=16, device="cpu", num_workers=0,
                               synthetic_n=24, seed=3)
    assert blob["features"].dtype == np.float32
    assert blob["features"].shape[0] == 24
    assert len(blob["indexes"]) == 24
    with open(tmp_path / "embedding.pkl", "rb") as fh:
        loaded = pickle.load(fh)
    assert set(loaded.keys()) == {"features", "indexes"}


def test_stream_top1_finds_planted_match(tmp_path):
    from dcr_amd.search import stream_top1, dump_matches
    rng = np.random.default_rng(0)
    D = 16
    # two chunks; plant exact query rows at known places
    q = rng.normal(size=(4, D)).astype(np.float32)
    q /= np.linalg.norm(q, axis=1, keepdims=True)
    c1 = rng.normal(size=(50, D)).astype(np.float32)
    c2 = rng.normal(size=(50, D)).astype(np.float32)
    c1 /= np.linalg.norm(c1, axis=1, keepdims=True)
    c2 /= np.linalg.norm(c2, axis=1, keepdims=True)
    c1[7] = q[0]
    c2[3] = q[2]
    for name, feats in [("embedding_0.pkl", c1), ("embedding_1.pkl", c2)]:
        with open(tmp_path / name, "wb") as fh:
            pickle.dump({"features": feats,
                         "indexes": [f"{name}:{i}" for i in range(50)]}, fh)
    scores, keys = stream_top1(torch.from_numpy(q),
                               sorted(tmp_path.glob("*.pkl")), device="cpu")
    assert scores[0] > 0.999 and keys[0] == "embedding_0.pkl:7"
    assert scores[2] > 0.999 and keys[2] == "embedding_1.pkl:3"
    dump_matches(scores, keys, tmp_path / "match.pkl")
    with open(tmp_path / "match.pkl", "rb") as fh:
        m = pickle.load(fh)
    assert list(m.keys()) == ["scores", "keys"]


def test_sharded_topk_matches_full():
    from dcr_amd.search import sharded_topk
    torch.manual_seed(0)
    q = torch.randn(5, 8)
    shard = torch.randn(100, 8)
    v, i = sharded_topk(q, shard, k=3, chunk=17, global_offset=1000)
    ref = (q @ shard.t()).topk(3, dim=1)
    assert torch.allclose(v, ref.values, atol=1e-5)
    assert torch.equal(i, ref.indices + 1000)


def test_compute_map():
    from dcr_amd.retrieval.map_eval import compute_map
    # 4 db items, 2 queries; query0: ok={0,1} ranked first -> AP 1.0
    ranks = np.array([[0, 3], [1, 2], [2, 1], [3, 0]])
    gnd = [{"ok": [0, 1], "junk": []}, {"ok": [3], "junk": [2]}]
    m, aps, _, _ = compute_map(ranks, gnd)
    assert aps[0] == 1.0
    assert aps[1] == 1.0  # junk(2) ranked above ok(3) is ignored
    assert m == 1.0


def test_fid_stats_caching(tmp_path):
    """save_fid_stats -> .npz consumed by calculate_fid_given_paths
    (reference fid.py:226-228,258-275)."""
    from dcr_amd.metrics import calculate_fid_given_paths, save_fid_stats
    rng = np.random.default_rng(1)
    d = tmp_path / "imgs"
    d.mkdir()
    for i in range(10):
        Image.fromarray(rng.integers(0, 255, (48, 48, 3)).astype(np.uint8)) \
            .save(d / f"{i}.png")
    npz = tmp_path / "stats.npz"
    save_fid_stats(str(d), str(npz), batch_size=5, dims=64)
    assert npz.exists()
    # identical statistics up to sqrtm noise on a low-rank covariance
    fid = calculate_fid_given_paths([str(npz), str(d)], batch_size=5,
                                    device="cpu", dims=64)
    assert fid < 0.5, fid


def test_ipr_realism_score():
    from dcr_amd.metrics.ipr import IPR
    torch.manual_seed(3)
    ipr = IPR(batch_size=8, k=3)
    ref = torch.rand(12, 3, 32, 32)
    ipr.compute_manifold_ref(ref)
    feats = ipr.extract_features(ref[:1])
    r = ipr.realism(feats[0])
    assert np.isfinite(r) and r > 0


def test_multiscale_features():
    """multiscale eval: 1x + 1/sqrt(2) + 1/2 scales, summed and renormed
    (reference utils_ret.py:676-698)."""
    from torch.utils.data import DataLoader
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.AdaptiveAvgPool2d(1), torch.nn.Flatten())
    imgs = torch.rand(6, 3, 32, 32)
    ds = [(imgs[i], i) for i in range(6)]
    feats = extract_features(model, DataLoader(ds, batch_size=3),
                             torch.device("cpu"), multiscale=True)
    assert feats.shape == (6, 3)
    norms = feats.norm(dim=-1)
    assert torch.allclose(norms, torch.ones(6), atol=1e-4)  # renormalized
