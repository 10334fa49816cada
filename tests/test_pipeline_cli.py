"""Pipeline + CLI golden-file tests (CPU, tiny models).

Covers the reference's CLI/artifact contracts: prompts.txt +
generations/{i}.png naming (diff_inference.py:179-201), savepath
derivation (:44-81), checkpoint_{step} consumption, retrieval CLI
end-to-end on synthetic data (BASELINE config 1 shape)."""
import json
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest
import torch
from PIL import Image

from dcr_amd.data import HashTokenizer
from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                            UNet2DConditionModel, UNetConfig, VAEConfig)
from dcr_amd.pipelines import StableDiffusionPipeline
from dcr_amd.schedulers import DDIMScheduler, DPMSolverMultistepScheduler


def tiny_pipe(embed_noise_lam=0.0, scheduler=None):
    torch.manual_seed(0)
    return StableDiffusionPipeline(
        UNet2DConditionModel(UNetConfig.tiny()),
        AutoencoderKL(VAEConfig.tiny()),
        CLIPTextModel(CLIPTextConfig.tiny()),
        HashTokenizer(),
        scheduler or DDIMScheduler(),
        embed_noise_lam=embed_noise_lam)


def test_pipeline_generates_images():
    pipe = tiny_pipe()
    out = pipe("a photo of a church", height=64, width=64,
               num_inference_steps=3, num_images_per_prompt=2,
               generator=torch.Generator().manual_seed(0))
    assert len(out.images) == 2
    assert out.images[0].size == (64, 64)


def test_pipeline_dpm_solver():
    pipe = tiny_pipe(scheduler=DPMSolverMultistepScheduler())
    out = pipe("hello", height=64, width=64, num_inference_steps=4)
    assert len(out.images) == 1


def test_pipeline_embed_noise_changes_output():
    g1 = torch.Generator().manual_seed(7)
    g2 = torch.Generator().manual_seed(7)
    p0 = tiny_pipe(0.0)
    imgs_a = p0("x", height=64, width=64, num_inference_steps=2, generator=g1,
                output_type="pt").images
    p1 = tiny_pipe(0.5)  # same seed for weights (manual_seed(0) inside)
    imgs_b = p1("x", height=64, width=64, num_inference_steps=2, generator=g2,
                output_type="pt").images
    assert not torch.allclose(imgs_a, imgs_b, atol=1e-4)


def test_pipeline_save_load_roundtrip(tmp_path):
    pipe = tiny_pipe()
    pipe.save_pretrained(tmp_path / "checkpoint")
    assert (tmp_path / "checkpoint" / "model_index.json").exists()
    pipe2 = StableDiffusionPipeline.from_pretrained(tmp_path / "checkpoint")
    g1 = torch.Generator().manual_seed(3)
    g2 = torch.Generator().manual_seed(3)
    a = pipe("same", height=64, width=64, num_inference_steps=2, generator=g1,
             output_type="pt").images
    b = pipe2("same", height=64, width=64, num_inference_steps=2, generator=g2,
              output_type="pt").images
    assert torch.allclose(a, b, atol=1e-4)


def test_savepath_derivation():
    import diff_inference as di

    class A:
        modelpath = "/models/myrun_imagenette_instancelevel_blip_nodup"
        dataset = None
        capstyle = None
        iternum = 2000
        modelstyle = "instancelevel_blip"
        rand_noise_lam = 0.1
        rand_augs = None
        rand_aug_repeats = 2

    sp = di.derive_savepath(A())
    assert sp == ("./inferences/imagenette10_frozentext/"
                  "myrun_imagenette_instancelevel_blip_nodup_2000/"
                  "instancelevel_blip_ginfer0.1")


def test_prompt_augmentation_modes():
    import diff_inference as di
    tok = HashTokenizer()
    np.random.seed(0)
    base = "a photo of a dog"
    p1 = di.prompt_augmentation(base, "rand_numb_add", tok, 2)
    assert len(p1.split()) == len(base.split()) + 2
    p2 = di.prompt_augmentation(base, "rand_word_add", tok, 2)
    assert len(p2.split()) >= len(base.split()) + 2
    p3 = di.prompt_augmentation(base, "rand_word_repeat", tok, 2)
    assert len(p3.split()) == len(base.split()) + 2
    with pytest.raises(Exception):
        di.prompt_augmentation(base, "nope", tok)


@pytest.mark.timeout(600)
def test_retrieval_cli_end_to_end(tmp_path):
    """diff_retrieval.py over synthetic generations+train dirs, CPU."""
    rng = np.random.default_rng(0)
    qdir, vdir = tmp_path / "gens", tmp_path / "train"
    qdir.mkdir(); vdir.mkdir()
    for i in range(6):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(qdir / f"{i}.png")
    for i in range(8):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(vdir / f"{i}.png")
    (qdir / "prompts.txt").write_text("\n".join(f"prompt {i}" for i in range(6)))
    out = tmp_path / "simscores"
    r = subprocess.run(
        [sys.executable, "diff_retrieval.py", "--query_dir", str(qdir),
         "--val_dir", str(vdir), "--pt_style", "sscd", "-b", "8", "-j", "0",
         "--imsize", "64", "-ssp", str(out), "--skip_fid", "--noeval"],
        capture_output=True, text=True, cwd=str(Path(__file__).parent.parent),
        timeout=570)
    assert r.returncode == 0, r.stderr[-2000:]
    assert (out / "similarity.pth").exists()
    assert (out / "similarity_wtrain.pth").exists()
    log = out / "imsimv2_retrieval_log.jsonl"
    recs = [json.loads(l) for l in log.read_text().splitlines()]
    assert any("sim_mean" in r_ for r_ in recs)


@pytest.mark.timeout(900)
def test_train_cli_end_to_end(tmp_path):
    """diff_train.py tiny synthetic run writes mangled output dir +
    checkpoint layout + jsonl logs."""
    r = subprocess.run(
        [sys.executable, "diff_train.py", "--synthetic_data",
         "--model_size", "tiny", "--resolution", "64",
         "--train_batch_size", "2", "--max_train_steps", "2",
         "--mixed_precision", "no", "--class_prompt", "classlevel",
         "--num_workers", "0", "--seed", "0", "--save_steps", "1000",
         "--modelsavesteps", "1000",
         "--output_dir", str(tmp_path / "m")],
        capture_output=True, text=True, cwd=str(Path(__file__).parent.parent),
        timeout=870)
    assert r.returncode == 0, r.stderr[-3000:]
    out = tmp_path / "m_classlevel_nodup"
    assert (out / "checkpoint" / "unet" / "config.json").exists(), \
        list(tmp_path.iterdir())
    assert (out / "checkpoint" / "state.pt").exists()


def test_train_cli_resume(tmp_path):
    """--resume auto picks the latest checkpoint_N and continues from its
    step count."""
    args = [sys.executable, "diff_train.py", "--synthetic_data",
            "--model_size", "tiny", "--resolution", "64",
            "--train_batch_size", "2", "--mixed_precision", "no",
            "--class_prompt", "classlevel", "--num_workers", "0",
            "--seed", "0", "--save_steps", "1000",
            "--output_dir", str(tmp_path / "m")]
    r = subprocess.run(args + ["--max_train_steps", "2",
                               "--modelsavesteps", "2"],
                       capture_output=True, text=True,
                       cwd=str(Path(__file__).parent.parent), timeout=870)
    assert r.returncode == 0, r.stderr[-3000:]
    out = tmp_path / "m_classlevel_nodup"
    assert (out / "checkpoint_2" / "state.pt").exists()
    r2 = subprocess.run(args + ["--max_train_steps", "4",
                                "--modelsavesteps", "1000",
                                "--resume", "auto"],
                        capture_output=True, text=True,
                        cwd=str(Path(__file__).parent.parent), timeout=870)
    assert r2.returncode == 0, r2.stderr[-3000:]
    assert "resumed from" in r2.stdout and "step 2" in r2.stdout, r2.stdout
    import torch as _t
    st = _t.load(out / "checkpoint" / "state.pt", map_location="cpu",
                 weights_only=False)
    assert st["global_step"] == 4


@pytest.mark.timeout(900)
def test_inference_cli_end_to_end(tmp_path, monkeypatch):
    """diff_inference.py from a tiny checkpoint: prompts.txt + numbered pngs."""
    ckpt_root = tmp_path / "run_imagenette_classlevel_nodup"
    pipe = tiny_pipe()
    pipe.save_pretrained(ckpt_root / "checkpoint")
    r = subprocess.run(
        [sys.executable,
         str(Path(__file__).parent.parent / "diff_inference.py"),
         "--modelpath", str(ckpt_root),
         "-nb", "2", "-imb", "2", "--resolution", "64", "--seed", "0"],
        capture_output=True, text=True, cwd=str(tmp_path),
        env={**__import__("os").environ,
             "PYTHONPATH": str(Path(__file__).parent.parent)},
        timeout=870)
    assert r.returncode == 0, r.stderr[-3000:]
    gen = tmp_path / "inferences" / "imagenette10_frozentext" / \
        "run_imagenette_classlevel_nodup" / "classlevel"
    assert (gen / "prompts.txt").exists(), r.stdout[-2000:]
    pngs = sorted((gen / "generations").glob("*.png"))
    assert [p.name for p in pngs] == ["0.png", "1.png", "2.png", "3.png"]


def test_convert_diffusers_checkpoint_roundtrip(tmp_path):
    """Our save_pretrained output IS diffusers-layout; the converter must
    round-trip it losslessly (keys match by construction)."""
    sys.path.insert(0, str(Path(__file__).parent.parent / "scripts"))
    import convert_diffusers_checkpoint as conv
    pipe = tiny_pipe()
    src = tmp_path / "src"
    pipe.save_pretrained(src)
    dst = tmp_path / "dst"
    conv.convert(src, dst, check=True)
    pipe2 = StableDiffusionPipeline.from_pretrained(dst)
    for k, v in pipe.unet.state_dict().items():
        assert torch.equal(v, pipe2.unet.state_dict()[k]), k
    for k, v in pipe.text_encoder.state_dict().items():
        assert torch.equal(v, pipe2.text_encoder.state_dict()[k]), k


@pytest.mark.timeout(600)
def test_retrieval_cli_splitloss(tmp_path):
    """--similarity_metric splitloss (patch-wise einsum path)."""
    rng = np.random.default_rng(0)
    qdir, vdir = tmp_path / "gens", tmp_path / "train"
    qdir.mkdir(); vdir.mkdir()
    for i in range(3):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(qdir / f"{i}.png")
    for i in range(4):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(vdir / f"{i}.png")
    out = tmp_path / "s"
    r = subprocess.run(
        [sys.executable, "diff_retrieval.py", "--query_dir", str(qdir),
         "--val_dir", str(vdir), "--pt_style", "sscd", "-b", "4", "-j", "0",
         "--imsize", "64", "-ssp", str(out), "--skip_fid", "--noeval",
         "--similarity_metric", "splitloss", "--einsum_chunks", "2"],
        capture_output=True, text=True, cwd=str(Path(__file__).parent.parent),
        timeout=570)
    assert r.returncode == 0, r.stderr[-2000:]
    sim = torch.load(out / "similarity.pth")
    assert sim.shape == (3, 4)


@pytest.mark.timeout(900)
def test_full_product_pipeline(tmp_path):
    """SURVEY §1 data flow end-to-end: diff_train writes checkpoint_{step}/
    -> diff_inference reads it and writes generations/{i}.png + prompts.txt
    -> diff_retrieval consumes (query=generations, val=train data)."""
    root = Path(__file__).parent.parent
    env = {**__import__("os").environ, "PYTHONPATH": str(root)}

    # 1) train (tiny, synthetic, 2 steps) — note 'imagenette' in the name
    #    so inference derives the imagenette10 savepath branch
    out_root = tmp_path / "run_imagenette"
    r = subprocess.run(
        [sys.executable, str(root / "diff_train.py"), "--synthetic_data",
         "--model_size", "tiny", "--resolution", "64", "--train_batch_size",
         "2", "--max_train_steps", "2", "--mixed_precision", "no",
         "--class_prompt", "classlevel", "--num_workers", "0", "--seed", "0",
         "--save_steps", "1000", "--modelsavesteps", "1000",
         "--output_dir", str(out_root)],
        capture_output=True, text=True, cwd=str(tmp_path), env=env, timeout=400)
    assert r.returncode == 0, r.stderr[-2000:]
    model_dir = tmp_path / "run_imagenette_classlevel_nodup"
    assert (model_dir / "checkpoint").is_dir()

    # 2) inference from the checkpoint
    r = subprocess.run(
        [sys.executable, str(root / "diff_inference.py"), "--modelpath",
         str(model_dir), "-nb", "3", "-imb", "1", "--resolution", "64",
         "--seed", "0"],
        capture_output=True, text=True, cwd=str(tmp_path), env=env, timeout=400)
    assert r.returncode == 0, r.stderr[-2000:]
    gen_dir = tmp_path / "inferences" / "imagenette10_frozentext" / \
        "run_imagenette_classlevel_nodup" / "classlevel"
    pngs = list((gen_dir / "generations").glob("*.png"))
    assert len(pngs) == 3 and (gen_dir / "prompts.txt").exists()

    # 3) retrieval: generations vs a synthetic "train" dir
    rng = np.random.default_rng(0)
    vdir = tmp_path / "train_imgs"
    vdir.mkdir()
    for i in range(5):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(vdir / f"{i}.png")
    out = tmp_path / "scores"
    r = subprocess.run(
        [sys.executable, str(root / "diff_retrieval.py"), "--query_dir",
         str(gen_dir / "generations"), "--val_dir", str(vdir), "--pt_style",
         "sscd", "-b", "4", "-j", "0", "--imsize", "64", "-ssp", str(out),
         "--skip_fid", "--noeval"],
        capture_output=True, text=True, cwd=str(tmp_path), env=env, timeout=400)
    assert r.returncode == 0, r.stderr[-2000:]
    sim = torch.load(out / "similarity.pth")
    assert sim.shape == (3, 5)


@pytest.mark.timeout(600)
def test_retrieval_cli_torchrun_world2(tmp_path):
    """distributed retrieval CLI (2 ranks, gloo): sharded extraction +
    all-gather must produce the same similarity artifacts."""
    import os as _os
    rng = np.random.default_rng(3)
    qdir, vdir = tmp_path / "gens", tmp_path / "train"
    qdir.mkdir(); vdir.mkdir()
    for i in range(5):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(qdir / f"{i}.png")
    for i in range(7):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(vdir / f"{i}.png")
    root = Path(__file__).parent.parent

    def run(out, extra):
        r = subprocess.run(
            extra + [str(root / "diff_retrieval.py"), "--query_dir", str(qdir),
                     "--val_dir", str(vdir), "--pt_style", "sscd", "-b", "4",
                     "-j", "0", "--imsize", "64", "-ssp", str(out),
                     "--skip_fid", "--noeval", "--dontsave"],
            capture_output=True, text=True, cwd=str(root),
            env={**_os.environ, "MASTER_ADDR": "127.0.0.1"}, timeout=560)
        assert r.returncode == 0, r.stderr[-2000:]
        log = out / "imsimv2_retrieval_log.jsonl"
        recs = [json.loads(l) for l in log.read_text().splitlines()]
        return next(r_ for r_ in recs if "sim_mean" in r_)

    single = run(tmp_path / "o1", [sys.executable])
    multi = run(tmp_path / "o2",
                [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                 "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
                 "--master-port", "29537"])
    assert abs(single["sim_mean"] - multi["sim_mean"]) < 1e-4
    assert abs(single["bg_mean"] - multi["bg_mean"]) < 1e-4


def test_inference_cli_torchrun_world2(tmp_path):
    """rank-sharded generation (diff_inference under torchrun, 2 ranks):
    disjoint batch slices, global numbering, complete output set."""
    import os as _os
    ckpt_root = tmp_path / "run_imagenette_classlevel_nodup"
    pipe = tiny_pipe()
    pipe.save_pretrained(ckpt_root / "checkpoint")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29539",
         str(Path(__file__).parent.parent / "diff_inference.py"),
         "--modelpath", str(ckpt_root),
         "-nb", "4", "-imb", "1", "--resolution", "64", "--seed", "0"],
        capture_output=True, text=True, cwd=str(tmp_path),
        env={**_os.environ, "PYTHONPATH": str(Path(__file__).parent.parent),
             "MASTER_ADDR": "127.0.0.1"},
        timeout=870)
    assert r.returncode == 0, r.stderr[-3000:]
    gen = tmp_path / "inferences" / "imagenette10_frozentext" / \
        "run_imagenette_classlevel_nodup" / "classlevel"
    assert (gen / "prompts.txt").exists(), r.stdout[-2000:]
    pngs = sorted((gen / "generations").glob("*.png"))
    assert [p.name for p in pngs] == ["0.png", "1.png", "2.png", "3.png"]


def test_pipeline_no_cfg_path():
    """guidance_scale <= 1 skips the CFG double batch."""
    pipe = tiny_pipe()
    out = pipe("x", height=64, width=64, num_inference_steps=2,
               guidance_scale=1.0, output_type="pt")
    assert out.images.shape == (1, 3, 64, 64)


@pytest.mark.timeout(600)
def test_retrieval_complexity_on_matched_train_images(tmp_path):
    """complexity metrics are computed on the matched TRAIN image and
    logged with the reference's cc_*/pval_* keys + artifact names."""
    rng = np.random.default_rng(5)
    qdir, vdir = tmp_path / "gens", tmp_path / "train"
    qdir.mkdir(); vdir.mkdir()
    for i in range(4):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(qdir / f"{i}.png")
    for i in range(6):
        Image.fromarray(rng.integers(0, 255, (64, 64, 3)).astype(np.uint8)) \
            .save(vdir / f"{i}.png")
    out = tmp_path / "s"
    r = subprocess.run(
        [sys.executable, "diff_retrieval.py", "--query_dir", str(qdir),
         "--val_dir", str(vdir), "--pt_style", "sscd", "-b", "4", "-j", "0",
         "--imsize", "64", "-ssp", str(out), "--skip_fid"],
        capture_output=True, text=True, cwd=str(Path(__file__).parent.parent),
        timeout=570)
    assert r.returncode == 0, r.stderr[-2000:]
    for name in ("entropies.pth", "totvar.pth", "compressions.pth",
                 "dbsims.pth"):
        assert (out / name).exists(), name
    assert torch.load(out / "dbsims.pth", weights_only=False).shape == (4,)
    recs = [json.loads(l) for l in
            (out / "imsimv2_retrieval_log.jsonl").read_text().splitlines()]
    comp = next(r_ for r_ in recs if "cc_ent" in r_)
    for k in ("cc_ent", "pval_ent", "cc_comp", "pval_comp", "cc_tvl",
              "pval_tvl", "cc_mixed", "pval_mixed"):
        assert k in comp, k


def test_sd_mitigation_prompt_list():
    """the 12 known-replicating LAION prompts (reference sd_mitigation.py:81)."""
    import sd_mitigation as sm
    assert len(sm.PROMPT_LIST) == 12
    assert sm.PROMPT_LIST[0] == "Wall View 002"
    assert "The No Limits Business Woman Podcast" in sm.PROMPT_LIST
    assert "Mothers influence on her young hippo" in sm.PROMPT_LIST


@pytest.mark.timeout(900)
def test_inference_cli_mitigations(tmp_path):
    """Newpipe embedding-noise + prompt augmentation via the CLI:
    savepath suffixes (reference :77-81) and different outputs."""
    ckpt_root = tmp_path / "run_imagenette_classlevel_nodup"
    pipe = tiny_pipe()
    pipe.save_pretrained(ckpt_root / "checkpoint")
    root = Path(__file__).parent.parent
    env = {**__import__("os").environ, "PYTHONPATH": str(root)}

    r = subprocess.run(
        [sys.executable, str(root / "diff_inference.py"), "--modelpath",
         str(ckpt_root), "-nb", "1", "-imb", "1", "--resolution", "64",
         "--seed", "0", "--rand_noise_lam", "0.5"],
        capture_output=True, text=True, cwd=str(tmp_path), env=env, timeout=430)
    assert r.returncode == 0, r.stderr[-2000:]
    g1 = tmp_path / "inferences" / "imagenette10_frozentext" / \
        "run_imagenette_classlevel_nodup" / "classlevel_ginfer0.5"
    assert (g1 / "generations" / "0.png").exists()

    r = subprocess.run(
        [sys.executable, str(root / "diff_inference.py"), "--modelpath",
         str(ckpt_root), "-nb", "1", "-imb", "1", "--resolution", "64",
         "--seed", "0", "--capstyle", "instancelevel_blip",
         "--rand_augs", "rand_word_add", "--rand_aug_repeats", "2"],
        capture_output=True, text=True, cwd=str(tmp_path), env=env, timeout=430)
    assert r.returncode == 0, r.stderr[-2000:]
    g2 = tmp_path / "inferences" / "imagenette10_frozentext" / \
        "run_imagenette_classlevel_nodup" / \
        "instancelevel_blip_auginfer_rand_word_add_2"
    assert (g2 / "prompts.txt").exists()
    # augmented prompt gained 2 words vs the synthetic pool's base form
    aug = (g2 / "prompts.txt").read_text().strip()
    assert len(aug.split()) >= 7  # "An image of X variant N" + 2 words
