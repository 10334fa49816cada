"""GPU end-to-end: tiny + flagship train step on MI355X, HIP path active."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_tiny_train_step_gpu(tmp_path):
    from dcr_amd.train import TrainConfig, Trainer
    cfg = TrainConfig(model_size="tiny", synthetic_data=True, synthetic_size=8,
                      resolution=64, train_batch_size=2, mixed_precision="bf16",
                      dataloader_num_workers=0, max_train_steps=4, seed=0,
                      learning_rate=1e-4, lr_warmup_steps=1,
                      output_dir=str(tmp_path / "out"))
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    batch = next(iter(tr.dataloader))
    losses = [tr.train_step(batch).item() for _ in range(6)]
    assert all(l == l for l in losses)
    assert losses[-1] < losses[0], losses


@pytest.mark.skipif(__import__("os").environ.get("DCR_DEV_ADAMW") != "1",
                    reason="device-state AdamW draft: validate in round 2 "
                           "(DCR_DEV_ADAMW=1)")
def test_tiny_train_step_device_state_adamw(tmp_path, monkeypatch):
    """Whole train step through the device-state optimizer path: hyper
    advances on device and the loss still goes down."""
    from dcr_amd.train import TrainConfig, Trainer
    cfg = TrainConfig(model_size="tiny", synthetic_data=True, synthetic_size=8,
                      resolution=64, train_batch_size=2, mixed_precision="bf16",
                      dataloader_num_workers=0, max_train_steps=4, seed=0,
                      learning_rate=1e-4, lr_warmup_steps=1,
                      output_dir=str(tmp_path / "out"))
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    assert tr.optimizer.hyper is not None
    batch = next(iter(tr.dataloader))
    losses = [tr.train_step(batch).item() for _ in range(6)]
    torch.cuda.synchronize()
    assert int(tr.optimizer.hyper[7].item()) == 6  # device step counter
    assert losses[-1] < losses[0], losses


def test_sd21_train_step_gpu(tmp_path):
    """One full-size flagship step must run and produce a finite loss."""
    from dcr_amd.train import TrainConfig, Trainer
    cfg = TrainConfig(model_size="sd21", synthetic_data=True, synthetic_size=4,
                      resolution=256, train_batch_size=2, mixed_precision="bf16",
                      dataloader_num_workers=0, max_train_steps=2, seed=0,
                      output_dir=str(tmp_path / "out"))
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    batch = next(iter(tr.dataloader))
    loss = tr.train_step(batch)
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_sd14_forward_gpu():
    """SD-1.4 family (sd_mitigation's model): head_dim 40/80/160 runs the
    generalized HIP attention kernel (round 2) — no composite
    rocBLAS+softmax fallback on the sampling path."""
    from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                                UNet2DConditionModel, UNetConfig, VAEConfig)
    from dcr_amd.ops import dispatch_counts
    torch.manual_seed(0)
    unet = UNet2DConditionModel(UNetConfig.sd14()).cuda().to(torch.bfloat16)
    te = CLIPTextModel(CLIPTextConfig.sd14()).cuda().to(torch.bfloat16)
    ids = torch.randint(0, 49408, (2, 77), device="cuda")
    gen0 = dispatch_counts["attention_gen"]
    math0 = dispatch_counts["attention_math"]
    with torch.no_grad():
        emb = te(ids)[0]
        out = unet(torch.randn(2, 4, 32, 32, device="cuda").bfloat16(),
                   torch.tensor([10, 500], device="cuda"), emb)
    assert out.shape == (2, 4, 32, 32)
    assert torch.isfinite(out.float()).all()
    assert dispatch_counts["attention_gen"] > gen0, \
        "SD-1.4 attention must dispatch the gen HIP kernel"
    assert dispatch_counts["attention_math"] == math0, \
        "no composite fallback on the SD-1.4 forward"


def test_unet_fwd_hip_matches_cpu_reference():
    """Tiny UNet forward on GPU (HIP kernels) vs CPU (torch fallback)."""
    from dcr_amd.models import UNet2DConditionModel, UNetConfig
    torch.manual_seed(0)
    unet = UNet2DConditionModel(UNetConfig.tiny()).eval()
    x = torch.randn(2, 4, 8, 8)
    t = torch.tensor([17, 503])
    ehs = torch.randn(2, 7, 32)
    with torch.no_grad():
        ref = unet(x, t, ehs)
        gpu = unet.cuda()(x.cuda(), t.cuda(), ehs.cuda())
    assert (gpu.cpu() - ref).abs().max().item() < 1e-3


def test_pipeline_sampling_gpu_channels_last(tmp_path):
    """tiny text2img on GPU: channels_last UNet/VAE + fused sampler kernels."""
    from dcr_amd.data import HashTokenizer
    from dcr_amd.models import (AutoencoderKL, CLIPTextModel, CLIPTextConfig,
                                UNet2DConditionModel, UNetConfig, VAEConfig)
    from dcr_amd.pipelines import StableDiffusionPipeline
    from dcr_amd.schedulers import DDIMScheduler, DPMSolverMultistepScheduler
    torch.manual_seed(0)
    for sched in (DDIMScheduler(), DPMSolverMultistepScheduler()):
        pipe = StableDiffusionPipeline(
            UNet2DConditionModel(UNetConfig.tiny()),
            AutoencoderKL(VAEConfig.tiny()),
            CLIPTextModel(CLIPTextConfig.tiny()),
            HashTokenizer(), sched).to("cuda")
        for m in (pipe.unet, pipe.vae, pipe.text_encoder):
            m.to(torch.bfloat16)
        out = pipe("a church", height=64, width=64, num_inference_steps=4,
                   num_images_per_prompt=2, output_type="pt").images
        assert out.shape == (2, 3, 64, 64)
        assert torch.isfinite(out).all()


def test_channels_last_train_step_matches_nchw(tmp_path):
    """same seed, channels_last vs NCHW: losses should agree closely."""
    from dcr_amd.train import TrainConfig, Trainer

    def run(cl):
        cfg = TrainConfig(model_size="tiny", synthetic_data=True,
                          synthetic_size=4, resolution=64, train_batch_size=2,
                          mixed_precision="bf16", channels_last=cl,
                          dataloader_num_workers=0, max_train_steps=2, seed=3,
                          output_dir=str(tmp_path / f"o{cl}"))
        tr = Trainer(cfg, device=torch.device("cuda", 0))
        b = next(iter(tr.dataloader))
        return tr.train_step(b).item()

    a, b = run(False), run(True)
    # NHWC vs NCHW pick different MIOpen conv algorithms; bf16 rounding
    # differences compound over ~50 layers — same order of magnitude is
    # the correctness bar here (exact parity is covered per-kernel in
    # test_ops_gpu.py::test_groupnorm_nhwc_fwd_bwd)
    assert abs(a - b) / max(abs(a), 1e-6) < 0.25, (a, b)


def test_native_conv_in_unet_shapes():
    """the native conv fwd at the UNet's concatenated up-block widths."""
    from dcr_amd import ops
    m = ops.ext()
    torch.manual_seed(2)
    for (C, K, H) in [(1920, 1280, 16), (2560, 1280, 8), (960, 640, 32),
                      (640, 320, 32)]:
        x = torch.randn(4, C, H, H, device="cuda").to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(K, C, 3, 3, device="cuda") * 0.03).to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        b = torch.randn(K, device="cuda")
        y = m.conv2d_nhwc_fwd_v2(x, w, b, 1, 1)
        ref = torch.nn.functional.conv2d(x.float(), w.float(), b, padding=1)
        err = (y.float() - ref).abs().max().item()
        assert err < 2e-2 * ref.abs().max().item(), (C, K, H, err)


def test_native_conv_train_step(tmp_path):
    """tiny channels_last train step with native conv dispatch active."""
    from dcr_amd.train import TrainConfig, Trainer
    cfg = TrainConfig(model_size="tiny", synthetic_data=True, synthetic_size=4,
                      resolution=64, train_batch_size=2,
                      mixed_precision="pure_bf16", channels_last=True,
                      dataloader_num_workers=0, max_train_steps=2, seed=0,
                      output_dir=str(tmp_path / "o"))
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    b = next(iter(tr.dataloader))
    l1 = tr.train_step(b)
    l2 = tr.train_step(b)
    assert torch.isfinite(l1) and torch.isfinite(l2)


def test_hip_kernels_are_the_executing_path(tmp_path):
    """every hot op of a bench-config train step must dispatch to the
    hand-written HIP kernels (fail-loud against silent eager fallback)."""
    from dcr_amd import ops
    from dcr_amd.train import TrainConfig, Trainer
    ops.dispatch_counts.clear()
    # sd21: head_dim 64 + K%64 channels so EVERY native path is eligible
    # (the tiny config's 32-wide heads legitimately use the math fallback)
    cfg = TrainConfig(model_size="sd21", synthetic_data=True, synthetic_size=2,
                      resolution=256, train_batch_size=1,
                      mixed_precision="pure_bf16", channels_last=True,
                      dataloader_num_workers=0, max_train_steps=1, seed=0,
                      output_dir=str(tmp_path / "o"))
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    tr.train_step(next(iter(tr.dataloader)))
    counts = dict(ops.dispatch_counts)
    for op in ("groupnorm_nhwc", "layernorm", "geglu", "attention",
               "add_noise", "adamw", "conv_nhwc"):
        assert counts.get(op, 0) > 0, (op, counts)


def test_hipgraph_captured_step():
    """Whole-step hipGraph capture: replays draw fresh RNG (losses
    differ), params update every replay, loss stays finite."""
    import os
    from dcr_amd.train import TrainConfig, Trainer
    cfg = TrainConfig(model_size="tiny", synthetic_data=True,
                      synthetic_size=8, resolution=64, train_batch_size=2,
                      mixed_precision="pure_bf16", dataloader_num_workers=0,
                      max_train_steps=8, seed=0, learning_rate=1e-3,
                      lr_warmup_steps=0, channels_last=True,
                      output_dir="/tmp/hipgraph_test_out")
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    batch = next(iter(tr.dataloader))
    tr.train_step(batch)  # eager warmup (MIOpen find)
    tr.enable_hipgraph(batch)
    p0 = tr.optimizer.flat_param.clone()
    losses = []
    for _ in range(4):
        losses.append(float(tr.train_step(batch)))
    assert all(l == l for l in losses), losses            # finite
    assert len(set(losses)) > 1, "replays must draw fresh noise"
    assert not torch.equal(tr.optimizer.flat_param, p0), "params must move"
