"""HIP kernel numerics vs plain-PyTorch fp32 references.

Every kernel is compared against the fp32 reference on random tensors, in
fp32 (tight tolerance) and bf16 (bf16 tolerance), across the SD-2.1
launch shapes (incl. odd sizes: 77-token rows, HW=64, non-wave-multiple
channel counts)."""
import os

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def _close(out, ref, rtol):
    """max-norm comparison relative to the reference's max magnitude —
    bf16 outputs quantize at ~0.8% of magnitude, so absolute tolerances
    are shape-dependent; relative ones are not."""
    scale = ref.abs().max().item() + 1e-6
    err = (out.float() - ref.float()).abs().max().item()
    assert err < rtol * scale, f"err={err:.4g} scale={scale:.4g} rtol={rtol}"


@pytest.fixture(scope="module")
def ext():
    from dcr_amd import ops
    m = ops.ext()
    assert m is not None, "HIP extension must be present on GPU box"
    return m


def _gn_shapes():
    # (N, C, H, W, groups) — SD-2.1 UNet/VAE shapes + odd cases
    return [
        (2, 320, 32, 32, 32),
        (2, 640, 16, 16, 32),
        (4, 1280, 8, 8, 32),
        (1, 512, 32, 32, 32),    # VAE mid
        (2, 128, 64, 64, 32),    # VAE early
        (3, 96, 7, 9, 8),        # odd HW (scalar path), odd N
    ]


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_groupnorm_silu_fwd(ext, dtype):
    torch.manual_seed(0)
    for (N, C, H, W, G) in _gn_shapes():
        x = torch.randn(N, C, H, W, device="cuda", dtype=dtype)
        w = torch.randn(C, device="cuda") * 0.5 + 1
        b = torch.randn(C, device="cuda") * 0.1
        y, mean, rstd = ext.groupnorm_silu_fwd(x, w, b, G, 1e-5, True)
        ref = F.silu(F.group_norm(x.float(), G, w, b, 1e-5))
        _close(y, ref, 1e-5 if dtype == torch.float32 else 1e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_groupnorm_silu_bwd(ext, dtype):
    torch.manual_seed(1)
    for (N, C, H, W, G) in _gn_shapes():
        x = torch.randn(N, C, H, W, device="cuda", dtype=dtype)
        w = (torch.randn(C, device="cuda") * 0.5 + 1).requires_grad_(True)
        b = (torch.randn(C, device="cuda") * 0.1).requires_grad_(True)
        dy = torch.randn(N, C, H, W, device="cuda", dtype=dtype)

        xr = x.float().detach().requires_grad_(True)
        ref = F.silu(F.group_norm(xr, G, w, b, 1e-5))
        ref.backward(dy.float())

        _, mean, rstd = ext.groupnorm_silu_fwd(x, w.detach(), b.detach(), G, 1e-5, True)
        dx, dw, db = ext.groupnorm_silu_bwd(dy.contiguous(), x, w.detach(), b.detach(),
                                            mean, rstd, G, True)
        if dtype == torch.float32:
            atol_x, atol_w = 1e-4, 1e-2
        else:
            atol_x, atol_w = 5e-2, 1.0  # bf16 inputs; dw/db sums over N*HW
        assert (dx.float() - xr.grad).abs().max().item() < atol_x, (N, C, H, W)
        rel_w = (dw.float() - w.grad).abs().max() / (w.grad.abs().max() + 1e-6)
        rel_b = (db.float() - b.grad).abs().max() / (b.grad.abs().max() + 1e-6)
        assert rel_w.item() < (1e-4 if dtype == torch.float32 else 3e-2)
        assert rel_b.item() < (1e-4 if dtype == torch.float32 else 3e-2)


@pytest.mark.parametrize("shape", [(16 * 1024, 320), (4 * 256, 640), (64, 1280),
                                   (32 * 77, 1024), (10, 52)])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layernorm_fwd_bwd(ext, shape, dtype):
    torch.manual_seed(2)
    M, N = shape
    x = torch.randn(M, N, device="cuda", dtype=dtype)
    w = (torch.randn(N, device="cuda") * 0.5 + 1).requires_grad_(True)
    b = torch.randn(N, device="cuda").requires_grad_(True)
    dy = torch.randn(M, N, device="cuda", dtype=dtype)

    y, mean, rstd = ext.layernorm_fwd(x, w.detach(), b.detach(), 1e-5)
    xr = x.float().detach().requires_grad_(True)
    ref = F.layer_norm(xr, (N,), w, b, 1e-5)
    _close(y, ref, 1e-5 if dtype == torch.float32 else 1e-2)

    ref.backward(dy.float())
    dx, dw, db = ext.layernorm_bwd(dy.contiguous(), x, w.detach(), mean, rstd)
    _close(dx, xr.grad, 1e-4 if dtype == torch.float32 else 3e-2)
    rel_w = (dw.float() - w.grad).abs().max() / (w.grad.abs().max() + 1e-6)
    assert rel_w.item() < (1e-4 if dtype == torch.float32 else 3e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_geglu_fwd_bwd(ext, dtype):
    torch.manual_seed(3)
    M, N = 2048, 1280  # inner dim of a 320-ch transformer FF
    x = torch.randn(M, 2 * N, device="cuda", dtype=dtype)
    dy = torch.randn(M, N, device="cuda", dtype=dtype)

    y = ext.geglu_fwd(x)
    a, g = x.float().chunk(2, dim=-1)
    ref = a * F.gelu(g)
    _close(y, ref, 1e-5 if dtype == torch.float32 else 1e-2)

    xr = x.float().detach().requires_grad_(True)
    ar, gr = xr.chunk(2, dim=-1)
    (ar * F.gelu(gr)).backward(dy.float())
    dx = ext.geglu_bwd(dy.contiguous(), x)
    _close(dx, xr.grad, 1e-5 if dtype == torch.float32 else 2e-2)


def test_adamw_matches_torch(ext):
    torch.manual_seed(4)
    n = 1_000_003  # odd size exercises the tail
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")

    pr = p.clone().requires_grad_(True)
    opt = torch.optim.AdamW([pr], lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=1e-2)
    for step in range(1, 4):
        pr.grad = g.clone()
        opt.step()
        ext.adamw_step(p, g, m, v, 1e-3, 0.9, 0.999, 1e-8, 1e-2, step)
    assert (p - pr.detach()).abs().max().item() < 1e-5


def test_add_noise_velocity(ext):
    from dcr_amd.schedulers import DDPMScheduler
    s = DDPMScheduler()
    ac = s.alphas_cumprod.cuda()
    for dtype in (torch.float32, torch.bfloat16):
        x0 = torch.randn(16, 4, 32, 32, device="cuda", dtype=dtype)
        noise = torch.randn_like(x0)
        t = torch.randint(0, 1000, (16,), device="cuda")
        xt = ext.add_noise(x0, noise, ac, t)
        acv = ac[t].view(-1, 1, 1, 1)
        ref = acv.sqrt() * x0.float() + (1 - acv).sqrt() * noise.float()
        _close(xt, ref, 1e-5 if dtype == torch.float32 else 1e-2)
        v = ext.get_velocity(x0, noise, ac, t)
        refv = acv.sqrt() * noise.float() - (1 - acv).sqrt() * x0.float()
        _close(v, refv, 1e-5 if dtype == torch.float32 else 1e-2)


def test_cfg_combine(ext):
    eu = torch.randn(4, 4, 32, 32, device="cuda", dtype=torch.bfloat16)
    et = torch.randn_like(eu)
    out = ext.cfg_combine(eu, et, 7.5)
    ref = eu.float() + 7.5 * (et.float() - eu.float())
    _close(out, ref, 1e-2)


def test_ops_dispatch_uses_hip_on_gpu():
    """group_norm_silu on CUDA must route through the extension (fail-loud)."""
    from dcr_amd import ops
    x = torch.randn(2, 64, 16, 16, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.ones(64, device="cuda")
    b = torch.zeros(64, device="cuda")
    y = ops.group_norm_silu(x, w, b, 8, 1e-5, True)
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


# ---------------------------------------------------------------- attention
def _attn_ref(q, k, v, scale, causal):
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    if causal:
        Lq, Lk = q.shape[-2], k.shape[-2]
        mask = torch.ones(Lq, Lk, dtype=torch.bool, device=q.device).tril_()
        s = s.masked_fill(~mask, float("-inf"))
    p = s.softmax(dim=-1)
    return p @ v.float()


def test_mfma_probe_layout(ext):
    """Pins the assumed gfx950 16x16x32 bf16 A/B/C fragment layouts."""
    torch.manual_seed(0)
    A = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    B = torch.randn(32, 16, device="cuda").to(torch.bfloat16)  # asymmetric
    C = ext.mfma_probe(A, B)
    ref = A.float() @ B.float()
    err = (C - ref).abs().max().item()
    assert err < 0.1 * ref.abs().max().item(), f"mfma layout wrong: err={err}"


@pytest.mark.parametrize("Lq,Lk,causal", [
    (1024, 1024, False),   # SD self-attn @ 32x32 latents
    (256, 256, False),
    (64, 64, False),
    (1024, 77, False),     # cross-attn, odd kv length
    (77, 77, True),        # CLIP text, causal, odd length
    (4096, 4096, False),   # 512px latents
    (100, 200, False),     # both odd
])
def test_flash_attention_fwd(ext, Lq, Lk, causal):
    torch.manual_seed(0)
    B, H, D = 2, 5, 64
    q = torch.randn(B, H, Lq, D, device="cuda").to(torch.bfloat16)
    k = torch.randn(B, H, Lk, D, device="cuda").to(torch.bfloat16)
    v = torch.randn(B, H, Lk, D, device="cuda").to(torch.bfloat16)
    scale = 1.0 / D ** 0.5
    o, lse = ext.attn_fwd(q.reshape(B * H, Lq, D), k.reshape(B * H, Lk, D),
                          v.reshape(B * H, Lk, D), scale, causal)
    ref = _attn_ref(q, k, v, scale, causal).reshape(B * H, Lq, D)
    _close(o, ref, 2e-2)
    # LSE check (non-masked rows)
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    if causal:
        mask = torch.ones(Lq, Lk, dtype=torch.bool, device=q.device).tril_()
        s = s.masked_fill(~mask, float("-inf"))
    lse_ref = s.logsumexp(dim=-1).reshape(B * H, Lq)
    assert (lse - lse_ref).abs().max().item() < 5e-2


@pytest.mark.parametrize("Lq,Lk,causal", [
    (256, 256, False),
    (1024, 77, False),
    (77, 77, True),
    (100, 200, False),
])
def test_flash_attention_bwd(ext, Lq, Lk, causal):
    torch.manual_seed(1)
    B, H, D = 1, 4, 64
    qf = torch.randn(B * H, Lq, D, device="cuda")
    kf = torch.randn(B * H, Lk, D, device="cuda")
    vf = torch.randn(B * H, Lk, D, device="cuda")
    dO = torch.randn(B * H, Lq, D, device="cuda")
    scale = 1.0 / D ** 0.5

    qr = qf.clone().requires_grad_(True)
    kr = kf.clone().requires_grad_(True)
    vr = vf.clone().requires_grad_(True)
    _attn_ref(qr, kr, vr, scale, causal).backward(dO)

    q = qf.to(torch.bfloat16)
    k = kf.to(torch.bfloat16)
    v = vf.to(torch.bfloat16)
    o, lse = ext.attn_fwd(q, k, v, scale, causal)
    dQ, dK, dV = ext.attn_bwd(q, k, v, o, dO.to(torch.bfloat16), lse,
                              scale, causal)
    _close(dQ, qr.grad, 5e-2)
    _close(dK, kr.grad, 5e-2)
    _close(dV, vr.grad, 5e-2)


def test_flash_attention_autograd_wrapper(ext):
    from dcr_amd import ops
    torch.manual_seed(2)
    q = torch.randn(2, 5, 256, 64, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(2, 5, 77, 64, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn_like(k, requires_grad=True)
    out = ops.attention(q, k, v)
    out.sum().backward()
    for g in (q.grad, k.grad, v.grad):
        assert g is not None and torch.isfinite(g.float()).all()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_groupnorm_nhwc_fwd_bwd(ext, dtype):
    """channels_last GroupNorm vs the fp32 reference."""
    torch.manual_seed(5)
    for (N, C, H, W, G) in [(2, 320, 32, 32, 32), (4, 1280, 8, 8, 32),
                            (2, 128, 64, 64, 32), (1, 512, 9, 7, 32)]:
        x = torch.randn(N, C, H, W, device="cuda", dtype=dtype) \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(C, device="cuda") * 0.5 + 1).requires_grad_(True)
        b = (torch.randn(C, device="cuda") * 0.1).requires_grad_(True)
        dy = torch.randn(N, C, H, W, device="cuda", dtype=dtype) \
            .to(memory_format=torch.channels_last)

        y, mean, rstd = ext.groupnorm_silu_nhwc_fwd(x, w.detach(), b.detach(),
                                                    G, 1e-5, True)
        xr = x.float().detach().requires_grad_(True)
        ref = F.silu(F.group_norm(xr, G, w, b, 1e-5))
        _close(y, ref, 1e-4 if dtype == torch.float32 else 1e-2)

        ref.backward(dy.float())
        dx, dw, db = ext.groupnorm_silu_nhwc_bwd(dy, x, w.detach(), b.detach(),
                                                 mean, rstd, G, True)
        _close(dx, xr.grad, 2e-3 if dtype == torch.float32 else 4e-2)
        rel_w = (dw.float() - w.grad).abs().max() / (w.grad.abs().max() + 1e-6)
        assert rel_w.item() < (1e-3 if dtype == torch.float32 else 3e-2)


def test_group_norm_dispatch_channels_last(ext):
    """ops.group_norm_silu must route channels_last input through the
    NHWC kernels and return a channels_last tensor."""
    from dcr_amd import ops
    x = torch.randn(2, 64, 16, 16, device="cuda", dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    w = torch.ones(64, device="cuda")
    b = torch.zeros(64, device="cuda")
    y = ops.group_norm_silu(x, w, b, 8, 1e-5, True)
    assert y.is_contiguous(memory_format=torch.channels_last)
    y.sum().backward()
    assert torch.isfinite(x.grad.float()).all()


def test_lincomb(ext):
    torch.manual_seed(6)
    x = torch.randn(4, 4, 32, 32, device="cuda", dtype=torch.bfloat16)
    y = torch.randn_like(x)
    z = torch.randn_like(x)
    out = ext.lincomb(x, y, z, 0.5, -1.25, 2.0)
    ref = 0.5 * x.float() - 1.25 * y.float() + 2.0 * z.float()
    _close(out, ref, 1e-2)
    out2 = ext.lincomb(x, y, None, 2.0, 3.0, 0.0)
    _close(out2, 2 * x.float() + 3 * y.float(), 1e-2)


def test_ddim_step_gpu_matches_cpu(ext):
    """GPU fused lincomb step vs the CPU float reference path."""
    from dcr_amd.schedulers import DDIMScheduler
    for pred in ("epsilon", "v_prediction"):
        s_gpu = DDIMScheduler(prediction_type=pred)
        s_cpu = DDIMScheduler(prediction_type=pred)
        s_gpu.set_timesteps(10)
        s_cpu.set_timesteps(10)
        torch.manual_seed(0)
        x = torch.randn(2, 4, 16, 16)
        eps = torch.randn_like(x) * 0.3
        t = int(s_gpu.timesteps[3])
        out_cpu = s_cpu.step(eps, t, x).prev_sample
        out_gpu = s_gpu.step(eps.cuda().bfloat16(), t, x.cuda().bfloat16()).prev_sample
        _close(out_gpu.cpu(), out_cpu, 1.5e-2)


def test_dpm_step_gpu_matches_cpu(ext):
    from dcr_amd.schedulers import DPMSolverMultistepScheduler
    s_gpu = DPMSolverMultistepScheduler()
    s_cpu = DPMSolverMultistepScheduler()
    s_gpu.set_timesteps(8)
    s_cpu.set_timesteps(8)
    torch.manual_seed(1)
    x_c = torch.randn(1, 4, 16, 16)
    x_g = x_c.cuda().bfloat16()
    for i, t in enumerate(s_cpu.timesteps):
        eps = torch.randn_like(x_c) * 0.2
        x_c = s_cpu.step(eps, int(t), x_c).prev_sample
        x_g = s_gpu.step(eps.cuda().bfloat16(), int(t), x_g).prev_sample
    err = (x_g.float().cpu() - x_c.float()).abs().max()
    scale = x_c.abs().max() + 1e-6
    assert err / scale < 0.08, (err, scale)  # bf16 accumulation over 8 steps


# ------------------------------------------------- implicit-GEMM conv (opt-in)
@pytest.mark.parametrize("shape", [
    (2, 320, 32, 32, 320, 3, 1),    # ResNet conv, stride 1
    (2, 320, 32, 32, 640, 1, 1),    # 1x1 shortcut
    (2, 640, 16, 16, 640, 3, 2),    # downsample
    (1, 128, 64, 64, 128, 3, 1),    # VAE early
    (2, 1280, 8, 8, 1280, 3, 1),
    (2, 96, 16, 16, 64, 3, 1),      # C % 64 != 0 -> BK=32 path
])
@pytest.mark.parametrize("version", ["v1", "v2"])
def test_conv_nhwc_fwd(ext, shape, version):
    N, C, H, W, K, R, stride = shape
    pad = 1 if R == 3 else 0
    torch.manual_seed(0)
    x = torch.randn(N, C, H, W, device="cuda").to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    w = (torch.randn(K, C, R, R, device="cuda") * 0.05).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    b = torch.randn(K, device="cuda")
    fn = ext.conv2d_nhwc_fwd if version == "v1" else ext.conv2d_nhwc_fwd_v2
    y = fn(x, w, b, stride, pad)
    ref = torch.nn.functional.conv2d(x.float(), w.float(), b, stride=stride,
                                     padding=pad)
    assert y.is_contiguous(memory_format=torch.channels_last)
    _close(y, ref, 2e-2)


@pytest.mark.skipif(os.environ.get("DCR_ATTN_V2") != "1",
                    reason="attn v2 masked-tail-skip draft: validate in "
                           "round 2 (DCR_ATTN_V2=1)")
@pytest.mark.parametrize("shape", [
    (2, 5, 1024, 77, False),    # SD cross-attn: tail tile has 13 valid keys
    (2, 4, 256, 256, False),
    (2, 16, 77, 77, True),      # CLIP causal
    (1, 2, 100, 37, False),     # odd everything, Lk < 64
    (2, 5, 4096, 4096, False),
])
def test_attn_fwd_v2_bitexact_vs_v1(ext, shape):
    """The skipped MFMAs only ever accumulate exact zeros, so v2 must be
    BIT-IDENTICAL to the production kernel."""
    B, H, Lq, Lk, causal = shape
    torch.manual_seed(0)
    D = 64
    q = torch.randn(B, Lq, H, D, device="cuda").to(torch.bfloat16)
    k = torch.randn(B, Lk, H, D, device="cuda").to(torch.bfloat16)
    v = torch.randn(B, Lk, H, D, device="cuda").to(torch.bfloat16)
    scale = D ** -0.5
    o1, l1 = ext.attn_fwd(q, k, v, scale, causal)
    o2, l2 = ext.attn_fwd_v2(q, k, v, scale, causal)
    assert torch.equal(o1, o2)
    assert torch.equal(l1, l2)


@pytest.mark.skipif(os.environ.get("DCR_DEV_ADAMW") != "1",
                    reason="device-state AdamW draft: validate in round 2 "
                           "(DCR_DEV_ADAMW=1)")
@pytest.mark.parametrize("bf16", [False, True])
def test_adamw_dev_matches_host(ext, bf16):
    """3-kernel device-state step == host-scalar clip+step over 6 steps,
    including one step where the clip triggers."""
    n = 10_001
    torch.manual_seed(0)
    lr, b1, b2, eps, wd, mx = 1e-3, 0.9, 0.999, 1e-8, 1e-2, 1.0
    master1 = torch.randn(n, device="cuda")
    master2 = master1.clone()
    if bf16:
        p1 = master1.bfloat16()
        p2 = master2.bfloat16()
    else:
        p1, p2 = master1, master2
    m1 = torch.zeros(n, device="cuda"); v1 = torch.zeros(n, device="cuda")
    m2 = m1.clone(); v2 = v1.clone()
    hyper = torch.tensor([lr, 1, 1, 1, 1, 1, 0, 0], device="cuda")
    for step in range(1, 7):
        g = torch.randn(n, device="cuda") * (10.0 if step == 3 else 0.01)
        gg = g.bfloat16() if bf16 else g
        # host path: explicit clip, then the host-scalar kernel
        total = gg.float().norm()
        coef = torch.clamp(mx / (total + 1e-6), max=1.0)
        gh = (gg.float() * coef).to(gg.dtype)
        if bf16:
            ext.adamw_step_bf16(p1, gh, master1, m1, v1, lr, b1, b2, eps, wd, step)
            ext.adamw_step_dev(p2, gg, master2, m2, v2, hyper, b1, b2, eps, wd, mx)
        else:
            ext.adamw_step(p1, gh, m1, v1, lr, b1, b2, eps, wd, step)
            ext.adamw_step_dev(p2, gg, None, m2, v2, hyper, b1, b2, eps, wd, mx)
    assert int(hyper[7].item()) == 6
    # atomic-order norm + bf16 clip rounding leave tiny differences
    tol = 2e-2 if bf16 else 1e-4
    _close(master2, master1, tol)
    _close(m2, m1, 2e-2 if bf16 else 1e-4)


@pytest.mark.skipif(os.environ.get("DCR_DEV_ADAMW") != "1",
                    reason="device-state AdamW draft: validate in round 2 "
                           "(DCR_DEV_ADAMW=1)")
def test_adamw_dev_graph_capture(ext):
    """The whole optimizer step replays from a hipGraph with zero host work;
    state (step counter, bias-correction powers) advances on device."""
    n = 4096
    torch.manual_seed(0)
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda") * 0.01
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    hyper = torch.tensor([1e-3, 1, 1, 1, 1, 1, 0, 0], device="cuda")
    args = (p, g, None, m, v, hyper, 0.9, 0.999, 1e-8, 1e-2, 1.0)
    # warmup on a side stream (capture requirement), then capture one step
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        ext.adamw_step_dev(*args)
    torch.cuda.current_stream().wait_stream(s)
    steps_before = int(hyper[7].item())
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        ext.adamw_step_dev(*args)
    for _ in range(3):
        graph.replay()
    torch.cuda.synchronize()
    # warmup + capture + 3 replays all advanced the device step counter
    assert int(hyper[7].item()) >= steps_before + 3
    assert torch.isfinite(p).all()


@pytest.mark.skipif(os.environ.get("DCR_NATIVE_CONV_BWD") != "1",
                    reason="conv bwd drafts: validate in round 2 "
                           "(DCR_NATIVE_CONV_BWD=1)")
@pytest.mark.parametrize("shape", [
    (2, 320, 32, 32, 320, 3, 1),
    (2, 640, 16, 16, 640, 3, 1),
    (2, 128, 64, 64, 128, 1, 1),
    (2, 640, 32, 32, 640, 3, 2),
])
def test_conv_nhwc_bwd(ext, shape):
    N, C, H, W, K, R, stride = shape
    pad = 1 if R == 3 else 0
    torch.manual_seed(0)
    x = torch.randn(N, C, H, W, device="cuda").to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    w = (torch.randn(K, C, R, R, device="cuda") * 0.05).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    P = (H + 2 * pad - R) // stride + 1
    dy = torch.randn(N, K, P, P, device="cuda").to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    dx, dW, db = ext.conv2d_nhwc_bwd(dy, x, w, stride, pad)
    rx, rw, rb = torch.ops.aten.convolution_backward(
        dy.float(), x.float(), w.float(), [w.shape[0]], [stride, stride],
        [pad, pad], [1, 1], False, [0, 0], 1, [True, True, True])
    _close(dx, rx, 3e-2)
    _close(dW, rw, 3e-2)
    _close(db, rb, 1e-2)


# ---------------------------------------------------------------------------
# MFMA GEMM for the transformer linears (gemm.hip, round 2)
# ---------------------------------------------------------------------------
def _gemm_shapes():
    # (M, N, K) — SD-2.1 linear shapes at bs16 + ragged/odd cases
    return [
        (16384, 320, 320),    # res32 qkv/out
        (16384, 2560, 320),   # res32 GEGLU proj
        (16384, 320, 1280),   # res32 ff.net.2
        (4096, 640, 1024),    # res16 cross k/v proj (ctx side is M=1232)
        (1024, 10240, 1280),  # res8 GEGLU proj
        (1232, 1280, 1024),   # cross-attn k/v: M = 16*77 (ragged M)
        (512, 192, 136),      # ragged K (%8 only) + small N
    ]


def test_gemm_bf16_fwd(ext):
    torch.manual_seed(11)
    for (M, N, K) in _gemm_shapes():
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
        b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        (y,) = ext.gemm_bf16(x, w, b, False, False, False)
        ref = F.linear(x.float(), w.float(), b.float())
        _close(y, ref, 2e-2)
        (y2,) = ext.gemm_bf16(x, w, None, False, False, False)
        _close(y2, ref - b.float()[None, :], 2e-2)


def test_gemm_bf16_dgrad(ext):
    torch.manual_seed(12)
    for (M, N, K) in _gemm_shapes():
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
        (dx,) = ext.gemm_bf16(dy, w, None, False, True, False)
        ref = dy.float() @ w.float()
        _close(dx, ref, 2e-2)


def test_gemm_bf16_wgrad_dbias(ext):
    torch.manual_seed(13)
    for (M, N, K) in _gemm_shapes():
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.5
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.1
        dw, db = ext.gemm_bf16(dy, x, None, True, True, True)
        ref_dw = dy.float().t() @ x.float()
        _close(dw, ref_dw, 2e-2)
        ref_db = dy.float().sum(0)
        _close(db, ref_db, 2e-2)
        # split-K over the contraction uses fp32 atomics, so two runs are
        # equal only to atomics-reordering rounding (like rocBLAS/MIOpen)
        (dw2,) = ext.gemm_bf16(dy, x, None, True, True, False)
        _close(dw2, ref_dw, 2e-2)


def test_dcr_linear_autograd_matches_f_linear(ext, monkeypatch):
    """Full fwd+bwd of the hybrid wrapper vs torch fp32 on a UNet shape:
    rocBLAS fwd/dgrad + native MFMA wgrad with fused bias-grad
    (opt-in: whole-model profile showed a net loss)."""
    monkeypatch.setenv("DCR_NATIVE_GEMM", "1")
    from dcr_amd.ops.linear import dcr_linear
    from dcr_amd.ops import dispatch_counts
    torch.manual_seed(14)
    M, N, K = 4096, 640, 320
    x = (torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.5
         ).requires_grad_(True)
    w = (torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
         ).requires_grad_(True)
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16).requires_grad_(True)
    before = dispatch_counts["gemm_wgrad"]
    y = dcr_linear(x, w, b)
    g = torch.randn_like(y)
    y.backward(g)
    assert dispatch_counts["gemm_wgrad"] == before + 1, \
        "native wgrad must run in its win regime"

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yf = F.linear(xf, wf, bf)
    yf.backward(g.float())

    _close(y, yf, 2e-2)
    _close(x.grad, xf.grad, 2e-2)
    _close(w.grad, wf.grad, 2e-2)
    _close(b.grad, bf.grad, 2e-2)


# ---------------------------------------------------------------------------
# Generalized head-dim attention forward (SD-1.4 40/80/160, round 2)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("D", [40, 80, 160])
@pytest.mark.parametrize("Lq,Lk", [(64, 64), (256, 77), (1024, 1024), (100, 33)])
def test_attn_fwd_gen(ext, D, Lq, Lk):
    torch.manual_seed(D + Lq)
    B, H = 2, 3
    q = torch.randn(B, Lq, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / D ** 0.5
    o = ext.attn_fwd_gen(q, k, v, scale, False)
    ref = F.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3).float(), k.permute(0, 2, 1, 3).float(),
        v.permute(0, 2, 1, 3).float(), scale=scale).permute(0, 2, 1, 3)
    _close(o, ref, 2e-2)


def test_attn_fwd_gen_causal(ext):
    torch.manual_seed(5)
    B, H, L, D = 2, 4, 128, 80
    q = torch.randn(B, L, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, L, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, L, H, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / D ** 0.5
    o = ext.attn_fwd_gen(q, k, v, scale, True)
    ref = F.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3).float(), k.permute(0, 2, 1, 3).float(),
        v.permute(0, 2, 1, 3).float(), is_causal=True,
        scale=scale).permute(0, 2, 1, 3)
    _close(o, ref, 2e-2)


def test_attention_dispatch_sd14_head_dims(ext):
    """ops.attention routes SD-1.4 inference shapes through the gen
    kernel, not the composite rocBLAS+softmax fallback."""
    from dcr_amd import ops
    from dcr_amd.ops import dispatch_counts
    before = dispatch_counts["attention_gen"]
    with torch.no_grad():
        q = torch.randn(2, 256, 8, 40, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(2, 77, 8, 40, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(2, 77, 8, 40, device="cuda", dtype=torch.bfloat16)
        out = ops.attention(q, k, v, layout="blhd")
    assert out.shape == q.shape
    assert dispatch_counts["attention_gen"] == before + 1


def test_conv_fused_res_temb_epilogue(ext):
    """conv v2 epilogue fusions: residual and [N,K] temb adds match the
    unfused ops bit-for-bit-ish (fp32 epilogue adds, bf16 store)."""
    torch.manual_seed(21)
    N, C, H, K = 4, 64, 16, 64
    x = torch.randn(N, C, H, H, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    w = (torch.randn(K, C, 3, 3, device="cuda") * 0.05).bfloat16() \
        .to(memory_format=torch.channels_last)
    b = torch.randn(K, device="cuda")
    res = torch.randn(N, K, H, H, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    temb = torch.randn(N, K, device="cuda").bfloat16()
    y = ext.conv2d_nhwc_fwd_v2(x, w, b, 1, 1, res, temb)
    base = F.conv2d(x.float(), w.float(), b, stride=1, padding=1)
    ref = base + res.float() + temb.float()[:, :, None, None]
    _close(y, ref, 2e-2)
    # no-extras call unchanged
    y0 = ext.conv2d_nhwc_fwd_v2(x, w, b, 1, 1)
    _close(y0, base, 2e-2)


def test_resnet_block_fused_matches_unfused(ext):
    """ResnetBlock2D on GPU (fused epilogues) vs the same block with
    DCR_NATIVE_CONV=0 (torch conv + separate adds)."""
    import os
    from dcr_amd.models.resnet import ResnetBlock2D
    torch.manual_seed(3)
    blk = ResnetBlock2D(64, 128, temb_channels=32).cuda().bfloat16() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(2, 64, 16, 16, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    temb = torch.randn(2, 32, device="cuda").bfloat16()
    y_fused = blk(x, temb)
    os.environ["DCR_NATIVE_CONV"] = "0"
    try:
        y_ref = blk(x, temb)
    finally:
        os.environ.pop("DCR_NATIVE_CONV", None)
    _close(y_fused, y_ref.float(), 2e-2)

    # gradients flow through both fused adds
    x.requires_grad_(False)
    xg = x.detach().clone().requires_grad_(True)
    tg = temb.detach().clone().requires_grad_(True)
    out = blk(xg, tg)
    out.float().pow(2).mean().backward()
    assert xg.grad is not None and torch.isfinite(xg.grad.float()).all()
    assert tg.grad is not None and torch.isfinite(tg.grad.float()).all()
    assert blk.time_emb_proj.weight.grad is not None


@pytest.mark.parametrize("Lq,Lk", [(64, 64), (256, 77), (1024, 1024),
                                   (100, 33)])
def test_attn_fwd_v3_matches_v1(ext, Lq, Lk):
    """v3 (async-stage + 1-barrier + setprio draft) computes the same
    math as v1 in the same order -> bit-equal outputs."""
    torch.manual_seed(Lq + Lk)
    B, H, D = 2, 3, 64
    q = torch.randn(B, Lq, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    o1, l1 = ext.attn_fwd(q, k, v, 0.125, False)
    o3, l3 = ext.attn_fwd_v3(q, k, v, 0.125, False)
    assert torch.equal(o1, o3)
    assert torch.equal(l1, l3)
    oc1, lc1 = ext.attn_fwd(q, k, v[:, :Lk], 0.125, True)
    oc3, lc3 = ext.attn_fwd_v3(q, k, v[:, :Lk], 0.125, True)
    assert torch.equal(oc1, oc3)


@pytest.mark.parametrize("Lq,Lk,causal", [(256, 256, False), (1024, 1024, False),
                                          (256, 77, False), (300, 100, False),
                                          (512, 512, True)])
def test_attn_fwd_v4_matches_ref(ext, Lq, Lk, causal):
    """v4 (swapped-QK^T in-register softmax, 32x32 MFMA): vs fp32 SDPA
    (random asymmetric data catches any fragment-layout transpose) and
    LSE vs v1."""
    torch.manual_seed(Lq + Lk)
    B, H, D = 2, 3, 64
    q = torch.randn(B, Lq, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / D ** 0.5
    o4, l4 = ext.attn_fwd_v4(q, k, v, scale, causal)
    ref = F.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3).float(), k.permute(0, 2, 1, 3).float(),
        v.permute(0, 2, 1, 3).float(), is_causal=causal,
        scale=scale).permute(0, 2, 1, 3)
    _close(o4, ref, 2e-2)
    o1, l1 = ext.attn_fwd(q, k, v, scale, causal)
    assert (l4 - l1).abs().max().item() < 1e-4


@pytest.mark.parametrize("Lq,Lk,causal", [(256, 256, False), (1024, 1024, False),
                                          (256, 77, False), (300, 100, False),
                                          (512, 512, True)])
def test_attn_bwd_v4_matches_ref(ext, Lq, Lk, causal):
    """v4 backward (swapped-operand, register-resident P/dS) vs torch
    fp32 autograd on random data."""
    torch.manual_seed(Lq * 3 + Lk)
    B, H, D = 2, 3, 64
    q = torch.randn(B, Lq, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Lk, H, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(B, Lq, H, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / D ** 0.5
    o, lse = ext.attn_fwd(q, k, v, scale, causal)
    dQ, dK, dV = ext.attn_bwd_v4(q, k, v, o, g, lse, scale, causal)

    qf = q.permute(0, 2, 1, 3).float().requires_grad_(True)
    kf = k.permute(0, 2, 1, 3).float().requires_grad_(True)
    vf = v.permute(0, 2, 1, 3).float().requires_grad_(True)
    of = F.scaled_dot_product_attention(qf, kf, vf, is_causal=causal,
                                        scale=scale)
    of.backward(g.permute(0, 2, 1, 3).float())
    _close(dQ, qf.grad.permute(0, 2, 1, 3), 3e-2)
    _close(dK, kf.grad.permute(0, 2, 1, 3), 3e-2)
    _close(dV, vf.grad.permute(0, 2, 1, 3), 3e-2)

    # and against the production v1 backward (same inputs/lse)
    dQ1, dK1, dV1 = ext.attn_bwd(q, k, v, o, g, lse, scale, causal)
    _close(dQ, dQ1.float(), 2e-2)
    _close(dK, dK1.float(), 2e-2)
    _close(dV, dV1.float(), 2e-2)
