"""CPU surface checks for the ops package (catch missing exports the GPU
fast paths depend on)."""
import torch

from dcr_amd import ops


def test_ops_exports():
    for name in ("group_norm_silu", "layer_norm", "geglu", "attention",
                 "add_noise", "get_velocity", "cfg_combine", "lincomb"):
        assert hasattr(ops, name), name


def test_lincomb_cpu_fallback():
    x = torch.randn(4, 8)
    y = torch.randn(4, 8)
    z = torch.randn(4, 8)
    out = ops.lincomb(x, y, 2.0, -1.0, z, 0.5)
    assert torch.allclose(out, 2 * x - y + 0.5 * z, atol=1e-6)


def test_attention_blhd_cpu():
    torch.manual_seed(0)
    q = torch.randn(2, 10, 3, 16)
    k = torch.randn(2, 12, 3, 16)
    v = torch.randn(2, 12, 3, 16)
    out = ops.attention(q, k, v, layout="blhd")
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3), v.permute(0, 2, 1, 3)
    ).permute(0, 2, 1, 3)
    assert torch.allclose(out, ref, atol=1e-5)


def test_native_conv_module_cpu_passthrough():
    """ops.conv.Conv2d == nn.Conv2d on CPU (not eligible there), same keys."""
    from dcr_amd.ops.conv import Conv2d
    torch.manual_seed(0)
    m = Conv2d(8, 16, 3, padding=1)
    ref = torch.nn.Conv2d(8, 16, 3, padding=1)
    ref.load_state_dict(m.state_dict())  # identical parameter surface
    x = torch.randn(2, 8, 6, 6, requires_grad=True)
    y = m(x)
    assert torch.allclose(y, ref(x), atol=1e-6)
    y.sum().backward()
    assert x.grad is not None


def test_aten_convolution_backward_contract():
    """the backward op the native conv path relies on matches autograd."""
    torch.manual_seed(1)
    x = torch.randn(2, 8, 6, 6, requires_grad=True)
    w = torch.randn(16, 8, 3, 3, requires_grad=True)
    b = torch.randn(16, requires_grad=True)
    y = torch.nn.functional.conv2d(x, w, b, stride=1, padding=1)
    dy = torch.randn_like(y)
    y.backward(dy)
    dx, dw, db = torch.ops.aten.convolution_backward(
        dy, x.detach(), w.detach(), [16], [1, 1], [1, 1], [1, 1], False,
        [0, 0], 1, [True, True, True])
    assert torch.allclose(dx, x.grad, atol=1e-5)
    assert torch.allclose(dw, w.grad, atol=1e-4)
    assert torch.allclose(db, b.grad, atol=1e-5)


def test_fused_adamw_cpu_matches_torch_adamw():
    """the CPU fallback math (which anchors resume + GPU-kernel tests)
    must match torch.optim.AdamW step-for-step."""
    torch.manual_seed(0)
    lin_a = torch.nn.Linear(16, 8)
    lin_b = torch.nn.Linear(16, 8)
    lin_b.load_state_dict(lin_a.state_dict())

    from dcr_amd.ops.adamw import FusedAdamW
    ours = FusedAdamW(lin_a.parameters(), lr=1e-3, betas=(0.9, 0.999),
                      eps=1e-8, weight_decay=1e-2)
    ref = torch.optim.AdamW(lin_b.parameters(), lr=1e-3, betas=(0.9, 0.999),
                            eps=1e-8, weight_decay=1e-2)
    for step in range(5):
        torch.manual_seed(100 + step)
        x = torch.randn(4, 16)
        la = lin_a(x).pow(2).mean()
        lb = lin_b(x).pow(2).mean()
        la.backward()
        lb.backward()
        ours.step()
        ours.zero_grad()
        ref.step()
        ref.zero_grad()
    for pa, pb in zip(lin_a.parameters(), lin_b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), (pa - pb).abs().max()


def test_clip_grad_norm_matches_torch():
    from dcr_amd.ops.adamw import FusedAdamW
    torch.manual_seed(1)
    lin = torch.nn.Linear(32, 32)
    opt = FusedAdamW(lin.parameters(), lr=1e-3)
    lin(torch.randn(8, 32)).pow(2).sum().backward()
    grads = [p.grad.clone() for p in lin.parameters()]
    total = torch.sqrt(sum(g.pow(2).sum() for g in grads))
    norm = opt.clip_grad_norm_(0.5)
    assert torch.allclose(norm, total, atol=1e-5)
    clipped = torch.linalg.vector_norm(opt.flat_grad)
    assert clipped <= 0.5 * (1 + 1e-4)


def test_dcr_linear_cpu_fallback_and_state_dict():
    """DcrLinear == nn.Linear on CPU (F.linear fallback) and state-dict
    compatible with nn.Linear (diffusers checkpoint naming)."""
    import torch
    import torch.nn as nn
    from dcr_amd.ops.linear import DcrLinear, dcr_linear

    torch.manual_seed(0)
    ref = nn.Linear(32, 48)
    mod = DcrLinear(32, 48)
    mod.load_state_dict(ref.state_dict())
    x = torch.randn(5, 7, 32)
    assert torch.equal(mod(x), ref(x))
    assert set(mod.state_dict()) == {"weight", "bias"}
    # functional fallback identical to F.linear on CPU
    y = dcr_linear(x, ref.weight, ref.bias)
    assert torch.equal(y, ref(x))


def test_fused_adamw_preserves_channels_last():
    """Flattening must keep channels_last conv weights channels_last —
    a plain view_as silently reverted them to NCHW, costing a per-call
    aten weight relayout and disabling the native conv for every trained
    module (found via scripts/profile_aten.py, round 2)."""
    import torch
    from dcr_amd.ops.adamw import FusedAdamW

    conv = torch.nn.Conv2d(8, 16, 3, padding=1)
    conv.to(memory_format=torch.channels_last)
    w0 = conv.weight.detach().clone()
    opt = FusedAdamW(conv.parameters(), lr=1e-2)
    assert conv.weight.is_contiguous(memory_format=torch.channels_last)
    assert conv.weight.data_ptr() == opt.flat_param.data_ptr()  # still a view
    assert torch.equal(conv.weight, w0)

    x = torch.randn(2, 8, 4, 4).to(memory_format=torch.channels_last)
    conv(x).pow(2).mean().backward()
    opt.step()
    assert conv.weight.is_contiguous(memory_format=torch.channels_last)
    assert not torch.equal(conv.weight, w0)

    # gathered grads land in the right arena slots (layout-aligned views)
    opt.zero_grad()
    conv(x).pow(2).mean().backward()
    g_auto = conv.weight.grad.clone()
    opt.gather_grads()
    assert torch.allclose(opt._grad_views[0], g_auto, atol=1e-7)


def test_fused_adamw_zero_skip_and_stale_slice_guard():
    """zero_grad skips the arena zero after a full-coverage cycle; a
    later partial-coverage cycle must still read 0 (not last cycle's
    grad) for params that produced no grad."""
    import torch
    from dcr_amd.ops.adamw import FusedAdamW

    a = torch.nn.Linear(4, 4)
    b = torch.nn.Linear(4, 4)
    opt = FusedAdamW(list(a.parameters()) + list(b.parameters()), lr=0.0,
                     weight_decay=0.0)
    x = torch.randn(2, 4)
    # cycle 1: both modules produce grads
    (a(x).sum() + b(x).sum()).backward()
    opt.step()
    opt.zero_grad()
    assert opt._skipped_zero  # full coverage -> skip
    # cycle 2: only module a runs
    a(x).pow(2).sum().backward()
    opt.gather_grads()
    opt._ensure_cold_slices()
    for p in b.parameters():
        v = opt._view_of[id(p)]
        assert torch.count_nonzero(v) == 0, "stale grads must be zeroed"
