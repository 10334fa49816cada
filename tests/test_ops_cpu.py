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
