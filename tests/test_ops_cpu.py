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
