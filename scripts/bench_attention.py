#!/usr/bin/env python3
"""A/B the hand-written flash attention vs torch SDPA at SD shapes.

Round-2 tuning baseline (guide techniques to apply: 8-phase interleave,
setprio, K/V register staging). Run on an MI355X box.
"""
from __future__ import annotations

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from dcr_amd import ops

SHAPES = [  # (B, H, Lq, Lk, causal) — SD-2.1 bs16 256px + 512px + CLIP
    (16, 5, 1024, 1024, False),
    (16, 10, 256, 256, False),
    (16, 20, 64, 64, False),
    (16, 5, 1024, 77, False),
    (16, 5, 4096, 4096, False),   # 512px latents
    (16, 16, 77, 77, True),       # CLIP text
]


def timeit(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    m = ops.ext()
    D = 64
    for (B, H, Lq, Lk, causal) in SHAPES:
        torch.manual_seed(0)
        q = torch.randn(B, Lq, H, D, device="cuda").to(torch.bfloat16)
        k = torch.randn(B, Lk, H, D, device="cuda").to(torch.bfloat16)
        v = torch.randn(B, Lk, H, D, device="cuda").to(torch.bfloat16)
        scale = D ** -0.5
        qp = q.permute(0, 2, 1, 3).contiguous()
        kp = k.permute(0, 2, 1, 3).contiguous()
        vp = v.permute(0, 2, 1, 3).contiguous()

        t_ours = timeit(lambda: m.attn_fwd(q, k, v, scale, causal))
        t_v4 = timeit(lambda: m.attn_fwd_v4(q, k, v, scale, causal))
        t_sdpa = timeit(lambda: F.scaled_dot_product_attention(
            qp, kp, vp, is_causal=causal, scale=scale))
        flops = 4 * B * H * Lq * Lk * D  # fwd QK^T + PV
        print(f"B{B} H{H} Lq{Lq} Lk{Lk} causal={int(causal)}: "
              f"ours {t_ours:.3f} ms ({flops / t_ours / 1e9:.0f} TF) "
              f"v4 {t_v4:.3f} ms ({flops / t_v4 / 1e9:.0f} TF) "
              f"sdpa {t_sdpa:.3f} ms ({flops / t_sdpa / 1e9:.0f} TF) "
              f"-> {t_sdpa / t_ours:.2f}x, v4/v1 {t_ours / t_v4:.2f}x")

        # backward A/B
        def ours_bwd():
            o, lse = m.attn_fwd(q, k, v, scale, causal)
            m.attn_bwd(q, k, v, o, o, lse, scale, causal)

        def ours_bwd_v4():
            o, lse = m.attn_fwd_v4(q, k, v, scale, causal)
            m.attn_bwd_v4(q, k, v, o, o, lse, scale, causal)

        def sdpa_bwd():
            q2 = qp.detach().requires_grad_(True)
            k2 = kp.detach().requires_grad_(True)
            v2 = vp.detach().requires_grad_(True)
            out = F.scaled_dot_product_attention(q2, k2, v2, is_causal=causal,
                                                 scale=scale)
            out.backward(out.detach())

        t_ob = timeit(ours_bwd, n=10)
        t_ob4 = timeit(ours_bwd_v4, n=10)
        t_sb = timeit(sdpa_bwd, n=10)
        print(f"    fwd+bwd: ours {t_ob:.3f} ms v4 {t_ob4:.3f} ms "
              f"sdpa {t_sb:.3f} ms -> {t_sb / t_ob:.2f}x, "
              f"bwd v4/v1 {t_ob / t_ob4:.2f}x")


if __name__ == "__main__":
    main()
