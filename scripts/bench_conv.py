#!/usr/bin/env python3
"""A/B the hand-written implicit-GEMM conv fwd vs MIOpen (channels_last)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from dcr_amd import ops

SHAPES = [  # (N, C, H, W, K, R, stride) — SD-2.1 bs16 256px conv shapes
    (16, 320, 32, 32, 320, 3, 1),
    (16, 640, 16, 16, 640, 3, 1),
    (16, 1280, 8, 8, 1280, 3, 1),
    (16, 320, 32, 32, 640, 1, 1),
    (16, 640, 32, 32, 640, 3, 2),
    (16, 128, 64, 64, 128, 3, 1),
]


def timeit(fn, n=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    torch.backends.cudnn.benchmark = True
    m = ops.ext()
    for (N, C, H, W, K, R, stride) in SHAPES:
        pad = 1 if R == 3 else 0
        x = torch.randn(N, C, H, W, device="cuda").to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        w = (torch.randn(K, C, R, R, device="cuda") * 0.05).to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        b = torch.randn(K, device="cuda")
        bb = b.to(torch.bfloat16)
        t_v1 = timeit(lambda: m.conv2d_nhwc_fwd(x, w, b, stride, pad))
        t_v2 = timeit(lambda: m.conv2d_nhwc_fwd_v2(x, w, b, stride, pad))
        t_miopen = timeit(lambda: F.conv2d(x, w, bb, stride=stride, padding=pad))
        P = (H + 2 * pad - R) // stride + 1
        flops = 2 * N * P * P * K * C * R * R
        print(f"N{N} C{C} H{H} K{K} R{R}s{stride}: "
              f"v1 {t_v1:.3f} ms ({flops / t_v1 / 1e9:.0f} TF) "
              f"v2 {t_v2:.3f} ms ({flops / t_v2 / 1e9:.0f} TF) "
              f"MIOpen {t_miopen:.3f} ms ({flops / t_miopen / 1e9:.0f} TF) "
              f"-> v2/MIOpen {t_miopen / t_v2:.2f}x")

        # backward: native (dcr conv_nhwc_bwd: dgrad + wgrad + fused
        # bias-grad) vs MIOpen via aten::convolution_backward (+ the
        # separate aten bias-grad reduce it needs)
        dy = torch.randn(N, K, P, P, device="cuda").to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        t_nb = timeit(lambda: m.conv2d_nhwc_bwd(dy, x, w, stride, pad))
        t_mb = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, [K], [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [True, True, True]))
        bflops = 2 * flops  # dgrad + wgrad
        print(f"    bwd: native {t_nb:.3f} ms ({bflops / t_nb / 1e9:.0f} TF) "
              f"MIOpen {t_mb:.3f} ms ({bflops / t_mb / 1e9:.0f} TF) "
              f"-> native/MIOpen {t_mb / t_nb:.2f}x")


if __name__ == "__main__":
    main()
