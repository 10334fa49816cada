#!/usr/bin/env python3
"""A/B the hand-written MFMA GEMM (gemm.hip) vs rocBLAS/Tensile
(F.linear / torch.matmul) on the SD-2.1 transformer linear shapes —
forward, dgrad and wgrad separately (VERDICT round-1 item 2)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from dcr_amd import ops

SHAPES = [  # (M, N, K) = x[M,K] @ W[N,K]^T, SD-2.1 bs16 256px
    (16384, 320, 320),    # res32 q/out proj
    (16384, 320, 1024),   # res32 cross k/v (per-token ctx: M=1232)
    (16384, 2560, 320),   # res32 GEGLU proj
    (16384, 320, 1280),   # res32 ff.net.2
    (4096, 640, 640),     # res16 q/out
    (4096, 5120, 640),    # res16 GEGLU
    (4096, 640, 2560),    # res16 ff.net.2
    (1024, 1280, 1280),   # res8 q/out
    (1024, 10240, 1280),  # res8 GEGLU
    (1024, 1280, 5120),   # res8 ff.net.2
    (1232, 1280, 1024),   # cross k/v proj (M = 16*77)
]


def timeit(fn, n=30):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    m = ops.ext()
    tot_n = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
    tot_r = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
    for (M, N, K) in SHAPES:
        x = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") * 0.1).to(torch.bfloat16)
        b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        dy = (torch.randn(M, N, device="cuda") * 0.5).to(torch.bfloat16)
        fl = 2.0 * M * N * K

        t_f = timeit(lambda: m.gemm_bf16(x, w, b, False, False, False))
        t_fr = timeit(lambda: F.linear(x, w, b))
        t_d = timeit(lambda: m.gemm_bf16(dy, w, None, False, True, False))
        t_dr = timeit(lambda: dy @ w)
        t_w = timeit(lambda: m.gemm_bf16(dy, x, None, True, True, True))
        t_wr = timeit(lambda: (dy.t() @ x, dy.float().sum(0)))
        tot_n["fwd"] += t_f; tot_r["fwd"] += t_fr
        tot_n["dgrad"] += t_d; tot_r["dgrad"] += t_dr
        tot_n["wgrad"] += t_w; tot_r["wgrad"] += t_wr
        print(f"M{M} N{N} K{K}: "
              f"fwd {t_f:.3f} ms ({fl/t_f/1e9:.0f} TF) vs {t_fr:.3f} "
              f"[{t_fr/t_f:.2f}x]  "
              f"dgrad {t_d:.3f} ({fl/t_d/1e9:.0f} TF) vs {t_dr:.3f} "
              f"[{t_dr/t_d:.2f}x]  "
              f"wgrad+db {t_w:.3f} ({fl/t_w/1e9:.0f} TF) vs {t_wr:.3f} "
              f"[{t_wr/t_w:.2f}x]")
    for k in tot_n:
        print(f"TOTAL {k}: native {tot_n[k]:.3f} ms vs rocBLAS "
              f"{tot_r[k]:.3f} ms -> {tot_r[k]/tot_n[k]:.2f}x")


if __name__ == "__main__":
    main()
