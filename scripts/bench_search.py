#!/usr/bin/env python3
"""Embedding-search benchmark (BASELINE config 5 shape): synthetic
SSCD-like index, chunked GEMM kNN + per-shard top-k.

Single process = one shard. Under torchrun (one rank per GPU) each rank
holds index_size vectors and the [k,2] candidate all-gather merges ranks
(dcr_amd.search.distributed_knn) — 8 GPUs => 8x index at the same wall
time plus one latency-bound gather.

Prints one JSON line: queries*index dot-products/sec (GEMM-bound).
"""
from __future__ import annotations

import argparse
import json
import time

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--index-size", type=int, default=1_000_000)
    ap.add_argument("--queries", type=int, default=10_000)
    ap.add_argument("--dim", type=int, default=512)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--chunk", type=int, default=1 << 20)
    ap.add_argument("--repeat", type=int, default=3)
    ap.add_argument("--bf16", action="store_true",
                    help="bf16 GEMM + fp32 re-rank (sharded_topk "
                         "compute_dtype) — expect ~2x the fp32 rate")
    args = ap.parse_args()

    from dcr_amd.parallel import dist as dist_utils
    from dcr_amd.search import distributed_knn

    rank, world, local = dist_utils.init_distributed_mode(gate_print=False)
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", local) if use_cuda else torch.device("cpu")

    g = torch.Generator().manual_seed(7 + rank)
    shard = torch.randn(args.index_size, args.dim, generator=g).to(device)
    shard = torch.nn.functional.normalize(shard, dim=-1)
    gq = torch.Generator().manual_seed(7)
    query = torch.randn(args.queries, args.dim, generator=gq).to(device)
    query = torch.nn.functional.normalize(query, dim=-1)

    cd = torch.bfloat16 if args.bf16 else None
    v, i = distributed_knn(query, shard, k=args.k, chunk=args.chunk,
                           compute_dtype=cd)  # warmup
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.repeat):
        v, i = distributed_knn(query, shard, k=args.k, chunk=args.chunk,
                               compute_dtype=cd)
    if use_cuda:
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.repeat

    if rank == 0:
        dots = args.queries * args.index_size * world
        print(json.dumps({
            "metric": "knn_dotproducts_per_sec",
            "value": round(dots / dt, 1),
            "tflops": round(2 * dots * args.dim / dt / 1e12, 2),
            "index_size": args.index_size * world,
            "queries": args.queries,
            "dim": args.dim,
            "k": args.k,
            "sec_per_search": round(dt, 4),
            "n_gpus": world,
            "dtype": "bf16_rerank" if args.bf16 else "fp32",
        }))


if __name__ == "__main__":
    main()
