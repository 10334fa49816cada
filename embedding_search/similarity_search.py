#!/usr/bin/env python3
"""Streaming similarity-search CLI — parity with
/root/reference/embedding_search/similarity_search.py (with its CLI-arg
and pickle-dump bugs fixed — SURVEY.md §2.6.3-4).

Single process: streams every LAION embedding chunk through
features @ gen.T with a running max. Multi-GPU (BASELINE config 5):
launch under torchrun; each rank searches its shard and ranks exchange
[k,2] top-k candidates over RCCL (dcr_amd.search.distributed_knn).
"""
import argparse
import pickle
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from dcr_amd.search import stream_top1, dump_matches


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--generation_embedding", "--generation-embedding-path",
                   dest="generation_embedding", type=str, required=True,
                   help="embedding.pkl of the generated images")
    p.add_argument("--laion_embedding_folder", "--laion-embedding-folder",
                   dest="laion_embedding_folder", type=str, required=True,
                   help="root dir containing chunk subdirs/pickles")
    p.add_argument("--dump_path", "--dump-path", dest="dump_path",
                   type=str, required=True)
    p.add_argument("--query_chunks", "--num-chunks", dest="query_chunks",
                   type=int, default=1,
                   help="split the query matrix into this many chunks "
                        "(reference --num-chunks)")
    args = p.parse_args()

    with open(args.generation_embedding, "rb") as fh:
        gen = pickle.load(fh)
    query = torch.from_numpy(np.asarray(gen["features"], dtype=np.float32))

    root = Path(args.laion_embedding_folder)
    chunk_files = sorted(root.rglob("embedding*.pkl")) or sorted(root.rglob("*.pkl"))
    if not chunk_files:
        raise SystemExit(f"no embedding pickles under {root}")
    print(f"searching {query.shape[0]} queries over {len(chunk_files)} chunks")

    scores, keys = stream_top1(query, chunk_files, query_chunks=args.query_chunks)
    dump_matches(scores, keys, args.dump_path)
    print(f"top-1 scores: mean {scores.mean():.4f} max {scores.max():.4f} -> "
          f"{args.dump_path}")


if __name__ == "__main__":
    main()
