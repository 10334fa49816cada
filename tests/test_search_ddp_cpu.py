"""Multi-process (gloo, world=2) sharded kNN test (BASELINE config 5
communication pattern: per-shard top-k + all-gather of [k,2] candidates)."""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dcr_amd.search import distributed_knn
        torch.manual_seed(0)  # same query on every rank
        D, N = 16, 40
        query = torch.randn(6, D)
        full = torch.randn(world * N, D)     # same full index on every rank
        shard = full[rank * N:(rank + 1) * N].clone()
        v, i = distributed_knn(query, shard, k=3, shard_sizes=[N] * world)
        ref = (query @ full.t()).topk(3, dim=1)
        assert torch.allclose(v, ref.values, atol=1e-5), (rank, v, ref.values)
        assert torch.equal(i, ref.indices)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_distributed_knn_world2():
    mp.spawn(_worker, args=(2, 29717), nprocs=2, join=True)


def test_distributed_knn_world4():
    """top-k all-gather at node-scale world sizes."""
    mp.spawn(_worker, args=(4, 29731), nprocs=4, join=True)


def test_distributed_knn_world8():
    mp.spawn(_worker, args=(8, 29733), nprocs=8, join=True)
