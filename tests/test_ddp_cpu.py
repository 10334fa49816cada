"""Multi-process (gloo, world=2) tests of the bucketed grad all-reduce
(SURVEY.md §2.3 N1 — RCCL path exercised on CPU via gloo)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(100 + rank)  # different init per rank (broadcast must fix)
        from dcr_amd.ops.adamw import FusedAdamW
        from dcr_amd.parallel.ddp import GradBucketAllReduce

        model = torch.nn.Sequential(
            torch.nn.Linear(16, 32), torch.nn.SiLU(), torch.nn.Linear(32, 4))
        opt = FusedAdamW(model.parameters(), lr=1e-3)
        ddp = GradBucketAllReduce(opt, bucket_mb=0.0001)  # tiny buckets => several

        # params must be identical after broadcast
        flat = opt.flat_param.clone()
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        assert all(torch.equal(gathered[0], g) for g in gathered), "broadcast failed"

        # different data per rank
        torch.manual_seed(500 + rank)
        x = torch.randn(8, 16)
        y = model(x).pow(2).mean()
        y.backward()
        ddp.finalize()

        # grads must now equal the cross-rank average
        gflat = opt.flat_grad.clone()
        ggath = [torch.empty_like(gflat) for _ in range(world)]
        dist.all_gather(ggath, gflat)
        assert all(torch.allclose(ggath[0], g, atol=1e-7) for g in ggath), "grad avg mismatch"

        opt.step()
        pflat = opt.flat_param.clone()
        pgath = [torch.empty_like(pflat) for _ in range(world)]
        dist.all_gather(pgath, pflat)
        assert all(torch.allclose(pgath[0], g, atol=1e-7) for g in pgath), "params diverged"

        # non-sync micro step launches no collectives and keeps buckets reset
        ddp.require_backward_grad_sync = False
        model(x).pow(2).mean().backward()
        ddp.finalize()
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_bucketed_allreduce_world2(tmp_path):
    port = 29711
    mp.spawn(_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)


def test_bucketed_allreduce_world4(tmp_path):
    """collective tests at node-scale world sizes (one MI355X
    node is 8 GPUs; gloo stands in for RCCL here)."""
    mp.spawn(_worker, args=(4, 29727, str(tmp_path)), nprocs=4, join=True)


def test_bucketed_allreduce_world8(tmp_path):
    mp.spawn(_worker, args=(8, 29729, str(tmp_path)), nprocs=8, join=True)


def _trainer_worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dcr_amd.train import TrainConfig, Trainer

        cfg = TrainConfig(model_size="tiny", synthetic_data=True, synthetic_size=4,
                          resolution=64, train_batch_size=2, mixed_precision="no",
                          dataloader_num_workers=0, max_train_steps=2, seed=0,
                          output_dir=os.path.join(tmpdir, "out"))
        tr = Trainer(cfg, device=torch.device("cpu"))
        batch = next(iter(tr.dataloader))
        tr.train_step(batch)
        flat = tr.optimizer.flat_param.clone()
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        assert torch.allclose(gathered[0], gathered[1], atol=1e-6), \
            "trainer ranks diverged after one step"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_trainer_ddp_world2(tmp_path):
    port = 29713
    mp.spawn(_trainer_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)


def _bf16_master_worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dcr_amd.train import TrainConfig, Trainer

        cfg = TrainConfig(model_size="tiny", synthetic_data=True, synthetic_size=4,
                          resolution=64, train_batch_size=2,
                          mixed_precision="pure_bf16",
                          dataloader_num_workers=0, max_train_steps=2, seed=0,
                          output_dir=os.path.join(tmpdir, "out"))
        tr = Trainer(cfg, device=torch.device("cpu"))
        assert tr.optimizer.master is not None
        # per-rank init seeds diverge the raw init; the ddp broadcast must
        # sync BOTH flat_param and the fp32 master (source of truth)
        mst = tr.optimizer.master.clone()
        gathered = [torch.empty_like(mst) for _ in range(world)]
        dist.all_gather(gathered, mst)
        assert torch.equal(gathered[0], gathered[1]), \
            "fp32 master diverged across ranks after init broadcast"
        batch = next(iter(tr.dataloader))
        tr.train_step(batch)
        mst = tr.optimizer.master.clone()
        dist.all_gather(gathered, mst)
        assert torch.allclose(gathered[0], gathered[1], atol=1e-7), \
            "fp32 master diverged after one pure-bf16 DDP step"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_pure_bf16_master_sync_world2(tmp_path):
    mp.spawn(_bf16_master_worker, args=(2, 29725, str(tmp_path)), nprocs=2,
             join=True)


def _dup_sampler_worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dcr_amd.train import TrainConfig, Trainer

        cfg = TrainConfig(model_size="tiny", instance_data_dir=os.path.join(
                              tmpdir, "imagenette"),
                          class_prompt="classlevel", duplication="dup_image",
                          resolution=64, train_batch_size=2,
                          mixed_precision="no", dataloader_num_workers=0,
                          max_train_steps=2, seed=0,
                          output_dir=os.path.join(tmpdir, f"out"))
        tr = Trainer(cfg, device=torch.device("cpu"))
        assert getattr(tr.dataset, "samplingweights", None) is not None or \
            hasattr(tr.dataset, "dataset"), "dup_image should set weights"
        # per-rank WeightedRandomSampler generators must differ
        idx = torch.tensor(list(iter(tr.dataloader.sampler))[:8])
        gathered = [torch.empty_like(idx) for _ in range(world)]
        dist.all_gather(gathered, idx)
        assert not torch.equal(gathered[0], gathered[1]), \
            "ranks drew identical duplication samples"
        # params still sync through a step
        batch = next(iter(tr.dataloader))
        tr.train_step(batch)
        flat = tr.optimizer.flat_param.clone()
        gflat = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gflat, flat)
        assert torch.allclose(gflat[0], gflat[1], atol=1e-6)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_dup_weighted_sampler_ddp_world2(tmp_path):
    from PIL import Image
    root = tmp_path / "imagenette"
    for cls in ["church", "tench"]:
        d = root / cls
        d.mkdir(parents=True)
        for i in range(6):
            Image.new("RGB", (80, 70), color=(i * 30, 100, 50)).save(d / f"{i}.png")
    mp.spawn(_dup_sampler_worker, args=(2, 29723, str(tmp_path)), nprocs=2,
             join=True)


def _feat_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from torch.utils.data import DataLoader
        from torch.utils.data.distributed import DistributedSampler
        from dcr_amd.retrieval import extract_features
        from dcr_amd.utils import MetricLogger

        # deterministic dataset shared by both ranks
        data = torch.arange(10, dtype=torch.float32).view(10, 1).repeat(1, 4)
        ds = [(data[i].view(1, 2, 2), i) for i in range(10)]
        sampler = DistributedSampler(ds, num_replicas=world, rank=rank,
                                     shuffle=False)
        loader = DataLoader(ds, batch_size=2, sampler=sampler)
        feats = extract_features(torch.nn.Flatten(), loader, torch.device("cpu"))
        # every rank must hold the complete, correctly-indexed matrix
        # (reference all_gather N4/N5 semantics, utils_ret.py:763-786)
        for i in range(10):
            assert torch.equal(feats[i], data[i]), (rank, i, feats[i])

        # SmoothedValue cross-rank sync (N6)
        ml = MetricLogger()
        ml.update(loss=float(rank + 1))  # rank0: 1.0, rank1: 2.0
        ml.synchronize_between_processes()
        assert abs(ml.meters["loss"].global_avg - 1.5) < 1e-6
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_extract_features_world2():
    mp.spawn(_feat_worker, args=(2, 29719), nprocs=2, join=True)


def _accum_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dcr_amd.ops.adamw import FusedAdamW
        from dcr_amd.parallel.ddp import GradBucketAllReduce

        torch.manual_seed(7)  # same init everywhere (broadcast also enforces)
        model = torch.nn.Linear(8, 4, bias=False)
        opt = FusedAdamW(model.parameters(), lr=1e-3)
        ddp = GradBucketAllReduce(opt, bucket_mb=0.0001)

        # two micro-batches per rank; only the second syncs
        torch.manual_seed(100 + rank)
        xa, xb = torch.randn(4, 8), torch.randn(4, 8)
        ddp.require_backward_grad_sync = False
        model(xa).pow(2).mean().backward()
        ddp.finalize()
        ddp.require_backward_grad_sync = True
        model(xb).pow(2).mean().backward()
        ddp.finalize()

        # expected: mean over ranks of (grad(xa) + grad(xb))
        ref = torch.nn.Linear(8, 4, bias=False)
        with torch.no_grad():
            ref.weight.copy_(model.weight)
        expected = torch.zeros_like(ref.weight)
        for r in range(world):
            torch.manual_seed(100 + r)
            ya, yb = torch.randn(4, 8), torch.randn(4, 8)
            ref.weight.grad = None
            ref(ya).pow(2).mean().backward()
            ref(yb).pow(2).mean().backward()
            expected += ref.weight.grad
        expected /= world

        got = opt.flat_grad.view_as(model.weight)
        assert torch.allclose(got, expected, atol=1e-6), \
            (rank, (got - expected).abs().max())
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_grad_accumulation_ddp_world2():
    """accumulated micro-steps + one synced step == averaged sum of all
    micro-grads (reference accelerate.accumulate semantics, N1)."""
    mp.spawn(_accum_worker, args=(2, 29721), nprocs=2, join=True)
