#!/usr/bin/env python3
"""Attribute the remaining aten glue (direct copies / fills / reduces) in
the train step to python-level ops via torch.profiler (rocprofv3 gives
kernel names only). Run on a GPU box; prints the top CPU-op table with
input shapes so the copy sources are identifiable."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
from torch.profiler import profile, ProfilerActivity

from dcr_amd.train import TrainConfig, Trainer


def main():
    torch.backends.cudnn.benchmark = True
    cfg = TrainConfig(
        model_size="sd21", synthetic_data=True, synthetic_size=64,
        resolution=256, train_batch_size=16, mixed_precision="pure_bf16",
        dataloader_num_workers=0, max_train_steps=10**9, seed=1234,
        learning_rate=5e-6, lr_warmup_steps=0, channels_last=True,
        output_dir="/tmp/prof_aten_out")
    tr = Trainer(cfg, device=torch.device("cuda", 0))
    batch = next(iter(tr.dataloader))
    for _ in range(4):
        tr.train_step(batch)
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        for _ in range(2):
            tr.train_step(batch)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="self_cuda_time_total", row_limit=45, max_name_column_width=55))


if __name__ == "__main__":
    main()
