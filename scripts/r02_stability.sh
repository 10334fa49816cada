#!/bin/bash
# 60-step SD-2.1 training at the bench config through the hipGraph
# replay path: loss must decrease like the round-1 eager run
# (profiles/r01_train60_sd21_loss.jsonl: 0.240 -> 0.088).
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
timeout 600 python - > gpurun_out/r02_train60_graph.log 2>&1 <<'PY'
import json, torch
from dcr_amd.train import TrainConfig, Trainer
torch.backends.cudnn.benchmark = True
cfg = TrainConfig(model_size="sd21", synthetic_data=True, synthetic_size=64,
                  resolution=256, train_batch_size=16,
                  mixed_precision="pure_bf16", channels_last=True,
                  dataloader_num_workers=0, max_train_steps=10**9, seed=7,
                  learning_rate=1e-4, lr_warmup_steps=0,
                  output_dir="/tmp/train60_out")
tr = Trainer(cfg, device=torch.device("cuda", 0))
it = iter(tr.dataloader)
batches = [next(it) for _ in range(4)]
for b in batches[:2]:
    tr.train_step(b)  # eager warmup / MIOpen find
tr.enable_hipgraph(batches[0])
losses = []
for i in range(60):
    l = float(tr.train_step(batches[i % 4]))
    losses.append(l)
    print(json.dumps({"step": i, "loss": round(l, 4)}), flush=True)
head = sum(losses[:5]) / 5
tail = sum(losses[-5:]) / 5
print(json.dumps({"head_mean": round(head, 4), "tail_mean": round(tail, 4)}))
assert all(l == l for l in losses)
assert tail < head, (head, tail)
print("GRAPH TRAINING OK")
PY
echo "train60=$?"
tail -4 gpurun_out/r02_train60_graph.log
cp gpurun_out/r02_train60_graph.log profiles/r02_train60_graph_loss.log 2>/dev/null || true
echo DONE
