"""Observability: windowed metric smoothing, cross-rank sync, trackers.

Capability parity: SmoothedValue/MetricLogger with cross-rank all-reduce
(/root/reference/utils_ret.py:526-674) and the wandb event schema
({loss, lr} per train step, diff_train.py:703-705; retrieval stats,
diff_retrieval.py:456-468). wandb is optional here; a JSONL tracker is
always on so runs are inspectable offline.
"""
from __future__ import annotations

import datetime
import json
import time
from collections import defaultdict, deque
from pathlib import Path

import torch

from ..parallel import dist as dist_utils
import torch.distributed as dist


class SmoothedValue:
    """Track a series with a smoothing window + global average."""

    def __init__(self, window_size: int = 20, fmt: str = "{median:.4f} ({global_avg:.4f})"):
        self.deque = deque(maxlen=window_size)
        self.total = 0.0
        self.count = 0
        self.fmt = fmt

    def update(self, value, n: int = 1):
        self.deque.append(value)
        self.count += n
        self.total += value * n

    def synchronize_between_processes(self):
        """all-reduce [count, total] (reference: utils_ret.py:550-555)."""
        if not dist_utils.is_dist():
            return
        t = torch.tensor([self.count, self.total], dtype=torch.float64)
        if torch.cuda.is_available():
            t = t.cuda()
        dist.barrier()
        dist.all_reduce(t)
        t = t.tolist()
        self.count = int(t[0])
        self.total = t[1]

    @property
    def median(self):
        return torch.tensor(list(self.deque)).median().item() if self.deque else 0.0

    @property
    def avg(self):
        return torch.tensor(list(self.deque), dtype=torch.float32).mean().item() \
            if self.deque else 0.0

    @property
    def global_avg(self):
        return self.total / max(self.count, 1)

    @property
    def max(self):
        return max(self.deque) if self.deque else 0.0

    @property
    def value(self):
        return self.deque[-1] if self.deque else 0.0

    def __str__(self):
        return self.fmt.format(median=self.median, avg=self.avg,
                               global_avg=self.global_avg, max=self.max,
                               value=self.value)


class MetricLogger:
    def __init__(self, delimiter: str = "  "):
        self.meters = defaultdict(SmoothedValue)
        self.delimiter = delimiter

    def update(self, **kwargs):
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            self.meters[k].update(float(v))

    def __getattr__(self, attr):
        if attr in self.meters:
            return self.meters[attr]
        raise AttributeError(attr)

    def __str__(self):
        return self.delimiter.join(f"{n}: {m}" for n, m in self.meters.items())

    def synchronize_between_processes(self):
        for m in self.meters.values():
            m.synchronize_between_processes()

    def add_meter(self, name, meter):
        self.meters[name] = meter

    def log_every(self, iterable, print_freq: int, header: str = ""):
        i = 0
        start = time.time()
        end = time.time()
        iter_time = SmoothedValue(fmt="{avg:.4f}")
        data_time = SmoothedValue(fmt="{avg:.4f}")
        n = len(iterable) if hasattr(iterable, "__len__") else None
        for obj in iterable:
            data_time.update(time.time() - end)
            yield obj
            iter_time.update(time.time() - end)
            if i % print_freq == 0 or (n is not None and i == n - 1):
                mem = ""
                if torch.cuda.is_available():
                    mem = f" max mem: {torch.cuda.max_memory_allocated() / 1024**2:.0f}MB"
                total = f"/{n}" if n else ""
                print(f"{header} [{i}{total}] {self} "
                      f"time: {iter_time} data: {data_time}{mem}")
            i += 1
            end = time.time()
        total_time = time.time() - start
        print(f"{header} Total time: {datetime.timedelta(seconds=int(total_time))}")


class Tracker:
    """wandb-compatible tracker with an always-on JSONL fallback.

    Reference schema: wandb project "diffrep_ft" for training
    (diff_train.py:551), "imsimv2_retrieval" for retrieval
    (diff_retrieval.py:382).
    """

    def __init__(self, project: str, name: str | None = None,
                 config: dict | None = None, out_dir: str | Path = ".",
                 enabled: bool = True):
        self.enabled = enabled and dist_utils.is_main_process()
        self.jsonl = None
        self.wandb = None
        if not self.enabled:
            return
        out_dir = Path(out_dir)
        out_dir.mkdir(parents=True, exist_ok=True)
        self.jsonl = open(out_dir / f"{project}_log.jsonl", "a")
        self.log({"_event": "init", "project": project, "name": name,
                  "config": config or {}})
        try:
            import wandb  # optional; absent in this image
            self.wandb = wandb.init(project=project, name=name, config=config)
        except Exception:
            self.wandb = None

    def log(self, metrics: dict, step: int | None = None):
        if not self.enabled:
            return
        rec = dict(metrics)
        if step is not None:
            rec["step"] = step
        rec["_t"] = time.time()
        def clean(v):
            if isinstance(v, torch.Tensor):
                return v.item() if v.numel() == 1 else v.tolist()
            return v
        rec = {k: clean(v) for k, v in rec.items()}
        self.jsonl.write(json.dumps(rec) + "\n")
        self.jsonl.flush()
        if self.wandb is not None:
            self.wandb.log(metrics, step=step)

    def finish(self):
        if self.jsonl:
            self.jsonl.close()
        if self.wandb is not None:
            self.wandb.finish()
