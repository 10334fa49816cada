"""hipEvent-based phase timing (SURVEY.md §5.1 — the reference has no
tracing; this is the framework's built-in per-phase profiler).

Usage:
    prof = PhaseProfiler(enabled=True)
    with prof.phase("vae_encode"):
        ...
    prof.summary()  # {phase: avg_ms}

Events are torch.cuda.Event (hipEvent under ROCm); timings resolve lazily
at summary() so the hot loop never synchronizes. Enable in the trainer
with DCR_PROFILE=1. For kernel-level data use rocprofv3 (see profiles/).
"""
from __future__ import annotations

import os
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Tuple

import torch


class PhaseProfiler:
    def __init__(self, enabled: bool | None = None):
        if enabled is None:
            enabled = os.environ.get("DCR_PROFILE") == "1"
        self.enabled = enabled and torch.cuda.is_available()
        self._pending: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []

    @contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        start = torch.cuda.Event(enable_timing=True)
        end = torch.cuda.Event(enable_timing=True)
        start.record()
        try:
            yield
        finally:
            end.record()
            self._pending.append((name, start, end))

    def summary(self, reset: bool = True) -> Dict[str, float]:
        if not self.enabled or not self._pending:
            return {}
        torch.cuda.synchronize()
        acc = defaultdict(list)
        for name, s, e in self._pending:
            acc[name].append(s.elapsed_time(e))
        out = {k: sum(v) / len(v) for k, v in acc.items()}
        if reset:
            self._pending.clear()
        return out
