"""utils coverage: profiler no-op on CPU, Tracker JSONL schema,
SmoothedValue statistics."""
import json

import torch

from dcr_amd.utils import MetricLogger, SmoothedValue, Tracker
from dcr_amd.utils.profiler import PhaseProfiler


def test_phase_profiler_cpu_noop():
    prof = PhaseProfiler(enabled=True)  # no CUDA -> disabled internally
    with prof.phase("x"):
        pass
    assert prof.summary() == {}


def test_smoothed_value_stats():
    v = SmoothedValue(window_size=3)
    for x in [1.0, 2.0, 3.0, 4.0]:
        v.update(x)
    assert v.value == 4.0
    assert v.median == 3.0          # window [2,3,4]
    assert abs(v.avg - 3.0) < 1e-6
    assert abs(v.global_avg - 2.5) < 1e-6
    assert v.max == 4.0


def test_metric_logger_format_and_meters():
    ml = MetricLogger()
    ml.update(loss=torch.tensor(0.5), lr=1e-4)
    assert "loss" in str(ml) and "lr" in str(ml)
    assert abs(ml.loss.value - 0.5) < 1e-9


def test_tracker_jsonl(tmp_path):
    tr = Tracker("proj", name="run", config={"a": 1}, out_dir=tmp_path)
    tr.log({"loss": torch.tensor(2.0), "vec": torch.tensor([1.0, 2.0])}, step=3)
    tr.finish()
    recs = [json.loads(l) for l in
            (tmp_path / "proj_log.jsonl").read_text().splitlines()]
    assert recs[0]["_event"] == "init" and recs[0]["config"] == {"a": 1}
    assert recs[1]["loss"] == 2.0 and recs[1]["step"] == 3
    assert recs[1]["vec"] == [1.0, 2.0]


def test_log_every_iterates():
    ml = MetricLogger()
    out = list(ml.log_every(range(5), print_freq=2, header="t"))
    assert out == [0, 1, 2, 3, 4]
