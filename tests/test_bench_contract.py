"""bench.py driver-contract tests: single JSON line, correct fields,
works under torch.distributed.run exactly as the driver launches it."""
import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).parent.parent


def _check_line(line: str, n_gpus: int):
    rec = json.loads(line)
    # every field the driver's contract names must be present
    required = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"}
    assert required <= rec.keys(), required - rec.keys()
    assert rec["metric"] == "sd21_256px_finetune_imgs_per_sec"
    assert rec["n_gpus"] == n_gpus
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["config"]["global_batch"] == 2 * n_gpus
    assert rec["config"]["parallelism"] == f"dp{n_gpus}"
    assert {"model", "global_batch", "seq_len"} <= rec["config"].keys()
    return rec


def test_bench_single_process(tmp_path):
    r = subprocess.run(
        [sys.executable, str(ROOT / "bench.py"), "--steps", "2", "--warmup",
         "1", "--model", "tiny", "--resolution", "64", "--batch-size", "2"],
        capture_output=True, text=True, cwd=str(ROOT), timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout  # exactly ONE JSON line
    _check_line(lines[0], 1)


def test_bench_torchrun_world2(tmp_path):
    """the driver's launch shape: torch.distributed.run, one rank/'GPU'."""
    env = {**os.environ, "MASTER_ADDR": "127.0.0.1"}
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533", str(ROOT / "bench.py"), "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--model", "tiny", "--resolution",
         "64", "--batch-size", "2"],
        capture_output=True, text=True, cwd=str(ROOT), env=env, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout  # only rank 0 prints
    _check_line(lines[0], 2)
