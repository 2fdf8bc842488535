"""The driver depends on bench.py's exact contract: flags, one JSON line on
stdout from rank 0, and the field schema.  Pin it."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = {
    "metric": str, "value": float, "unit": str, "n_gpus": int, "steps": int,
    "warmup": int, "ms_per_step": float, "higher_is_better": bool,
    "scaling": str, "dtype": str, "data": str, "config": dict,
}


def _last_json_line(text):
    for line in reversed(text.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{text[-2000:]}")


def test_bench_single_process_schema():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch", "8", "--image", "64"],
        cwd=ROOT, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    out = _last_json_line(r.stdout)
    for field, typ in REQUIRED_FIELDS.items():
        assert field in out, f"missing {field}"
        assert isinstance(out[field], typ), (field, type(out[field]))
    assert "vs_baseline" in out  # null allowed (no published baseline)
    assert out["n_gpus"] == 1
    assert out["steps"] == 1
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["config"]["global_batch"] == 8
    assert out["config"]["parallelism"] == "dp1"


@pytest.mark.slow
def test_bench_torchrun_two_ranks():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29551", "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--batch", "4", "--image", "64"],
        cwd=ROOT, capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    out = _last_json_line(r.stdout)
    assert out["n_gpus"] == 2
    assert out["config"]["global_batch"] == 8
    assert out["config"]["parallelism"] == "dp2"
    # exactly ONE json line (rank 0 only prints)
    njson = sum(1 for l in r.stdout.splitlines() if l.strip().startswith("{"))
    assert njson == 1


def test_bench_data_pipeline_schema():
    """--data-pipeline keeps the contract: same JSON schema, data field
    marks the timed input pipeline."""
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch", "8", "--image", "64", "--data-pipeline"],
        cwd=ROOT, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    out = _last_json_line(r.stdout)
    for field in REQUIRED_FIELDS:
        assert field in out, f"missing {field}"
    assert out["data"] == "synthetic+pipeline"
    assert out["config"]["data_pipeline"] is True
