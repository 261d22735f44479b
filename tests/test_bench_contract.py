"""Driver-contract guard: `python bench.py` must emit exactly one JSON
line with the agreed fields, at CPU-debug scale."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cpu",
         "--steps", "1", "--warmup", "0", "--gb-per-gpu", "0.005",
         "--partitions-per-executor", "16"],
        capture_output=True, text=True, timeout=300, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in j, field
    assert j["n_gpus"] == 1 and j["steps"] == 1
    assert j["higher_is_better"] is True and j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    assert isinstance(j["value"], (int, float)) and j["value"] > 0
    cfg = j["config"]
    for field in ("model", "global_batch", "partitions", "parallelism"):
        assert field in cfg, field


import pytest


@pytest.mark.parametrize("workload", ["pagerank", "join", "reducebykey",
                                      "groupby"])
def test_bench_all_workloads_emit_json(tmp_path, workload):
    """Every BASELINE workload's bench path runs end-to-end on CPU and
    emits the JSON line (guards attr regressions like r02's seq_len)."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cpu",
         "--workload", workload, "--steps", "1", "--warmup", "0",
         "--gb-per-gpu", "0.002", "--partitions-per-executor", "16"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert j["value"] > 0 and j["config"]["model"] == workload
