"""The driver depends on bench.py's CLI + JSON contract; lock it down."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--scale", "8",
         "--steps", "2", "--warmup", "1", "--backend", "torch"],
        capture_output=True, text=True, timeout=300, cwd=tmp_path, env=env)
    assert r.returncode == 0, r.stderr
    lines = [ln for ln in r.stdout.strip().splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line on stdout: {r.stdout!r}"
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["metric"] == "louvain_edges_per_sec"
    assert out["n_gpus"] == 1
    assert out["steps"] == 2
    assert out["warmup"] == 1
    assert out["higher_is_better"] is True
    assert out["data"] == "synthetic"
    assert out["value"] > 0
    assert out["ms_per_step"] > 0
    assert out["config"]["scale"] == 8
    assert out["config"]["ne_directed"] > 0
