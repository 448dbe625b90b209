"""The driver depends on bench.py's CLI and JSON-line contract — protect it."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def test_bench_json_contract_cpu():
    env = dict(os.environ, MAML355_NO_HIP="1")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "1",
         "--warmup", "0", "--tasks_per_gpu", "2", "--inner_steps", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert REQUIRED_KEYS.issubset(d.keys()), REQUIRED_KEYS - set(d.keys())
    assert d["metric"] == "meta-tasks/sec"
    assert d["value"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["global_batch"] == 2
    assert d["config"]["parallelism"] == "task-dp1"
