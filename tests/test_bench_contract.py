"""bench.py driver contract: one JSON line on stdout with the required
fields (the round driver parses exactly this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        env.pop(k, None)
    env["MASTER_PORT"] = "29733"
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "1",
         "--warmup", "0", "--batch-size", "2", "--image-size", "64",
         "--model", "resnet18"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [ln for ln in r.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, r.stdout
    d = json.loads(json_lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1
    assert d["data"] == "synthetic"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["config"]["parallelism"] == "dp1"
    assert d["value"] > 0
