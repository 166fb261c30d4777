"""bench.py driver contract: one JSON line on stdout with the required
fields (the round driver parses exactly this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    from tests.conftest import free_port
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        env.pop(k, None)
    env["MASTER_PORT"] = str(free_port())
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


def test_bench_world4_gloo_end_to_end():
    """The driver's multi-GPU launch pattern, on CPU/gloo at world 4:
    torch.distributed.run -> bench.py --gpus 4, rank-0 JSON aggregate.
    Proves the launch/aggregation path the 8x MI355X SCALE run uses."""
    from tests.conftest import free_port
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), os.path.join(REPO, "bench.py"),
         "--gpus", "4", "--steps", "2", "--warmup", "1",
         "--batch-size", "2", "--image-size", "64",
         "--model", "resnet18", "--kfac-name", "inverse_dp",
         "--dtype", "fp32"],
        capture_output=True, text=True, timeout=900, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    json_lines = [ln for ln in r.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, r.stdout
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 4
    assert d["config"]["parallelism"] == "dp4"
    # whole-job aggregate: 4 ranks x bs 2 x 2 steps
    assert d["config"]["global_batch"] == 8
    assert d["value"] > 0


def test_bench_world2_gloo_mpd_eigen():
    """The comm-heavy MPD 'eigen' algorithm (factor allreduce +
    eigenbasis broadcasts on rotating groups) through bench.py's own
    launch path at world 2 -- the SCALE run flips --kfac-name eigen."""
    from tests.conftest import free_port
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--batch-size", "2", "--image-size", "64",
         "--model", "resnet18", "--kfac-name", "eigen",
         "--dtype", "fp32"],
        capture_output=True, text=True, timeout=900, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    json_lines = [ln for ln in r.stdout.splitlines()
                  if ln.startswith("{")]
    d = json.loads(json_lines[0])
    assert d["config"]["kfac"] == "eigen" and d["n_gpus"] == 2
    assert d["value"] > 0
