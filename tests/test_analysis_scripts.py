"""Analysis-tier scripts (parse_logs / reader /
time_breakdown / dp_block_partition / comm_models / inverse_model
equivalents) -- unit tests for the pure helpers plus end-to-end CLI
smoke runs on synthetic inputs, so the offline tooling stays as
covered as the training path."""

import os
import subprocess
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCRIPTS = os.path.join(REPO, "scripts")
sys.path.insert(0, REPO)


# ---------------------------------------------------------------- pure helpers

def test_block_partition_beats_round_robin_and_is_valid():
    sys.path.insert(0, SCRIPTS)
    from dp_block_partition import (block_partition, eig_cost,
                                    round_robin_cost)
    rng = np.random.default_rng(0)
    dims = rng.choice([64, 128, 256, 576, 1152, 2304, 4608], size=30)
    weights = [eig_cost(int(m)) for m in dims]
    for P in (2, 4, 8):
        best, blocks = block_partition(weights, P)
        # contiguous cover of [0, n)
        assert blocks[0][0] == 0 and blocks[-1][1] == len(weights)
        for (a, b), (c, d) in zip(blocks, blocks[1:]):
            assert b == c and a < b
        # reported cost == realized max block sum
        realized = max(sum(weights[a:b]) for a, b in blocks)
        assert abs(best - realized) < 1e-6 * max(1.0, realized)
        # optimal contiguous <= round robin, >= ideal share
        assert best <= round_robin_cost(weights, P) + 1e-9
        assert best >= sum(weights) / P - 1e-9


def test_block_partition_degenerate_more_workers_than_layers():
    sys.path.insert(0, SCRIPTS)
    from dp_block_partition import block_partition
    best, blocks = block_partition([3.0, 1.0], 2)
    assert best == 3.0 and blocks == [(0, 1), (1, 2)]


def test_fit_alpha_beta_recovers_line():
    sys.path.insert(0, SCRIPTS)
    from comm_models import fit_alpha_beta
    sizes = np.array([1e3, 1e4, 1e5, 1e6, 4e6])
    times = 5e-6 + 2e-9 * sizes
    alpha, beta = fit_alpha_beta(sizes, times)
    assert abs(alpha - 5e-6) < 1e-8
    assert abs(beta - 2e-9) < 1e-12


def test_fit_cubic_recovers_coefficients():
    sys.path.insert(0, SCRIPTS)
    from inverse_model import fit_cubic
    ms = [64, 128, 256, 512, 1024]
    ts = [1e-4 + 3e-12 * m ** 3 for m in ms]
    c0, c3 = fit_cubic(ms, ts)
    assert abs(c0 - 1e-4) < 1e-7
    assert abs(c3 - 3e-12) < 1e-15


def test_parse_logs_extracts_iters_json_and_epoch_lines(tmp_path):
    sys.path.insert(0, SCRIPTS)
    from parse_logs import parse
    log = tmp_path / "run.log"
    log.write_text(
        "noise line\n"
        "step 1 iter=123.4ms [io=1ms]\n"
        "step 2 iter=125.6ms [io=1ms]\n"
        "epoch=1 acc=0.5 img/s=456.7\n"
        '{"metric": "images/sec", "value": 460.0, "ms_per_step": 120.0,'
        ' "config": {"model": "resnet50", "kfac": "eigen_dp"}}\n')
    iters, ips, cfg = parse(str(log))
    assert iters == [123.4, 125.6, 120.0]
    assert ips == [456.7, 460.0]
    assert cfg["config"]["model"] == "resnet50"


def test_time_breakdown_parses_phase_lines(tmp_path):
    sys.path.insert(0, SCRIPTS)
    from time_breakdown import parse_file
    log = tmp_path / "run.log"
    log.write_text(
        "iter=100.0ms [io=1.0ms fwbw=20.0ms comm=3.0ms kfac=70.0ms "
        "update=6.0ms]\n"
        "iter=110.0ms [io=1.0ms fwbw=22.0ms comm=3.0ms kfac=78.0ms "
        "update=6.0ms]\n"
        'KFAC_PHASES(ms/step): {"compute_factor": 2.0, "eigh": 420.0}\n')
    iters, phases = parse_file(str(log))
    assert iters["iter"] == [100.0, 110.0]
    assert iters["kfac"] == [70.0, 78.0]
    assert phases == {"compute_factor": 2.0, "eigh": 420.0}


# ---------------------------------------------------------------- CLI smokes

def _run(args, timeout=240):
    return subprocess.run([sys.executable] + args, capture_output=True,
                          text=True, timeout=timeout, cwd=REPO)


def test_parse_logs_cli(tmp_path):
    log = tmp_path / "a.log"
    log.write_text("iter=50.0ms [x]\niter=52.0ms [x]\n")
    r = _run([os.path.join(SCRIPTS, "parse_logs.py"), str(log)])
    assert r.returncode == 0, r.stderr[-500:]
    assert "51.0" in r.stdout


def test_reader_cli_resnet18():
    r = _run([os.path.join(SCRIPTS, "reader.py"), "resnet18",
              "--world", "4"])
    assert r.returncode == 0, r.stderr[-500:]
    # per-layer factor dims and the comm byte totals must be reported
    assert "4608" in r.stdout and "MiB" in r.stdout
    assert "factor-dim histogram" in r.stdout


def test_dp_block_partition_cli():
    r = _run([os.path.join(SCRIPTS, "dp_block_partition.py")])
    assert r.returncode == 0, r.stderr[-500:]
    assert "block" in r.stdout.lower() or "partition" in r.stdout.lower()


def test_shell_tier_syntax():
    """Every launcher/driver shell script must at least parse
    (bash -n): launch_torch.sh, train_*.sh, batch*.sh, smoke script."""
    import glob
    scripts = (glob.glob(os.path.join(REPO, "*.sh"))
               + glob.glob(os.path.join(SCRIPTS, "*.sh")))
    assert len(scripts) >= 8
    for s in scripts:
        r = subprocess.run(["bash", "-n", s], capture_output=True,
                           text=True)
        assert r.returncode == 0, (s, r.stderr)
