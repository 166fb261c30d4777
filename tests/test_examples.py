"""Example trainers run end-to-end (CPU, tiny synthetic configs).

Guards the example-script tier the reference ships (examples/*.py) --
argument parsing, distributed init, K-FAC wiring, phase timers and the
speed report all execute.
"""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_example(script, args, timeout=420):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "0"
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, script)] + args,
        capture_output=True, text=True, timeout=timeout, env=env,
        cwd=REPO)
    assert r.returncode == 0, f"{script} failed:\n{r.stdout}\n{r.stderr}"
    return r.stdout + r.stderr


@pytest.mark.parametrize("extra", [[], ["--kfac-name", "inverse_dp"]])
def test_train_cifar_speed_mode(extra):
    out = run_example(
        "examples/train_cifar.py",
        ["--model", "resnet20", "--batch-size", "4", "--iters-per-epoch",
         "3", "--epochs", "1", "--speed", "--display", "2",
         "--kfac-update-freq", "2"] + extra)
    assert "images/s" in out


def test_train_wikitext_rnn():
    out = run_example(
        "examples/train_wikitext_rnn.py",
        ["--vocab-size", "300", "--bptt", "8", "--batch-size", "4",
         "--iters-per-epoch", "3", "--epochs", "1", "--speed",
         "--display", "2"])
    assert "images/s" in out


def test_train_transformer_small():
    out = run_example(
        "examples/train_transformer.py",
        ["--batch-size", "4", "--iters-per-epoch", "2", "--epochs", "1",
         "--speed", "--display", "1"])
    assert "images/s" in out


def test_train_imagenet_checkpoint_resume(tmp_path):
    """Save a checkpoint at epoch 0, resume from it: the resume path
    (utils.load_checkpoint + KFACParamScheduler epoch) must work
    (reference: examples/pytorch_imagenet_resnet.py:162-167,305-312)."""
    ckpt = str(tmp_path / "ck-{epoch}.pth.tar")
    common_args = ["--model", "resnet18", "--num-classes", "7",
                   "--image-size", "64", "--batch-size", "2",
                   "--iters-per-epoch", "2", "--display", "1",
                   "--kfac-update-freq", "2"]
    run_example("examples/train_imagenet.py",
                common_args + ["--epochs", "1",
                               "--checkpoint-format", ckpt])
    saved = str(tmp_path / "ck-0.pth.tar")
    assert os.path.exists(saved)
    out = run_example("examples/train_imagenet.py",
                      common_args + ["--epochs", "2",
                                     "--resume-from", saved])
    assert "epoch 1" in out  # resumed into epoch 1
