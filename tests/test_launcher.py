"""End-to-end launcher integration test: main.py via subprocess on a tiny
synthetic dataset, 2 partitions, gloo/CPU — the reference's de-facto
integration strategy (runnable example scripts as tests)."""
import os
import subprocess
import sys

import torch

from tests.conftest import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_main(tmp_path, extra, timeout=240):
    cmd = [sys.executable, os.path.join(REPO, "main.py"),
           "--dataset", "synth-tiny", "--n-partitions", "2",
           "--n-epochs", "10", "--n-layers", "2", "--n-hidden", "8",
           "--log-every", "3", "--fix-seed", "--seed", "3",
           "--backend", "gloo", "--port", str(free_port())] + extra
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(cmd, cwd=tmp_path, env=env, capture_output=True,
                       text=True, timeout=timeout)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    return r.stdout


def test_main_vanilla_with_eval(tmp_path):
    out = run_main(tmp_path, [])
    assert "Validation Accuracy" in out
    assert "model saved" in out
    # checkpoint in the reference format and location
    ckpt = tmp_path / "model" / "synth-tiny-2-metis-vol-trans_final.pth.tar"
    assert ckpt.exists()
    sd = torch.load(ckpt, weights_only=True)
    assert any(k.startswith("layers.0.linear1") for k in sd)


def test_main_pipelined_corrections(tmp_path):
    out = run_main(tmp_path, ["--enable-pipeline", "--feat-corr",
                              "--grad-corr", "--no-eval"])
    assert "Epoch" in out
    res = tmp_path / "results"
    assert res.exists()


def test_main_random_partition_use_pp(tmp_path):
    out = run_main(tmp_path, ["--partition-method", "random", "--use-pp",
                              "--no-eval"])
    assert "Epoch" in out


def test_main_gcn_model(tmp_path):
    out = run_main(tmp_path, ["--model", "gcn", "--no-eval",
                              "--enable-pipeline"])
    assert "Epoch" in out


def test_checkpoint_resume(tmp_path):
    run_main(tmp_path, ["--no-eval", "--checkpoint-every", "4"])
    out = run_main(tmp_path, ["--no-eval", "--checkpoint-every", "4",
                              "--resume"])
    assert "resumed from epoch 7" in out
