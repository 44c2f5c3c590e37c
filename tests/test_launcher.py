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


def test_main_yelp_style_config(tmp_path):
    """The Yelp canonical shape: multilabel BCE, conv+linear tail layers,
    sync batch norm, inductive, pipelined (BASELINE config #4 pattern)."""
    out = run_main(tmp_path, ["--dataset", "synth-tinyml", "--n-layers", "3",
                              "--n-linear", "1", "--norm", "batch",
                              "--inductive", "--enable-pipeline"])
    assert "Accuracy" in out  # micro-F1 eval path


def test_main_reddit_recipe(tmp_path):
    """The reference's headline reddit.sh recipe flag-for-flag (4 layers,
    hidden, dropout 0.5, --inductive --enable-pipeline --use-pp) at
    synth-small scale (scripts/reddit.sh; /root/reference/scripts/reddit.sh)."""
    out = run_main(tmp_path, ["--n-layers", "4", "--dropout", "0.5",
                              "--lr", "0.01", "--inductive",
                              "--enable-pipeline", "--use-pp"])
    assert "Epoch" in out
    assert "Accuracy" in out


def test_multi_node_launcher(tmp_path):
    """Simulate 2 nodes on localhost: each runs main.py with its own
    --node-rank and --parts-per-node 1; rendezvous over MASTER_ADDR/PORT
    (reference scripts/reddit_multi_node.sh pattern). Node 0 partitions
    first; node 1 waits for the partition dir (shared FS assumption, same
    as the reference)."""
    import time

    port = free_port()
    base = [sys.executable, os.path.join(REPO, "main.py"),
            "--dataset", "synth-tiny", "--n-partitions", "2",
            "--parts-per-node", "1", "--n-epochs", "10", "--n-layers", "2",
            "--n-hidden", "8", "--fix-seed", "--seed", "3", "--no-eval",
            "--backend", "gloo", "--port", str(port),
            "--master-addr", "127.0.0.1"]
    env = dict(os.environ, PYTHONPATH=REPO)
    p0 = subprocess.Popen(base + ["--node-rank", "0"], cwd=tmp_path, env=env,
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                          text=True)
    # node 1 skips partitioning (node_rank != 0) and needs the files;
    # wait for node 0 to write them
    deadline = time.time() + 60
    meta = tmp_path / "partitions" / "synth-tiny-2-metis-vol-trans" / \
        "meta.json"
    while not meta.exists() and time.time() < deadline:
        time.sleep(0.5)
    assert meta.exists(), "node 0 never wrote the partition"
    p1 = subprocess.Popen(base + ["--node-rank", "1"], cwd=tmp_path, env=env,
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                          text=True)
    out0, _ = p0.communicate(timeout=180)
    out1, _ = p1.communicate(timeout=180)
    assert p0.returncode == 0, out0
    assert p1.returncode == 0, out1
    assert "Process 000" in out0 and "Process 001" in out1


def test_main_bf16(tmp_path):
    out = run_main(tmp_path, ["--dtype", "bf16", "--no-eval",
                              "--enable-pipeline"])
    assert "Epoch" in out and "nan" not in out


def test_partition_tool(tmp_path):
    r = subprocess.run(
        [sys.executable, "-m", "pipegcn_amd.tools.partition_tool",
         "--dataset", "synth-tiny", "--n-partitions", "2"],
        cwd=tmp_path, env=dict(os.environ, PYTHONPATH=REPO),
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    assert (tmp_path / "partitions" / "synth-tiny-2-metis-vol-trans" /
            "meta.json").exists()


def test_skip_partition_with_hints(tmp_path):
    """--skip-partition reuses an existing partition dir; with
    --n-feat/--n-class/--n-train hints it must not reload the dataset
    (reference main.py:25-31)."""
    run_main(tmp_path, ["--no-eval"])  # creates partitions/
    out = run_main(tmp_path, ["--no-eval", "--skip-partition",
                              "--n-feat", "16", "--n-class", "4",
                              "--n-train", "79"])
    assert "Epoch" in out
