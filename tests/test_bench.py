"""bench.py contract tests — the driver launches this file directly
(`python bench.py --gpus N --steps K --warmup W`) and parses ONE JSON
line; breakage here silently kills the round-end benchmark."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def _run(args, timeout=240):
    r = subprocess.run([sys.executable, "bench.py", "--shape", "small",
                        "--steps", "2", "--warmup", "1"] + args,
                       cwd=REPO, capture_output=True, text=True,
                       timeout=timeout)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def test_bench_json_contract():
    d = _run([])
    for k in REQUIRED:
        assert k in d, k
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is False
    assert d["data"] == "synthetic"
    assert d["config"]["parallelism"] == "graph-partition dp1"
    assert d["value"] > 0


@pytest.mark.parametrize("extra", [["--model", "gcn"],
                                   ["--use-pp"],
                                   ["--no-pipeline"],
                                   ["--feat-corr", "--grad-corr"],
                                   ["--norm", "batch"]])
def test_bench_variants(extra):
    d = _run(extra)
    assert d["value"] > 0


def test_bench_world2_gloo():
    """The exact launch pattern the driver uses for N>1 (CPU/gloo here)."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29591", "bench.py", "--gpus", "2", "--shape",
         "small", "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["config"]["boundary_comm_overlap_pct"] is not None
