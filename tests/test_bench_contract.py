"""The driver depends on bench.py's CLI + JSON contract; pin it.

Runs the real script (tiny sizes, CPU) in a subprocess and validates the
single JSON line the driver parses.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_bench_json_contract():
    out = subprocess.run(
        [
            sys.executable, os.path.join(REPO, "bench.py"),
            "--steps", "2", "--warmup", "1", "--batch", "128",
            "--train-n", "512", "--setup-epochs", "1",
        ],
        capture_output=True, text=True, timeout=540, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)

    assert d["metric"] == "inputs_per_sec_prioritized"
    assert d["unit"] == "inputs/s"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    # value is the WHOLE-JOB aggregate: inputs/s * step time = global batch
    assert d["value"] * d["ms_per_step"] / 1000 == pytest.approx(
        d["config"]["global_batch"], rel=1e-6
    )
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["dtype"] == "bf16" and d["data"] == "synthetic"
    assert d["config"]["model"] == "cifar10_resnet20"
    assert d["config"]["parallelism"] == "dp1"
    assert "vs_baseline" in d  # null (no published number), but present


def test_bench_gpus_flag_validated():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "8",
         "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=120, cwd=REPO,
    )
    assert out.returncode != 0
    assert "WORLD_SIZE" in (out.stderr + out.stdout)


@pytest.mark.timeout(900)
def test_bench_shard_train_world2_contract():
    """The strong-scaling mode end-to-end on CPU (gloo world 2): sharded
    forward + train-AT-sharded scoring must produce the contract JSON with
    scaling=strong and the whole-job (not per-rank) aggregate."""
    import numpy as np

    port = int(np.random.RandomState(os.getpid()).randint(20000, 40000))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
            "--master-port", str(port),
            os.path.join(REPO, "bench.py"), "--gpus", "2",
            "--steps", "2", "--warmup", "1", "--batch", "128",
            "--train-n", "512", "--setup-epochs", "1", "--shard-train",
        ],
        capture_output=True, text=True, timeout=840, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["scaling"] == "strong"
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "train-shard2"
    assert d["config"]["global_batch"] == 128  # total work fixed, not x2
