"""bench.py's distributed branch, end-to-end on CPU.

VERDICT r1 item 3: the driver's 8-GPU SCALE run must succeed blind, so
the exact multi-rank path it launches — ``python -m torch.distributed.run
--nnodes=1 --nproc-per-node N bench.py`` — is exercised here with gloo at
world_size 4 (and 2): real rendezvous, init_distributed, Trainer,
BucketedDataParallel all-reduce, max-over-ranks timing, the JSON line.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench_dist(nproc, extra=()):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", "0",
        "--standalone", "--local-addr", "127.0.0.1",
        os.path.join(REPO, "bench.py"),
        "--gpus", str(nproc), "--steps", "2", "--warmup", "1",
        "--batch-size", "4", "--image", "32", *extra,
    ]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                         text=True, timeout=600)
    assert out.returncode == 0, f"stdout:\n{out.stdout}\nstderr:\n{out.stderr}"
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in:\n{out.stdout}\n{out.stderr}"
    return json.loads(lines[-1])


@pytest.mark.timeout(600)
def test_bench_torchrun_4rank_gloo():
    res = _run_bench_dist(4)
    assert res["metric"] == "train_images_per_sec"
    assert res["n_gpus"] == 4
    assert res["config"]["parallelism"] == "dp4"
    assert res["config"]["global_batch"] == 16
    assert res["value"] > 0
    assert res["steps"] == 2 and res["warmup"] == 1


@pytest.mark.timeout(600)
def test_bench_torchrun_8rank_gloo():
    # same world size the driver's 8-GPU SCALE run uses
    res = _run_bench_dist(8)
    assert res["n_gpus"] == 8
    assert res["config"]["parallelism"] == "dp8"
    assert res["config"]["global_batch"] == 32
    assert res["value"] > 0


@pytest.mark.timeout(600)
def test_bench_torchrun_2rank_gloo_no_kurt():
    res = _run_bench_dist(2, ("--no-kurt",))
    assert res["n_gpus"] == 2
    assert res["config"]["kurtosis"] is False
    assert res["value"] > 0
