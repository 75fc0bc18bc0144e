"""bench.py contract tests: single-process CPU run and the driver's exact
torchrun launch shape (2 CPU ranks over gloo) must both emit one valid JSON
line with the whole-job aggregate."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json_line(out):
    for line in reversed(out.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


def test_bench_single_process_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu", "--steps", "2", "--warmup", "1",
         "--batch-size", "2", "--seq-len", "64"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 1 and j["steps"] == 2
    assert j["value"] > 0 and j["scaling"] == "weak"
    assert j["config"]["parallelism"] == "dp1"
    assert j["data"] == "synthetic"


def test_bench_torchrun_two_ranks_cpu():
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29741", "bench.py", "--gpus", "2", "--cpu",
         "--steps", "2", "--warmup", "1", "--batch-size", "2",
         "--seq-len", "64"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    j = _last_json_line(r.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["global_batch"] == 4
    assert j["config"]["parallelism"] == "dp2"
