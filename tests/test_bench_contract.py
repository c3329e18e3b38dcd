"""bench.py contract tests (the driver's BENCH/SCALE entry point): the
JSON line matches BASELINE.json's metric/config fields, and the exact
multi-rank launch shape the driver uses (torch.distributed.run, gloo on
CPU here / RCCL on the GPU box) completes with n_gpus=WORLD_SIZE."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    raise AssertionError(f"no metric JSON in output:\n{stdout[-2000:]}")


def _check_contract(rec: dict, n: int):
    assert rec["metric"] == "lab1_price_match_agent_decisions_per_sec"
    assert rec["unit"] == "decisions/s"
    assert rec["n_gpus"] == n
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["config"]["model"]
    assert rec["vs_baseline"] is None or rec["vs_baseline"] > 0


@pytest.mark.timeout(300)
def test_bench_single_rank_stub_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--stub-llm", "--steps", "2",
         "--warmup", "1", "--batch", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    _check_contract(_last_json(out.stdout), 1)


@pytest.mark.timeout(300)
def test_bench_two_rank_driver_launch_shape():
    port = str(29700 + os.getpid() % 200)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", port, "bench.py", "--gpus", "2", "--stub-llm",
         "--steps", "1", "--warmup", "0", "--batch", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json(out.stdout)
    _check_contract(rec, 2)
    # whole-job aggregate: 2 ranks of fixed per-rank work
    assert rec["steps"] == 1
