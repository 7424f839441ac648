"""Driver-contract tests for bench.py.

The round-end driver may invoke `python bench.py --gpus N` DIRECTLY
(BENCH_r01.json `cmd`), without torchrun.  bench.py must then self-exec
under torch.distributed.run so N ranks actually run and the reported
`n_gpus` equals N — otherwise an 8-GPU scaling measurement silently
becomes a 1-rank run (round-1 VERDICT.md, top priority fix).
"""
from __future__ import annotations

import json
import os
import subprocess
import sys

import pytest

BENCH = os.path.join(os.path.dirname(__file__), os.pardir, "bench.py")


def _run_bench(extra_args, timeout=600):
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    return subprocess.run(
        [sys.executable, BENCH] + extra_args,
        capture_output=True,
        text=True,
        timeout=timeout,
        env=env,
    )


def _parse_json_line(stdout: str) -> dict:
    for line in stdout.splitlines():
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in bench stdout:\n{stdout}")


@pytest.mark.slow
def test_gpus_flag_spawns_ranks():
    """`python bench.py --gpus 2` with no torchrun env must run 2 ranks."""
    r = _run_bench(
        ["--gpus", "2", "--steps", "1", "--warmup", "0", "--num-envs", "4",
         "--batch-per-gpu", "64", "--env", "cpu"]
    )
    assert r.returncode == 0, r.stderr[-3000:]
    result = _parse_json_line(r.stdout)
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "dp2"
    assert result["config"]["global_batch"] == 128  # world * batch_per_gpu


def test_world_size_mismatch_fails_loudly():
    """--gpus 4 under a WORLD_SIZE=1 env must abort, not report n_gpus=4."""
    env = dict(os.environ)
    env.update({"WORLD_SIZE": "1", "RANK": "0", "LOCAL_RANK": "0"})
    r = subprocess.run(
        [sys.executable, BENCH, "--gpus", "4", "--steps", "1", "--warmup", "0"],
        capture_output=True,
        text=True,
        timeout=300,
        env=env,
    )
    assert r.returncode != 0
    assert "refusing" in (r.stderr + r.stdout)


def test_single_gpu_json_contract():
    r = _run_bench(
        ["--steps", "2", "--warmup", "1", "--num-envs", "4",
         "--batch-per-gpu", "64", "--env", "cpu"]
    )
    assert r.returncode == 0, r.stderr[-3000:]
    result = _parse_json_line(r.stdout)
    for key in (
        "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
        "higher_is_better", "scaling", "vs_baseline", "dtype", "data",
        "timed_region_s", "avg_return", "config",
    ):
        assert key in result, key
    assert result["n_gpus"] == 1
    assert result["steps"] == 2 and result["warmup"] == 1
    assert result["value"] > 0
    assert result["avg_return"]["last_epoch"] is not None
