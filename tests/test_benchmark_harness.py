"""Benchmark harness smoke tests: run scripts + converter round trip."""
import csv
import os
import subprocess
import sys

import pytest

BENCH = os.path.join(os.path.dirname(__file__), "..", "benchmarks")


def test_run_benchmarks_and_convert(tmp_path):
    res = tmp_path / "results"
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(BENCH, "run_benchmarks.py"),
            "--algorithms", "vpg",
            "--envs", "CartPole-v1",
            "--seeds", "0",
            "--num-epochs", "2",
            "--num-envs", "10",
            "--outdir", str(res),
        ],
        capture_output=True,
        text=True,
        timeout=300,
    )
    assert out.returncode == 0, out.stderr
    run_dir = res / "CartPole-v1" / "vpg" / "seed-0"
    assert (run_dir / "experiment.log").exists()
    assert (run_dir / "metrics.csv").exists()

    csv_dir = tmp_path / "csv"
    out2 = subprocess.run(
        [
            sys.executable,
            os.path.join(BENCH, "convert.py"),
            "--indir", str(res),
            "--outdir", str(csv_dir),
        ],
        capture_output=True,
        text=True,
        timeout=120,
    )
    assert out2.returncode == 0, out2.stderr
    rows = list(csv.DictReader(open(csv_dir / "CartPole-v1.csv")))
    assert len(rows) == 2  # one smoothed point per epoch
    assert rows[0]["algorithm"] == "vpg"
    assert int(rows[0]["step"]) == 4000
