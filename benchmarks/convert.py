"""Convert benchmark runs into per-env csv curves.

Parity with the reference converter (reference: benchmarks/convert.py:
28-113), reading this stack's metrics.csv files instead of tfevents:
for each env, gather every algorithm's seed runs, take the per-step
mean/std across seeds of the return tag (on-policy:
`sampling/average_episode_return`; off-policy:
`evaluation/average_episode_return` — reference convert.py:19-25),
apply a 10-point rolling mean, and write <csvdir>/<env>.csv with
columns: algorithm, step, mean_return, std_return.

    python benchmarks/convert.py --indir results --outdir csv
"""
from __future__ import annotations

import argparse
import csv
import os
from collections import defaultdict
from typing import Dict, List

import numpy as np

# return tag per algorithm (reference convert.py RETURN_TAGS)
RETURN_TAGS = {
    "vpg": "sampling/average_episode_return",
    "trpo": "sampling/average_episode_return",
    "ppo": "sampling/average_episode_return",
    "ddpg": "evaluation/average_episode_return",
    "td3": "evaluation/average_episode_return",
}
SMOOTHING_WINDOW = 10


def read_metric_curve(metrics_csv: str, tag: str) -> Dict[int, float]:
    curve: Dict[int, float] = {}
    with open(metrics_csv) as f:
        for row in csv.DictReader(f):
            if row["tag"] == tag and row["step"] not in ("", "None"):
                curve[int(row["step"])] = float(row["value"])
    return curve


def read_tfevents_curve(tb_dir: str, tag: str) -> Dict[int, float]:
    """Read a tag's scalar curve from events.out.tfevents.* files
    (reference convert.py reads tfevents via EventAccumulator; this
    uses the native reader in rl_replicas_amd.tfevents)."""
    from rl_replicas_amd.tfevents import read_scalar_events

    curve: Dict[int, float] = {}
    if not os.path.isdir(tb_dir):
        return curve
    for fname in sorted(os.listdir(tb_dir)):
        if not fname.startswith("events.out.tfevents"):
            continue
        for ev_tag, value, step in read_scalar_events(os.path.join(tb_dir, fname)):
            if ev_tag == tag:
                curve[step] = value
    return curve


def rolling_mean(values: np.ndarray, window: int) -> np.ndarray:
    out = np.empty_like(values, dtype=np.float64)
    for i in range(len(values)):
        lo = max(0, i - window + 1)
        out[i] = values[lo : i + 1].mean()
    return out


def convert_env(env_dir: str, env_id: str, outdir: str) -> None:
    rows: List[List] = []
    for algorithm in sorted(os.listdir(env_dir)):
        tag = RETURN_TAGS.get(algorithm)
        if tag is None:
            continue
        seed_curves = []
        algo_dir = os.path.join(env_dir, algorithm)
        for seed_dir in sorted(os.listdir(algo_dir)):
            path = os.path.join(algo_dir, seed_dir, "metrics.csv")
            if os.path.exists(path):
                seed_curves.append(read_metric_curve(path, tag))
            else:
                # tfevents input path (the reference converter's native
                # format; written here by rl_replicas_amd/tfevents.py)
                curve = read_tfevents_curve(
                    os.path.join(algo_dir, seed_dir, "tensorboard"), tag
                )
                if curve:
                    seed_curves.append(curve)
        if not seed_curves:
            continue
        common_steps = sorted(set.intersection(*(set(c) for c in seed_curves)))
        if not common_steps:
            continue
        values = np.array([[c[s] for s in common_steps] for c in seed_curves])
        mean = rolling_mean(values.mean(axis=0), SMOOTHING_WINDOW)
        std = values.std(axis=0)
        for s, m, sd in zip(common_steps, mean, std):
            rows.append([algorithm, s, m, sd])

    os.makedirs(outdir, exist_ok=True)
    out_path = os.path.join(outdir, f"{env_id}.csv")
    with open(out_path, "w", newline="") as f:
        writer = csv.writer(f)
        writer.writerow(["algorithm", "step", "mean_return", "std_return"])
        writer.writerows(rows)
    print(f"wrote {out_path} ({len(rows)} rows)")


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--indir", default="results")
    parser.add_argument("--outdir", default="csv")
    args = parser.parse_args()
    for env_id in sorted(os.listdir(args.indir)):
        env_dir = os.path.join(args.indir, env_id)
        if os.path.isdir(env_dir):
            convert_env(env_dir, env_id, args.outdir)


if __name__ == "__main__":
    main()
