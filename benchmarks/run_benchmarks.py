"""Benchmark runner: algorithms x envs x seeds.

Parity with the reference harness (reference: benchmarks/
run_benchmarks.py:22-51): loops the requested algorithms over the
requested environments and seeds, writing each run into
<outdir>/<env>/<algorithm>/seed-<seed>/ (experiment.log + metrics.csv +
model.pt), following the Spinning Up benchmark protocol.

The MuJoCo-v4 env ids resolve to the synthetic MuJoCo-shaped envs
(rl_replicas_amd.envs.synthetic) in this stack; CartPole-v1/Pendulum-v1
are exact classic-control dynamics.

    python benchmarks/run_benchmarks.py --algorithms ppo td3 \
        --envs HalfCheetah-v4 --seeds 0 1 2 --outdir results \
        [--device cuda] [--num-envs 20]
"""
from __future__ import annotations

import argparse
import contextlib
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from run_ddpg import run_ddpg  # noqa: E402
from run_ppo import run_ppo  # noqa: E402
from run_td3 import run_td3  # noqa: E402
from run_trpo import run_trpo  # noqa: E402
from run_vpg import run_vpg  # noqa: E402

RUNNERS = {
    "vpg": run_vpg,
    "trpo": run_trpo,
    "ppo": run_ppo,
    "ddpg": run_ddpg,
    "td3": run_td3,
}


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--algorithms", nargs="+", default=list(RUNNERS), choices=list(RUNNERS))
    parser.add_argument("--envs", nargs="+", default=["HalfCheetah-v4"])
    parser.add_argument("--seeds", nargs="+", type=int, default=[0, 1, 2])
    parser.add_argument("--outdir", default="results")
    parser.add_argument("--device", default=None, help="cuda / cpu (default: auto)")
    parser.add_argument("--num-envs", type=int, default=20, help="vectorized env instances")
    parser.add_argument("--num-epochs", type=int, default=None, help="override epoch count (for smoke runs)")
    parser.add_argument("--env-mode", choices=["cpu", "device"], default="cpu",
                        help="env residency for the synthetic envs (device = GPU-resident rollouts)")
    args = parser.parse_args()

    for env_id in args.envs:
        for algorithm in args.algorithms:
            for seed in args.seeds:
                out = os.path.join(args.outdir, env_id, algorithm, f"seed-{seed}")
                os.makedirs(out, exist_ok=True)
                print(f"=== {algorithm} / {env_id} / seed {seed} -> {out}")
                # off-policy epochs sample batch_size=50 steps: the env
                # count must divide it
                num_envs = args.num_envs
                if algorithm in ("ddpg", "td3"):
                    num_envs = max(d for d in (1, 2, 5, 10, 25, 50) if d <= num_envs)
                # stdout -> experiment.log (reference run_vpg.py:49-56)
                with open(os.path.join(out, "experiment.log"), "w") as log:
                    with contextlib.redirect_stdout(log):
                        RUNNERS[algorithm](
                            env_id,
                            seed,
                            out,
                            device=args.device,
                            num_envs=num_envs,
                            num_epochs=args.num_epochs,
                            env_mode=args.env_mode,
                        )


if __name__ == "__main__":
    main()
