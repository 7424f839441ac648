"""TRPO benchmark run (reference config — CG optimizer defaults,
750 epochs x 4000 steps)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from common import build_on_policy  # noqa: E402


def run_trpo(env_id, seed, outdir, device=None, num_envs=20, num_epochs=None, env_mode='cpu'):
    from rl_replicas_amd.algorithms import TRPO

    env, sampler, policy, value_function = build_on_policy(env_id, seed, device, num_envs, "cg", env_mode=env_mode)
    model = TRPO(policy, value_function, env, sampler)
    model.learn(num_epochs=num_epochs or 750, batch_size=4000, output_dir=outdir)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--env", default="HalfCheetah-v4")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--outdir", default=".")
    p.add_argument("--device", default=None)
    p.add_argument("--num-envs", type=int, default=20)
    p.add_argument("--num-epochs", type=int, default=None)
    p.add_argument("--env-mode", choices=["cpu", "device"], default="cpu")
    a = p.parse_args()
    run_trpo(a.env, a.seed, a.outdir, a.device, a.num_envs, a.num_epochs, a.env_mode)
