"""DDPG benchmark run (reference config: 20000 epochs x 50 steps = 1M,
minibatch 100, buffer 1e6, 10k warm-up)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from common import build_off_policy  # noqa: E402


def run_ddpg(env_id, seed, outdir, device=None, num_envs=1, num_epochs=None, env_mode='cpu'):
    from rl_replicas_amd.algorithms import DDPG

    env, sampler, policy, exploration, qs, buffer, evaluator = build_off_policy(
        env_id, seed, device, num_envs, twin=False, env_mode=env_mode
    )
    model = DDPG(policy, exploration, qs[0], env, sampler, buffer, evaluator)
    model.learn(num_epochs=num_epochs or 20000, batch_size=50, output_dir=outdir)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--env", default="HalfCheetah-v4")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--outdir", default=".")
    p.add_argument("--device", default=None)
    p.add_argument("--num-envs", type=int, default=1)
    p.add_argument("--num-epochs", type=int, default=None)
    p.add_argument("--env-mode", choices=["cpu", "device"], default="cpu")
    a = p.parse_args()
    run_ddpg(a.env, a.seed, a.outdir, a.device, a.num_envs, a.num_epochs, a.env_mode)
