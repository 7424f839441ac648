"""Flagship benchmark: PPO HalfCheetah-v4, whole-node env-steps/sec.

BASELINE.json metric: "env-steps/sec (whole node) + avg return @1M
steps, PPO HalfCheetah-v4 1/2/4/8 GPU" on synthetic-obs rollouts with
random-init weights (no MuJoCo/network in this stack — BASELINE.md).

One bench step = one full PPO epoch at the reference's published
config (BASELINE.md "Config behind the numbers"): 4000 env steps per
GPU (weak scaling) sampled from vectorized synthetic HalfCheetah-shaped
envs through the policy, then GAE + advantage normalization + up to 80
clipped-surrogate policy gradient steps (KL early stop) + 80 value
steps — nothing skipped inside the timed region.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1 via torch.distributed.run, one rank per GPU over RCCL

vs_baseline: the reference's implied serial throughput is ~946
env-steps/s (PPO, 3M steps / 3170 s) on its CPU-only
single-env design; that is the only published throughput number.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch
import torch.nn as nn


def _relaunch_distributed(gpus: int) -> int:
    """Self-exec under torch.distributed.run when --gpus N>1 is invoked
    without a torchrun environment (WORLD_SIZE unset).

    The driver may call `python bench.py --gpus 8` directly; without
    this, that command would silently run ONE rank.  One rank per GPU
    over RCCL; rendezvous on 127.0.0.1 (container hostnames may not
    resolve)."""
    import socket
    import subprocess

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={gpus}",
        "--master-addr=127.0.0.1",
        f"--master-port={port}",
        os.path.abspath(__file__),
    ] + sys.argv[1:]
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    return subprocess.run(cmd, env=env).returncode


def build_model(device: str, num_envs: int, seed: int, env_mode: str = "device"):
    from rl_replicas_amd import envs, ops
    from rl_replicas_amd.algorithms import PPO
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.policies import GaussianPolicy
    from rl_replicas_amd.samplers import DeviceSampler, VectorSampler
    from rl_replicas_amd.value_function import ValueFunction

    obs_dim, act_dim = envs.MUJOCO_SHAPES["HalfCheetah-v4"]
    # reference on-policy config: policy [obs,64,32,act] tanh,
    # log_std -0.5, Adam 3e-4; value [obs,64,32,1], Adam 1e-3
    pnet = MLP([obs_dim, 64, 32, act_dim]).to(device)
    log_std = nn.Parameter(-0.5 * torch.ones(act_dim, device=device))
    policy = GaussianPolicy(
        pnet, ops.make_adam(list(pnet.parameters()) + [log_std], lr=3e-4), log_std
    )
    vnet = MLP([obs_dim, 64, 32, 1]).to(device)
    vf = ValueFunction(vnet, ops.make_adam(vnet.parameters(), lr=1e-3))

    if env_mode == "device":
        # GPU-resident envs: rollout (policy fwd + Philox sample + env
        # dynamics) never leaves the device (envs/device.py)
        venv = envs.DeviceVectorEnv("HalfCheetah-v4", num_envs=num_envs, device=device)
        sampler = DeviceSampler(venv, seed=seed)
    else:
        venv = envs.VectorEnv("HalfCheetah-v4", num_envs=num_envs)
        sampler = VectorSampler(venv, seed=seed)
    model = PPO(policy, vf, venv, sampler)  # gamma .99, lambda .97, eps .2, 80/80
    return model, sampler


def build_model_td3(device: str, num_envs: int, seed: int):
    """BASELINE config #3: TD3 Ant-v4 on one MI355X — the reference
    run_td3.py config (policy [obs,256,256,act] ReLU+Tanh, twin Q
    [obs+act,256,256,1] ReLU, Adam 1e-3, buffer 1e6, 50 env steps + 50
    train iterations of minibatch 100 per epoch)."""
    from rl_replicas_amd import envs, ops
    from rl_replicas_amd.algorithms import TD3
    from rl_replicas_amd.evaluator import Evaluator
    from rl_replicas_amd.networks import MLP
    from rl_replicas_amd.policies import DeterministicPolicy, RandomPolicy
    from rl_replicas_amd.q_function import QFunction
    from rl_replicas_amd.replay_buffer import ReplayBuffer
    from rl_replicas_amd.samplers import DeviceSampler

    obs_dim, act_dim = envs.MUJOCO_SHAPES["Ant-v4"]
    pnet = MLP(
        [obs_dim, 256, 256, act_dim],
        activation_function=nn.ReLU,
        output_activation_function=nn.Tanh,
    ).to(device)
    policy = DeterministicPolicy(pnet, ops.make_adam(pnet.parameters(), lr=1e-3))
    qs = []
    for _ in range(2):
        qn = MLP([obs_dim + act_dim, 256, 256, 1], activation_function=nn.ReLU).to(device)
        qs.append(QFunction(qn, ops.make_adam(qn.parameters(), lr=1e-3)))
    venv = envs.DeviceVectorEnv("Ant-v4", num_envs=num_envs, device=device)
    sampler = DeviceSampler(venv, seed=seed, is_continuous=True)
    model = TD3(
        policy,
        RandomPolicy(venv.action_space),
        qs[0],
        qs[1],
        venv,
        sampler,
        ReplayBuffer(int(1e6), device=device),
        Evaluator(seed=seed),
    )
    return model, sampler


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument(
        "--algo",
        choices=["ppo", "td3"],
        default="ppo",
        help="flagship PPO HalfCheetah (driver contract) or TD3 Ant-v4 "
        "(BASELINE config #3; step = 50 env steps + 50 train iterations)",
    )
    parser.add_argument("--steps", type=int, default=None, help="timed PPO epochs (default: 300 on GPU so the timed region is seconds long, 3 on CPU)")
    parser.add_argument("--warmup", type=int, default=None, help="untimed warmup epochs (default: 20 on GPU, 1 on CPU)")
    parser.add_argument("--batch-per-gpu", type=int, default=4000)
    parser.add_argument("--num-envs", type=int, default=200)
    parser.add_argument("--phase-timing", action="store_true", help="print sample/train ms split (rank 0, stderr)")
    parser.add_argument(
        "--env",
        choices=["device", "cpu"],
        default="device",
        help="env residency: GPU-resident synthetic envs (default; same "
        "dynamics, rollout stays on device) or numpy CPU VectorEnv",
    )
    parser.add_argument(
        "--dtype",
        choices=["fp32", "bf16"],
        default="fp32",
        help="MLP GEMM compute dtype: exact-fp32 MFMA (default, exceeds the "
        "reference's fp32) or bf16 MFMA with fp32 accumulate",
    )
    args = parser.parse_args()
    os.environ["RL_REPLICAS_AMD_COMPUTE_DTYPE"] = args.dtype

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        sys.exit(_relaunch_distributed(args.gpus))

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if args.gpus > 1 and world != args.gpus:
        raise SystemExit(
            f"bench.py: --gpus {args.gpus} but WORLD_SIZE={world}; refusing to "
            f"report n_gpus={args.gpus} from {world} rank(s). Launch with "
            f"torchrun --nproc-per-node {args.gpus} or let bench.py self-exec."
        )

    use_gpu = torch.cuda.is_available()
    if args.steps is None:
        args.steps = 300 if use_gpu else 3
    if args.warmup is None:
        args.warmup = 20 if use_gpu else 1
    # modulo lets multi-rank smoke tests share one GPU (gloo backend)
    dev_idx = local_rank % max(1, torch.cuda.device_count()) if use_gpu else 0
    device = f"cuda:{dev_idx}" if use_gpu else "cpu"
    if use_gpu:
        torch.cuda.set_device(dev_idx)

    from rl_replicas_amd.parallel import enable_data_parallel, init_from_env

    if world > 1:
        init_from_env()

    torch.manual_seed(1234 + rank)
    np.random.seed(1234 + rank)

    if args.algo == "td3":
        # reference off-policy protocol: 50 env steps + 50 train
        # iterations (minibatch 100) per epoch (run_td3.py, td3.py:94-105)
        td3_batch = 50
        if args.num_envs > td3_batch:
            args.num_envs = td3_batch
        model, sampler = build_model_td3(device, args.num_envs, seed=1234 + rank)
    else:
        model, sampler = build_model(device, args.num_envs, seed=1234 + rank, env_mode=args.env)
    if world > 1:
        enable_data_parallel(model)

    import tempfile

    model._begin_learn(tempfile.mkdtemp())
    model.metrics_manager.stdout = False  # keep the JSON line clean

    phase_ms = {"sample": 0.0, "train": 0.0}
    epoch_returns = []  # per-epoch mean episode return (learning evidence)

    def one_epoch_ppo():
        t0 = time.perf_counter()
        experience = sampler.sample(args.batch_per_gpu, model.policy)
        if use_gpu and args.phase_timing:
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        model.current_total_steps += sum(experience.episode_lengths)
        model.train(experience)
        if use_gpu and args.phase_timing:
            torch.cuda.synchronize()
        phase_ms["sample"] += (t1 - t0) * 1000.0
        phase_ms["train"] += (time.perf_counter() - t1) * 1000.0
        if experience.episode_returns:
            epoch_returns.append(float(np.mean(experience.episode_returns)))

    def one_epoch_td3():
        t0 = time.perf_counter()
        experience = sampler.sample(50, model.noised_policy)
        model.replay_buffer.add_experience(experience)
        if use_gpu and args.phase_timing:
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        model.current_total_steps += sum(experience.episode_lengths)
        model.train(model.replay_buffer, num_train_steps=50, minibatch_size=100)
        if use_gpu and args.phase_timing:
            torch.cuda.synchronize()
        phase_ms["sample"] += (t1 - t0) * 1000.0
        phase_ms["train"] += (time.perf_counter() - t1) * 1000.0
        if experience.episode_returns:
            epoch_returns.append(float(np.mean(experience.episode_returns)))

    one_epoch = one_epoch_td3 if args.algo == "td3" else one_epoch_ppo
    if args.algo == "td3":
        args.batch_per_gpu = 50  # env steps per epoch, reference protocol

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_epoch()

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_epoch()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks -> whole-job time
    if world > 1:
        t = torch.tensor([elapsed], device=device if use_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    total_env_steps = world * args.batch_per_gpu * args.steps
    value = total_env_steps / elapsed
    if rank == 0 and args.phase_timing:
        n = args.warmup + args.steps
        print(
            f"phase ms/epoch: sample={phase_ms['sample']/n:.2f} train={phase_ms['train']/n:.2f}",
            file=sys.stderr,
        )
    if rank == 0:
        n_epochs = args.warmup + args.steps
        result = {
            "metric": "env_steps_per_sec",
            "value": value,
            "unit": "env-steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / 946.0,
            "dtype": args.dtype,
            "data": "synthetic",
            "timed_region_s": elapsed,
            "phase_ms_per_epoch": (
                {
                    "sample": phase_ms["sample"] / n_epochs,
                    "train": phase_ms["train"] / n_epochs,
                }
                if args.phase_timing
                else None  # exact split needs --phase-timing (adds syncs)
            ),
            "avg_return": {
                "first_epoch": epoch_returns[0] if epoch_returns else None,
                "last_epoch": epoch_returns[-1] if epoch_returns else None,
                "note": "mean episode return on the synthetic HalfCheetah-shaped env "
                "(rank 0), first vs last bench epoch; real training happens inside "
                "the timed region so the return rises. MuJoCo-comparable returns "
                "are impossible offline (BASELINE.md); algorithm math is instead "
                "pinned to the reference by tests/test_reference_equivalence.py.",
            },
            "config": (
                {
                    "model": "PPO HalfCheetah-v4 (policy MLP [17,64,32,6] tanh, value [17,64,32,1])",
                    "global_batch": world * args.batch_per_gpu,
                    "seq_len": 1000,
                    "parallelism": f"dp{world}",
                    "num_envs_per_gpu": args.num_envs,
                    "env_residency": args.env,
                    "policy_grads_per_epoch": 80,
                    "value_grads_per_epoch": 80,
                    "note": "step = one PPO epoch (sample batch + full update); vs_baseline is the reference's implied ~946 env-steps/s serial CPU throughput (BASELINE.md)",
                }
                if args.algo == "ppo"
                else {
                    "model": "TD3 Ant-v4 (policy MLP [27,256,256,8] ReLU+Tanh, twin Q [35,256,256,1] ReLU)",
                    "global_batch": world * args.batch_per_gpu,
                    "seq_len": 1000,
                    "parallelism": f"dp{world}",
                    "num_envs_per_gpu": args.num_envs,
                    "env_residency": "device",
                    "train_iterations_per_epoch": 50,
                    "minibatch_size": 100,
                    "replay_buffer": 1_000_000,
                    "note": "step = one TD3 epoch (50 env steps + 50 twin-critic train iterations, reference run_td3.py protocol); no published reference throughput for TD3 (BASELINE.md) — vs_baseline uses the same 946 serial-CPU figure as a floor",
                }
            ),
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
