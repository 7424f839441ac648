"""Microbenchmark of the hot HIP kernels at the flagship bench shapes.

Run on a GPU box:  python tools/kernbench.py
Prints per-kernel microseconds (median of repeats, event-timed) for the
PPO HalfCheetah config (policy [17,64,32,6], value [17,64,32,1],
B=4000) under both ROWS tilings, plus the loss/scan/update kernels.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def time_fn(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    from rl_replicas_amd import ops
    from rl_replicas_amd.networks import MLP

    assert ops.hip_available()
    ext = ops._load_extension()
    dev = "cuda"
    torch.manual_seed(0)

    B = 4000
    results = {}

    for rows in (16, 32, 64):
        os.environ["RL_REPLICAS_AMD_MLP_ROWS"] = str(rows)
        mlp = MLP([17, 64, 32, 6]).to(dev)
        from rl_replicas_amd.ops.fused_mlp import _extract_layers

        weights, biases, acts = _extract_layers(mlp)
        x = torch.randn(B, 17, device=dev)
        results[f"fwd_nosave_rows{rows}"] = time_fn(
            lambda: ext.mlp_forward(x, list(weights), list(biases), acts, False)
        )
        results[f"fwd_save_rows{rows}"] = time_fn(
            lambda: ext.mlp_forward(x, list(weights), list(biases), acts, True)
        )
        outs = ext.mlp_forward(x, list(weights), list(biases), acts, True)
        dy = torch.randn_like(outs[0])
        results[f"bwd_rows{rows}"] = time_fn(
            lambda: ext.mlp_backward(dy, x, list(weights), list(biases),
                                     list(outs[1:]), outs[0], acts)
        )
        # off-policy width
        q = MLP([23, 256, 256, 1]).to(dev)
        wq, bq, aq = _extract_layers(q)
        xq = torch.randn(100, 23, device=dev)
        results[f"qfwd100_rows{rows}"] = time_fn(
            lambda: ext.mlp_forward(xq, list(wq), list(bq), aq, False)
        )
    os.environ.pop("RL_REPLICAS_AMD_MLP_ROWS")

    # losses
    mean = torch.randn(B, 6, device=dev)
    actions = torch.randn(B, 6, device=dev)
    logstd = -0.5 * torch.ones(6, device=dev)
    old_logp = ext.gaussian_logp(mean, actions, logstd)
    adv = torch.randn(B, device=dev)
    results["gauss_loss"] = time_fn(
        lambda: ext.gaussian_policy_loss(mean, actions, old_logp, adv, logstd, 0.2, 1)
    )
    results["gauss_kl"] = time_fn(lambda: ext.gaussian_kl(mean, actions, logstd, old_logp))
    results["gauss_sample"] = time_fn(lambda: ext.gaussian_sample(mean, logstd, 1, 2, -1.0, -1.0))
    v = torch.randn(B, device=dev)
    results["value_mse"] = time_fn(lambda: ext.value_mse_loss(v, adv))

    # scan
    lens = [1000, 1000, 1000, 1000]
    offs = torch.tensor([0, 1000, 2000, 3000, 4000], dtype=torch.int32, device=dev)
    dn = torch.zeros(4, dtype=torch.int32, device=dev)
    rw = torch.randn(B, device=dev)
    vals = torch.randn(B, device=dev)
    lv = torch.randn(4, device=dev)
    results["segmented_gae"] = time_fn(lambda: ext.segmented_gae(rw, vals, lv, offs, dn, 0.99, 0.97))
    results["normalize"] = time_fn(lambda: ext.normalize(rw))

    # device-env transition (bench shape: 200 envs, HalfCheetah dims)
    st = torch.randn(200, 17, device=dev)
    at = torch.randn(200, 6, device=dev)
    A = torch.randn(17, 17, device=dev)
    Bm = torch.randn(6, 17, device=dev)
    w = torch.randn(17, device=dev)
    results["env_step"] = time_fn(
        lambda: ext.synthetic_env_step(st, at, A, Bm, w, 0.05, 1, 2, False)
    )
    results["env_reset"] = time_fn(lambda: ext.synthetic_env_reset(200, 17, st, 1, 2))

    # adam on the policy params
    mlp = MLP([17, 64, 32, 6]).to(dev)
    params = [p.data for p in mlp.parameters()]
    grads = [torch.randn_like(p) for p in params]
    ms = [torch.zeros_like(p) for p in params]
    vs = [torch.zeros_like(p) for p in params]
    steps = [torch.zeros((), device=dev) for _ in params]
    results["fused_adam"] = time_fn(
        lambda: ext.fused_adam_(params, grads, ms, vs, steps, 3e-4, 0.9, 0.999, 1e-8, 0.0)
    )

    for k, v_ in sorted(results.items()):
        print(f"{k:24s} {v_:8.2f} us")


if __name__ == "__main__":
    main()
