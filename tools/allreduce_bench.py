"""All-reduce latency microbench over the DP message sizes.

The DP design sends ONE fused flat fp32 buffer per optimizer step
(parallel/ddp.py): policy+value grads are ~13 KB at the reference PPO
config, so collectives are pure latency — this measures that latency
across message sizes to give docs/ARCHITECTURE.md's "single flat
message, no buckets" claim a measured basis (round-1 VERDICT item 2).

Launch (any world size / backend):
  torchrun --nproc-per-node N --master-addr 127.0.0.1 tools/allreduce_bench.py
  python tools/allreduce_bench.py            # world=1 self-init

Rank 0 prints one JSON line with per-size mean/p50 latency (us) and the
effective algorithm bandwidth.
"""
from __future__ import annotations

import json
import os
import time

import torch
import torch.distributed as dist

SIZES = [
    13 * 1024,       # reference PPO policy+value flat grads (~3.4K params)
    64 * 1024,
    256 * 1024,
    1024 * 1024,
    8 * 1024 * 1024,  # off-policy twin-critic scale
]
ITERS = 200
WARMUP = 20


def main() -> None:
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")) % torch.cuda.device_count())
    backend = os.environ.get("RL_REPLICAS_AMD_DIST_BACKEND") or ("nccl" if use_gpu else "gloo")
    if world == 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group(backend=backend)
    device = "cuda" if use_gpu else "cpu"

    results = []
    for size_bytes in SIZES:
        n = size_bytes // 4
        buf = torch.randn(n, device=device)
        for _ in range(WARMUP):
            dist.all_reduce(buf)
        if use_gpu:
            torch.cuda.synchronize()
        lat_us = []
        for _ in range(ITERS):
            t0 = time.perf_counter()
            dist.all_reduce(buf)
            if use_gpu:
                torch.cuda.synchronize()
            lat_us.append((time.perf_counter() - t0) * 1e6)
        lat_us.sort()
        mean = sum(lat_us) / len(lat_us)
        p50 = lat_us[len(lat_us) // 2]
        # ring all-reduce moves 2(N-1)/N * size per GPU
        moved = 2 * (world - 1) / max(1, world) * size_bytes
        results.append(
            {
                "size_bytes": size_bytes,
                "mean_us": round(mean, 2),
                "p50_us": round(p50, 2),
                "p99_us": round(lat_us[int(len(lat_us) * 0.99)], 2),
                "alg_bw_GBps": round(moved / (p50 * 1e-6) / 1e9, 3) if world > 1 else None,
            }
        )

    if rank == 0:
        print(
            json.dumps(
                {
                    "bench": "all_reduce_latency",
                    "backend": backend,
                    "world_size": world,
                    "device": device,
                    "iters": ITERS,
                    "sizes": results,
                }
            )
        )
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
