"""Worker-process env vectorization (shared-memory transport).

The native envs in this package are numpy-batched in-process
(vector.py) — fastest when stepping IS vectorizable math.  Third-party
simulator envs (real gymnasium MuJoCo) only expose the serial step API
and cost ~1 ms/step on a core: feeding a 400K-steps/s learner from one
process is impossible, so this module provides the worker-process
analogue: N envs sharded over W worker
processes, observations/rewards/flags exchanged through POSIX shared
memory (one write per worker per step, no per-step pickling of
observation arrays), with the same step/reset/autoreset/final_obs
contract as VectorEnv — drop-in under VectorSampler.

Workers never touch the GPU; use the default "fork" start method on
Linux (CUDA must not be initialized before construction if fork is
used — construct your envs before moving models to the GPU, or pass
context="spawn").
"""
from __future__ import annotations

import multiprocessing as mp
from multiprocessing import shared_memory
from typing import Callable, List, Optional

import numpy as np


def _worker_loop(remote, env_fns, start: int, shm_names, shapes, dtypes):
    """One worker: owns envs [start, start+len(env_fns)); reads its
    action slice and writes its obs/reward/flag slices in shared memory."""
    shms = {k: shared_memory.SharedMemory(name=n) for k, n in shm_names.items()}
    arrs = {
        k: np.ndarray(shapes[k], dtype=dtypes[k], buffer=shms[k].buf) for k in shms
    }
    envs = [fn() for fn in env_fns]
    n = len(envs)
    sl = slice(start, start + n)
    elapsed = np.zeros(n, dtype=np.int64)

    try:
        while True:
            cmd, payload = remote.recv()
            if cmd == "reset":
                seeds = payload
                for i, env in enumerate(envs):
                    obs, _ = env.reset(seed=None if seeds is None else seeds + start + i)
                    arrs["obs"][start + i] = obs
                elapsed[:] = 0
                remote.send(("ok", None))
            elif cmd == "step":
                actions = arrs["act"][sl]
                for i, env in enumerate(envs):
                    obs, reward, term, trunc, _ = env.step(actions[i])
                    arrs["final"][start + i] = obs
                    arrs["rew"][start + i] = reward
                    arrs["term"][start + i] = term
                    arrs["trunc"][start + i] = trunc
                    if term or trunc:
                        obs, _ = env.reset()
                    arrs["obs"][start + i] = obs
                remote.send(("ok", None))
            elif cmd == "close":
                for env in envs:
                    env.close()
                remote.send(("ok", None))
                break
    finally:
        for shm in shms.values():
            shm.close()


class SubprocVectorEnv:
    """VectorEnv-compatible vectorization over worker processes."""

    def __init__(
        self,
        env_fns: List[Callable],
        num_workers: Optional[int] = None,
        context: str = "fork",
    ):
        self.num_envs = len(env_fns)
        probe = env_fns[0]()
        self.observation_space = probe.observation_space
        self.action_space = probe.action_space
        self.spec = probe.spec
        probe.close()

        obs_shape = self.observation_space.shape
        act_shape = self.action_space.shape or (1,)
        n = self.num_envs
        shapes = {
            "obs": (n, *obs_shape),
            "final": (n, *obs_shape),
            "act": (n, *act_shape),
            "rew": (n,),
            "term": (n,),
            "trunc": (n,),
        }
        dtypes = {
            "obs": np.float32,
            "final": np.float32,
            "act": np.float64,
            "rew": np.float64,
            "term": np.bool_,
            "trunc": np.bool_,
        }
        self._shms = {}
        self._arrs = {}
        for k, shape in shapes.items():
            nbytes = int(np.prod(shape)) * np.dtype(dtypes[k]).itemsize
            shm = shared_memory.SharedMemory(create=True, size=max(1, nbytes))
            self._shms[k] = shm
            self._arrs[k] = np.ndarray(shape, dtype=dtypes[k], buffer=shm.buf)
        shm_names = {k: s.name for k, s in self._shms.items()}

        workers = num_workers or min(self.num_envs, mp.cpu_count() or 8)
        chunks = np.array_split(np.arange(n), workers)
        ctx = mp.get_context(context)
        self._remotes = []
        self._procs = []
        for chunk in chunks:
            if len(chunk) == 0:
                continue
            parent, child = ctx.Pipe()
            proc = ctx.Process(
                target=_worker_loop,
                args=(
                    child,
                    [env_fns[i] for i in chunk],
                    int(chunk[0]),
                    shm_names,
                    shapes,
                    dtypes,
                ),
                daemon=True,
            )
            proc.start()
            child.close()
            self._remotes.append(parent)
            self._procs.append(proc)
        self._closed = False

    # -- VectorEnv API ------------------------------------------------------
    def reset(self, *, seed: Optional[int] = None) -> np.ndarray:
        for r in self._remotes:
            r.send(("reset", seed))
        self._sync()
        return self._arrs["obs"].copy()

    def step(self, actions: np.ndarray):
        a = np.asarray(actions)
        self._arrs["act"][:] = a.reshape(self._arrs["act"].shape)
        for r in self._remotes:
            r.send(("step", None))
        self._sync()
        return (
            self._arrs["obs"].copy(),
            self._arrs["rew"].copy(),
            self._arrs["term"].copy(),
            self._arrs["trunc"].copy(),
            self._arrs["final"].copy(),
        )

    def _sync(self) -> None:
        for r in self._remotes:
            status, _ = r.recv()
            if status != "ok":  # pragma: no cover
                raise RuntimeError("env worker failed")

    def close(self) -> None:
        if self._closed:
            return
        self._closed = True
        for r in self._remotes:
            try:
                r.send(("close", None))
                r.recv()
            except (BrokenPipeError, EOFError):  # pragma: no cover
                pass
        for p in self._procs:
            p.join(timeout=5)
        for shm in self._shms.values():
            shm.close()
            shm.unlink()

    def __del__(self):  # pragma: no cover
        try:
            self.close()
        except Exception:
            pass
