"""Metrics / observability.

API parity: reference src/rl_replicas/metrics_manager.py:11-42 —
`record_scalar(tag, scalar, total_steps=None, tensorboard=False)`
prints `"{tag}: {:<8.3g}"` to stdout and optionally logs to
TensorBoard under `<log_dir>/tensorboard`; `dump()` flushes,
`close()` closes.  Tag families (epoch, total_steps, sampling/*,
policy/*, value_function/*, q-function*/*, evaluation/*, time —
including the reference's historical `avarage` spellings) are emitted
by the algorithms unchanged so downstream tooling keyed on the
reference's tags keeps working.

Scalars marked `tensorboard=True` are persisted to BOTH a CSV file
(`<log_dir>/metrics.csv`: tag,step,value — read by the benchmark
converter) and a real `events.out.tfevents.*` file under
`<log_dir>/tensorboard` written by the native tfevents writer
(rl_replicas_amd/tfevents.py) — so downstream TensorBoard tooling
keyed on the reference's event files keeps working even though the
tensorboard package isn't installed here.  If tensorboard IS
importable, its SummaryWriter is used instead of the native writer.

MI355X addition: `record_phase_ms` aggregates per-phase HIP-event /
wall timings (sample / h2d / forward / backward / allreduce / optimizer)
for the phase breakdown the reference never had.
"""
from __future__ import annotations

import csv
import logging
import os
from typing import Dict, Optional

logger = logging.getLogger(__name__)

try:  # tensorboard is optional
    from torch.utils.tensorboard import SummaryWriter  # type: ignore

    _HAS_TB = True
except Exception:  # pragma: no cover
    SummaryWriter = None  # type: ignore
    _HAS_TB = False


class MetricsManager:
    def __init__(self, log_dir: str = ".", stdout: bool = True):
        self.log_dir = log_dir
        self.stdout = stdout
        os.makedirs(log_dir, exist_ok=True)
        self._csv_path = os.path.join(log_dir, "metrics.csv")
        self._csv_file = open(self._csv_path, "a", newline="")
        self._csv = csv.writer(self._csv_file)
        if self._csv_file.tell() == 0:
            self._csv.writerow(["tag", "step", "value"])
        tb_dir = os.path.join(log_dir, "tensorboard")
        if _HAS_TB:
            self.tensorboard_writer = SummaryWriter(tb_dir)
        else:
            from rl_replicas_amd.tfevents import EventFileWriter

            self.tensorboard_writer = EventFileWriter(tb_dir)
        self._phase_ms: Dict[str, float] = {}

    def record_scalar(
        self,
        tag: str,
        scalar: float,
        total_steps: Optional[int] = None,
        tensorboard: bool = False,
    ) -> None:
        if self.stdout:
            print("{}: {:<8.3g}".format(tag, scalar))
        if tensorboard:
            if total_steps is None:
                logger.warning("total_steps argument is required for tensorboard")
            self._csv.writerow([tag, total_steps, scalar])
            if self.tensorboard_writer is not None:
                self.tensorboard_writer.add_scalar(tag, scalar, total_steps)

    def record_phase_ms(self, phase: str, ms: float) -> None:
        """Accumulate per-phase timing; emitted/cleared by dump_phases()."""
        self._phase_ms[phase] = self._phase_ms.get(phase, 0.0) + ms

    def dump_phases(self, total_steps: Optional[int] = None) -> None:
        for phase, ms in sorted(self._phase_ms.items()):
            self.record_scalar(f"phase_ms/{phase}", ms, total_steps, tensorboard=True)
        self._phase_ms.clear()

    def dump(self) -> None:
        self._csv_file.flush()
        if self.tensorboard_writer is not None:
            self.tensorboard_writer.flush()

    def close(self) -> None:
        self._csv_file.close()
        if self.tensorboard_writer is not None:
            self.tensorboard_writer.close()
