"""Opt-in wall-clock section profiler.

Capability parity with the reference's ``Timing``
(common/timing_utils.py:17-48): accumulate per-named-section wall time and
report per task. Additionally understands GPU sections — when a section is
marked ``gpu=True`` the timer synchronizes the device before sampling, so
numbers reflect device time rather than launch time.
"""

import time
from typing import Dict

import torch


class Timing:
    def __init__(self, enabled: bool = True):
        self.enabled = enabled
        self._start: Dict[str, float] = {}
        self.acc: Dict[str, float] = {}

    def start_record_time(self, name: str, gpu: bool = False) -> None:
        if not self.enabled:
            return
        if gpu and torch.cuda.is_available():
            torch.cuda.synchronize()
        self._start[name] = time.monotonic()

    def end_record_time(self, name: str, gpu: bool = False) -> None:
        if not self.enabled or name not in self._start:
            return
        if gpu and torch.cuda.is_available():
            torch.cuda.synchronize()
        self.acc[name] = self.acc.get(name, 0.0) + (
            time.monotonic() - self._start.pop(name)
        )

    def report_timing(self, reset: bool = False) -> str:
        msg = ", ".join(f"{k}: {v * 1e3:.1f} ms" for k, v in sorted(self.acc.items()))
        if reset:
            self.acc.clear()
        return msg
