"""Structured per-round metrics: JSONL + phase timers.

The reference logs per-round loss/accuracy via prints; here every round
appends one JSON line (round, per-client losses, global eval, samples/sec,
per-phase times) so BASELINE.md numbers come straight from the log.
"""

from __future__ import annotations

import json
import time
from contextlib import contextmanager
from typing import Any, Dict, Optional

import torch


class PhaseTimer:
    """Wall-clock per-phase timer; cuda-synchronizing on GPU for honesty."""

    def __init__(self, device: torch.device) -> None:
        self.device = device
        self.times: Dict[str, float] = {}

    @contextmanager
    def phase(self, name: str):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.device.type == "cuda":
                torch.cuda.synchronize(self.device)
            self.times[name] = self.times.get(name, 0.0) + (
                time.perf_counter() - t0
            )

    def reset(self) -> None:
        self.times.clear()


class MetricsLogger:
    def __init__(self, path: Optional[str]) -> None:
        self.path = path
        self._fh = open(path, "a") if path else None

    def log(self, record: Dict[str, Any]) -> None:
        record = dict(record, ts=time.time())
        if self._fh:
            self._fh.write(json.dumps(record) + "\n")
            self._fh.flush()

    def close(self) -> None:
        if self._fh:
            self._fh.close()
            self._fh = None
