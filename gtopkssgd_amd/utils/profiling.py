"""GPU phase timing + stats.

The reference logs wall-clock spans for compress/comm phases from its
comm thread and optionally feeds a tensorboardX writer.  Here the
equivalent is HIP-event bracketed spans (stream-ordered, no host sync
until you read them) plus a JSONL stats sink.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import torch


class GpuPhaseTimer:
    """Event-pair phase timer.  `start(name)`/`stop(name)` enqueue HIP
    events on the current stream; `collect()` synchronises pending pairs
    and folds elapsed ms into running totals.  Near-zero overhead when
    disabled."""

    def __init__(self, enabled: bool = True):
        self.enabled = enabled and torch.cuda.is_available()
        self._pending: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []
        self._open: Dict[str, torch.cuda.Event] = {}
        self.totals_ms: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)

    def start(self, name: str):
        if not self.enabled:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        self._open[name] = ev

    def stop(self, name: str):
        if not self.enabled:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        self._pending.append((name, self._open.pop(name), ev))

    def collect(self) -> Dict[str, float]:
        """Synchronise and fold; returns mean ms per phase."""
        if self.enabled and self._pending:
            torch.cuda.synchronize()
            for name, a, b in self._pending:
                self.totals_ms[name] += a.elapsed_time(b)
                self.counts[name] += 1
            self._pending.clear()
        return {k: self.totals_ms[k] / max(1, self.counts[k])
                for k in self.totals_ms}

    def reset(self):
        self._pending.clear()
        self._open.clear()
        self.totals_ms.clear()
        self.counts.clear()

    def summary(self) -> str:
        means = self.collect()
        return " ".join(f"{k}={v:.3f}ms" for k, v in sorted(means.items()))


class StatsWriter:
    """Minimal SummaryWriter-compatible sink writing JSONL (the
    reference passes a tensorboardX writer into its AllReducer; this
    mirrors add_scalar without the tensorboard dependency)."""

    def __init__(self, path: Optional[str] = None):
        from . import JsonlLogger
        self._sink = JsonlLogger(path) if path else None

    def add_scalar(self, tag: str, value: float, global_step: int = 0):
        if self._sink:
            self._sink.log(tag=tag, value=float(value), step=global_step)

    def close(self):
        pass
