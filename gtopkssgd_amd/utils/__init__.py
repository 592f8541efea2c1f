"""Misc helpers (reference parity: `upstream:utils.py`)."""
from __future__ import annotations

import json
import os
import random
import string
import time
from typing import Dict


def gen_random_id(n: int = 8) -> str:
    return "".join(random.choice(string.ascii_lowercase + string.digits)
                   for _ in range(n))


class Timer:
    """Accumulating phase timer (io/fwd/bwd/update spans)."""

    def __init__(self):
        self.totals: Dict[str, float] = {}
        self._t0: Dict[str, float] = {}

    def start(self, phase: str):
        self._t0[phase] = time.perf_counter()

    def stop(self, phase: str):
        self.totals[phase] = self.totals.get(phase, 0.0) + \
            (time.perf_counter() - self._t0[phase])

    def reset(self):
        self.totals.clear()

    def summary(self) -> str:
        return " ".join(f"{k}={v*1000:.1f}ms" for k, v in
                        sorted(self.totals.items()))


class JsonlLogger:
    """Append-only JSONL metrics writer."""

    def __init__(self, path: str):
        self.path = path
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)

    def log(self, **record):
        record.setdefault("ts", time.time())
        with open(self.path, "a") as f:
            f.write(json.dumps(record) + "\n")


def force_insert(d: dict, key, value):
    """dict force-insert helper (reference `utils.py` parity)."""
    d[key] = value
    return d
