"""Checkpoint save/load.

Reference parity: per-epoch
``<prefix>/<run-tag>/model-<epoch>.pth`` state_dict saves from rank 0.
Documented extension: the optimizer + compressor (error-feedback
residual, flat momentum) state is saved alongside so training — and the
EF residual mass — survives resume, which the reference lacked.
"""
from __future__ import annotations

import os
import re
from typing import Optional

import torch

from .settings import logger


def save_checkpoint(prefix: str, run_tag: str, epoch: int, model,
                    optimizer=None) -> str:
    """rank-0 per-epoch checkpoint; returns the path written."""
    d = os.path.join(prefix, run_tag)
    os.makedirs(d, exist_ok=True)
    path = os.path.join(d, f"model-{epoch}.pth")
    payload = {"epoch": epoch, "state_dict": model.state_dict()}
    if optimizer is not None:
        payload["optimizer"] = optimizer.state_dict()
    torch.save(payload, path)
    logger.info("checkpoint saved: %s", path)
    return path


def load_checkpoint(path: str, model, optimizer=None,
                    map_location="cpu") -> int:
    """Load a checkpoint; returns the stored epoch."""
    payload = torch.load(path, map_location=map_location,
                         weights_only=False)
    model.load_state_dict(payload["state_dict"])
    if optimizer is not None and "optimizer" in payload:
        optimizer.load_state_dict(payload["optimizer"])
    return payload.get("epoch", 0)


def latest_checkpoint(prefix: str, run_tag: str) -> Optional[str]:
    d = os.path.join(prefix, run_tag)
    if not os.path.isdir(d):
        return None
    best, best_ep = None, -1
    for name in os.listdir(d):
        m = re.fullmatch(r"model-(\d+)\.pth", name)
        if m and int(m.group(1)) > best_ep:
            best_ep = int(m.group(1))
            best = os.path.join(d, name)
    return best
