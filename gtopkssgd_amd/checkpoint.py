"""Checkpoint save/load.

Reference parity: per-epoch
``<prefix>/<run-tag>/model-<epoch>.pth`` state_dict saves from rank 0.
Documented extension: the optimizer state (inner optimizer, flat
momentum — identical on every rank) is saved alongside, and the
error-feedback residuals — which are RANK-LOCAL state — are saved as
one sidecar file per rank (``ef-<epoch>-rank<r>.pth``) so EF mass
survives resume at any world size. On resume each rank restores its own
sidecar; a rank without one (e.g. the world size changed) starts with a
zero residual instead of silently adopting rank 0's.
"""
from __future__ import annotations

import os
import re
from typing import Optional

import torch

from .settings import logger


def _ef_sidecar(d: str, epoch: int, rank: int) -> str:
    return os.path.join(d, f"ef-{epoch}-rank{rank}.pth")


def save_checkpoint(prefix: str, run_tag: str, epoch: int, model,
                    optimizer=None, rank: int = 0) -> str:
    """Per-epoch checkpoint.  Call from EVERY rank: rank 0 writes the
    model+optimizer file; every rank writes its EF-residual sidecar.
    Returns the model file path."""
    d = os.path.join(prefix, run_tag)
    os.makedirs(d, exist_ok=True)
    path = os.path.join(d, f"model-{epoch}.pth")
    if rank == 0:
        payload = {"epoch": epoch, "state_dict": model.state_dict()}
        if optimizer is not None:
            payload["optimizer"] = optimizer.state_dict()
        torch.save(payload, path)
        logger.info("checkpoint saved: %s", path)
    comp = getattr(optimizer, "compression", None)
    if comp is not None and getattr(comp, "residuals", None):
        torch.save({"compression": comp.state_dict()},
                   _ef_sidecar(d, epoch, rank))
    return path


def load_checkpoint(path: str, model, optimizer=None,
                    map_location="cpu", rank: int = 0) -> int:
    """Load a checkpoint; returns the stored epoch.

    The model/optimizer payload is rank-0's file; the EF residual is
    then restored rank-locally from this rank's sidecar (zeroed when
    the sidecar is missing, with a warning — never rank 0's copy)."""
    payload = torch.load(path, map_location=map_location,
                         weights_only=False)
    model.load_state_dict(payload["state_dict"])
    if optimizer is not None and "optimizer" in payload:
        optimizer.load_state_dict(payload["optimizer"])
    epoch = payload.get("epoch", 0)
    comp = getattr(optimizer, "compression", None)
    if comp is not None and hasattr(comp, "residuals"):
        side = _ef_sidecar(os.path.dirname(path), epoch, rank)
        if rank != 0:
            # drop rank-0's residual that came in via the main payload
            for v in comp.residuals.values():
                v.zero_()
        if os.path.exists(side):
            comp.load_state_dict(
                torch.load(side, map_location=map_location,
                           weights_only=False)["compression"])
        elif rank != 0 and comp.residuals:
            logger.warning(
                "no EF sidecar for rank %d at epoch %d — residual "
                "restarted at zero (world size changed?)", rank, epoch)
    return epoch


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
