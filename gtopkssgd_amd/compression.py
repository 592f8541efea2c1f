"""Gradient compressors with error-feedback residual state.

Reference parity: `upstream:compression.py` —
`NoneCompressor`, `TopKCompressor` with a per-name residual dict
(error feedback), `add_residuals` re-insertion of globally-rejected
values, and the `compressors` registry.

MI355X-native design: the compressor is device-agnostic at this layer;
the hot ops (`residual += grad`, magnitude top-k select with residual
zeroing, rejected-value re-insertion) dispatch through
:mod:`gtopkssgd_amd.ops`, which runs hand-written HIP kernels on gfx950
and a plain-PyTorch path on CPU (used by gloo tests as well as the
numerics oracle).
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from . import ops
from .settings import MAX_INT32_NUMEL


class NoneCompressor:
    """Identity compressor — dense path (density = 1)."""

    name = "none"

    def __init__(self, density: float = 1.0):
        self.density = 1.0
        self.residuals: Dict[str, torch.Tensor] = {}

    def compress(self, tensor: torch.Tensor, name: str = "flat",
                 ratio: Optional[float] = None) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        return tensor, None

    def add_residuals(self, included_indexes, name: str = "flat"):
        return

    def state_dict(self):
        return {"name": self.name, "density": self.density, "residuals": {}}

    def load_state_dict(self, state):
        return


class TopKCompressor:
    """Magnitude top-k over the flattened gradient with error feedback.

    Per-name residual tensors persist across iterations:

        acc      = grad + residual          (error feedback)
        (V, I)   = topk_by_magnitude(acc, k),  k = max(1, int(density * n))
        residual = acc;  residual[I] = 0    (unselected mass carries over)

    After the global merge, :meth:`add_residuals` re-inserts the values of
    locally-selected-but-globally-rejected coordinates back into the
    residual.
    """

    name = "topk"

    def __init__(self, density: float = 0.001):
        assert 0.0 < density <= 1.0, density
        self.density = float(density)
        self.residuals: Dict[str, torch.Tensor] = {}
        # Stash of the last local selection per name, needed by
        # add_residuals after the merge.
        self._last: Dict[str, Tuple[torch.Tensor, torch.Tensor]] = {}

    def k_for(self, numel: int, ratio: Optional[float] = None) -> int:
        ratio = self.density if ratio is None else ratio
        return max(1, int(ratio * numel))

    def compress(self, tensor: torch.Tensor, name: str = "flat",
                 ratio: Optional[float] = None) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (values fp32, indexes int32) of the local top-k of
        `tensor + residual[name]`, updating the residual in place.

        `tensor` must be a flat fp32 tensor; it is left untouched (the
        accumulated gradient lives in the residual buffer).
        """
        assert tensor.dim() == 1, "compress expects a flat gradient"
        n = tensor.numel()
        assert n <= MAX_INT32_NUMEL, "int32 index space exceeded"
        k = self.k_for(n, ratio)
        residual = self.residuals.get(name)
        if residual is None:
            residual = torch.zeros_like(tensor)
            self.residuals[name] = residual
        values, indexes = ops.ef_compress(residual, tensor, k)
        self._last[name] = (values, indexes)
        return values, indexes

    def ensure_residual(self, numel: int, device, name: str = "flat"
                        ) -> torch.Tensor:
        """Pre-allocate the residual (the backward-overlap path
        accumulates into it from hooks before compress is ever called)."""
        residual = self.residuals.get(name)
        if residual is None or residual.numel() != numel:
            residual = torch.zeros(numel, dtype=torch.float32, device=device)
            self.residuals[name] = residual
        return residual

    def compress_finish(self, name: str = "flat",
                        ratio: Optional[float] = None,
                        out_val: Optional[torch.Tensor] = None,
                        out_idx: Optional[torch.Tensor] = None
                        ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Select-only compress: the EF accumulation (and on GPU the
        level-1 histogram) already ran via ops.ef_accum_hist_range over
        every range of the flat gradient (backward-overlap path)."""
        residual = self.residuals[name]
        k = self.k_for(residual.numel(), ratio)
        values, indexes = ops.ef_compress_finish(residual, k, out_val,
                                                 out_idx)
        self._last[name] = (values, indexes)
        return values, indexes

    def add_residuals(self, included_indexes: torch.Tensor, name: str = "flat"):
        """Re-add locally-selected values whose index is NOT in
        `included_indexes` (the global top-k index set) to the residual."""
        lvals, lidx = self._last[name]
        residual = self.residuals[name]
        ops.readd_rejected_(residual, lidx, lvals, included_indexes)

    def state_dict(self):
        return {
            "name": self.name,
            "density": self.density,
            "residuals": {k: v for k, v in self.residuals.items()},
        }

    def load_state_dict(self, state):
        self.density = state.get("density", self.density)
        for key, value in state.get("residuals", {}).items():
            if key in self.residuals:
                self.residuals[key].copy_(value)
            else:
                self.residuals[key] = value.clone()


class GTopKCompressor(TopKCompressor):
    """Alias class: local selection identical to TopKCompressor; the
    'g' (global) part is the merge collective's job (allreducer)."""

    name = "gtopk"


compressors = {
    "none": NoneCompressor,
    None: NoneCompressor,
    "topk": TopKCompressor,
    "gtopk": GTopKCompressor,
}
