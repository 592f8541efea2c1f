"""Hot-path ops: HIP/CDNA4 kernels on GPU, plain PyTorch on CPU.

Each public function dispatches on the device of its first tensor:

* **CUDA (= ROCm/gfx950)**: calls into the in-tree HIP extension
  ``gtopkssgd_amd._C`` (built by ``setup.py build_ext --inplace`` with
  ``PYTORCH_ROCM_ARCH=gfx950``).  If the extension is missing the call
  RAISES — there is no silent eager fallback on GPU.
* **CPU**: a plain-PyTorch implementation, used by the gloo test path
  and as the numerics reference.

Kernel inventory (SURVEY.md §2.2 K1–K8):

==================  =====================================================
``ef_compress``     K1+K2+K3 fused: ``residual += grad``; bucketed
                    magnitude top-k select over the flat vector
                    (histogram of |v| bit-keys + threshold + compaction);
                    selected positions zeroed in the residual.
``scatter_add_``    K4: ``dense[idx] += val`` (fp32 atomics on GPU).
``gather_zero``     K4 epilogue: gather candidate values and zero the
                    dense buffer at those positions; duplicate indices
                    gather the value once (others read 0) so the dense
                    buffer is self-cleaning and candidates are deduped.
``topk_among``      K5: top-k by magnitude among m candidates
                    (value,index) pairs, m ≈ 2k.
``readd_rejected_`` K6: residual[i] += v for locally-selected (i,v) whose
                    index is absent from the global top-k index set.
``densify_``        scatter ``dense[idx] = val * scale``.
``sgd_step_``       K7: fused flat SGD (momentum, weight decay, optional
                    nesterov) reading a dense grad buffer and zeroing it
                    as it goes (self-cleaning for the next iteration).
==================  =====================================================
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

_EXT = None
_EXT_ERR: Optional[BaseException] = None

HIST_BINS = 8192  # |fp32| bit-key >> 18 → [exp:8][mantissa-high:5] bits


def _ext():
    """Import the in-tree HIP extension, raising loudly if unavailable."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    if _EXT_ERR is not None:
        raise RuntimeError(
            "gtopkssgd_amd._C HIP extension failed to import earlier"
        ) from _EXT_ERR
    try:
        from gtopkssgd_amd import _C  # type: ignore

        _EXT = _C
        return _EXT
    except ImportError as exc:  # pragma: no cover - GPU-only path
        _EXT_ERR = exc
        raise RuntimeError(
            "gtopkssgd_amd._C HIP extension is not built. Build it in-tree "
            "with `PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext "
            "--inplace`. Eager fallback on GPU is disabled by design."
        ) from exc


def ext_available() -> bool:
    try:
        _ext()
        return True
    except RuntimeError:
        return False


class _Scratch:
    """Per-device scratch tensors for the GPU select kernels."""

    def __init__(self, device: torch.device):
        self.hist = torch.empty(2 * HIST_BINS, dtype=torch.int32,
                                device=device)
        # info: [threshold_bin, count_above, k, pad]
        self.info = torch.empty(4, dtype=torch.int32, device=device)
        self.counters = torch.empty(2, dtype=torch.int32, device=device)
        self.marks: Dict[int, torch.Tensor] = {}
        self.cands: Dict[int, Tuple[torch.Tensor, torch.Tensor]] = {}

    def mark_buffer(self, n: int, device: torch.device) -> torch.Tensor:
        buf = self.marks.get(n)
        if buf is None:
            buf = torch.zeros(n, dtype=torch.uint8, device=device)
            self.marks[n] = buf
        return buf

    def cand_buffers(self, k: int, device: torch.device
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
        bufs = self.cands.get(k)
        if bufs is None:
            bufs = (torch.empty(2 * k, dtype=torch.float32, device=device),
                    torch.empty(2 * k, dtype=torch.int32, device=device))
            self.cands[k] = bufs
        return bufs


_scratch: Dict[int, _Scratch] = {}


def _get_scratch(device: torch.device) -> _Scratch:
    key = device.index if device.index is not None else -1
    s = _scratch.get(key)
    if s is None:
        s = _Scratch(device)
        _scratch[key] = s
    return s


# ---------------------------------------------------------------------------
# ef_compress — K1+K2+K3
# ---------------------------------------------------------------------------

def ef_compress(residual: torch.Tensor, grad: torch.Tensor, k: int
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual += grad; (val, idx int32) = topk_by_magnitude(residual, k);
    residual[idx] = 0.  Returns (values fp32, indexes int32)."""
    n = residual.numel()
    assert grad.numel() == n and 0 < k <= n
    if residual.is_cuda:
        s = _get_scratch(residual.device)
        out_val = torch.empty(k, dtype=torch.float32, device=residual.device)
        out_idx = torch.empty(k, dtype=torch.int32, device=residual.device)
        _ext().ef_compress(residual, grad, k, s.hist, s.info, s.counters,
                           out_val, out_idx)
        return out_val, out_idx
    residual.add_(grad)
    if k == n:
        values = residual.clone()
        indexes = torch.arange(n, dtype=torch.int32)
        residual.zero_()
        return values, indexes
    _, idx = torch.topk(residual.abs(), k, sorted=False)
    values = residual[idx].clone()
    residual[idx] = 0.0
    return values, idx.to(torch.int32)


# --- backward-overlap variants: the EF accumulate + level-1 histogram
# is additive over disjoint ranges, so the optimizer launches it
# per-bucket from post-accumulate-grad hooks on a side HIP stream
# during backward; ef_compress_finish runs the remaining threshold
# scan + refine + compact at synchronize() time. ---

def ef_hist_zero(device: torch.device):
    """Zero the histogram scratch before the first bucket of an
    iteration (GPU overlap path only)."""
    s = _get_scratch(device)
    _ext().ef_hist_zero(s.hist)


def ef_accum_hist_range(residual: torch.Tensor, grad: torch.Tensor,
                        start: int, end: int):
    """residual[start:end] += grad[start:end] (+ level-1 histogram on
    GPU).  Disjoint ranges covering [0, n) compose to ef_compress's
    first pass."""
    if residual.is_cuda:
        s = _get_scratch(residual.device)
        _ext().ef_accum_hist_range(residual, grad, int(start), int(end),
                                   s.hist)
        return
    residual[start:end] += grad[start:end]


def ef_compress_finish(residual: torch.Tensor, k: int,
                       out_val: Optional[torch.Tensor] = None,
                       out_idx: Optional[torch.Tensor] = None
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Select after accumulation: (val, idx) = topk_by_magnitude
    (residual, k); residual[idx] = 0.  On GPU the histogram from the
    per-range passes must cover the whole vector."""
    n = residual.numel()
    assert 0 < k <= n
    if residual.is_cuda:
        s = _get_scratch(residual.device)
        if out_val is None:
            out_val = torch.empty(k, dtype=torch.float32,
                                  device=residual.device)
        if out_idx is None:
            out_idx = torch.empty(k, dtype=torch.int32,
                                  device=residual.device)
        _ext().ef_compress_finish(residual, k, s.hist, s.info, s.counters,
                                  out_val, out_idx)
        return out_val, out_idx
    if k == n:
        values = residual.clone()
        indexes = torch.arange(n, dtype=torch.int32)
        residual.zero_()
        return values, indexes
    _, idx = torch.topk(residual.abs(), k, sorted=False)
    values = residual[idx].clone()
    residual[idx] = 0.0
    return values, idx.to(torch.int32)


# ---------------------------------------------------------------------------
# merge primitives — K4, K5
# ---------------------------------------------------------------------------

def scatter_add_(dense: torch.Tensor, idx: torch.Tensor, val: torch.Tensor):
    """dense[idx] += val (duplicate indices accumulate)."""
    if dense.is_cuda:
        _ext().scatter_add(dense, idx, val)
        return
    dense.index_add_(0, idx.long(), val)


def gather_zero(dense: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """val = dense[idx]; dense[idx] = 0.  For duplicate indices exactly one
    occurrence receives the stored value, the rest read 0 (dedup)."""
    if dense.is_cuda:
        out = torch.empty(idx.numel(), dtype=torch.float32,
                          device=dense.device)
        _ext().gather_zero(dense, idx, out)
        return out
    lidx = idx.long()
    vals = dense[lidx].clone()
    sorted_idx, order = torch.sort(lidx, stable=True)
    dup = sorted_idx[1:] == sorted_idx[:-1]
    vals[order[1:][dup]] = 0.0
    dense[lidx] = 0.0
    return vals


def topk_among(values: torch.Tensor, indexes: torch.Tensor, k: int
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-k by |value| among m candidates; returns (values, indexes)."""
    m = values.numel()
    assert indexes.numel() == m and 0 < k <= m
    if values.is_cuda:
        s = _get_scratch(values.device)
        out_val = torch.empty(k, dtype=torch.float32, device=values.device)
        out_idx = torch.empty(k, dtype=torch.int32, device=values.device)
        _ext().topk_among(values, indexes, k, s.hist, s.info, s.counters,
                          out_val, out_idx)
        return out_val, out_idx
    if k == m:
        return values.clone(), indexes.clone()
    _, pos = torch.topk(values.abs(), k, sorted=False)
    return values[pos].clone(), indexes[pos].clone()


def merge_round_(dense: torch.Tensor, send_buf: torch.Tensor,
                 recv_buf: torch.Tensor, k: int
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """One fused gtopk merge round (C1 bookkeeping): union-sum both
    packed sparse sets into ``dense``, gather-and-zero the ≤2k
    candidates, re-select top-k.  The merged set is written back into
    ``send_buf`` (wire-ready for the next round); returns (val, idx)
    views of it.  ``dense`` ends zeroed again.  On GPU this is ONE ext
    call (7 kernel launches) instead of ~12 — the messages are ~200 KB,
    latency-bound, so launch count per round is what matters."""
    sidx = send_buf[:k]
    sval = send_buf[k:2 * k].view(torch.float32)
    if dense.is_cuda:
        s = _get_scratch(dense.device)
        cand_val, cand_idx = s.cand_buffers(k, dense.device)
        _ext().merge_round(dense, send_buf, recv_buf, k, s.hist, s.info,
                           s.counters, cand_val, cand_idx, sval, sidx)
        return sval, sidx
    ridx = recv_buf[:k]
    rval = recv_buf[k:2 * k].view(torch.float32)
    scatter_add_(dense, sidx, sval)
    scatter_add_(dense, ridx, rval)
    cand_idx = torch.cat([sidx, ridx])
    cand_val = gather_zero(dense, cand_idx)
    mval, midx = topk_among(cand_val, cand_idx, k)
    sidx.copy_(midx)
    sval.copy_(mval)
    return sval, sidx


# ---------------------------------------------------------------------------
# residual re-insertion — K6
# ---------------------------------------------------------------------------

def readd_rejected_(residual: torch.Tensor, lidx: torch.Tensor,
                    lval: torch.Tensor, gidx: torch.Tensor):
    """residual[lidx[j]] += lval[j] for every j with lidx[j] ∉ gidx."""
    if residual.is_cuda:
        s = _get_scratch(residual.device)
        marks = s.mark_buffer(residual.numel(), residual.device)
        _ext().readd_rejected(residual, lidx, lval, gidx, marks)
        return
    rejected = ~torch.isin(lidx, gidx)
    ridx = lidx[rejected].long()
    residual.index_add_(0, ridx, lval[rejected])


# ---------------------------------------------------------------------------
# update — K7
# ---------------------------------------------------------------------------

def densify_(dense: torch.Tensor, idx: torch.Tensor, val: torch.Tensor,
             scale: float = 1.0):
    """dense[idx] = val * scale (dense assumed zero elsewhere).

    Zero values are skipped: after `gather_zero` dedup, a duplicate
    index appears as (i, 0.0) and must not clobber the real (i, v)."""
    if dense.is_cuda:
        _ext().densify(dense, idx, val, float(scale))
        return
    mask = val != 0
    dense[idx[mask].long()] = val[mask] * scale


def sgd_step_(param: torch.Tensor, momentum_buf: torch.Tensor,
              dense_grad: torch.Tensor, lr: float, momentum: float,
              weight_decay: float, dampening: float = 0.0,
              nesterov: bool = False, zero_grad: bool = True,
              grad_scale: float = 1.0):
    """Fused flat SGD matching torch.optim.SGD semantics (momentum buffer
    pre-initialised to zeros; identical to torch for dampening == 0).

        g   = grad * grad_scale + wd * p
        buf = mu * buf + (1 - dampening) * g
        p  -= lr * (g + mu * buf)   if nesterov else   lr * buf

    When ``zero_grad`` the dense grad buffer is zeroed in the same pass
    (self-cleaning scratch for the next iteration).
    """
    if param.is_cuda:
        _ext().sgd_step(param, momentum_buf, dense_grad, float(lr),
                        float(momentum), float(weight_decay),
                        float(dampening), bool(nesterov), bool(zero_grad),
                        float(grad_scale))
        return
    g = dense_grad if grad_scale == 1.0 else dense_grad * grad_scale
    if weight_decay != 0.0:
        g = g.add(param, alpha=weight_decay)
    if momentum != 0.0:
        momentum_buf.mul_(momentum).add_(g, alpha=1.0 - dampening)
        if nesterov:
            g = g.add(momentum_buf, alpha=momentum)
        else:
            g = momentum_buf
    param.add_(g, alpha=-lr)
    if zero_grad:
        dense_grad.zero_()
