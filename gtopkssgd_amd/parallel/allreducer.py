"""Sparse collectives: the gTop-k recursive-halving merge, the Top-k
allgather baseline, and the dense fallback.

Reference parity: `upstream:allreducer.py` (SURVEY.md §2.2 C1–C6, §2.3,
§3.3).  The reference stages gradients to CPU numpy and speaks raw MPI;
here every buffer is GPU-resident and the collectives are RCCL
point-to-point send/recv + broadcast over xGMI (``torch.distributed``
with the "nccl" backend IS RCCL on ROCm).  On CPU the same code runs
over gloo — that path is the plumbing/oracle configuration
(BASELINE.json config 1).

Merge schedule (recursive halving + broadcast):
in round r the upper half of each active pair sends its k-sparse set to
the lower half and goes idle; the receiver merges (scatter-add both
sets into a dense scratch, gather-and-zero the ≤2k candidate positions,
re-select top-k).  After log2(P) rounds rank 0 holds the global top-k of
the P-way sum, which is broadcast.  Message size stays O(k) per round;
total cost O(k·log P).

Wire format (K8): ONE int32 buffer of length 2k per message —
``buf[:k]`` = indices (int32), ``buf[k:]`` = fp32 values bitcast to
int32.  The select kernels write directly into views of this buffer, so
packing costs nothing.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist

from .. import ops


def _pack_views(buf: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """(idx int32 view, val fp32 view) of a 2k int32 wire buffer."""
    return buf[:k], buf[k:2 * k].view(torch.float32)


def gtopk_sparse_allreduce(
    values: torch.Tensor,
    indexes: torch.Tensor,
    dense_scratch: torch.Tensor,
    k: int,
    send_buf: Optional[torch.Tensor] = None,
    recv_buf: Optional[torch.Tensor] = None,
    group: Optional[dist.ProcessGroup] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Global top-k of the element-wise SUM of all ranks' sparse sets.

    Returns (values, indexes) — identical on every rank.  The dense
    scratch (length n, fp32, zero on entry) is returned zeroed.
    """
    device = values.device
    if send_buf is None:
        send_buf = torch.empty(2 * k, dtype=torch.int32, device=device)
    if recv_buf is None:
        recv_buf = torch.empty(2 * k, dtype=torch.int32, device=device)

    sidx, sval = _pack_views(send_buf, k)
    sidx.copy_(indexes)
    sval.copy_(values)

    if not (dist.is_available() and dist.is_initialized()):
        return sval, sidx
    world = dist.get_world_size(group)
    if world == 1:
        return sval, sidx
    rank_ = dist.get_rank(group)

    ridx, rval = _pack_views(recv_buf, k)
    group_size = 1
    alive = True
    while group_size < world:
        if alive:
            if rank_ % (2 * group_size) < group_size:
                peer = rank_ + group_size
                if peer < world:
                    dist.recv(recv_buf, src=peer, group=group)
                    # merge: union-sum then re-select k of ≤2k candidates
                    ops.scatter_add_(dense_scratch, sidx, sval)
                    ops.scatter_add_(dense_scratch, ridx, rval)
                    cand_idx = torch.cat([sidx, ridx])
                    cand_val = ops.gather_zero(dense_scratch, cand_idx)
                    mval, midx = ops.topk_among(cand_val, cand_idx, k)
                    sidx.copy_(midx)
                    sval.copy_(mval)
            else:
                peer = rank_ - group_size
                dist.send(send_buf, dst=peer, group=group)
                alive = False
        group_size *= 2

    dist.broadcast(send_buf, src=dist.get_global_rank(group, 0) if group
                   else 0, group=group)
    return sval, sidx


def topk_sparse_allreduce(
    values: torch.Tensor,
    indexes: torch.Tensor,
    dense_scratch: torch.Tensor,
    k: int,
    gather_bufs: Optional[list] = None,
    group: Optional[dist.ProcessGroup] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-k S-SGD baseline (C3): allgather every rank's local top-k and
    sum the union INTO ``dense_scratch`` (no re-selection — the update
    is the ≤ P·k-sparse union sum).

    Returns (dense_scratch, all_indexes) where all_indexes is the
    concatenation of every rank's index set (with duplicates).
    """
    device = values.device
    k = int(k)
    send = torch.empty(2 * k, dtype=torch.int32, device=device)
    sidx, sval = _pack_views(send, k)
    sidx.copy_(indexes)
    sval.copy_(values)

    if not (dist.is_available() and dist.is_initialized()) or \
            dist.get_world_size(group) == 1:
        ops.scatter_add_(dense_scratch, sidx, sval)
        return dense_scratch, sidx.clone()

    world = dist.get_world_size(group)
    if gather_bufs is None:
        gather_bufs = [torch.empty(2 * k, dtype=torch.int32, device=device)
                       for _ in range(world)]
    dist.all_gather(gather_bufs, send, group=group)
    idx_list = []
    for buf in gather_bufs:
        gidx, gval = _pack_views(buf, k)
        ops.scatter_add_(dense_scratch, gidx, gval)
        idx_list.append(gidx)
    return dense_scratch, torch.cat(idx_list)


def dense_allreduce(flat_grad: torch.Tensor,
                    group: Optional[dist.ProcessGroup] = None,
                    average: bool = True) -> torch.Tensor:
    """C4: dense fallback — ncclAllReduce(sum) over the flat gradient."""
    if dist.is_available() and dist.is_initialized() and \
            dist.get_world_size(group) > 1:
        dist.all_reduce(flat_grad, op=dist.ReduceOp.SUM, group=group)
        if average:
            flat_grad.div_(dist.get_world_size(group))
    return flat_grad


class GradReducer:
    """Owns the communication state for one flat gradient: the dense
    merge scratch, the wire buffers, and the sparse-mode dispatch.

    MI355X-native note: unlike the reference's background comm thread +
    CPU staging (`upstream:allreducer.py` class AllReducer), everything
    here is stream-ordered GPU work — a Python thread would serialise
    on the GIL and add D2H copies for nothing on xGMI.
    """

    def __init__(self, numel: int, k: int, device: torch.device,
                 mode: str = "gtopk",
                 group: Optional[dist.ProcessGroup] = None):
        assert mode in ("gtopk", "topk", "dense"), mode
        self.numel = numel
        self.k = k
        self.mode = mode
        self.group = group
        self.device = device
        self.dense_scratch = torch.zeros(numel, dtype=torch.float32,
                                         device=device)
        self.send_buf = torch.empty(2 * k, dtype=torch.int32, device=device)
        self.recv_buf = torch.empty(2 * k, dtype=torch.int32, device=device)
        self._gather_bufs: Optional[list] = None

    def world_size(self) -> int:
        if dist.is_available() and dist.is_initialized():
            return dist.get_world_size(self.group)
        return 1

    def gtopk(self, values, indexes):
        return gtopk_sparse_allreduce(
            values, indexes, self.dense_scratch, self.k,
            self.send_buf, self.recv_buf, self.group)

    def topk(self, values, indexes):
        if self._gather_bufs is None and self.world_size() > 1:
            self._gather_bufs = [
                torch.empty(2 * self.k, dtype=torch.int32, device=self.device)
                for _ in range(self.world_size())]
        return topk_sparse_allreduce(
            values, indexes, self.dense_scratch, self.k,
            self._gather_bufs, self.group)

    def dense(self, flat_grad):
        return dense_allreduce(flat_grad, self.group)
