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

    group_size = 1
    alive = True
    while group_size < world:
        if alive:
            if rank_ % (2 * group_size) < group_size:
                peer = rank_ + group_size
                if peer < world:
                    dist.recv(recv_buf, src=peer, group=group)
                    # fused round: union-sum + gather/zero + re-select,
                    # merged set written back into send_buf (wire-ready)
                    ops.merge_round_(dense_scratch, send_buf, recv_buf, k)
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


def _self_benchmark():  # pragma: no cover - exercised on GPU boxes
    """Collective self-benchmark (reference parity: the ``__main__``
    block of `upstream:allreducer.py` benchmarks gtopk_sparse_allreduce
    on random vectors under mpirun).

        python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 -m gtopkssgd_amd.parallel.allreducer \
            --n 25557032 --density 0.001 --iters 50
    """
    import argparse
    import json
    import os
    import time

    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=25_557_032)  # ResNet-50
    p.add_argument("--density", type=float, default=0.001)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--mode", default="gtopk",
                   choices=["gtopk", "topk", "dense", "torchtopk"])
    p.add_argument("--device", default=None)
    p.add_argument("--simulate-rounds", type=int, default=0,
                   help="world=1: run R fused merge-round bookkeeping "
                        "passes per iteration (per-round kernel cost "
                        "without comm — the xGMI p2p time adds on top)")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_gpu = torch.cuda.is_available() and args.device != "cpu"
    if world > 1:
        dist.init_process_group("nccl" if use_gpu else "gloo")
        if use_gpu:
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    device = torch.device("cuda" if use_gpu else "cpu")

    from ..compression import TopKCompressor
    n = args.n
    k = max(1, int(args.density * n))
    comp = TopKCompressor(args.density)
    comp.residuals["b"] = torch.zeros(n, device=device)
    reducer = GradReducer(
        n, k, device,
        mode=args.mode if args.mode != "torchtopk" else "gtopk")
    g = torch.Generator(device="cpu").manual_seed(rank)
    grad = torch.randn(n, generator=g).to(device)

    if args.simulate_rounds:
        # a synthetic peer message: a valid packed sparse set from a
        # different random vector
        gp = torch.Generator(device="cpu").manual_seed(rank + 777)
        peer = torch.randn(n, generator=gp)
        pidx = torch.topk(peer.abs(), k, sorted=False).indices
        reducer.recv_buf[:k].copy_(pidx.to(torch.int32).to(device))
        reducer.recv_buf[k:].view(torch.float32).copy_(
            peer[pidx].to(device))

    def one_iter():
        if args.mode == "dense":
            reducer.dense(grad.clone())
            return
        if args.mode == "torchtopk":
            # the reference's select path: torch.topk over the flat vec
            res = comp.residuals["b"]
            res.add_(grad)
            _, idx = torch.topk(res.abs(), k, sorted=False)
            vals = res[idx]
            res[idx] = 0
            return
        vals, idx = comp.compress(grad, "b")
        if args.mode == "gtopk":
            if args.simulate_rounds and world == 1:
                sb = reducer.send_buf
                sb[:k].copy_(idx)
                sb[k:].view(torch.float32).copy_(vals)
                for _ in range(args.simulate_rounds):
                    ops.merge_round_(reducer.dense_scratch, sb,
                                     reducer.recv_buf, k)
                comp.add_residuals(sb[:k], "b")
                return
            gval, gidx = reducer.gtopk(vals, idx)
            comp.add_residuals(gidx, "b")
        else:
            reducer.topk(vals, idx)
            reducer.dense_scratch.zero_()

    for _ in range(args.warmup):
        one_iter()
    if use_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        one_iter()
    if use_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    dt = (time.perf_counter() - t0) / args.iters
    if rank == 0:
        print(json.dumps({
            "collective": args.mode, "n": n, "k": k, "world": world,
            "ms_per_iter": round(dt * 1000, 3),
            "simulate_rounds": args.simulate_rounds,
            "device": str(device)}), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":  # pragma: no cover
    _self_benchmark()


# Reference-familiar alias (`upstream:allreducer.py` class AllReducer).
AllReducer = GradReducer
