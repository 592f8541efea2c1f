"""gTop-k merge collective vs a dense oracle (gloo, multi-process CPU).

The oracle implements SURVEY.md §2.3's recursive-halving schedule on
dense vectors; the implementation must match it exactly (continuous
random values — no magnitude ties).
"""
import torch

from gtopkssgd_amd import ops
from gtopkssgd_amd.parallel.allreducer import (GradReducer,
                                               gtopk_sparse_allreduce,
                                               topk_sparse_allreduce)
from utils_dist import run_in_procs

N = 2000
K = 40


def _local_dense(rank: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(100 + rank)
    return torch.randn(N, generator=g)


def _local_sparse(rank: int):
    dense = _local_dense(rank)
    idx = torch.topk(dense.abs(), K).indices
    sparse = torch.zeros(N)
    sparse[idx] = dense[idx]
    return dense[idx].clone(), idx.to(torch.int32), sparse


def oracle_gtopk(world: int) -> torch.Tensor:
    dense = [_local_sparse(r)[2] for r in range(world)]
    group_size = 1
    while group_size < world:
        for r in range(world):
            if r % (2 * group_size) < group_size and r + group_size < world:
                merged = dense[r] + dense[r + group_size]
                keep = torch.topk(merged.abs(), K).indices
                out = torch.zeros(N)
                out[keep] = merged[keep]
                dense[r] = out
        group_size *= 2
    return dense[0]


def _worker_gtopk(rank, world):
    val, idx, _ = _local_sparse(rank)
    scratch = torch.zeros(N)
    gval, gidx = gtopk_sparse_allreduce(val, idx, scratch, K)
    result = torch.zeros(N)
    ops.densify_(result, gidx, gval, 1.0)
    # identical across ranks
    import torch.distributed as dist
    gathered = [torch.zeros(N) for _ in range(world)]
    dist.all_gather(gathered, result)
    for other in gathered:
        assert torch.equal(other, result), "ranks diverged"
    # matches the dense oracle
    expect = oracle_gtopk(world)
    assert torch.allclose(result, expect, atol=1e-6), \
        (result - expect).abs().max()
    # dense scratch self-cleaned
    assert torch.count_nonzero(scratch) == 0


def test_gtopk_world2():
    run_in_procs(_worker_gtopk, 2)


def test_gtopk_world4():
    run_in_procs(_worker_gtopk, 4)


def test_gtopk_world3_nonpow2():
    run_in_procs(_worker_gtopk, 3)


def _worker_topk(rank, world):
    val, idx, sparse = _local_sparse(rank)
    scratch = torch.zeros(N)
    dense, all_idx = topk_sparse_allreduce(val, idx, scratch, K)
    expect = torch.zeros(N)
    for r in range(world):
        expect += _local_sparse(r)[2]
    assert torch.allclose(dense, expect, atol=1e-6)
    assert all_idx.numel() == world * K


def test_topk_union_world2():
    run_in_procs(_worker_topk, 2)


def _worker_density1(rank, world):
    """rho=1: the merge must equal the dense sum (SURVEY.md §2.3
    invariant), since k=n keeps everything at every level."""
    g = _local_dense(rank)
    idx = torch.arange(N, dtype=torch.int32)
    scratch = torch.zeros(N)
    gval, gidx = gtopk_sparse_allreduce(g.clone(), idx, scratch, N)
    result = torch.zeros(N)
    ops.densify_(result, gidx, gval, 1.0)
    expect = sum(_local_dense(r) for r in range(world))
    assert torch.allclose(result, expect, atol=1e-5)


def test_density_one_equals_dense_allreduce():
    run_in_procs(_worker_density1, 2)


def test_single_process_passthrough():
    val, idx, sparse = _local_sparse(0)
    scratch = torch.zeros(N)
    gval, gidx = gtopk_sparse_allreduce(val, idx, scratch, K)
    assert torch.allclose(gval, val)
    assert torch.equal(gidx, idx)


def test_grad_reducer_wrapper():
    r = GradReducer(N, K, torch.device("cpu"), mode="gtopk")
    val, idx, _ = _local_sparse(0)
    gval, gidx = r.gtopk(val, idx)
    assert gval.numel() == K


def test_gtopk_world8():
    run_in_procs(_worker_gtopk, 8, timeout=240)


def test_gtopk_world5_nonpow2():
    run_in_procs(_worker_gtopk, 5, timeout=240)


def test_halving_schedule_is_deadlock_free():
    """Simulate the send/recv schedule for P up to 64: every send must
    have exactly one matching recv in the same round, each rank is in
    at most one operation per round, and rank 0 ends with the result."""
    for world in list(range(2, 20)) + [32, 33, 64]:
        alive = set(range(world))
        group_size = 1
        while group_size < world:
            sends, recvs = {}, {}
            for r in range(world):
                if r not in alive:
                    continue
                if r % (2 * group_size) < group_size:
                    peer = r + group_size
                    if peer < world:
                        recvs[r] = peer
                else:
                    sends[r] = r - group_size
            for sender, dst in sends.items():
                assert recvs.get(dst) == sender, (world, group_size,
                                                  sender, dst)
                alive.discard(sender)
            for receiver, src in recvs.items():
                if src in sends:
                    assert sends[src] == receiver
                # recv only posted when the peer exists and is alive
                assert src in sends or src >= world or src not in alive
            group_size *= 2
        assert 0 in alive
