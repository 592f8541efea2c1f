"""torchrun worker for the self-arming multi-GPU (RCCL) tests.

Launched by tests/test_multigpu_gpu.py as

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --master-port P \
        tests/workers_multigpu.py --check {merge,train,dense}

Each check exercises the real RCCL/xGMI code path (backend "nccl") that
the gloo CPU tests cannot: p2p send/recv halving rounds, broadcast,
bucketed async all-reduce, and the NCCL-backend metric sync.
"""
import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

N = 200_000
K = 200


def _local_dense(rank: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(100 + rank)
    return torch.randn(N, generator=g)


def _local_sparse(rank: int):
    dense = _local_dense(rank)
    idx = torch.topk(dense.abs(), K).indices
    sparse = torch.zeros(N)
    sparse[idx] = dense[idx]
    return dense[idx].clone(), idx.to(torch.int32), sparse


def _oracle_gtopk(world: int) -> torch.Tensor:
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


def check_merge(rank, world, device):
    """gtopk merge over RCCL p2p vs the dense oracle; HIP kernels on."""
    from gtopkssgd_amd import ops
    from gtopkssgd_amd.parallel.allreducer import gtopk_sparse_allreduce
    assert ops.ext_available(), "HIP extension must be loaded on GPU"
    val, idx, _ = _local_sparse(rank)
    val, idx = val.to(device), idx.to(device)
    scratch = torch.zeros(N, device=device)
    gval, gidx = gtopk_sparse_allreduce(val, idx, scratch, K)
    result = torch.zeros(N, device=device)
    ops.densify_(result, gidx, gval, 1.0)
    gathered = [torch.zeros(N, device=device) for _ in range(world)]
    dist.all_gather(gathered, result)
    for other in gathered:
        assert torch.equal(other, result), "ranks diverged"
    expect = _oracle_gtopk(world).to(device)
    err = (result - expect).abs().max().item()
    assert torch.allclose(result, expect, atol=1e-6), err
    assert torch.count_nonzero(scratch) == 0, "scratch not self-cleaned"


def _train_steps(rank, world, device, mode, steps=3):
    import gtopkssgd_amd as gt
    from gtopkssgd_amd.trainer import DLTrainer
    trainer = DLTrainer(rank=rank, size=world, dnn="resnet20",
                        dataset="cifar10", batch_size=32, lr=0.01,
                        device=str(device), n_samples=256, num_workers=0)
    density = 1.0 if mode == "dense" else 0.001
    comp = (gt.compressors["none"]() if mode == "dense"
            else gt.compressors[mode](density))
    opt = gt.DistributedOptimizer(
        trainer.optimizer, trainer.net.named_parameters(),
        compression=comp, density=density, warmup_iters=1)
    for _ in range(steps):
        loss = trainer.train(1, dist_optimizer=opt)
        trainer.update_model(dist_optimizer=opt)
        assert loss == loss, "NaN loss"
    metrics = trainer.test(0)  # exercises NCCL-backend metric sync (C6)
    assert "top1" in metrics
    # parameters must remain identical across ranks (sync training)
    flat = opt.flat_param
    gathered = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    for g in gathered:
        assert torch.equal(g, flat), "parameters diverged across ranks"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--check", required=True,
                   choices=["merge", "train", "dense"])
    args = p.parse_args()
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ["LOCAL_RANK"])
    torch.cuda.set_device(local)
    device = torch.device("cuda", local)
    dist.init_process_group("nccl")
    try:
        if args.check == "merge":
            check_merge(rank, world, device)
        elif args.check == "train":
            _train_steps(rank, world, device, "gtopk")
        else:
            _train_steps(rank, world, device, "dense")
        torch.cuda.synchronize()
        dist.barrier()
        if rank == 0:
            print(f"MULTIGPU OK {args.check}", flush=True)
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
