#!/usr/bin/env python3
"""End-to-end data-parallel training sample: a torch MLP trained on a
synthetic regression task with mlsl_amd's DistributedData wrapper
(bucketed non-blocking gradient allreduce overlapped with backward).

Run:
    python samples/train_ddp.py                       # 1 process
    torchrun --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 samples/train_ddp.py  # 8 GPUs over RCCL

Prints the final loss; identical across ranks (same seed for data, params
broadcast from rank 0, averaged gradients) — the sample asserts that and
prints PASSED, mirroring the reference sample protocol.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import mlsl_amd as mx
from mlsl_amd.parallel import DistributedData


def main():
    steps = int(os.environ.get("STEPS", "20"))
    mx.init()
    rank, world = mx.rank(), mx.world_size()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) %
                              torch.cuda.device_count())
    dev = "cuda" if use_cuda else "cpu"

    torch.manual_seed(1234)          # same model init everywhere
    model = torch.nn.Sequential(
        torch.nn.Linear(256, 512), torch.nn.GELU(),
        torch.nn.Linear(512, 512), torch.nn.GELU(),
        torch.nn.Linear(512, 64)).to(dev)
    dist = mx.Distribution(world, 1)
    dd = DistributedData(model, dist)
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)

    # rank-local shard of a fixed synthetic dataset
    g = torch.Generator().manual_seed(42 + rank)
    x = torch.randn(64, 256, generator=g).to(dev)
    w_true = torch.randn(256, 64, generator=torch.Generator().manual_seed(7)).to(dev)
    y = x @ w_true

    loss = None
    for _ in range(steps):
        out = model(x)
        loss = torch.nn.functional.mse_loss(out, y)
        loss.backward()
        dd.finish_gradients()        # waits bucketed allreduce, averages
        opt.step()
        opt.zero_grad(set_to_none=False)

    # parameters must be bit-identical across ranks after averaged updates
    ref = torch.cat([p.detach().float().flatten().cpu()
                     for p in model.parameters()])
    lo = ref.clone()
    hi = ref.clone()
    mx.wait(dist.all_reduce(lo, lo, lo.numel(), op="min", group="data"))
    mx.wait(dist.all_reduce(hi, hi, hi.numel(), op="max", group="data"))
    drift = (hi - lo).abs().max().item()
    print(f"rank {rank}/{world} final loss {loss.item():.6f} "
          f"cross-rank param drift {drift:.3e}")
    assert drift < 1e-6, "ranks diverged"
    if rank == 0:
        print("PASSED")
    mx.finalize()


if __name__ == "__main__":
    main()
