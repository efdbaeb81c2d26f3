"""Synchronous data-parallel SGD via Ring — reference examples/ring.py
analog, with the collectives owned by the Ring itself (bucketed
allreduce) instead of user-wired Gloo.

CPU demo: python examples/ring_sgd.py            (gloo, world 2)
GPU:      python examples/ring_sgd.py --gpus 4   (RCCL over xGMI)
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse
import functools


def train(rank, size, backend=None, steps=20):
    import torch

    from fiber_amd.ring import RingContext

    ctx = RingContext(rank, size, backend=backend)
    ctx.init()
    device = ctx.device

    torch.manual_seed(1234)  # identical init on every rank
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.Tanh(),
        torch.nn.Linear(128, 10),
    ).to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)

    gen = torch.Generator().manual_seed(rank)  # rank-local shard
    for step in range(steps):
        x = torch.randn(256, 64, generator=gen).to(device)
        y = (x.sum(dim=1) > 0).long().to(device)
        loss = torch.nn.functional.cross_entropy(model(x), y)
        opt.zero_grad()
        loss.backward()
        ctx.allreduce_grads(model.parameters(), average=True)
        opt.step()
        if rank == 0 and step % 5 == 0:
            print("step %3d loss %.4f" % (step, loss.item()))
    ctx.barrier()
    ctx.shutdown()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=0)
    parser.add_argument("--world", type=int, default=2)
    args = parser.parse_args()

    import fiber_amd

    if args.gpus:
        ring = fiber_amd.Ring(
            args.gpus, functools.partial(train, backend="nccl"),
            gpu_per_rank=1,
        )
    else:
        ring = fiber_amd.Ring(
            args.world, functools.partial(train, backend="gloo"),
            gpu_per_rank=0,
        )
    ring.run(timeout=300)
    print("done")


if __name__ == "__main__":
    main()
