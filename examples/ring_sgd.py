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


def make_convnet():
    """The reference example's MNIST ConvNet (uber/fiber
    examples/ring.py:89-105: conv 20@5x5 -> conv 50@5x5 -> fc500 -> 10)."""
    import torch

    return torch.nn.Sequential(
        torch.nn.Conv2d(1, 20, 5), torch.nn.MaxPool2d(2),
        torch.nn.ReLU(),
        torch.nn.Conv2d(20, 50, 5), torch.nn.MaxPool2d(2),
        torch.nn.ReLU(),
        torch.nn.Flatten(),
        torch.nn.Linear(800, 500), torch.nn.ReLU(),
        torch.nn.Linear(500, 10),
    )


def train(rank, size, backend=None, steps=20, batch=64):
    import torch

    from fiber_amd.ring import RingContext

    ctx = RingContext(rank, size, backend=backend)
    ctx.init()
    device = ctx.device

    torch.manual_seed(1234)  # identical init on every rank
    model = make_convnet().to(device)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)

    gen = torch.Generator().manual_seed(rank)  # rank-local shard
    for step in range(steps):
        # synthetic MNIST-shaped data with a learnable rule
        x = torch.randn(batch, 1, 28, 28, generator=gen).to(device)
        y = (x.mean(dim=(1, 2, 3)) > 0).long().to(device)
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
