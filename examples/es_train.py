"""Train the flagship ES MLP policy — BASELINE config 3.

Single GPU:  python examples/es_train.py --iters 100
Multi GPU:   python -m torch.distributed.run --nproc-per-node 8 \
                 --master-addr 127.0.0.1 examples/es_train.py --iters 100
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse
import os
import time


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--iters", type=int, default=100)
    parser.add_argument("--pop-per-gpu", type=int, default=4096)
    parser.add_argument("--horizon", type=int, default=256)
    parser.add_argument("--sigma", type=float, default=0.05)
    parser.add_argument("--lr", type=float, default=0.02)
    args = parser.parse_args()

    import torch

    from fiber_amd.es import ESConfig, ESEngine

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0")))
    torch.cuda.set_device(device)

    ctx = None
    if world > 1:
        from fiber_amd.ring import RingContext

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        ctx = RingContext(rank, world, backend="nccl", device=device)
        ctx.init()

    cfg = ESConfig(pop_per_gpu=args.pop_per_gpu, horizon=args.horizon,
                   sigma=args.sigma, lr=args.lr)
    engine = ESEngine(cfg, ctx=ctx, device=device)
    t0 = time.perf_counter()
    for i in range(args.iters):
        stats = engine.step()
        if rank == 0 and (i % 10 == 0 or i == args.iters - 1):
            print("iter %4d  fitness mean %+8.3f  max %+8.3f  |g| %.4f"
                  % (i, stats["fitness_mean"], stats["fitness_max"],
                     stats["grad_norm"]))
    torch.cuda.synchronize()
    if rank == 0:
        elapsed = time.perf_counter() - t0
        total = stats["rollouts"] * args.iters
        print("%.1fs, %.1fM rollouts/s" % (elapsed, total / elapsed / 1e6))
    if ctx is not None:
        ctx.shutdown()


if __name__ == "__main__":
    main()
