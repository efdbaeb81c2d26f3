#!/usr/bin/env python3
"""Flagship benchmark: OpenAI-ES MLP policy rollouts on MI355X.

BASELINE.json config 3: 2-layer(-hidden) MLP policy on synthetic
CartPole-dim observations, antithetic-pair perturbations from an on-chip
Philox noise table, Ring all-reduce over xGMI (RCCL) at N>1.  One "step"
is one full ES iteration (rollout shard -> all-gather fitness ->
centered rank -> noise-weighted gradient -> all-reduce -> Adam update).

Run (single GPU):      python bench.py
Run (N ranks, driver): python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...

Metric: es_rollouts_per_sec — completed env episodes per second, whole
job (a rollout = one episode of one env instance: pop_total x 64 envs
per iteration).  Synthetic data, random-init weights (no datasets in
this environment); weak scaling (per-GPU population fixed).
"""

import argparse
import json
import os
import time


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=3)
    parser.add_argument("--pop-per-gpu", type=int, default=None)
    parser.add_argument("--horizon", type=int, default=None)
    parser.add_argument("--model", choices=["mlp", "conv"], default="mlp",
                        help="flagship MLP config (default) or the "
                             "ConvNet pixel-policy config")
    parser.add_argument("--mode", choices=["engine", "pool"],
                        default="engine",
                        help="engine: one rank per GPU drives ESEngine "
                             "directly (the driver's torchrun contract); "
                             "pool: the SAME workload dispatched through "
                             "the framework — Pool(collective=True) "
                             "SPMD fan-out + pool-communicator "
                             "collectives (single-process launch only)")
    args = parser.parse_args()
    if args.pop_per_gpu is None:
        # Big shards are the MI355X-first choice (288 GB HBM, fixed costs
        # amortize, collectives stay large); per-GPU work is fixed as N
        # grows (weak scaling).  32768 members = 21.3 occupancy rounds of
        # the rollout kernel (256 CU x 6 wg) — the 16384 shard wasted
        # 6.7% in the last-round tail (measured ladder:
        # profiles/r02_rollout_pmc.md: 83.6M @16k, 90.2M @32k, 91.2M
        # @128k; 32k keeps ms/step driver-friendly).
        args.pop_per_gpu = 32768 if args.model == "mlp" else 2048
    if args.horizon is None:
        args.horizon = 256 if args.model == "mlp" else 64

    import torch

    if args.mode == "pool":
        return bench_through_pool(args)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # FAM_BENCH_CPU=1: dress rehearsal of the EXACT driver invocation
    # (torchrun rendezvous, env parsing, barriers, MAX-reduce, JSON
    # contract) on a GPU-less box — HIP kernels stubbed, gloo backend.
    # tests/test_bench_rehearsal.py runs it at world 2 so the 8-GPU
    # SCALE path has been executed end-to-end before the driver tries.
    cpu_rehearsal = bool(os.environ.get("FAM_BENCH_CPU"))
    if cpu_rehearsal:
        from fiber_amd import ops as _ops
        from tests.test_es_distributed import _cpu_stub_ops

        _cpu_stub_ops(_ops)
        device = torch.device("cpu")
    else:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)

    ctx = None
    if world > 1:
        from fiber_amd.ring import RingContext

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if os.environ.get("FAM_NCCL_DEBUG"):
            # RCCL-level diagnostics for ring/xGMI setup issues
            os.environ.setdefault("NCCL_DEBUG", "WARN")
        ctx = RingContext(rank, world,
                          backend="gloo" if cpu_rehearsal else "nccl",
                          device=device)
        ctx.init()

    if args.model == "mlp":
        from fiber_amd.es import ESConfig, ESEngine

        cfg = ESConfig(pop_per_gpu=args.pop_per_gpu, horizon=args.horizon)
        engine = ESEngine(cfg, ctx=ctx, device=device)
        model_name = "es-mlp-obs4-h64x64-act2"
        dtype = "bf16"
    else:
        from fiber_amd.es.conv_policy import ConvESConfig, ConvESEngine

        cfg = ConvESConfig(pop_per_gpu=args.pop_per_gpu,
                           horizon=args.horizon)
        engine = ConvESEngine(cfg, ctx=ctx, device=device)
        model_name = "es-conv-84x84x4-dqn-act6"
        dtype = "bf16+fp8(e4m3)"  # conv1/fc ride OCP fp8, rest bf16

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    for i in range(args.warmup):
        engine.step(iteration=i)

    if ctx is not None:
        ctx.barrier()
    sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        engine.step(iteration=args.warmup + i)
    if ctx is not None:
        ctx.barrier()
    sync()
    elapsed = time.perf_counter() - t0

    if ctx is not None:
        t = torch.tensor([elapsed], device=device)
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    rollouts_per_step = cfg.pop_per_gpu * world * cfg.envs_per_member
    value = rollouts_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "es_rollouts_per_sec",
            "value": value,
            "unit": "rollouts/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": cfg.pop_per_gpu * world
                * cfg.envs_per_member,
                "seq_len": args.horizon,
                "parallelism": "dp%d" % world,
                "pop_per_gpu": cfg.pop_per_gpu,
                "envs_per_member": cfg.envs_per_member,
                "env_steps_per_sec": value * args.horizon,
                "sigma": cfg.sigma,
            },
        }))

    if ctx is not None:
        ctx.shutdown()


def bench_through_pool(args):
    """The same flagship metric, dispatched THROUGH the framework:
    Pool(collective=True) with one GPU-pinned worker per device, SPMD
    fan-out per block of iterations, fitness/grad collectives on the
    pool communicator over xGMI.  Overhead vs engine mode is the cost of
    the framework (measured: profiles/r02_es_pool_overhead.md)."""
    import json as _json
    import time as _time

    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        raise SystemExit("--mode pool is a single-process launch: the "
                         "pool spawns its own per-GPU workers")
    from fiber_amd.pool import ZPool
    from examples.es_pool import es_steps, init_es_worker

    cfg = dict(pop_per_gpu=args.pop_per_gpu, horizon=args.horizon)
    pool = ZPool(processes=args.gpus, gpu_per_worker=1, collective=True,
                 initializer=init_es_worker, initargs=(cfg, args.model))
    try:
        pool.run_on_all(es_steps, (0, args.warmup, True), timeout=1800)
        t0 = _time.perf_counter()
        per_rank = pool.run_on_all(
            es_steps, (args.warmup, args.steps, True), timeout=3600
        )
        elapsed = _time.perf_counter() - t0
    finally:
        pool.terminate()
        pool.join()

    envs = 64 if args.model == "mlp" else 16
    rollouts_per_step = args.pop_per_gpu * args.gpus * envs
    value = rollouts_per_step * args.steps / elapsed
    print(_json.dumps({
        "metric": "es_rollouts_per_sec",
        "value": value,
        "unit": "rollouts/s",
        "n_gpus": args.gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if args.model == "mlp" else "bf16+fp8(e4m3)",
        "data": "synthetic",
        "config": {
            "model": ("es-mlp-obs4-h64x64-act2" if args.model == "mlp"
                      else "es-conv-84x84x4-dqn-act6"),
            "global_batch": args.pop_per_gpu * args.gpus * envs,
            "seq_len": args.horizon,
            "parallelism": "pool-dp%d" % args.gpus,
            "pop_per_gpu": args.pop_per_gpu,
            "envs_per_member": envs,
            "dispatch": "Pool.run_on_all + pool RCCL communicator",
            "theta_sum": per_rank[0]["theta_sum"],
        },
    }))


if __name__ == "__main__":
    main()
