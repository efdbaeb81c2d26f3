"""POET-style outer loop — BASELINE config 5.

Population of 8 (policy, env) pairs with a Manager-shared archive and
Pool.starmap_async, mirroring the reference's POET framing
(uber/fiber mkdocs/introduction.md POET section): each worker evaluates
one policy on one env variant; the archive records per-pair bests; envs
whose policies plateau get mutated.

Runs on CPU plumbing (the per-pair payload here is a small numpy ES
step) or with @fiber_amd.meta(gpu=1) workers when GPUs are present.
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse

import numpy as np

import fiber_amd


def evaluate_pair(policy_seed, env_difficulty, iteration):
    """One ES-style inner step for a (policy, env) pair (synthetic)."""
    rng = np.random.default_rng(policy_seed + iteration * 977)
    w = rng.standard_normal(64)
    # fitness landscape whose optimum shifts with difficulty
    target = np.sin(np.arange(64) * env_difficulty)
    fitness = float(-np.mean((w * 0.1 - target) ** 2))
    return policy_seed, env_difficulty, fitness


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--pairs", type=int, default=8)
    parser.add_argument("--generations", type=int, default=5)
    args = parser.parse_args()

    manager = fiber_amd.Manager()
    archive = manager.dict()
    pool = fiber_amd.Pool(args.pairs, error_handling=True)
    try:
        difficulties = [0.1 * (i + 1) for i in range(args.pairs)]
        seeds = list(range(args.pairs))
        for gen in range(args.generations):
            tasks = [
                (seeds[i], difficulties[i], gen)
                for i in range(args.pairs)
            ]
            result = pool.starmap_async(evaluate_pair, tasks)
            for seed, diff, fitness in result.get(120):
                key = "pair-%d" % seed
                prev = archive.get(key) if key in archive else None
                if prev is None or fitness > prev[0]:
                    archive[key] = (fitness, diff, gen)
            # mutate the hardest env of a plateaued pair
            worst = min(archive[k][0] for k in list(archive))
            for i in range(args.pairs):
                if archive["pair-%d" % i][0] == worst:
                    difficulties[i] *= 1.1
            print("gen %d  archive best %.4f  worst %.4f"
                  % (gen,
                     max(archive[k][0] for k in list(archive)),
                     worst))
        return dict((k, archive[k]) for k in list(archive))
    finally:
        pool.terminate()
        pool.join()
        manager.shutdown()


if __name__ == "__main__":
    main()
