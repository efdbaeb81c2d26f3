"""Numpy mirror of the device Philox4x32-10 + Box-Muller noise generator
(fiber_amd/csrc/ops/philox.h) — used by CPU tests and by the fp32
reference rollout to reproduce the exact per-member perturbations."""

import numpy as np

_M0 = np.uint32(0xD2511F53)
_M1 = np.uint32(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)

TAG_NOISE = np.uint32(0x45530001)
TAG_ENV = np.uint32(0x45530002)


def philox4x32_10(k0, k1, c0, c1, c2, c3, rounds=10):
    """Vectorized over arrays of counters.  All args uint32 arrays/scalars."""
    k0 = np.uint32(k0) * np.ones_like(np.asarray(c0, np.uint32))
    k1 = np.uint32(k1) * np.ones_like(np.asarray(c0, np.uint32))
    c0 = np.asarray(c0, np.uint32).copy()
    c1 = np.asarray(c1, np.uint32) * np.ones_like(c0)
    c2 = np.asarray(c2, np.uint32) * np.ones_like(c0)
    c3 = np.asarray(c3, np.uint32) * np.ones_like(c0)
    with np.errstate(over="ignore"):
        for _ in range(rounds):
            prod0 = c0.astype(np.uint64) * np.uint64(_M0)
            prod1 = c2.astype(np.uint64) * np.uint64(_M1)
            hi0 = (prod0 >> np.uint64(32)).astype(np.uint32)
            lo0 = prod0.astype(np.uint32)
            hi1 = (prod1 >> np.uint64(32)).astype(np.uint32)
            lo1 = prod1.astype(np.uint32)
            n0 = hi1 ^ c1 ^ k0
            n1 = lo1
            n2 = hi0 ^ c3 ^ k1
            n3 = lo0
            c0, c1, c2, c3 = n0, n1, n2, n3
            k0 = k0 + _W0
            k1 = k1 + _W1
    return c0, c1, c2, c3


def normal4(k0, k1, c0, c1, c2, c3):
    """4 standard normals per counter (matches fam_normal4 bit-for-bit up
    to float32 libm differences)."""
    x0, x1, x2, x3 = philox4x32_10(k0, k1, c0, c1, c2, c3)
    inv = np.float32(2.3283064365386963e-10)
    twopi = np.float32(6.283185307179586)
    u0 = (x0.astype(np.float32) + np.float32(1.0)) * inv
    u1 = (x1.astype(np.float32) + np.float32(1.0)) * inv
    u2 = (x2.astype(np.float32) + np.float32(1.0)) * inv
    u3 = (x3.astype(np.float32) + np.float32(1.0)) * inv
    r0 = np.sqrt(np.float32(-2.0) * np.log(u0, dtype=np.float32))
    r1 = np.sqrt(np.float32(-2.0) * np.log(u2, dtype=np.float32))
    z0 = r0 * np.cos(twopi * u1, dtype=np.float32)
    z1 = r0 * np.sin(twopi * u1, dtype=np.float32)
    z2 = r1 * np.cos(twopi * u3, dtype=np.float32)
    z3 = r1 * np.sin(twopi * u3, dtype=np.float32)
    return np.stack([z0, z1, z2, z3], axis=-1).astype(np.float32)


def uniform4(k0, k1, c0, c1, c2, c3):
    """4 uniforms in [-1, 1) per counter (matches fam_uniform4:
    Philox4x32-7, the minimum BigCrush-passing round count)."""
    x0, x1, x2, x3 = philox4x32_10(k0, k1, c0, c1, c2, c3, rounds=7)
    inv31 = np.float32(4.656612873077393e-10)
    z = [
        x.astype(np.int32).astype(np.float32) * inv31
        for x in (x0, x1, x2, x3)
    ]
    return np.stack(z, axis=-1).astype(np.float32)


def noise_for_pair(seed, iteration, pair, nparams):
    """The full eps vector (len nparams) for one antithetic pair."""
    nblocks = (nparams + 3) // 4
    jb = np.arange(nblocks, dtype=np.uint32)
    z = normal4(seed, iteration, np.uint32(pair) * np.ones_like(jb), jb,
                TAG_NOISE, np.uint32(0))
    return z.reshape(-1)[:nparams]


def env_init_state(seed, iteration, n_envs, obs_dim=4):
    """Initial env states (member-independent, matches the kernel)."""
    e = np.arange(n_envs, dtype=np.uint32)
    z = normal4(seed, iteration, e, np.uint32(0), TAG_ENV, np.uint32(0))
    return 0.3 * z[:, :obs_dim].astype(np.float32)
