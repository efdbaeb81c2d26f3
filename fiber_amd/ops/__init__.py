"""Python wrappers for the CDNA4 ES kernels (fiber_amd._ops).

Fail-loud policy: on a machine with a GPU, a missing/unbuilt extension is
an error — there is deliberately NO eager/PyTorch fallback for the hot
ops, so a silently-degraded bench is impossible.
"""

import torch

_OPS = None
_IMPORT_ERROR = None

try:
    from fiber_amd import _ops as _OPS  # built in-tree by setup.py
except ImportError as exc:  # pragma: no cover
    _IMPORT_ERROR = exc


NPARAMS = _OPS.NPARAMS if _OPS else 4610
ENVS_PER_MEMBER = _OPS.ENVS_PER_MEMBER if _OPS else 64
OBS_DIM = 4
ACT_DIM = 2
HIDDEN = 64


def _require_ops():
    if _OPS is None:
        raise RuntimeError(
            "fiber_amd._ops (gfx950 HIP extension) is not built: %r. "
            "Run `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950)." % (_IMPORT_ERROR,)
        )
    return _OPS


def _stream():
    return torch.cuda.current_stream().cuda_stream


def _check(t, name, dtype=torch.float32, device=True):
    if t.dtype != dtype:
        raise TypeError("%s must be %s, got %s" % (name, dtype, t.dtype))
    if device and not t.is_cuda:
        raise TypeError("%s must be a CUDA (ROCm) tensor" % name)
    if not t.is_contiguous():
        raise TypeError("%s must be contiguous" % name)
    return t


def es_rollout_mlp(theta, sigma, seed, iteration, horizon, member_offset,
                   pop_shard, obs_mu, obs_nu, env_A, env_B):
    """One persistent-kernel launch rolling out ``pop_shard`` perturbed
    members x 64 envs for ``horizon`` steps.  Returns (fitness[pop_shard],
    obs_stat[2*OBS+1]) on device."""
    ops = _require_ops()
    _check(theta, "theta")
    assert theta.numel() == NPARAMS
    fitness = torch.empty(pop_shard, dtype=torch.float32,
                          device=theta.device)
    obs_stat = torch.zeros(2 * OBS_DIM + 1, dtype=torch.float32,
                           device=theta.device)
    ops.es_rollout_mlp(
        theta.data_ptr(), float(sigma), int(seed) & 0xFFFFFFFF,
        int(iteration) & 0xFFFFFFFF, int(horizon), int(member_offset),
        int(pop_shard), _check(obs_mu, "obs_mu").data_ptr(),
        _check(obs_nu, "obs_nu").data_ptr(),
        _check(env_A, "env_A").data_ptr(),
        _check(env_B, "env_B").data_ptr(), fitness.data_ptr(),
        obs_stat.data_ptr(), _stream())
    return fitness, obs_stat


def es_grad(wpair, pair_begin, pair_end, seed, iteration, device,
            nparams=None):
    """Noise-weighted gradient over local pairs; eps regenerated on-chip."""
    ops = _require_ops()
    _check(wpair, "wpair")
    if nparams is None:
        nparams = NPARAMS
    grad = torch.zeros(nparams, dtype=torch.float32, device=device)
    ops.es_grad(wpair.data_ptr(), int(pair_begin), int(pair_end),
                int(seed) & 0xFFFFFFFF, int(iteration) & 0xFFFFFFFF,
                int(nparams), grad.data_ptr(), _stream())
    return grad


_rank_workspace = {}  # device -> (n, uint8 tensor); grow-only cache


def centered_rank(fitness):
    """Centered rank transform in [-0.5, 0.5].

    The O(n^2) comparison kernel wins for small populations (one launch,
    no sort); past 16k members (e.g. the named 8-GPU config at pop
    131,072) a stable device radix sort (rocPRIM) + scatter takes over —
    identical output incl. the index-order tie-break."""
    _check(fitness, "fitness")
    n = fitness.numel()
    ops = _require_ops()
    out = torch.empty_like(fitness)
    if n > 16384:
        dev = fitness.device
        cached = _rank_workspace.get(dev)
        if cached is None or cached[0] < n:
            need = ops.centered_rank_sorted_workspace(n)
            cached = (n, torch.empty(need, dtype=torch.uint8, device=dev))
            _rank_workspace[dev] = cached
        ws = cached[1]
        ops.centered_rank_sorted(fitness.data_ptr(), n, out.data_ptr(),
                                 ws.data_ptr(), ws.numel(), _stream())
        return out
    ops.centered_rank(fitness.data_ptr(), n, out.data_ptr(), _stream())
    return out


def centered_rank_ref(fitness):
    """Pure-torch fp32 reference (any device) for tests."""
    n = fitness.numel()
    order = torch.argsort(torch.argsort(fitness, stable=True))
    return order.float() / (n - 1) - 0.5


def mlp_policy_forward(theta, x):
    """Batched policy forward logits via the MFMA path."""
    ops = _require_ops()
    _check(theta, "theta")
    _check(x, "x")
    batch = x.shape[0]
    logits = torch.empty(batch, ACT_DIM, dtype=torch.float32,
                         device=x.device)
    ops.mlp_policy_forward(theta.data_ptr(), x.data_ptr(), batch,
                           logits.data_ptr(), _stream())
    return logits


def mlp_policy_forward_ref(theta, x):
    """fp32 torch reference of the same policy (tests compare this to the
    bf16 MFMA kernel with a loose tolerance)."""
    w1 = theta[:256].view(HIDDEN, OBS_DIM)
    b1 = theta[256:320]
    w2 = theta[320:4416].view(HIDDEN, HIDDEN)
    b2 = theta[4416:4480]
    w3 = theta[4480:4608].view(ACT_DIM, HIDDEN)
    b3 = theta[4608:4610]
    # the kernel rounds every operand to bf16 before the matmuls
    bf = lambda t: t.to(torch.bfloat16).to(torch.float32)
    h1 = torch.tanh(bf(x) @ bf(w1).T + b1)
    h2 = torch.tanh(bf(h1) @ bf(w2).T + b2)
    return bf(h2) @ bf(w3).T + b3


def mfma_gemm64_probe(a, b):
    """C = tanh(A @ B) for 64x64 fp32 inputs through the MFMA strip path
    (hardware verification of the fragment layout assumptions)."""
    ops = _require_ops()
    _check(a, "a")
    _check(b, "b")
    out_t = torch.empty(64, 64, dtype=torch.float32, device=a.device)
    ops.mfma_gemm64_probe(a.data_ptr(), b.data_ptr(), out_t.data_ptr(),
                          _stream())
    # kernel stores C transposed ([col][row]); undo here
    return out_t.T.contiguous()


def ops_available():
    return _OPS is not None
