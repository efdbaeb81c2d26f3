"""OpenAI-ES engine on MI355X: persistent-rollout kernel + RCCL ring.

The flagship workload (BASELINE.json config 3; reference analog:
uber/fiber examples/gecco-2020/es.py pool.map of perturbation rollouts).
Each rank owns one MI355X and a population shard; one ES iteration is:

  1. es_rollout_mlp       — ONE kernel launch: shard x 64 envs x horizon
                            synthetic-env rollouts with on-chip Philox
                            perturbations (antithetic pairs)
  2. all_gather(fitness)  — tiny (pop floats) over xGMI
  3. centered_rank        — device kernel on the full population
  4. es_grad              — noise-weighted gradient, eps regenerated
  5. all_reduce(grad)     — 4610 floats
  6. Adam update + obs-normalization stats merge (identical on all ranks
     => identical theta without broadcasting)
"""

import dataclasses

import torch

from .. import ops, tracing


@dataclasses.dataclass
class ESConfig:
    pop_per_gpu: int = 2048       # must be even (antithetic pairs)
    horizon: int = 200
    sigma: float = 0.05
    lr: float = 0.02
    seed: int = 1234
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    # fixed by the compiled kernel:
    obs_dim: int = ops.OBS_DIM
    act_dim: int = ops.ACT_DIM
    hidden: int = ops.HIDDEN
    envs_per_member: int = ops.ENVS_PER_MEMBER


def make_env_params(seed, device):
    """Fixed synthetic-env dynamics, identical on every rank."""
    gen = torch.Generator().manual_seed(seed ^ 0xE17)
    env_A = (torch.randn(4, 4, generator=gen) * 0.5).to(device)
    env_B = torch.randn(4, generator=gen).to(device)
    return env_A.contiguous(), env_B.contiguous()


def init_theta(seed, device):
    gen = torch.Generator().manual_seed(seed)
    theta = torch.zeros(ops.NPARAMS)
    # orthogonal-ish scaled init per layer, zero biases
    w1 = torch.randn(64, 4, generator=gen) / (4 ** 0.5)
    w2 = torch.randn(64, 64, generator=gen) / (64 ** 0.5)
    w3 = torch.randn(2, 64, generator=gen) / (64 ** 0.5)
    theta[:256] = w1.reshape(-1)
    theta[320:4416] = w2.reshape(-1)
    theta[4480:4608] = w3.reshape(-1)
    return theta.to(device).contiguous()


class ESEngine:
    def __init__(self, config: ESConfig, ctx=None, device=None):
        self.cfg = config
        self.ctx = ctx  # fiber_amd.ring.RingContext or None
        self.rank = ctx.rank if ctx else 0
        self.world = ctx.size if ctx else 1
        if device is None:
            device = torch.device("cuda", 0)
        self.device = device
        if config.pop_per_gpu % 2:
            raise ValueError("pop_per_gpu must be even (antithetic pairs)")
        self.pop_total = config.pop_per_gpu * self.world

        self.theta = init_theta(config.seed, device)
        self.adam_m = torch.zeros_like(self.theta)
        self.adam_v = torch.zeros_like(self.theta)
        self.t_step = 0

        self.env_A, self.env_B = make_env_params(config.seed, device)
        self.obs_sum = torch.zeros(config.obs_dim, device=device)
        self.obs_sumsq = torch.zeros(config.obs_dim, device=device)
        self.obs_count = torch.zeros(1, device=device)
        self.obs_mu = torch.zeros(config.obs_dim, device=device)
        self.obs_nu = torch.ones(config.obs_dim, device=device)

        # static workspace so repeated steps allocate nothing
        self._fitness_all = torch.empty(self.pop_total, device=device)
        # grad (NPARAMS) and obs_stat (2*obs_dim+1) share ONE flat buffer
        # so the per-step reduction is a single RCCL all-reduce: xGMI ring
        # latency is paid once, not twice, per iteration (matters at 8
        # ranks where both payloads are tiny).
        nstat = 2 * config.obs_dim + 1
        self._reduce_buf = torch.empty(ops.NPARAMS + nstat, device=device)
        self._grad_view = self._reduce_buf[: ops.NPARAMS]
        self._stat_view = self._reduce_buf[ops.NPARAMS :]

    # -- one ES iteration --------------------------------------------------
    def step(self, iteration=None):
        cfg = self.cfg
        if iteration is None:
            iteration = self.t_step
        shard = cfg.pop_per_gpu
        member_offset = self.rank * shard

        with tracing.range("es.rollout"):
            fitness, obs_stat = ops.es_rollout_mlp(
                self.theta, cfg.sigma, cfg.seed, iteration, cfg.horizon,
                member_offset, shard, self.obs_mu, self.obs_nu, self.env_A,
                self.env_B,
            )

        if self.ctx is not None:
            with tracing.range("es.allgather_fitness"):
                self.ctx.all_gather_into(self._fitness_all, fitness)
            fitness_all = self._fitness_all
        else:
            fitness_all = fitness

        ranks = ops.centered_rank(fitness_all)
        wpair = (ranks[0::2] - ranks[1::2]).contiguous()

        pair_begin = member_offset // 2
        pair_end = (member_offset + shard) // 2
        with tracing.range("es.grad"):
            grad = ops.es_grad(wpair, pair_begin, pair_end, cfg.seed,
                               iteration, self.device)
        if self.ctx is not None:
            with tracing.range("es.allreduce_grad"):
                # one fused all-reduce for grad + obs stats
                self._grad_view.copy_(grad)
                self._stat_view.copy_(obs_stat)
                self.ctx.allreduce(self._reduce_buf)
                grad = self._grad_view
                obs_stat = self._stat_view
        grad /= float(self.pop_total) * cfg.sigma

        # maximize fitness => ascend
        self.t_step += 1
        b1, b2 = cfg.adam_beta1, cfg.adam_beta2
        self.adam_m.mul_(b1).add_(grad, alpha=1 - b1)
        self.adam_v.mul_(b2).addcmul_(grad, grad, value=1 - b2)
        mhat = self.adam_m / (1 - b1 ** self.t_step)
        vhat = self.adam_v / (1 - b2 ** self.t_step)
        self.theta.add_(cfg.lr * mhat / (vhat.sqrt() + cfg.adam_eps))

        # running obs normalization (applied next iteration)
        self.obs_sum += obs_stat[: cfg.obs_dim]
        self.obs_sumsq += obs_stat[cfg.obs_dim : 2 * cfg.obs_dim]
        self.obs_count += obs_stat[2 * cfg.obs_dim]
        count = self.obs_count.clamp_min(1.0)
        self.obs_mu = (self.obs_sum / count).contiguous()
        var = self.obs_sumsq / count - self.obs_mu ** 2
        self.obs_nu = var.clamp_min(1e-2).contiguous()

        return {
            "fitness_mean": float(fitness_all.mean()),
            "fitness_max": float(fitness_all.max()),
            "grad_norm": float(grad.norm()),
            "rollouts": self.pop_total * cfg.envs_per_member,
            "env_steps": self.pop_total * cfg.envs_per_member * cfg.horizon,
        }

    # -- checkpoint / resume ----------------------------------------------
    # (the reference has no checkpointing — SURVEY §5; this is additive.)
    def state_dict(self):
        return {
            "theta": self.theta.cpu(),
            "adam_m": self.adam_m.cpu(),
            "adam_v": self.adam_v.cpu(),
            "t_step": self.t_step,
            "obs_sum": self.obs_sum.cpu(),
            "obs_sumsq": self.obs_sumsq.cpu(),
            "obs_count": self.obs_count.cpu(),
            "config": dataclasses.asdict(self.cfg),
        }

    def load_state_dict(self, state):
        self.theta.copy_(state["theta"].to(self.device))
        self.adam_m.copy_(state["adam_m"].to(self.device))
        self.adam_v.copy_(state["adam_v"].to(self.device))
        self.t_step = int(state["t_step"])
        self.obs_sum.copy_(state["obs_sum"].to(self.device))
        self.obs_sumsq.copy_(state["obs_sumsq"].to(self.device))
        self.obs_count.copy_(state["obs_count"].to(self.device))
        count = self.obs_count.clamp_min(1.0)
        self.obs_mu = (self.obs_sum / count).contiguous()
        var = self.obs_sumsq / count - self.obs_mu ** 2
        self.obs_nu = var.clamp_min(1e-2).contiguous()

    def save(self, path):
        import torch as _torch

        _torch.save(self.state_dict(), path)

    def load(self, path):
        import torch as _torch

        self.load_state_dict(_torch.load(path, weights_only=False))


# ---------------------------------------------------------------------------
# fp32 torch reference of one rollout batch (mirrors the kernel exactly,
# including bf16 rounding points and the Philox noise) — the numerics
# oracle for tests/gpu/test_es.py.
# ---------------------------------------------------------------------------


def rollout_reference(theta, sigma, seed, iteration, horizon, members,
                      obs_mu, obs_nu, env_A, env_B):
    """Returns fitness[len(members)] computed on CPU in fp32 (+bf16
    rounding at the same points as the kernel)."""

    from . import philox_ref

    def bf(t):
        return t.to(torch.bfloat16).to(torch.float32)

    theta = theta.detach().cpu()
    obs_mu = obs_mu.detach().cpu()
    obs_nu = obs_nu.detach().cpu()
    env_A = env_A.detach().cpu()
    env_B = env_B.detach().cpu()
    E = ops.ENVS_PER_MEMBER
    fitness = []
    s0 = torch.from_numpy(
        philox_ref.env_init_state(seed, iteration, E)
    )  # [E][4]

    for m in members:
        pair = m // 2
        sgn = -sigma if (m % 2) else sigma
        eps = torch.from_numpy(
            philox_ref.noise_for_pair(seed, iteration, pair, ops.NPARAMS)
        )
        th = theta + sgn * eps
        w1 = bf(th[:256].view(64, 4))
        b1 = th[256:320]
        w2 = bf(th[320:4416].view(64, 64))
        b2 = th[4416:4480]
        w3 = bf(th[4480:4608].view(2, 64))
        b3 = th[4608:4610]

        s = s0.clone()
        racc = torch.zeros(E)
        rstd = torch.rsqrt(obs_nu + 1e-4)
        for _ in range(horizon):
            x = ((s - obs_mu) * rstd).clamp(-5, 5)
            h1 = bf(torch.tanh(bf(x) @ w1.T + b1))
            # layer-2 activations stay fp32 in the kernel (fused-logits
            # epilogue, no bf16 store)
            h2 = torch.tanh(h1 @ w2.T + b2)
            logits = h2 @ w3.T + b3
            asign = torch.where(logits[:, 1] > logits[:, 0], 1.0, -1.0)
            drive = torch.tanh(s @ env_A.T)
            snew = 0.97 * s + 0.08 * drive + 0.05 * env_B * asign[:, None]
            racc += 1.0 - 0.1 * (snew * snew).sum(dim=1)
            s = snew
        fitness.append(float(racc.mean()))
    return torch.tensor(fitness)
