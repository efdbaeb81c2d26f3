"""GPU numerics tests for the CDNA4 ES kernels.

Every kernel is compared against a plain PyTorch fp32 reference (with
bf16 rounding applied at the same points the kernel rounds).
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from fiber_amd import ops
    from fiber_amd.es import philox_ref
else:  # collected but skipped on CPU boxes
    ops = None

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@requires_gpu
class TestMfmaLayout:
    def test_mfma_gemm_against_torch(self):
        """Random ASYMMETRIC inputs (guide ERRATA #3: symmetric inputs
        cannot detect a transposed C-write)."""
        torch.manual_seed(0)
        a = torch.randn(64, 64, device="cuda") * 0.3
        b = torch.randn(64, 64, device="cuda") * 0.3
        got = ops.mfma_gemm64_probe(a, b)
        bf = lambda t: t.to(torch.bfloat16).to(torch.float32)
        want = torch.tanh(bf(a) @ bf(b))
        err = (got - want).abs().max().item()
        if err > 2e-2:
            # diagnose which transposition matches
            candidates = {
                "A@B": torch.tanh(bf(a) @ bf(b)),
                "A.T@B": torch.tanh(bf(a).T @ bf(b)),
                "A@B.T": torch.tanh(bf(a) @ bf(b).T),
                "(A@B).T": torch.tanh(bf(a) @ bf(b)).T,
                "B@A": torch.tanh(bf(b) @ bf(a)),
            }
            diag = {
                k: float((got - v).abs().max())
                for k, v in candidates.items()
            }
            pytest.fail("MFMA layout mismatch; per-variant max errs: %r"
                        % diag)

    def test_mfma_identity(self):
        eye = torch.eye(64, device="cuda")
        m = torch.randn(64, 64, device="cuda") * 0.2
        got = ops.mfma_gemm64_probe(eye, m)
        want = torch.tanh(m.to(torch.bfloat16).to(torch.float32))
        assert (got - want).abs().max().item() < 2e-2


@requires_gpu
class TestPolicyForward:
    def test_vs_torch_reference(self):
        torch.manual_seed(1)
        theta = (torch.randn(ops.NPARAMS, device="cuda") * 0.2).contiguous()
        x = torch.randn(192, ops.OBS_DIM, device="cuda").contiguous()
        got = ops.mlp_policy_forward(theta, x)
        want = ops.mlp_policy_forward_ref(theta, x)
        assert (got - want).abs().max().item() < 3e-2

    def test_ragged_batch(self):
        theta = (torch.randn(ops.NPARAMS, device="cuda") * 0.2).contiguous()
        x = torch.randn(70, ops.OBS_DIM, device="cuda").contiguous()
        got = ops.mlp_policy_forward(theta, x)
        want = ops.mlp_policy_forward_ref(theta, x)
        assert (got - want).abs().max().item() < 3e-2


@requires_gpu
class TestCenteredRank:
    def test_vs_reference(self):
        torch.manual_seed(2)
        f = torch.randn(4096, device="cuda").contiguous()
        got = ops.centered_rank(f)
        want = ops.centered_rank_ref(f)
        assert torch.allclose(got, want, atol=1e-6)

    def test_with_ties(self):
        f = torch.tensor([1.0, 1.0, 0.0, 2.0], device="cuda")
        got = ops.centered_rank(f)
        want = ops.centered_rank_ref(f)
        assert torch.allclose(got, want, atol=1e-6)

    def _assert_rank_exact(self, got, want, n):
        """Integer ranks (incl. index-order tie-breaks) must match
        EXACTLY; the float encoding rank/(n-1)-0.5 may differ by 1 ulp
        (torch lowers tensor/scalar to reciprocal-multiply, the kernel
        divides — measured: 96 one-ulp diffs at n=131072, zero rank
        diffs)."""
        assert torch.equal(
            torch.round((got + 0.5) * (n - 1)),
            torch.round((want + 0.5) * (n - 1)),
        ), "rank/tie-break mismatch"
        assert torch.allclose(got, want, atol=2e-7, rtol=0)

    def test_large_population_sorted_path(self):
        """pop 131,072 (the named 8-GPU config) takes the rocPRIM radix
        sort path; ranks must match the stable-argsort reference
        exactly, including index-order tie-breaks and negative zeros."""
        torch.manual_seed(3)
        n = 131072
        # quantized values force heavy ties
        f = (torch.randn(n, device="cuda") * 4).round().contiguous()
        got = ops.centered_rank(f)
        want = ops.centered_rank_ref(f)
        self._assert_rank_exact(got, want, n)

    def test_sorted_path_boundary(self):
        """Just past the n^2/sort switch (16384): both paths agree."""
        torch.manual_seed(4)
        n = 16385
        f = torch.randn(n, device="cuda").contiguous()
        got = ops.centered_rank(f)
        want = ops.centered_rank_ref(f)
        self._assert_rank_exact(got, want, n)


@requires_gpu
class TestPhiloxDevice:
    def test_es_grad_regenerates_reference_noise(self):
        """With a one-hot pair weight, es_grad returns exactly the eps
        vector of that pair — compares device Philox+BoxMuller against
        the numpy mirror."""
        device = torch.device("cuda")
        npairs = 8
        for hot in (0, 5):
            wpair = torch.zeros(npairs, device=device)
            wpair[hot] = 1.0
            grad = ops.es_grad(wpair, 0, npairs, seed=1234, iteration=3,
                               device=device)
            ref = torch.from_numpy(
                philox_ref.noise_for_pair(1234, 3, hot, ops.NPARAMS)
            ).to(device)
            assert (grad - ref).abs().max().item() < 1e-5

    def test_grad_linearity(self):
        device = torch.device("cuda")
        w = torch.tensor([0.7, -1.3, 0.2, 0.0], device=device)
        total = ops.es_grad(w, 0, 4, seed=7, iteration=1, device=device)
        acc = torch.zeros_like(total)
        for k in range(4):
            onehot = torch.zeros(4, device=device)
            onehot[k] = w[k]
            acc += ops.es_grad(onehot, 0, 4, seed=7, iteration=1,
                               device=device)
        assert (total - acc).abs().max().item() < 1e-4
