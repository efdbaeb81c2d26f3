// Python bindings for the ES CDNA4 kernels (fiber_amd._ops).
//
// Deliberately torch-header-free: tensors cross as raw device pointers +
// the current HIP stream (integers from the Python wrapper,
// fiber_amd/ops/__init__.py), which keeps the extension a plain hipcc
// build and the launch path thin.

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <stdexcept>
#include <string>

namespace py = pybind11;

#define NPARAMS_HOST 4610
#define ENVS_HOST 64

extern "C" __global__ void es_rollout_mlp(
    const float*, float, uint32_t, uint32_t, int, int, const float*,
    const float*, const float*, const float*, float*, float*);
extern "C" __global__ void es_grad(const float*, int, int, int, uint32_t,
                                   uint32_t, float*);
extern "C" __global__ void centered_rank(const float*, int, float*);
extern "C" __global__ void mlp_policy_forward(const float*, const float*,
                                              int, float*);
extern "C" __global__ void mfma_gemm64_probe(const float*, const float*,
                                             float*);

static void check(hipError_t err, const char* what) {
  if (err != hipSuccess) {
    throw std::runtime_error(std::string(what) + ": " +
                             hipGetErrorString(err));
  }
}

static void launch_rollout(uintptr_t theta, double sigma, uint32_t seed,
                           uint32_t iter, int horizon, int member_offset,
                           int pop_shard, uintptr_t obs_mu, uintptr_t obs_nu,
                           uintptr_t env_A, uintptr_t env_B,
                           uintptr_t fitness, uintptr_t obs_stat,
                           uintptr_t stream) {
  hipLaunchKernelGGL(es_rollout_mlp, dim3(pop_shard), dim3(256), 0,
                     (hipStream_t)stream, (const float*)theta, (float)sigma,
                     seed, iter, horizon, member_offset,
                     (const float*)obs_mu, (const float*)obs_nu,
                     (const float*)env_A, (const float*)env_B,
                     (float*)fitness, (float*)obs_stat);
  check(hipGetLastError(), "es_rollout_mlp launch");
}

static void launch_grad(uintptr_t wpair, int pair_begin, int pair_end,
                        uint32_t seed, uint32_t iter, uintptr_t grad,
                        uintptr_t stream) {
  const int jblocks = (NPARAMS_HOST + 3) / 4;
  const int bx = (jblocks + 255) / 256;
  int pairs = pair_end - pair_begin;
  if (pairs <= 0) return;
  // Split pairs so enough workgroups exist to fill the chip.
  int chunks = pairs < 64 ? 1 : 32;
  int per_chunk = (pairs + chunks - 1) / chunks;
  hipLaunchKernelGGL(es_grad, dim3(bx, chunks), dim3(256), 0,
                     (hipStream_t)stream, (const float*)wpair, pair_begin,
                     pair_end, per_chunk, seed, iter, (float*)grad);
  check(hipGetLastError(), "es_grad launch");
}

static void launch_centered_rank(uintptr_t f, int n, uintptr_t out,
                                 uintptr_t stream) {
  const int bx = (n + 255) / 256;
  hipLaunchKernelGGL(centered_rank, dim3(bx), dim3(256), 0,
                     (hipStream_t)stream, (const float*)f, n, (float*)out);
  check(hipGetLastError(), "centered_rank launch");
}

static void launch_mlp_forward(uintptr_t theta, uintptr_t x, int batch,
                               uintptr_t logits, uintptr_t stream) {
  const int bx = (batch + ENVS_HOST - 1) / ENVS_HOST;
  hipLaunchKernelGGL(mlp_policy_forward, dim3(bx), dim3(256), 0,
                     (hipStream_t)stream, (const float*)theta,
                     (const float*)x, batch, (float*)logits);
  check(hipGetLastError(), "mlp_policy_forward launch");
}

static void launch_gemm64_probe(uintptr_t a, uintptr_t b, uintptr_t c,
                                uintptr_t stream) {
  hipLaunchKernelGGL(mfma_gemm64_probe, dim3(1), dim3(256), 0,
                     (hipStream_t)stream, (const float*)a, (const float*)b,
                     (float*)c);
  check(hipGetLastError(), "mfma_gemm64_probe launch");
}

PYBIND11_MODULE(_ops, m) {
  m.doc() = "fiber_amd CDNA4 ES kernels (gfx950)";
  m.attr("NPARAMS") = NPARAMS_HOST;
  m.attr("ENVS_PER_MEMBER") = ENVS_HOST;
  m.def("es_rollout_mlp", &launch_rollout, py::arg("theta"),
        py::arg("sigma"), py::arg("seed"), py::arg("iter"),
        py::arg("horizon"), py::arg("member_offset"), py::arg("pop_shard"),
        py::arg("obs_mu"), py::arg("obs_nu"), py::arg("env_A"),
        py::arg("env_B"), py::arg("fitness"), py::arg("obs_stat"),
        py::arg("stream"));
  m.def("es_grad", &launch_grad, py::arg("wpair"), py::arg("pair_begin"),
        py::arg("pair_end"), py::arg("seed"), py::arg("iter"),
        py::arg("grad"), py::arg("stream"));
  m.def("centered_rank", &launch_centered_rank, py::arg("f"), py::arg("n"),
        py::arg("out"), py::arg("stream"));
  m.def("mlp_policy_forward", &launch_mlp_forward, py::arg("theta"),
        py::arg("x"), py::arg("batch"), py::arg("logits"),
        py::arg("stream"));
  m.def("mfma_gemm64_probe", &launch_gemm64_probe, py::arg("a"),
        py::arg("b"), py::arg("c"), py::arg("stream"));
}
