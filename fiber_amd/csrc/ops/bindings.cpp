// Python bindings for the ES CDNA4 kernels (fiber_amd._ops).
//
// Deliberately torch-header-free: tensors cross as raw device pointers +
// the current HIP stream (integers from the Python wrapper,
// fiber_amd/ops/__init__.py), which keeps the extension a plain hipcc
// build and the launch path thin.

#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <rocprim/rocprim.hpp>

#include <stdexcept>
#include <string>

namespace py = pybind11;

#define NPARAMS_HOST 4610
#define ENVS_HOST 64

extern "C" __global__ void es_rollout_mlp(
    const float*, float, uint32_t, uint32_t, int, int, const float*,
    const float*, const float*, const float*, float*, float*);
extern "C" __global__ void es_grad(const float*, int, int, int, uint32_t,
                                   uint32_t, int, float*);
extern "C" __global__ void centered_rank(const float*, int, float*);
extern "C" __global__ void rank_pack_keys(const float*, int, unsigned*,
                                          unsigned*);
extern "C" __global__ void rank_scatter(const unsigned*, int, float*);
extern "C" __global__ void mlp_policy_forward(const float*, const float*,
                                              int, float*);
extern "C" __global__ void mfma_gemm64_probe(const float*, const float*,
                                             float*);
// conv policy pipeline (conv_kernels.hip)
extern "C" __global__ void es_perturb(const float*, int, int, float,
                                      uint32_t, const uint32_t*, int,
                                      __hip_bfloat16*, unsigned char*,
                                      unsigned char*);
extern "C" __global__ void conv_env_init(uint32_t, const uint32_t*, int,
                                         float*, float*);
extern "C" __global__ void conv_noisegen(uint32_t, const uint32_t*,
                                         uint32_t, unsigned char*);
extern "C" __global__ void conv_layer1(const __hip_bfloat16*,
                                       const unsigned char*, const float*,
                                       const __hip_bfloat16*,
                                       const unsigned char*, int,
                                       __hip_bfloat16*);
extern "C" __global__ void conv_layer2(const __hip_bfloat16*,
                                       const __hip_bfloat16*, int,
                                       unsigned char*);
extern "C" __global__ void conv_fc(const __hip_bfloat16*,
                                   const unsigned char*,
                                   const unsigned char*, int,
                                   __hip_bfloat16*);
extern "C" __global__ void conv_head_env(const __hip_bfloat16*,
                                         const __hip_bfloat16*, int,
                                         const float*, const float*, float*,
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
                        uint32_t seed, uint32_t iter, int nparams,
                        uintptr_t grad, uintptr_t stream) {
  const int jblocks = (nparams + 3) / 4;
  const int bx = (jblocks + 255) / 256;
  int pairs = pair_end - pair_begin;
  if (pairs <= 0) return;
  // Split pairs so enough workgroups exist to fill the chip (fewer
  // chunks for large nparams — bx already provides parallelism).
  int chunks = pairs < 64 ? 1 : (bx >= 32 ? 4 : 32);
  int per_chunk = (pairs + chunks - 1) / chunks;
  hipLaunchKernelGGL(es_grad, dim3(bx, chunks), dim3(256), 0,
                     (hipStream_t)stream, (const float*)wpair, pair_begin,
                     pair_end, per_chunk, seed, iter, nparams,
                     (float*)grad);
  check(hipGetLastError(), "es_grad launch");
}

static void launch_centered_rank(uintptr_t f, int n, uintptr_t out,
                                 uintptr_t stream) {
  const int bx = (n + 255) / 256;
  hipLaunchKernelGGL(centered_rank, dim3(bx), dim3(256), 0,
                     (hipStream_t)stream, (const float*)f, n, (float*)out);
  check(hipGetLastError(), "centered_rank launch");
}

// Sort-based rank for large populations: pack order-preserving uint keys,
// stable device radix sort (rocPRIM — AMD-native), scatter ranks.
// Workspace layout: [keys_in][keys_out][vals_in][vals_out][rocprim temp],
// each array 256-byte aligned; caller allocates (torch caching allocator).
static inline uint64_t rank_array_stride(int n) {
  return (((uint64_t)n * 4) + 255) & ~(uint64_t)255;
}

static uint64_t centered_rank_sorted_workspace(int n) {
  size_t temp_bytes = 0;
  check(rocprim::radix_sort_pairs(nullptr, temp_bytes,
                                  (const unsigned*)nullptr,
                                  (unsigned*)nullptr,
                                  (const unsigned*)nullptr,
                                  (unsigned*)nullptr, (size_t)n),
        "radix_sort_pairs size query");
  return 4 * rank_array_stride(n) + (uint64_t)temp_bytes + 256;
}

static void launch_centered_rank_sorted(uintptr_t f, int n, uintptr_t out,
                                        uintptr_t work, uint64_t work_bytes,
                                        uintptr_t stream) {
  const uint64_t stride = rank_array_stride(n);
  if (work_bytes < 4 * stride)
    throw std::runtime_error("centered_rank_sorted workspace too small");
  unsigned* keys_in = (unsigned*)work;
  unsigned* keys_out = (unsigned*)(work + stride);
  unsigned* vals_in = (unsigned*)(work + 2 * stride);
  unsigned* vals_out = (unsigned*)(work + 3 * stride);
  void* temp = (void*)(work + 4 * stride);
  size_t temp_bytes = (size_t)(work_bytes - 4 * stride);
  const int bx = (n + 255) / 256;
  hipLaunchKernelGGL(rank_pack_keys, dim3(bx), dim3(256), 0,
                     (hipStream_t)stream, (const float*)f, n, keys_in,
                     vals_in);
  check(hipGetLastError(), "rank_pack_keys launch");
  check(rocprim::radix_sort_pairs(temp, temp_bytes, keys_in, keys_out,
                                  vals_in, vals_out, (size_t)n, 0, 32,
                                  (hipStream_t)stream),
        "radix_sort_pairs");
  hipLaunchKernelGGL(rank_scatter, dim3(bx), dim3(256), 0,
                     (hipStream_t)stream, vals_out, n, (float*)out);
  check(hipGetLastError(), "rank_scatter launch");
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

// ---- conv pipeline launchers ---------------------------------------------

static void launch_perturb(uintptr_t theta, int nparams, int np_pad,
                           double sigma, uint32_t seed, uintptr_t iterp,
                           int member_offset, int pop, uintptr_t wpert,
                           uintptr_t w3_fp8, uintptr_t w1_fp8,
                           uintptr_t stream) {
  if ((member_offset | pop) & 1)
    throw std::runtime_error(
        "es_perturb needs pair-aligned member_offset/pop");
  const int bx = 64;  // grid-stride over param blocks; grid.y = PAIRS
  hipLaunchKernelGGL(es_perturb, dim3(bx, pop / 2), dim3(256), 0,
                     (hipStream_t)stream, (const float*)theta, nparams,
                     np_pad, (float)sigma, seed, (const uint32_t*)iterp,
                     member_offset, (__hip_bfloat16*)wpert,
                     (unsigned char*)w3_fp8, (unsigned char*)w1_fp8);
  check(hipGetLastError(), "es_perturb launch");
}

static void launch_conv_env_init(uint32_t seed, uintptr_t iterp,
                                 int nmembers, uintptr_t state,
                                 uintptr_t racc, uintptr_t stream) {
  const int n = nmembers * 16;
  hipLaunchKernelGGL(conv_env_init, dim3((n + 255) / 256), dim3(256), 0,
                     (hipStream_t)stream, seed, (const uint32_t*)iterp,
                     nmembers, (float*)state, (float*)racc);
  check(hipGetLastError(), "conv_env_init launch");
}

static void launch_conv_noisegen(uint32_t seed, uintptr_t iterp, uint32_t t,
                                 int nenv, uintptr_t znoise,
                                 uintptr_t stream) {
  // flattened (env, position-quad) grid; 7056/4 quads per env
  const int total = nenv * (84 * 84 / 4);
  hipLaunchKernelGGL(conv_noisegen, dim3((total + 255) / 256), dim3(256), 0,
                     (hipStream_t)stream, seed, (const uint32_t*)iterp, t,
                     (unsigned char*)znoise);
  check(hipGetLastError(), "conv_noisegen launch");
}

static void launch_conv_forward(uintptr_t wpert, uintptr_t w3_fp8,
                                uintptr_t w1_fp8, uintptr_t state,
                                uintptr_t gtab, uintptr_t znoise,
                                uintptr_t act1, uintptr_t act2,
                                uintptr_t act3, int nmembers,
                                uintptr_t stream) {
  const int nenv = nmembers * 16;
  hipLaunchKernelGGL(conv_layer1, dim3(nenv), dim3(256), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)wpert,
                     (const unsigned char*)w1_fp8, (const float*)state,
                     (const __hip_bfloat16*)gtab,
                     (const unsigned char*)znoise, nenv,
                     (__hip_bfloat16*)act1);
  check(hipGetLastError(), "conv_layer1 launch");
  hipLaunchKernelGGL(conv_layer2, dim3(nenv), dim3(256), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)wpert,
                     (const __hip_bfloat16*)act1, nenv,
                     (unsigned char*)act2);
  check(hipGetLastError(), "conv_layer2 launch");
  hipLaunchKernelGGL(conv_fc, dim3(nmembers, 4), dim3(256), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)wpert,
                     (const unsigned char*)w3_fp8,
                     (const unsigned char*)act2, nmembers,
                     (__hip_bfloat16*)act3);
  check(hipGetLastError(), "conv_fc launch");
}

static void launch_conv_head_env(uintptr_t wpert, uintptr_t act3,
                                 int nmembers, uintptr_t env_A,
                                 uintptr_t env_B, uintptr_t state,
                                 uintptr_t racc, uintptr_t stream) {
  hipLaunchKernelGGL(conv_head_env, dim3(nmembers), dim3(256), 0,
                     (hipStream_t)stream, (const __hip_bfloat16*)wpert,
                     (const __hip_bfloat16*)act3, nmembers,
                     (const float*)env_A, (const float*)env_B,
                     (float*)state, (float*)racc);
  check(hipGetLastError(), "conv_head_env launch");
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
        py::arg("nparams"), py::arg("grad"), py::arg("stream"));
  m.def("centered_rank", &launch_centered_rank, py::arg("f"), py::arg("n"),
        py::arg("out"), py::arg("stream"));
  m.def("centered_rank_sorted_workspace", &centered_rank_sorted_workspace,
        py::arg("n"));
  m.def("centered_rank_sorted", &launch_centered_rank_sorted, py::arg("f"),
        py::arg("n"), py::arg("out"), py::arg("work"),
        py::arg("work_bytes"), py::arg("stream"));
  m.def("mlp_policy_forward", &launch_mlp_forward, py::arg("theta"),
        py::arg("x"), py::arg("batch"), py::arg("logits"),
        py::arg("stream"));
  m.def("mfma_gemm64_probe", &launch_gemm64_probe, py::arg("a"),
        py::arg("b"), py::arg("c"), py::arg("stream"));

  m.attr("NP_CONV") = 677686;
  m.attr("NP_CONV_PAD") = 677688;
  m.attr("CONV_ENVS") = 16;
  m.def("es_perturb", &launch_perturb, py::arg("theta"), py::arg("nparams"),
        py::arg("np_pad"), py::arg("sigma"), py::arg("seed"),
        py::arg("iterp"), py::arg("member_offset"), py::arg("pop"),
        py::arg("wpert"), py::arg("w3_fp8"), py::arg("w1_fp8"),
        py::arg("stream"));
  m.def("conv_env_init", &launch_conv_env_init, py::arg("seed"),
        py::arg("iterp"), py::arg("nmembers"), py::arg("state"),
        py::arg("racc"), py::arg("stream"));
  m.def("conv_noisegen", &launch_conv_noisegen, py::arg("seed"),
        py::arg("iterp"), py::arg("t"), py::arg("nenv"),
        py::arg("znoise"), py::arg("stream"));
  m.def("conv_forward", &launch_conv_forward, py::arg("wpert"),
        py::arg("w3_fp8"), py::arg("w1_fp8"), py::arg("state"),
        py::arg("gtab"), py::arg("znoise"), py::arg("act1"),
        py::arg("act2"), py::arg("act3"), py::arg("nmembers"),
        py::arg("stream"));
  m.def("conv_head_env", &launch_conv_head_env, py::arg("wpert"),
        py::arg("act3"), py::arg("nmembers"), py::arg("env_A"),
        py::arg("env_B"), py::arg("state"), py::arg("racc"),
        py::arg("stream"));
}
