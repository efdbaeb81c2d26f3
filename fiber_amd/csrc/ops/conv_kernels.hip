// CDNA4 kernels for the ConvNet-policy ES path (BASELINE config 4):
// DQN-shaped policy on 84x84x4 synthetic pixel observations, batched
// rollouts resident in HBM (288 GB budget), every conv/fc layer on MFMA.
//
//   obs 84x84x4 --conv 16@8x8 s4--> 20x20x16 --conv 32@4x4 s2--> 9x9x32
//       --fc 256--> --head 6-->
//
// Data layouts (all channel-last so im2col K-runs are memory-contiguous):
//   obs  [member][env][y][x][ic=4]    e4m3 (fp8)
//   act1 [member][env][pos=20x20][16] bf16
//   act2 [member][env][pos=9x9][32]   e4m3 (flat k for fc = (y*9+x)*32+oc)
//   act3 [member][env][256]           bf16
// Weights: one flat fp32 master theta; es_perturb materializes per-member
// bf16 perturbed copies (antithetic Philox pairs, same counter scheme as
// the MLP path) into wpert[member][NP_CONV_PAD].
//
// K-ordering of each weight row matches the B-side layout exactly:
//   W1 row k = (ky*8+kx)*4+ic  -> one K-tile(32) = one kernel row ky
//   W2 row k = (ky*4+kx)*16+ic -> one K-tile(32) = two kx pixels
//   W3 row k = flat act2 index
// so every MFMA B-fragment (8 consecutive k, 16 B) is one aligned
// contiguous load from obs/act buffers.  E = 16 envs per member makes the
// fc layer a full 256x16x2592 MFMA GEMM (weights read once per member
// step, amortized over all 16 envs).

#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>

#include "philox.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef long fp8x8;  // 8 packed OCP e4m3 bytes (MFMA fp8 operand)

#define FAM_TAG_NOISE 0x45530001u
#define FAM_TAG_ENV 0x45530002u
#define FAM_TAG_OBS 0x45530003u

__device__ inline float fast_tanh_c(float x) {
  float e = __expf(2.0f * x);
  return 1.0f - 2.0f * __builtin_amdgcn_rcpf(e + 1.0f);
}

// geometry
#define CIN 4
#define IMG 84
#define C1 16
#define O1 20
#define C2 32
#define O2 9
#define FCU 256
#define NFLAT (O2 * O2 * C2)  // 2592
#define NACT 6
#define CENV 16  // envs per member
#define SDIM 4

// flat theta offsets (fp32 master / bf16 perturbed share the layout)
#define COFF_W1 0
#define COFF_B1 (C1 * 256)                  // 4096
#define COFF_W2 (COFF_B1 + C1)              // 4112
#define COFF_B2 (COFF_W2 + C2 * 256)        // 12304
#define COFF_W3 (COFF_B2 + C2)              // 12336
#define COFF_B3 (COFF_W3 + FCU * NFLAT)     // 675888
#define COFF_W4 (COFF_B3 + FCU)             // 676144
#define COFF_B4 (COFF_W4 + NACT * FCU)      // 677680
#define NP_CONV (COFF_B4 + NACT)            // 677686
#define NP_CONV_PAD 677688                  // 16B-aligned member stride

// ---------------------------------------------------------------------------
// es_perturb: wpert[member][NP_CONV_PAD] = bf16(theta +/- sigma*eps_pair)
// ---------------------------------------------------------------------------
extern "C" __global__ void es_perturb(const float* __restrict__ theta,
                                      int nparams, int np_pad, float sigma,
                                      uint32_t seed,
                                      const uint32_t* __restrict__ iterp,
                                      int member_offset,
                                      __hip_bfloat16* __restrict__ wpert,
                                      unsigned char* __restrict__ w3_fp8,
                                      unsigned char* __restrict__ w1_fp8) {
  // One block materializes BOTH members of an antithetic pair: the
  // Philox + Box-Muller draw (the kernel's measured bottleneck,
  // NOTES_ROUND2 #2) runs once per pair instead of once per member.
  // Identical outputs: member 2k writes theta + sigma*eps, member 2k+1
  // writes theta - sigma*eps from the same eps.
  const uint32_t iter = *iterp;
  const int member0 = member_offset + 2 * blockIdx.y;  // even member
  const uint32_t pair = (uint32_t)(member0 >> 1);
  __hip_bfloat16* outp = wpert + (size_t)(2 * blockIdx.y) * np_pad;
  __hip_bfloat16* outm = outp + np_pad;
  unsigned char* out8p =
      w3_fp8 ? w3_fp8 + (size_t)(2 * blockIdx.y) * (FCU * NFLAT) : nullptr;
  unsigned char* out8m = out8p ? out8p + (FCU * NFLAT) : nullptr;
  unsigned char* out8w1p =
      w1_fp8 ? w1_fp8 + (size_t)(2 * blockIdx.y) * (C1 * 256) : nullptr;
  unsigned char* out8w1m = out8w1p ? out8w1p + (C1 * 256) : nullptr;
  const int jb0 = blockIdx.x * blockDim.x + threadIdx.x;
  const int stride = gridDim.x * blockDim.x;
  for (int jb = jb0; jb * 4 < nparams; jb += stride) {
    float z[4];
    fam_normal4(seed, iter, pair, (uint32_t)jb, FAM_TAG_NOISE, 0u, z);
    const int j0 = jb * 4;
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int j = j0 + u;
      if (j < nparams) {
        const float sz = sigma * z[u];
        const float vp = theta[j] + sz;
        const float vm = theta[j] - sz;
        outp[j] = __float2bfloat16(vp);
        outm[j] = __float2bfloat16(vm);
        // the fc layer consumes W3 in OCP e4m3 (fp8) — halves the
        // dominant HBM term of the rollout (profiles: conv_fc is
        // BW-bound on single-use weights)
        if (out8p && j >= COFF_W3 && j < COFF_B3) {
          out8p[j - COFF_W3] = __hip_fp8_e4m3(vp).__x;
          out8m[j - COFF_W3] = __hip_fp8_e4m3(vm).__x;
        }
        if (out8w1p && j < COFF_B1) {
          out8w1p[j] = __hip_fp8_e4m3(vp).__x;
          out8w1m[j] = __hip_fp8_e4m3(vm).__x;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// conv_env_init: state[b][e][d] = 0.3*z  (member-independent, like MLP)
// ---------------------------------------------------------------------------
extern "C" __global__ void conv_env_init(uint32_t seed,
                                         const uint32_t* __restrict__ iterp,
                                         int nmembers,
                                         float* __restrict__ state,
                                         float* __restrict__ racc) {
  const uint32_t iter = *iterp;
  const int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= nmembers * CENV) return;
  const int e = idx % CENV;
  float z[4];
  fam_normal4(seed, iter, (uint32_t)e, 0u, FAM_TAG_ENV, 0u, z);
#pragma unroll
  for (int d = 0; d < SDIM; ++d) state[idx * SDIM + d] = 0.3f * z[d];
  racc[idx] = 0.f;
}

// ---------------------------------------------------------------------------
// conv_noisegen + conv_obsgen: obs[b][e][y][x][ic] =
//     fp8( 0.52*noise(e,t,pos,ic) + state[b][e][ic] * gtab[y][x] )
//
// The noise is keyed by (env, position, t, iter) ONLY — common random
// numbers across the whole ES population, so every member of a chunk
// sees the same scenery draw.  Exploit that: conv_noisegen runs the
// philox-7 chains ONCE per (chunk, t) into a 16-env fp32 buffer
// (1/nmembers of the RNG work obsgen used to redo per member), and
// conv_obsgen becomes a pure bandwidth kernel that combines the shared
// noise with each member's state.  The arithmetic expression and
// rounding points are unchanged, so results are bit-identical to the
// fused version (and to conv_rollout_reference).
// ---------------------------------------------------------------------------
// Work is FLATTENED across (env, position-quad) so the kernel's duration
// is one 4-draw philox chain, not a 1764-quad serial loop: this kernel
// sits on the critical path ahead of every obsgen, so its latency (not
// its total work) is what matters.  4 positions per thread -> 4
// independent philox chains in flight (the mad_u64 round chain is
// serial; cross-position ILP fills the pipe).  IMG*IMG = 7056 = 1764
// quads per env.
//
// The scaled noise 0.52*z is staged as e4m3 (one byte per channel): the
// obs themselves are e4m3-quantized right after the state term is
// added, so the extra rounding is below the obs quantization step, and
// it keeps the whole staging field at E x 7056 x 4 B = 113 KB — fully
// L2-resident for every obsgen workgroup of the chunk.  The reference
// mirror (conv_rollout_reference) applies the same double rounding.
#define NOISE_QUADS (IMG * IMG / 4)
extern "C" __global__ void conv_noisegen(uint32_t seed,
                                         const uint32_t* __restrict__ iterp,
                                         uint32_t t,
                                         unsigned char* __restrict__ znoise) {
  const uint32_t iter = *iterp;
  const int gid = blockIdx.x * blockDim.x + threadIdx.x;
  if (gid >= CENV * NOISE_QUADS) return;
  const int e = gid / NOISE_QUADS;
  const int p = (gid % NOISE_QUADS) * 4;
  unsigned char* out = znoise + (size_t)e * (IMG * IMG * CIN);
  union {
    unsigned char b[16];
    uint32_t w[4];
  } pk;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    float z[4];
    fam_uniform4(seed, iter, (uint32_t)e, (uint32_t)(p + q), FAM_TAG_OBS,
                 t, z);
#pragma unroll
    for (int c = 0; c < CIN; ++c)
      pk.b[q * 4 + c] = __hip_fp8_e4m3(0.52f * z[c]).__x;
  }
  *reinterpret_cast<fam_uint4*>(&out[p * CIN]) =
      *reinterpret_cast<fam_uint4*>(pk.w);
}

// ---------------------------------------------------------------------------
// conv_layer1: act1 = tanh(conv(obs, W1) + b1); one wg per (member, env).
// M=16 (one tile), N=400 (25 tiles), K=256 (8 tiles = 8 kernel rows).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
conv_layer1(const __hip_bfloat16* __restrict__ wpert,
            const unsigned char* __restrict__ w1_fp8,
            const float* __restrict__ state,
            const __hip_bfloat16* __restrict__ gtab,
            const unsigned char* __restrict__ znoise, int nenv_total,
            __hip_bfloat16* __restrict__ act1) {
  __shared__ alignas(16) unsigned char w1[C1][256];
  __shared__ float b1[C1];
  // The observation image is GENERATED straight into LDS: the per-pixel
  // obs = fp8(noise + state*gtab) expression is computed once here from
  // the shared (L2-resident) e4m3 noise field — the 28 KB obs image
  // never exists in HBM at all (it used to cost a write + a read per
  // workgroup plus a whole kernel).  The 8x8-stride-4 im2col fragment
  // reads then hit LDS.
  __shared__ alignas(16) unsigned char obsh[IMG * IMG * CIN];
  const int be = blockIdx.x;
  const int member = be / CENV;
  const int e = be % CENV;
  const __hip_bfloat16* wm = wpert + (size_t)member * NP_CONV_PAD;
  const unsigned char* zn = znoise + (size_t)e * (IMG * IMG * CIN);
  __hip_bfloat16* out = act1 + (size_t)be * (O1 * O1 * C1);

  const int tid = threadIdx.x;
  for (int i = tid; i < C1 * 256 / 8; i += blockDim.x) {
    reinterpret_cast<fp8x8*>(&w1[0][0])[i] =
        reinterpret_cast<const fp8x8*>(w1_fp8 +
                                       (size_t)member * (C1 * 256))[i];
  }
  {
    float s0 = state[be * SDIM + 0], s1 = state[be * SDIM + 1];
    float s2 = state[be * SDIM + 2], s3 = state[be * SDIM + 3];
    for (int p = tid * 4; p + 3 < IMG * IMG; p += blockDim.x * 4) {
      fam_uint4 pk;
      uint32_t* pw = &pk.x;
      const fam_uint4 zraw =
          *reinterpret_cast<const fam_uint4*>(&zn[p * CIN]);
      const uint32_t* zw = &zraw.x;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const float g = __bfloat162float(gtab[p + q]);
        const float r0 = __builtin_amdgcn_cvt_f32_fp8(zw[q], 0) + s0 * g;
        const float r1 = __builtin_amdgcn_cvt_f32_fp8(zw[q], 1) + s1 * g;
        const float r2 = __builtin_amdgcn_cvt_f32_fp8(zw[q], 2) + s2 * g;
        const float r3 = __builtin_amdgcn_cvt_f32_fp8(zw[q], 3) + s3 * g;
        uint32_t w = __builtin_amdgcn_cvt_pk_fp8_f32(r0, r1, 0u, false);
        pw[q] = __builtin_amdgcn_cvt_pk_fp8_f32(r2, r3, w, true);
      }
      *reinterpret_cast<fam_uint4*>(&obsh[p * CIN]) = pk;
    }
  }
  if (tid < C1) b1[tid] = __bfloat162float(wm[COFF_B1 + tid]);
  __syncthreads();

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kgrp = lane >> 4;
  const int arow = lane & 15;  // output channel

  fp8x8 afrag[8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk)
    afrag[kk] =
        *reinterpret_cast<const fp8x8*>(&w1[arow][kk * 32 + kgrp * 8]);

  // two N-tiles in flight per wave so the scattered 8 B obs loads of one
  // tile hide under the other's MFMAs (latency-bound otherwise)
  const int c = lane & 15;
  for (int nt = wave; nt < 25; nt += 8) {
    const int nt2 = nt + 4;
    const bool two = nt2 < 25;
    const int pos0 = nt * 16 + c, oy0 = pos0 / O1, ox0 = pos0 % O1;
    const int p1 = two ? nt2 * 16 + c : pos0;
    const int oy1 = p1 / O1, ox1 = p1 % O1;
    f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
    f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {  // ky = kk
      const fp8x8 b0 = *reinterpret_cast<const fp8x8*>(
          &obsh[(((oy0 * 4 + kk) * IMG) + ox0 * 4 + kgrp * 2) * CIN]);
      const fp8x8 b1f = *reinterpret_cast<const fp8x8*>(
          &obsh[(((oy1 * 4 + kk) * IMG) + ox1 * 4 + kgrp * 2) * CIN]);
      acc0 = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag[kk], b0,
                                                        acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag[kk], b1f,
                                                        acc1, 0, 0, 0);
    }
    const int drow = kgrp * 4;  // output-channel base for this lane's D
    union {
      __hip_bfloat16 h[4];
      unsigned long long u;
    } pk;
#pragma unroll
    for (int ri = 0; ri < 4; ++ri)
      pk.h[ri] = __float2bfloat16(fast_tanh_c(acc0[ri] + b1[drow + ri]));
    *reinterpret_cast<unsigned long long*>(&out[pos0 * C1 + drow]) = pk.u;
    if (two) {
#pragma unroll
      for (int ri = 0; ri < 4; ++ri)
        pk.h[ri] =
            __float2bfloat16(fast_tanh_c(acc1[ri] + b1[drow + ri]));
      *reinterpret_cast<unsigned long long*>(&out[p1 * C1 + drow]) = pk.u;
    }
  }
}

// ---------------------------------------------------------------------------
// conv_layer2: act2 = tanh(conv(act1, W2) + b2); one wg per (member, env).
// M=32 (2 tiles), N=81 (6 tiles, padded), K=256 (8 tiles = 2 pixels each).
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
conv_layer2(const __hip_bfloat16* __restrict__ wpert,
            const __hip_bfloat16* __restrict__ act1, int nenv_total,
            unsigned char* __restrict__ act2) {
  __shared__ alignas(16) __hip_bfloat16 w2[C2][256];
  __shared__ float b2[C2];
  // act1 staged in LDS: the 4x4-stride-2 im2col pattern touches each
  // input pixel from up to 4 windows and BOTH M-tiles — ~7.7x read
  // amplification that was going to L2/HBM (12.8 KB staged once
  // instead; w2 16 KB + act1 12.8 KB still leaves 4 wg/CU).
  __shared__ alignas(16) __hip_bfloat16 a1sh[O1 * O1 * C1];
  const int be = blockIdx.x;
  const int member = be / CENV;
  const __hip_bfloat16* wm = wpert + (size_t)member * NP_CONV_PAD;
  const __hip_bfloat16* in = act1 + (size_t)be * (O1 * O1 * C1);
  unsigned char* out = act2 + (size_t)be * NFLAT;

  const int tid = threadIdx.x;
  for (int i = tid; i < C2 * 256 / 8; i += blockDim.x) {
    reinterpret_cast<bf16x8*>(&w2[0][0])[i] =
        reinterpret_cast<const bf16x8*>(wm + COFF_W2)[i];
  }
  for (int i = tid; i < O1 * O1 * C1 / 8; i += blockDim.x)
    reinterpret_cast<bf16x8*>(a1sh)[i] =
        reinterpret_cast<const bf16x8*>(in)[i];
  if (tid < C2) b2[tid] = __bfloat162float(wm[COFF_B2 + tid]);
  __syncthreads();

  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kgrp = lane >> 4;

  // 12 tile jobs: job = mt*6 + ntile
  for (int job = wave; job < 12; job += 4) {
    const int mt = job / 6, nt = job % 6;
    const int arow = mt * 16 + (lane & 15);
    int pos = nt * 16 + (lane & 15);
    const bool valid = pos < O2 * O2;
    if (!valid) pos = 0;
    const int oy = pos / O2, ox = pos % O2;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      // K-tile kk covers pixels (ky = kk/2, kx = (kk%2)*2 + {0,1})
      const int ky = kk >> 1;
      const int kx0 = (kk & 1) * 2;
      const int pix = ((oy * 2 + ky) * O1) + ox * 2 + kx0 + (kgrp >> 1);
      const bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
          &w2[arow & 31][kk * 32 + kgrp * 8]);
      const bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
          &a1sh[pix * C1 + (kgrp & 1) * 8]);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0,
                                                    0, 0);
    }
    const int drow = mt * 16 + kgrp * 4;
    const int dpos = nt * 16 + (lane & 15);
    if (dpos < O2 * O2) {
      union {
        unsigned char b[4];
        uint32_t u;
      } pk;
#pragma unroll
      for (int ri = 0; ri < 4; ++ri)
        pk.b[ri] = __hip_fp8_e4m3(
            fast_tanh_c(acc[ri] + b2[(drow + ri) & 31])).__x;
      *reinterpret_cast<uint32_t*>(&out[dpos * C2 + (drow & 31)]) = pk.u;
    }
  }
}

// ---------------------------------------------------------------------------
// conv_fc: act3 = tanh(W3 @ act2_flat + b3); one wg per MEMBER.
// M=256 (16 tiles), N=16 (the member's envs), K=2592 (81 tiles).
// Weights are read straight from HBM (single use), activations too.
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
conv_fc(const __hip_bfloat16* __restrict__ wpert,
        const unsigned char* __restrict__ w3_fp8,
        const unsigned char* __restrict__ act2, int nmembers,
        __hip_bfloat16* __restrict__ act3) {
  const int member = blockIdx.x;
  const __hip_bfloat16* wm = wpert + (size_t)member * NP_CONV_PAD;
  const unsigned char* w3 = w3_fp8 + (size_t)member * (FCU * NFLAT);
  const unsigned char* in = act2 + (size_t)member * CENV * NFLAT;
  __hip_bfloat16* out = act3 + (size_t)member * CENV * FCU;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kgrp = lane >> 4;
  const int env = lane & 15;  // B column

  // grid = (members, 4): each workgroup owns 4 of the 16 M-tiles, one
  // per wave — 4x the workgroups of the one-wg-per-member mapping.
  // Profiling showed this kernel latency-bound on the small 8 B
  // fragment loads at 2 wg/CU, not HBM-BW-bound; more resident waves +
  // 4-deep load batching is the fix.
  const int mt = blockIdx.y * 4 + wave;
  const int arow = mt * 16 + (lane & 15);
  const unsigned char* ap = &w3[(size_t)arow * NFLAT + kgrp * 8];
  // NOTE: register double-buffering and dual accumulators were tried
  // and measured SLOWER (369.5K / 368.3K vs 373.4K whole-bench) — at
  // 4 wg/CU the load bursts are wave-parallel-hidden.  What DOES pay
  // is LDS-staging the B operand (act2): all 16 waves of a member used
  // to re-pull the same 41.5 KB from L2/HBM while the W stream was
  // evicting it.  B is staged in 27-tile k-chunks (16 envs x 864 k,
  // padded to 872 B rows so the 16 env-columns of a fragment read hit
  // 16 distinct LDS banks), double-buffered: 28 KB LDS keeps 4 wg/CU.
  constexpr int kChunkK = 864;          // 27 of 81 k-tiles
  constexpr int kRowPad = 872;          // bank-staggered env row stride
  __shared__ alignas(16) unsigned char bsh[2][CENV * kRowPad];
  auto stage = [&](int buf, int c0) {
    // 16 envs x 864 B = 1728 8-byte words, cooperatively
    for (int w8 = tid; w8 < CENV * (kChunkK / 8); w8 += 256) {
      const int e8 = w8 / (kChunkK / 8);
      const int j8 = w8 % (kChunkK / 8);
      *reinterpret_cast<fp8x8*>(&bsh[buf][e8 * kRowPad + j8 * 8]) =
          *reinterpret_cast<const fp8x8*>(
              &in[(size_t)e8 * NFLAT + c0 + j8 * 8]);
    }
  };
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  stage(0, 0);
  __syncthreads();
  for (int c = 0; c < 3; ++c) {
    if (c + 1 < 3) stage((c + 1) & 1, (c + 1) * kChunkK);
    const unsigned char* bl = &bsh[c & 1][env * kRowPad + kgrp * 8];
    const int kt0 = c * 27;
#pragma unroll 1
    for (int kk = 0; kk < 24; kk += 8) {
      fp8x8 a[8], b[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        a[u] = *reinterpret_cast<const fp8x8*>(ap + (kt0 + kk + u) * 32);
        b[u] = *reinterpret_cast<const fp8x8*>(bl + (kk + u) * 32);
      }
#pragma unroll
      for (int u = 0; u < 8; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a[u], b[u], acc,
                                                         0, 0, 0);
    }
    {  // chunk tail: tiles 24..26
      fp8x8 a[3], b[3];
#pragma unroll
      for (int u = 0; u < 3; ++u) {
        a[u] = *reinterpret_cast<const fp8x8*>(ap + (kt0 + 24 + u) * 32);
        b[u] = *reinterpret_cast<const fp8x8*>(bl + (24 + u) * 32);
      }
#pragma unroll
      for (int u = 0; u < 3; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a[u], b[u], acc,
                                                         0, 0, 0);
    }
    __syncthreads();
  }
  const int drow = mt * 16 + kgrp * 4;
  union {
    __hip_bfloat16 h[4];
    unsigned long long u;
  } pk;
#pragma unroll
  for (int ri = 0; ri < 4; ++ri) {
    const float bias = __bfloat162float(wm[COFF_B3 + drow + ri]);
    pk.h[ri] = __float2bfloat16(fast_tanh_c(acc[ri] + bias));
  }
  // D col = env, rows drow..drow+3 -> act3[env][drow..]
  *reinterpret_cast<unsigned long long*>(&out[env * FCU + drow]) = pk.u;
}

// ---------------------------------------------------------------------------
// conv_head_env: logits -> argmax action -> synthetic env step -> reward.
// One wg per member; one wave per 4 envs (16 lanes per env).
// ---------------------------------------------------------------------------
extern "C" __global__ void conv_head_env(
    const __hip_bfloat16* __restrict__ wpert,
    const __hip_bfloat16* __restrict__ act3, int nmembers,
    const float* __restrict__ env_A, const float* __restrict__ env_B,
    float* __restrict__ state, float* __restrict__ racc) {
  const int member = blockIdx.x;
  const __hip_bfloat16* wm = wpert + (size_t)member * NP_CONV_PAD;
  const __hip_bfloat16* a3 = act3 + (size_t)member * CENV * FCU;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int env = wave * 4 + (lane >> 4);       // 0..15
  const int usub = lane & 15;                   // 16 unit groups of 16
  const int be = member * CENV + env;

  float part[NACT];
#pragma unroll
  for (int a = 0; a < NACT; ++a) part[a] = 0.f;
  for (int uu = 0; uu < 16; ++uu) {
    const int u = usub * 16 + uu;
    const float hv = __bfloat162float(a3[env * FCU + u]);
#pragma unroll
    for (int a = 0; a < NACT; ++a)
      part[a] += __bfloat162float(wm[COFF_W4 + a * FCU + u]) * hv;
  }
  // reduce across the 16 lanes of this env group
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) {
#pragma unroll
    for (int a = 0; a < NACT; ++a)
      part[a] += __shfl_down(part[a], off, 16);
  }
  if (usub == 0) {
    int best = 0;
    float bestv = part[0] + __bfloat162float(wm[COFF_B4]);
#pragma unroll
    for (int a = 1; a < NACT; ++a) {
      const float v = part[a] + __bfloat162float(wm[COFF_B4 + a]);
      if (v > bestv) {
        bestv = v;
        best = a;
      }
    }
    const float force = ((float)best - 2.5f) * 0.4f;  // in [-1, 1]
    float s[SDIM], snew[SDIM];
#pragma unroll
    for (int d = 0; d < SDIM; ++d) s[d] = state[be * SDIM + d];
    float sq = 0.f;
#pragma unroll
    for (int d = 0; d < SDIM; ++d) {
      float drive = 0.f;
#pragma unroll
      for (int e2 = 0; e2 < SDIM; ++e2) drive += env_A[d * SDIM + e2] * s[e2];
      snew[d] = 0.97f * s[d] + 0.08f * fast_tanh_c(drive) +
                0.05f * env_B[d] * force;
      sq += snew[d] * snew[d];
    }
#pragma unroll
    for (int d = 0; d < SDIM; ++d) state[be * SDIM + d] = snew[d];
    racc[be] += 1.f - 0.1f * sq;
  }
}
