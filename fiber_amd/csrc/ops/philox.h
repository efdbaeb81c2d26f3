// Counter-based RNG for the ES noise table (device + host identical).
//
// Philox4x32-10 (public algorithm, Salmon et al. 2011) + Box-Muller.
// The ES engine never materializes the noise table in HBM: perturbations
// are regenerated on the fly from (seed, iteration, pair, index-block)
// counters, in both the rollout kernel (apply +/- sigma*eps) and the
// gradient kernel (accumulate rank-weighted eps).  This is the
// MI355X-native replacement for the reference workload's shared noise
// table.  A numpy mirror lives in
// fiber_amd/es/philox_ref.py for CPU-side verification.

#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define FAM_PHILOX_M0 0xD2511F53u
#define FAM_PHILOX_M1 0xCD9E8D57u
#define FAM_PHILOX_W0 0x9E3779B9u
#define FAM_PHILOX_W1 0xBB67AE85u

struct fam_uint4 {
  uint32_t x, y, z, w;
};

__device__ __host__ inline uint32_t fam_mulhi(uint32_t a, uint32_t b) {
#ifdef __HIP_DEVICE_COMPILE__
  return __umulhi(a, b);
#else
  return (uint32_t)(((uint64_t)a * (uint64_t)b) >> 32);
#endif
}

template <int ROUNDS>
__device__ __host__ inline fam_uint4 philox4x32_r(uint32_t k0, uint32_t k1,
                                                  uint32_t c0, uint32_t c1,
                                                  uint32_t c2, uint32_t c3) {
#pragma unroll
  for (int r = 0; r < ROUNDS; ++r) {
    // Single 64-bit multiplies: the GPU backend lowers these to one
    // v_mad_u64_u32 each (hi+lo together) instead of a v_mul_hi_u32 +
    // v_mul_lo_u32 pair — the mulhi chains dominate the philox-bound
    // kernels (obsgen PMC: 102% VALUBusy).
    uint64_t p0 = (uint64_t)FAM_PHILOX_M0 * c0;
    uint64_t p1 = (uint64_t)FAM_PHILOX_M1 * c2;
    uint32_t hi0 = (uint32_t)(p0 >> 32);
    uint32_t lo0 = (uint32_t)p0;
    uint32_t hi1 = (uint32_t)(p1 >> 32);
    uint32_t lo1 = (uint32_t)p1;
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0;
    c1 = n1;
    c2 = n2;
    c3 = n3;
    k0 += FAM_PHILOX_W0;
    k1 += FAM_PHILOX_W1;
  }
  return {c0, c1, c2, c3};
}

__device__ __host__ inline fam_uint4 philox4x32_10(uint32_t k0, uint32_t k1,
                                                   uint32_t c0, uint32_t c1,
                                                   uint32_t c2, uint32_t c3) {
  return philox4x32_r<10>(k0, k1, c0, c1, c2, c3);
}

// 4 uniforms in [-1, 1) from one philox draw (no transcendentals —
// used for the pixel-observation noise where gaussianity is not needed
// and Box-Muller's log/sqrt/sincos dominated the obsgen kernel).
// 7 rounds: the minimum Philox4x32 configuration that passes BigCrush
// (Salmon et al. 2011, Table 2) — used for the bulk pixel-obs noise
// where the 32-bit mulhi chains dominate the obsgen kernel's VALU time.
// ES perturbations keep the default 10 rounds.
__device__ __host__ inline void fam_uniform4(uint32_t k0, uint32_t k1,
                                             uint32_t c0, uint32_t c1,
                                             uint32_t c2, uint32_t c3,
                                             float z[4]) {
  fam_uint4 u = philox4x32_r<7>(k0, k1, c0, c1, c2, c3);
  const float inv31 = 4.656612873077393e-10f;  // 2^-31
  z[0] = (float)(int32_t)u.x * inv31;
  z[1] = (float)(int32_t)u.y * inv31;
  z[2] = (float)(int32_t)u.z * inv31;
  z[3] = (float)(int32_t)u.w * inv31;
}

// 4 standard normals from one philox draw (two Box-Muller pairs).
// u in (0,1]: (x + 1) * 2^-32.
__device__ __host__ inline void fam_normal4(uint32_t k0, uint32_t k1,
                                            uint32_t c0, uint32_t c1,
                                            uint32_t c2, uint32_t c3,
                                            float z[4]) {
  fam_uint4 u = philox4x32_10(k0, k1, c0, c1, c2, c3);
  const float two32_inv = 2.3283064365386963e-10f;  // 2^-32
  const float twopi = 6.2831853071795864769f;
  float u0 = ((float)u.x + 1.0f) * two32_inv;
  float u1 = ((float)u.y + 1.0f) * two32_inv;
  float u2 = ((float)u.z + 1.0f) * two32_inv;
  float u3 = ((float)u.w + 1.0f) * two32_inv;
  float r0 = sqrtf(-2.0f * logf(u0));
  float r1 = sqrtf(-2.0f * logf(u2));
  z[0] = r0 * cosf(twopi * u1);
  z[1] = r0 * sinf(twopi * u1);
  z[2] = r1 * cosf(twopi * u3);
  z[3] = r1 * sinf(twopi * u3);
}
