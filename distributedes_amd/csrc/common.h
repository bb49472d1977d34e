// Shared device helpers for the distributedes_amd CDNA4 (gfx950) kernels.
//
// Philox4x32-10 + Box-Muller: bit-compatible with the CPU oracle in
// core/noise.py (verified by tests/test_gpu_kernels.py).  splitmix64 and the
// member seed/offset derivation mirror core/noise.py exactly so every rank
// (and the host) can derive any member's noise slice without communication.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#define DES_WAVE 64

namespace des {

// ---------------- Philox4x32-10 ----------------
struct alignas(16) U32x4 { uint32_t x, y, z, w; };

__device__ __forceinline__ U32x4 philox4x32_10(uint64_t block, uint64_t key64) {
  uint32_t c0 = (uint32_t)(block & 0xFFFFFFFFu);
  uint32_t c1 = (uint32_t)(block >> 32);
  uint32_t c2 = 0u, c3 = 0u;
  uint32_t k0 = (uint32_t)(key64 & 0xFFFFFFFFu);
  uint32_t k1 = (uint32_t)(key64 >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
    uint32_t hi0 = __umulhi(c0, M0), lo0 = c0 * M0;
    uint32_t hi1 = __umulhi(c2, M1), lo1 = c2 * M1;
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
  }
  return {c0, c1, c2, c3};
}

// 4 standard normals for Philox block `b` (matches _normals_from_blocks).
__device__ __forceinline__ float4 philox_normal4(uint64_t block, uint64_t key64) {
  U32x4 r = philox4x32_10(block, key64);
  const float inv = 2.3283064365386963e-10f;  // 2^-32
  float u0 = ((float)r.x + 0.5f) * inv;
  float u1 = ((float)r.y + 0.5f) * inv;
  float u2 = ((float)r.z + 0.5f) * inv;
  float u3 = ((float)r.w + 0.5f) * inv;
  float r0 = sqrtf(-2.0f * logf(u0));
  float r1 = sqrtf(-2.0f * logf(u2));
  const float tau = 6.2831853071795864f;
  float s0, c0, s1, c1;
  __sincosf(tau * u1, &s0, &c0);
  __sincosf(tau * u3, &s1, &c1);
  return {r0 * c0, r0 * s0, r1 * c1, r1 * s1};
}

// ---------------- splitmix64 (== core/noise.py) ----------------
__device__ __host__ __forceinline__ uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

__device__ __host__ __forceinline__ uint64_t member_seed(uint64_t run_seed,
                                                         uint64_t gen,
                                                         uint64_t pair) {
  uint64_t a = splitmix64(run_seed ^ (gen * 0x9E3779B97F4A7C15ull));
  return splitmix64(a ^ (pair * 0xBF58476D1CE4E5B9ull));
}

__device__ __host__ __forceinline__ int64_t member_offset(uint64_t run_seed,
                                                          uint64_t gen,
                                                          uint64_t pair,
                                                          int64_t table_size,
                                                          int64_t dim) {
  int64_t span = table_size - dim;
  uint64_t s = member_seed(run_seed, gen, pair);
  if (span >= 16) return (int64_t)(s % (uint64_t)(span / 16 + 1)) * 16;
  return (int64_t)(s % (uint64_t)span);
}

// ---------------- env hash (== envs/base.py uniform_from_counter) ----------
__device__ __forceinline__ float hash_uniform(int64_t counter, float lo, float hi) {
  uint64_t x = (uint64_t)counter;
  x = (x ^ (x >> 17)) & 0x3FFFFFFFull;
  x = (x * 0x9E3779B1ull) & 0x3FFFFFFFFFFFull;
  x = (x ^ (x >> 13)) & 0x3FFFFFFFull;
  x = (x * 0x85EBCA77ull) & 0x3FFFFFFFFFFFull;
  float u = (float)((x >> 10) & 0xFFFFFFull) * (1.0f / 16777216.0f);
  return lo + (hi - lo) * u;
}

// full-wave sum reduction; every lane returns the sum
__device__ __forceinline__ float wave_reduce_sum_f(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

// ---------------- bf16 helpers ----------------
__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  uint32_t b = __float_as_uint(f);
  // round-to-nearest-even (matches torch's float->bfloat16 cast)
  uint32_t rounded = b + 0x7FFFu + ((b >> 16) & 1u);
  return (uint16_t)(rounded >> 16);
}

__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
  return __uint_as_float(((uint32_t)h) << 16);
}

// packed bf16 pair + v_dot2c_f32_bf16 (2 MACs/instruction, no unpack)
typedef __attribute__((ext_vector_type(2))) short bf16x2;

__device__ __forceinline__ float dot2(bf16x2 a, bf16x2 b, float c) {
  return __builtin_amdgcn_fdot2_f32_bf16(a, b, c, false);
}

__device__ __forceinline__ bf16x2 pack_bf16(float lo, float hi) {
  bf16x2 r;
  r.x = (short)f32_to_bf16(lo);
  r.y = (short)f32_to_bf16(hi);
  return r;
}

}  // namespace des

#define HIP_CHECK(expr)                                                  \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, \
             __LINE__);                                                  \
    }                                                                    \
  } while (0)
