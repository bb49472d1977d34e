#include "hip/hip_runtime.h"
// K2: perturb — theta_i = theta + sigma * sign_i * table[off_i : off_i+D]
// K10: weighted_noise_sum — acc[d] = sum_i w_i * s_i * table[off_i + d]
//                           (+ optional sum_i w_i * eps^2 for the CEM refit)
//
// Memory-bound streaming kernels (HBM ~8 TB/s peak / ~6.3 achievable):
// float4-vectorized table reads (offsets are 16-element aligned by
// core/noise.py member_offset), bf16x4 packed stores for the population
// weights.  Launch shapes sized so grid >> 256 CUs (8 XCDs fill).
#include <algorithm>

#include "common.h"
#include "gemv.h"

namespace {

// --------------------------------------------------------------------------
// perturb: 2D grid (chunks, members).  Each thread writes 4 elements.
// sigma_vec == nullptr -> scalar sigma (ES); else per-coordinate (CEM).
template <typename OutT, typename TT>
__global__ void perturb_kernel(OutT* __restrict__ out,
                               const float* __restrict__ theta,
                               const TT* __restrict__ table,
                               const int64_t* __restrict__ offsets,
                               const float* __restrict__ signs, float sigma,
                               const float* __restrict__ sigma_vec, int64_t P,
                               int64_t D, int64_t out_stride) {
  int64_t i = blockIdx.y;  // member
  if (i >= P) return;
  const float s = signs[i] * (sigma_vec ? 1.0f : sigma);
  const TT* src = table + offsets[i];
  OutT* dst = out + i * out_stride;
  // rows are normally padded to 16B (out_stride % 8 == 0) so vector stores
  // stay aligned; unpadded odd-D rows fall back to element stores
  const bool row_aligned = ((i * out_stride) & 3) == 0;
  int64_t d4 = D >> 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < d4;
       v += stride) {
    int64_t d = v << 2;
    float4 eps = des::load4(src + d);
    float4 th = *reinterpret_cast<const float4*>(theta + d);
    float4 sv = sigma_vec ? *reinterpret_cast<const float4*>(sigma_vec + d)
                          : float4{1.f, 1.f, 1.f, 1.f};
    float r0 = th.x + s * sv.x * eps.x;
    float r1 = th.y + s * sv.y * eps.y;
    float r2 = th.z + s * sv.z * eps.z;
    float r3 = th.w + s * sv.w * eps.w;
    if constexpr (sizeof(OutT) == 2) {
      uint16_t h0 = des::f32_to_bf16(r0), h1 = des::f32_to_bf16(r1);
      uint16_t h2 = des::f32_to_bf16(r2), h3 = des::f32_to_bf16(r3);
      if (row_aligned) {  // one 8-byte store
        union { uint16_t h[4]; uint64_t u; } pk = {{h0, h1, h2, h3}};
        *reinterpret_cast<uint64_t*>(dst + d) = pk.u;
      } else {
        dst[d] = h0; dst[d + 1] = h1; dst[d + 2] = h2; dst[d + 3] = h3;
      }
    } else {
      if (row_aligned) {
        *reinterpret_cast<float4*>(dst + d) = float4{r0, r1, r2, r3};
      } else {
        float* f = reinterpret_cast<float*>(dst);
        f[d] = r0; f[d + 1] = r1; f[d + 2] = r2; f[d + 3] = r3;
      }
    }
  }
  // tail (D % 4) by thread 0 of block (0, i)
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t d = d4 << 2; d < D; ++d) {
      float sv = sigma_vec ? sigma_vec[d] : 1.0f;
      float r = theta[d] + s * sv * des::load1(src + d);
      if constexpr (sizeof(OutT) == 2)
        reinterpret_cast<uint16_t*>(dst)[d] = des::f32_to_bf16(r);
      else
        reinterpret_cast<float*>(dst)[d] = r;
    }
  }
}

// --------------------------------------------------------------------------
// weighted_noise_sum: each block owns a contiguous D-chunk of the output and
// loops over all P members, streaming their noise slices (coalesced float4).
// Members with w_i == 0 are skipped (wave-uniform branch -> CEM elite subset
// costs only elite traffic).  fp32 accumulate in registers.
template <bool WANT_SQ, typename TT>
__global__ void wsum_kernel(float* __restrict__ acc, float* __restrict__ acc_sq,
                            const TT* __restrict__ table,
                            const int64_t* __restrict__ offsets,
                            const float* __restrict__ signs,
                            const float* __restrict__ weights, int64_t P,
                            int64_t D) {
  int64_t d4 = D >> 2;
  int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (v < d4) {
    int64_t d = v << 2;
    float4 a = {0.f, 0.f, 0.f, 0.f};
    float4 q = {0.f, 0.f, 0.f, 0.f};
    for (int64_t i = 0; i < P; ++i) {
      float w = weights[i];
      if (w == 0.0f) continue;
      float ws = w * signs[i];
      float4 eps = des::load4(table + offsets[i] + d);
      a.x += ws * eps.x; a.y += ws * eps.y;
      a.z += ws * eps.z; a.w += ws * eps.w;
      if constexpr (WANT_SQ) {
        q.x += w * eps.x * eps.x; q.y += w * eps.y * eps.y;
        q.z += w * eps.z * eps.z; q.w += w * eps.w * eps.w;
      }
    }
    *reinterpret_cast<float4*>(acc + d) = a;
    if constexpr (WANT_SQ) *reinterpret_cast<float4*>(acc_sq + d) = q;
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (int64_t d = d4 << 2; d < D; ++d) {
      float a = 0.f, q = 0.f;
      for (int64_t i = 0; i < P; ++i) {
        float w = weights[i];
        if (w == 0.0f) continue;
        float eps = des::load1(table + offsets[i] + d);
        a += w * signs[i] * eps;
        if constexpr (WANT_SQ) q += w * eps * eps;
      }
      acc[d] = a;
      if constexpr (WANT_SQ) acc_sq[d] = q;
    }
  }
}

}  // namespace

template <typename OutT, typename TT>
static void launch_perturb(OutT* out, const float* theta, const TT* table,
                           const int64_t* offsets, const float* signs,
                           float sigma, const float* sigma_vec, int64_t P,
                           int64_t D, int64_t out_stride, hipStream_t stream) {
  int threads = 256;
  int chunks = (int)std::min<int64_t>(((D >> 2) + threads - 1) / threads, 64);
  hipLaunchKernelGGL((perturb_kernel<OutT, TT>), dim3(chunks, (uint32_t)P),
                     dim3(threads), 0, stream, out, theta, table, offsets,
                     signs, sigma, sigma_vec, P, D, out_stride);
}

extern "C" void des_launch_perturb(void* out, bool out_bf16, const float* theta,
                        const void* table, bool table_bf16,
                        const int64_t* offsets, const float* signs,
                        float sigma, const float* sigma_vec, int64_t P,
                        int64_t D, int64_t out_stride, hipStream_t stream) {
  if (out_bf16 && table_bf16)
    launch_perturb((uint16_t*)out, theta, (const uint16_t*)table, offsets,
                   signs, sigma, sigma_vec, P, D, out_stride, stream);
  else if (out_bf16)
    launch_perturb((uint16_t*)out, theta, (const float*)table, offsets, signs,
                   sigma, sigma_vec, P, D, out_stride, stream);
  else if (table_bf16)
    launch_perturb((float*)out, theta, (const uint16_t*)table, offsets, signs,
                   sigma, sigma_vec, P, D, out_stride, stream);
  else
    launch_perturb((float*)out, theta, (const float*)table, offsets, signs,
                   sigma, sigma_vec, P, D, out_stride, stream);
}

template <typename TT>
static void launch_wsum(float* acc, float* acc_sq, const TT* table,
                        const int64_t* offsets, const float* signs,
                        const float* weights, int64_t P, int64_t D,
                        bool want_sq, hipStream_t stream) {
  int threads = 256;
  int64_t d4 = D >> 2;
  int grid = (int)std::max<int64_t>((d4 + threads - 1) / threads, 1);
  if (want_sq)
    hipLaunchKernelGGL((wsum_kernel<true, TT>), dim3(grid), dim3(threads), 0,
                       stream, acc, acc_sq, table, offsets, signs, weights, P,
                       D);
  else
    hipLaunchKernelGGL((wsum_kernel<false, TT>), dim3(grid), dim3(threads), 0,
                       stream, acc, acc_sq, table, offsets, signs, weights, P,
                       D);
}

extern "C" void des_launch_wsum(float* acc, float* acc_sq, const void* table,
                     bool table_bf16, const int64_t* offsets,
                     const float* signs, const float* weights, int64_t P,
                     int64_t D, bool want_sq, hipStream_t stream) {
  if (table_bf16)
    launch_wsum(acc, acc_sq, (const uint16_t*)table, offsets, signs, weights,
                P, D, want_sq, stream);
  else
    launch_wsum(acc, acc_sq, (const float*)table, offsets, signs, weights, P,
                D, want_sq, stream);
}


