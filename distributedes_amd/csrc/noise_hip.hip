#include "hip/hip_runtime.h"
// K1: noise_table_fill — counter-based Philox N(0,1) fill.
//
// Bit-identical across ranks given the same seed, so the table never crosses
// the wire (C2 broadcasts only the 8-byte seed).  Memory-bound fill:
// one thread = one Philox block = 4 floats, float4 stores, grid-stride.
#include <algorithm>

#include "common.h"

namespace {

__global__ void noise_fill_kernel(float* __restrict__ out, int64_t n,
                                  uint64_t seed) {
  int64_t nblocks = (n + 3) >> 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < nblocks;
       b += stride) {
    float4 z = des::philox_normal4((uint64_t)b, seed);
    int64_t e = b << 2;
    if (e + 4 <= n) {
      *reinterpret_cast<float4*>(out + e) = z;
    } else {  // tail
      float v[4] = {z.x, z.y, z.z, z.w};
      for (int i = 0; i < 4 && e + i < n; ++i) out[e + i] = v[i];
    }
  }
}

__global__ void noise_fill_bf16_kernel(uint16_t* __restrict__ out, int64_t n,
                                       uint64_t seed) {
  int64_t nblocks = (n + 3) >> 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < nblocks;
       b += stride) {
    float4 z = des::philox_normal4((uint64_t)b, seed);
    int64_t e = b << 2;
    if (e + 4 <= n) {
      union { uint16_t h[4]; uint64_t u; } pk;
      pk.h[0] = des::f32_to_bf16(z.x); pk.h[1] = des::f32_to_bf16(z.y);
      pk.h[2] = des::f32_to_bf16(z.z); pk.h[3] = des::f32_to_bf16(z.w);
      *reinterpret_cast<uint64_t*>(out + e) = pk.u;
    } else {
      float v[4] = {z.x, z.y, z.z, z.w};
      for (int i = 0; i < 4 && e + i < n; ++i)
        out[e + i] = des::f32_to_bf16(v[i]);
    }
  }
}

}  // namespace

extern "C" void des_launch_noise_fill(float* out, int64_t n, uint64_t seed,
                                      hipStream_t stream) {
  int threads = 256;
  int64_t nblocks = (n + 3) >> 2;
  int grid = (int)std::min<int64_t>((nblocks + threads - 1) / threads, 2048 * 4);
  hipLaunchKernelGGL(noise_fill_kernel, dim3(grid), dim3(threads), 0, stream,
                     out, n, seed);
}

extern "C" void des_launch_noise_fill_bf16(uint16_t* out, int64_t n,
                                           uint64_t seed, hipStream_t stream) {
  int threads = 256;
  int64_t nblocks = (n + 3) >> 2;
  int grid = (int)std::min<int64_t>((nblocks + threads - 1) / threads, 2048 * 4);
  hipLaunchKernelGGL(noise_fill_bf16_kernel, dim3(grid), dim3(threads), 0,
                     stream, out, n, seed);
}
