// K12: normalizer_update — running obs mean/var accumulation (Welford-style
// moment sums) in two stages:
//   stage 1: per-block fp32 partials (float4 row segments, no atomics)
//   stage 2: 256-partial tree reduce -> ONE fp64 add per dim
// The round-1 single-kernel version issued 256 fp64 HBM atomics per dim
// (contention-serialized) and scalar loads; at pop=10000x376 it ran ~11x
// over its bandwidth floor.
#include <algorithm>

#include "common.h"

namespace {

constexpr int NBLOCKS = 256;

// partials layout: [NBLOCKS][2][dim] fp32 (sum, sumsq)
__global__ void observe_partial_kernel(const float* __restrict__ obs,
                                       float* __restrict__ partials, int P,
                                       int dim) {
  int b = blockIdx.x;
  int rows_per = (P + gridDim.x - 1) / gridDim.x;
  int r0 = b * rows_per, r1 = min(r0 + rows_per, P);
  float* ps = partials + (int64_t)b * 2 * dim;
  float* pq = ps + dim;
  int d4n = dim >> 2;
  // vector path: thread t owns float4 column t (dims 4t..4t+3)
  for (int d4 = threadIdx.x; d4 < d4n; d4 += blockDim.x) {
    float4 s = {0.f, 0.f, 0.f, 0.f}, q = {0.f, 0.f, 0.f, 0.f};
    for (int r = r0; r < r1; ++r) {
      float4 v = *reinterpret_cast<const float4*>(obs + (int64_t)r * dim +
                                                  (d4 << 2));
      s.x += v.x; s.y += v.y; s.z += v.z; s.w += v.w;
      q.x += v.x * v.x; q.y += v.y * v.y; q.z += v.z * v.z; q.w += v.w * v.w;
    }
    // accumulate across the generation's steps; the reduce kernel runs
    // ONCE per generation (flush), not per step
    float4 o0 = *reinterpret_cast<const float4*>(ps + (d4 << 2));
    float4 o1 = *reinterpret_cast<const float4*>(pq + (d4 << 2));
    s.x += o0.x; s.y += o0.y; s.z += o0.z; s.w += o0.w;
    q.x += o1.x; q.y += o1.y; q.z += o1.z; q.w += o1.w;
    *reinterpret_cast<float4*>(ps + (d4 << 2)) = s;
    *reinterpret_cast<float4*>(pq + (d4 << 2)) = q;
  }
  // scalar tail dims
  for (int d = (d4n << 2) + threadIdx.x; d < dim; d += blockDim.x) {
    float s = 0.f, q = 0.f;
    for (int r = r0; r < r1; ++r) {
      float v = obs[(int64_t)r * dim + d];
      s += v;
      q += v * v;
    }
    ps[d] += s;
    pq[d] += q;
  }
}

// block = 8 b-groups x 32 dims: consecutive threads read consecutive dims
// of one partial row (coalesced), fp64 accumulation, LDS cross-group reduce
__global__ void observe_reduce_kernel(const float* __restrict__ partials,
                                      double* __restrict__ sum,
                                      double* __restrict__ sumsq, int nblocks,
                                      int dim) {
  __shared__ double red[8][32];
  int phase = blockIdx.y;  // 0 = sum, 1 = sumsq
  int dl = threadIdx.x & 31, bg = threadIdx.x >> 5;
  int d = blockIdx.x * 32 + dl;
  double s = 0.0;
  if (d < dim)
    for (int b = bg; b < nblocks; b += 8)
      s += (double)partials[((int64_t)b * 2 + phase) * dim + d];
  red[bg][dl] = s;
  __syncthreads();
  if (bg == 0 && d < dim) {
    double tot = 0.0;
#pragma unroll
    for (int g = 0; g < 8; ++g) tot += red[g][dl];
    (phase ? sumsq : sum)[d] += tot;
  }
}

}  // namespace

extern "C" void des_launch_normalizer_observe(const float* obs, float* scratch,
                                              int P, int dim,
                                              hipStream_t stream) {
  int g = std::min(NBLOCKS, P);
  hipLaunchKernelGGL(observe_partial_kernel, dim3(g), dim3(256), 0, stream,
                     obs, scratch, P, dim);
}

extern "C" void des_launch_normalizer_observe_flush(const float* scratch,
                                                    double* sum, double* sumsq,
                                                    int P, int dim,
                                                    hipStream_t stream) {
  int g = std::min(NBLOCKS, P);
  hipLaunchKernelGGL(observe_reduce_kernel, dim3((dim + 31) / 32, 2),
                     dim3(256), 0, stream, scratch, sum, sumsq, g, dim);
}
