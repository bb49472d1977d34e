// K12: normalizer_update — running obs mean/var accumulation (Welford-style
// moment sums), fp32 block partials -> one fp64 atomic per dim per block
// (Guideline 12: per-block partial reduction before atomics).
#include <algorithm>

#include "common.h"

namespace {

constexpr int NBLOCKS = 256;

__global__ void normalizer_observe_kernel(const float* __restrict__ obs,
                                          double* __restrict__ sum,
                                          double* __restrict__ sumsq, int P,
                                          int dim) {
  // block b owns rows [b*rows_per, ...); thread t accumulates dims t, t+256
  int b = blockIdx.x;
  int rows_per = (P + gridDim.x - 1) / gridDim.x;
  int r0 = b * rows_per, r1 = min(r0 + rows_per, P);
  for (int d = threadIdx.x; d < dim; d += blockDim.x) {
    float s = 0.f, sq = 0.f;
    for (int r = r0; r < r1; ++r) {
      float v = obs[(int64_t)r * dim + d];
      s += v;
      sq += v * v;
    }
    if (r1 > r0) {
      atomicAdd(sum + d, (double)s);
      atomicAdd(sumsq + d, (double)sq);
    }
  }
}

}  // namespace

extern "C" void des_launch_normalizer_observe(const float* obs, double* sum,
                                              double* sumsq, int P, int dim,
                                              hipStream_t stream) {
  int g = std::min(NBLOCKS, P);
  hipLaunchKernelGGL(normalizer_observe_kernel, dim3(g), dim3(256), 0, stream,
                     obs, sum, sumsq, P, dim);
}
