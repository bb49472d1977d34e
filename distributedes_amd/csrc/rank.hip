// K9: centered_rank — fitness -> centered-rank weights u in [-0.5, 0.5]
// (BASELINE north star names this a hand-written kernel).
//
// Instead of a sort, each member's rank is COUNTED:
//     rank_i = #{ j : f_j < f_i  or  (f_j == f_i and j < i) }
// which is exactly the stable-argsort rank the torch/numpy reference
// computes (ties broken by index) — so GPU and CPU agree BIT-FOR-BIT (the
// count is integer; no float reassociation).  O(P^2) compares are ~1 G ops
// at the pop=32768 stress config: trivial for 256 CUs, and every rank
// computes the identical result with no cross-rank communication.
//
// Shape: block = 256 threads, thread i owns one member; j loops over
// LDS-staged fitness tiles (coalesced loads, conflict-free broadcast-ish
// reads).  u_i = rank_i / (P-1) - 0.5 written directly.
#include "common.h"

namespace {

constexpr int TILE = 1024;

__global__ __launch_bounds__(256) void centered_rank_kernel(
    const float* __restrict__ fitness, float* __restrict__ u, int P) {
  __shared__ float ftile[TILE];
  int i = blockIdx.x * 256 + threadIdx.x;
  float fi = i < P ? fitness[i] : 0.f;
  int rank = 0;
  for (int j0 = 0; j0 < P; j0 += TILE) {
    int n = min(TILE, P - j0);
    __syncthreads();
    for (int e = threadIdx.x; e < n; e += 256) ftile[e] = fitness[j0 + e];
    __syncthreads();
    if (i < P) {
      for (int jj = 0; jj < n; ++jj) {
        float fj = ftile[jj];
        rank += (fj < fi) || (fj == fi && (j0 + jj) < i);
      }
    }
  }
  if (i < P) u[i] = (float)rank / (float)(P - 1) - 0.5f;
}

}  // namespace

extern "C" void des_launch_centered_rank(const float* fitness, float* u,
                                         int P, hipStream_t stream) {
  int grid = (P + 255) / 256;
  hipLaunchKernelGGL(centered_rank_kernel, dim3(grid), dim3(256), 0, stream,
                     fitness, u, P);
}
