// Ablation probe for the IMPALA res-conv shape (CI=16, CO=16, HW=32):
// which phase of the MFMA tiled conv dominates its wall time?
//   hipcc --offload-arch=gfx950 -O3 tools/probe_impala_ablate.hip -o /tmp/pa && /tmp/pa
// Variants (template V): 0 full | 1 no A-gather | 2 no MFMA | 3 no epilogue
// | 4 halo+B staging only.  Guide §5.4 rule 17: stubbed values are kept
// live with asm so DCE cannot delete upstream work.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(8))) short mfma_ab;
typedef __attribute__((ext_vector_type(4))) float mfma_cd;

constexpr int CI = 16, CO = 16, HW = 32, TS = 16;
constexpr int TH = TS + 2, M = TS * TS, K = CI * 9, KSTEPS = (K + 31) / 32;
constexpr int NWAVES = M / 64, NTILES = (M / 16) * (CO / 16);
constexpr int MAXT = NTILES / NWAVES;

template <int V>
__global__ __launch_bounds__(256) void conv_ablate(
    const unsigned short* __restrict__ in, unsigned short* __restrict__ out,
    const unsigned short* __restrict__ w, long rs, int P) {
  __shared__ unsigned short a_lds[CI * TH * TH];
  __shared__ unsigned short sa[M * 32];
  __shared__ unsigned short sb[CO * 32];
  __shared__ int tbl[32];
  const int i = blockIdx.x;
  const int ty0 = (blockIdx.y / (HW / TS)) * TS, tx0 = (blockIdx.y % (HW / TS)) * TS;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const unsigned short* wp = w + (long)i * rs;
  const unsigned short* src = in + (long)i * CI * HW * HW;
  for (int e = tid; e < CI * TH * TH; e += M) {
    int ci = e / (TH * TH), r = e % (TH * TH);
    int y = ty0 + r / TH - 1, x = tx0 + r % TH - 1;
    unsigned short v = 0;
    if (y >= 0 && y < HW && x >= 0 && x < HW) v = src[(ci * HW + y) * HW + x];
    a_lds[e] = v;
  }
  mfma_cd acc[MAXT];
#pragma unroll
  for (int u = 0; u < MAXT; ++u) acc[u] = 0.f;
  for (int ks = 0; ks < KSTEPS; ++ks) {
    if (tid < 32) {
      int k = ks * 32 + tid, off = -1;
      if (k < K) { int ci = k / 9, rm = k - ci * 9; off = (ci * TH + rm / 3) * TH + rm % 3; }
      tbl[tid] = off;
    }
    __syncthreads();
    if (V != 1 && V != 4) {   // A-gather
      for (int e = tid; e < M * 8; e += M) {
        int m = e >> 3, kk0 = (e & 7) * 4;
        int mybase = (m / TS) * TH + (m % TS);
        const int4 tq = *reinterpret_cast<const int4*>(tbl + kk0);
        unsigned short v0 = tq.x < 0 ? 0 : a_lds[tq.x + mybase];
        unsigned short v1 = tq.y < 0 ? 0 : a_lds[tq.y + mybase];
        unsigned short v2 = tq.z < 0 ? 0 : a_lds[tq.z + mybase];
        unsigned short v3 = tq.w < 0 ? 0 : a_lds[tq.w + mybase];
        *reinterpret_cast<uint2*>(sa + m * 32 + kk0) =
            uint2{(unsigned)v0 | ((unsigned)v1 << 16),
                  (unsigned)v2 | ((unsigned)v3 << 16)};
      }
    }
    for (int e = tid; e < CO * 32; e += M) {   // B staging (all variants)
      int n = e >> 5, kk = e & 31, k = ks * 32 + kk;
      sb[n * 32 + kk] = k < K ? wp[n * K + k] : (unsigned short)0;
    }
    __syncthreads();
    if (V != 2 && V != 4) {   // MFMA
#pragma unroll
      for (int u = 0; u < MAXT; ++u) {
        int t = wid + NWAVES * u;
        int mt = t / (CO / 16), nt = t % (CO / 16);
        mfma_ab af = *reinterpret_cast<const mfma_ab*>(
            sa + (mt * 16 + (lane & 15)) * 32 + (lane >> 4) * 8);
        mfma_ab bf = *reinterpret_cast<const mfma_ab*>(
            sb + (nt * 16 + (lane & 15)) * 32 + (lane >> 4) * 8);
        acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[u], 0, 0, 0);
      }
    } else {
      // keep slabs live so the compiler cannot delete staging (rule 17)
      float x = (float)sa[tid] + (float)sb[tid & 255];
      asm volatile("" :: "v"(x));
    }
    __syncthreads();
  }
  if (V != 3) {   // epilogue (LDS-staged coalesced store)
#pragma unroll
    for (int u = 0; u < MAXT; ++u) {
      int t = wid + NWAVES * u;
      int mt = t / (CO / 16), nt = t % (CO / 16);
      int n = nt * 16 + (lane & 15);
      int m0 = mt * 16 + ((lane >> 4) << 2);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        union { float f; unsigned u_; } cv{acc[u][r]};
        sa[n * M + m0 + r] = (unsigned short)(cv.u_ >> 16);
      }
    }
    __syncthreads();
    unsigned short* dst = out + (long)i * CO * HW * HW;
    for (int e = tid; e < CO * M; e += M) {
      int n = e / M, m = e - n * M;
      dst[((long)n * HW + ty0 + m / TS) * HW + tx0 + m % TS] = sa[e];
    }
  } else {
    float x = acc[0][0];
    asm volatile("" :: "v"(x));
  }
}

template <int V>
float bench(const unsigned short* in, unsigned short* out,
            const unsigned short* w, long rs, int P, int iters) {
  dim3 grid(P, (HW / TS) * (HW / TS));
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL(conv_ablate<V>, grid, dim3(256), 0, 0, in, out, w, rs, P);
  hipDeviceSynchronize();
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  hipEventRecord(a);
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(conv_ablate<V>, grid, dim3(256), 0, 0, in, out, w, rs, P);
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms; hipEventElapsedTime(&ms, a, b);
  return ms / iters;
}

int main() {
  const int P = 8192; const long rs = 4096;
  unsigned short *in, *out, *w;
  hipMalloc(&in, (long)P * CI * HW * HW * 2);
  hipMalloc(&out, (long)P * CO * HW * HW * 2);
  hipMalloc(&w, (long)P * rs * 2);
  hipMemset(in, 0x3c, (long)P * CI * HW * HW * 2);
  hipMemset(w, 0x3c, (long)P * rs * 2);
  printf("full       %.3f ms\n", bench<0>(in, out, w, rs, P, 20));
  printf("no-gather  %.3f ms\n", bench<1>(in, out, w, rs, P, 20));
  printf("no-mfma    %.3f ms\n", bench<2>(in, out, w, rs, P, 20));
  printf("no-epilog  %.3f ms\n", bench<3>(in, out, w, rs, P, 20));
  printf("stage-only %.3f ms\n", bench<4>(in, out, w, rs, P, 20));
  return 0;
}
