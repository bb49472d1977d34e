// Probe: streaming-GEMV variants for the Humanoid MLP forward
// (pop=10000, 376->256->256->17 bf16).  Which loop structure gets closest
// to the pure-stream bandwidth of reading each member's 333 KB row?
//   hipcc --offload-arch=gfx950 -O3 tools/probe_mlp_gemv.hip -o /tmp/pm && /tmp/pm
// V0 pure-read upper bound | V1 w8 4-row (round-1 shape) | V2 V1+nontemporal
// | V3 nt + 8-row ILP | V4 nt + xreg 16B full-K-regs
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

#define CHK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); return 1; } } while (0)

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

__device__ __forceinline__ float b2f(unsigned short h) {
  union { unsigned int u; float f; } c;
  c.u = (unsigned int)h << 16;
  return c.f;
}
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

constexpr int IN = 376, H = 256, A = 17;
constexpr int64_t D = 376 * 256 + 256 + 256 * 256 + 256 + 17 * 256 + 17;
constexpr int64_t STRIDE = (D + 7) & ~7;  // 166680 -> pad: 166680? D=166673 -> 166680
constexpr int64_t OW1 = 0, OB1 = IN * H, OW2 = OB1 + H, OB2 = OW2 + H * H;
constexpr int64_t OWH = OB2 + H, OBH = OWH + A * H;

// V0: pure stream read (upper bound)
__global__ __launch_bounds__(256) void v0_stream(const unsigned short* __restrict__ popw,
                                                 float* __restrict__ sink, int P) {
  int i = blockIdx.x;
  const unsigned short* w = popw + (int64_t)i * STRIDE;
  int tid = threadIdx.x;
  float acc = 0.f;
  for (int64_t ib = tid * 8; ib + 8 <= D; ib += 256 * 8) {
    u32x4 v = __builtin_nontemporal_load(reinterpret_cast<const u32x4*>(w + ib));
    acc += (float)(v.x ^ v.y ^ v.z ^ v.w);
  }
  if (acc == 12345.678f) sink[i] = acc;  // keep live
}

template <bool NT>
__device__ __forceinline__ u32x4 ld16(const unsigned short* p) {
  if constexpr (NT)
    return __builtin_nontemporal_load(reinterpret_cast<const u32x4*>(p));
  else
    return *reinterpret_cast<const u32x4*>(p);
}

// 4-row w8 gemv (round-1 structure), template NT
template <bool NT, int ACT>
__device__ void gemv4(const unsigned short* w, const unsigned short* b,
                      const float* x, float* y, int in_dim, int out_dim,
                      int lane, int wid) {
  for (int r0 = wid * 4; r0 < out_dim; r0 += 16) {
    int rc = out_dim - 1;
    const unsigned short* w0 = w + (int64_t)r0 * in_dim;
    const unsigned short* w1 = w + (int64_t)min(r0 + 1, rc) * in_dim;
    const unsigned short* w2 = w + (int64_t)min(r0 + 2, rc) * in_dim;
    const unsigned short* w3 = w + (int64_t)min(r0 + 3, rc) * in_dim;
    float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
    for (int i = lane * 8; i < in_dim; i += 512) {
      u32x4 q0 = ld16<NT>(w0 + i), q1 = ld16<NT>(w1 + i);
      u32x4 q2 = ld16<NT>(w2 + i), q3 = ld16<NT>(w3 + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xv = x[i + j];
        a0 += xv * b2f(reinterpret_cast<const unsigned short*>(&q0)[j]);
        a1 += xv * b2f(reinterpret_cast<const unsigned short*>(&q1)[j]);
        a2 += xv * b2f(reinterpret_cast<const unsigned short*>(&q2)[j]);
        a3 += xv * b2f(reinterpret_cast<const unsigned short*>(&q3)[j]);
      }
    }
    a0 = wave_sum(a0); a1 = wave_sum(a1); a2 = wave_sum(a2); a3 = wave_sum(a3);
    if (lane < 4 && r0 + lane < out_dim) {
      float acc = lane == 0 ? a0 : lane == 1 ? a1 : lane == 2 ? a2 : a3;
      acc += b2f(b[r0 + lane]);
      if (ACT) acc = tanhf(acc);
      y[r0 + lane] = acc;
    }
  }
}

// 8-row nt gemv
template <int ACT>
__device__ void gemv8(const unsigned short* w, const unsigned short* b,
                      const float* x, float* y, int in_dim, int out_dim,
                      int lane, int wid) {
  for (int r0 = wid * 8; r0 < out_dim; r0 += 32) {
    int rc = out_dim - 1;
    const unsigned short* wr[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      wr[j] = w + (int64_t)min(r0 + j, rc) * in_dim;
    float a[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int i = lane * 8; i < in_dim; i += 512) {
#pragma unroll
      for (int r = 0; r < 8; ++r) {
        u32x4 q = ld16<true>(wr[r] + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          a[r] += x[i + j] * b2f(reinterpret_cast<const unsigned short*>(&q)[j]);
      }
    }
#pragma unroll
    for (int r = 0; r < 8; ++r) a[r] = wave_sum(a[r]);
    if (lane < 8 && r0 + lane < out_dim) {
      float acc = a[lane] + b2f(b[r0 + lane]);
      if (ACT) acc = tanhf(acc);
      y[r0 + lane] = acc;
    }
  }
}

// K4 A/B: FUSED perturb-forward gemv — weights materialize on the fly as
// theta[k] + s*sigma*table[off + k].  theta (651 KB fp32) is shared by all
// members -> L2-resident regular loads; the noise slice streams nt like the
// materialized population rows would.  HBM traffic is identical to the
// materialized bf16 population (bf16 table), so this measures the VALU/L2
// overhead fusion adds — and what removing the perturb kernel + 3.3 GB of
// population residency would buy.
template <int ACT>
__device__ void gemv4_fused(const float* theta, const unsigned short* noise,
                            float ssig, const float* x, float* y, int in_dim,
                            int out_dim, int lane, int wid) {
  for (int r0 = wid * 4; r0 < out_dim; r0 += 16) {
    int rc = out_dim - 1;
    float acc[4] = {0, 0, 0, 0};
    int rows[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) rows[j] = min(r0 + j, rc);
    for (int i = lane * 8; i < in_dim; i += 512) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t base = (int64_t)rows[rr] * in_dim + i;
        u32x4 q = ld16<true>(noise + base);
        float4 t0 = *reinterpret_cast<const float4*>(theta + base);
        float4 t1 = *reinterpret_cast<const float4*>(theta + base + 4);
        const unsigned short* nh = reinterpret_cast<const unsigned short*>(&q);
        const float* tf = reinterpret_cast<const float*>(&t0);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[rr] += x[i + j] * (tf[j] + ssig * b2f(nh[j]));
        const float* tg = reinterpret_cast<const float*>(&t1);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[rr] += x[i + 4 + j] * (tg[j] + ssig * b2f(nh[4 + j]));
      }
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) acc[rr] = wave_sum(acc[rr]);
    if (lane < 4 && r0 + lane < out_dim) {
      float a = acc[lane];
      if (ACT) a = tanhf(a);
      y[r0 + lane] = a;
    }
  }
}

__global__ __launch_bounds__(256) void fwd_fused_kernel(
    const float* __restrict__ theta, const unsigned short* __restrict__ table,
    const long* __restrict__ offs, const float* __restrict__ signs,
    const float* __restrict__ obs, float* __restrict__ out, int P) {
  __shared__ float xb[512];
  __shared__ float yb[512];
  int i = blockIdx.x;
  const unsigned short* noise = table + offs[i];
  float ssig = signs[i] * 0.02f;
  int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  for (int d = tid; d < IN; d += 256) xb[d] = obs[(int64_t)i * IN + d];
  __syncthreads();
  gemv4_fused<1>(theta + OW1, table + offs[i] + OW1, ssig, xb, yb, IN, H, lane, wid);
  __syncthreads();
  gemv4_fused<1>(theta + OW2, noise + OW2, ssig, yb, xb, H, H, lane, wid);
  __syncthreads();
  gemv4_fused<0>(theta + OWH, noise + OWH, ssig, xb, yb, H, A, lane, wid);
  __syncthreads();
  for (int a = tid; a < A; a += 256) out[(int64_t)i * A + a] = tanhf(yb[a]);
}

template <int V>
__global__ __launch_bounds__(256) void fwd_kernel(const unsigned short* __restrict__ popw,
                                                  const float* __restrict__ obs,
                                                  float* __restrict__ out, int P) {
  __shared__ float xb[512];
  __shared__ float yb[512];
  int i = blockIdx.x;
  const unsigned short* w = popw + (int64_t)i * STRIDE;
  int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  for (int d = tid; d < IN; d += 256) xb[d] = obs[(int64_t)i * IN + d];
  __syncthreads();
  if (V == 1) gemv4<false, 1>(w + OW1, w + OB1, xb, yb, IN, H, lane, wid);
  if (V == 2) gemv4<true, 1>(w + OW1, w + OB1, xb, yb, IN, H, lane, wid);
  if (V == 3) gemv8<1>(w + OW1, w + OB1, xb, yb, IN, H, lane, wid);
  __syncthreads();
  if (V == 1) gemv4<false, 1>(w + OW2, w + OB2, yb, xb, H, H, lane, wid);
  if (V == 2) gemv4<true, 1>(w + OW2, w + OB2, yb, xb, H, H, lane, wid);
  if (V == 3) gemv8<1>(w + OW2, w + OB2, yb, xb, H, H, lane, wid);
  __syncthreads();
  if (V == 1) gemv4<false, 0>(w + OWH, w + OBH, xb, yb, H, A, lane, wid);
  if (V == 2) gemv4<true, 0>(w + OWH, w + OBH, xb, yb, H, A, lane, wid);
  if (V == 3) gemv8<0>(w + OWH, w + OBH, xb, yb, H, A, lane, wid);
  __syncthreads();
  for (int a = tid; a < A; a += 256) out[(int64_t)i * A + a] = tanhf(yb[a]);
}

int main() {
  const int P = 10000;
  unsigned short* popw;
  float *obs, *out;
  CHK(hipMalloc(&popw, (int64_t)P * STRIDE * 2));
  CHK(hipMalloc(&obs, (int64_t)P * IN * 4));
  CHK(hipMalloc(&out, (int64_t)P * A * 4));
  CHK(hipMemset(popw, 0x3C, (int64_t)P * STRIDE * 2));  // small bf16 values
  CHK(hipMemset(obs, 0x3D, (int64_t)P * IN * 4));
  double gb = (double)P * D * 2 / 1e9;
  hipEvent_t e0, e1;
  CHK(hipEventCreate(&e0)); CHK(hipEventCreate(&e1));
  auto bench = [&](const char* name, auto launch) {
    for (int r = 0; r < 3; ++r) launch();            // warmup
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(e0));
    const int REP = 30;
    for (int r = 0; r < REP; ++r) launch();
    CHK(hipEventRecord(e1));
    CHK(hipEventSynchronize(e1));
    float ms;
    CHK(hipEventElapsedTime(&ms, e0, e1));
    printf("%-14s %8.1f us/iter  %7.2f GB -> %6.2f TB/s\n", name,
           ms * 1000 / REP, gb, gb / (ms / REP) / 1e9 * 1e3);
    return 0;
  };
  bench("v0_stream", [&] { hipLaunchKernelGGL(v0_stream, dim3(P), dim3(256), 0, 0, popw, obs, P); });
  bench("v1_gemv4", [&] { hipLaunchKernelGGL(fwd_kernel<1>, dim3(P), dim3(256), 0, 0, popw, obs, out, P); });
  bench("v2_gemv4_nt", [&] { hipLaunchKernelGGL(fwd_kernel<2>, dim3(P), dim3(256), 0, 0, popw, obs, out, P); });
  bench("v3_gemv8_nt", [&] { hipLaunchKernelGGL(fwd_kernel<3>, dim3(P), dim3(256), 0, 0, popw, obs, out, P); });

  // K4 fused perturb-forward A/B (theta fp32 + bf16 noise table)
  float* theta;
  unsigned short* table;
  long* offsv;
  float* signs;
  const int64_t TBL = 1 << 28;  // 256M bf16 = 512 MB table
  CHK(hipMalloc(&theta, D * 4));
  CHK(hipMalloc(&table, TBL * 2));
  CHK(hipMalloc(&offsv, P * 8));
  CHK(hipMalloc(&signs, P * 4));
  CHK(hipMemset(theta, 0x3D, D * 4));
  CHK(hipMemset(table, 0x3C, TBL * 2));
  std::vector<long> ho(P);
  std::vector<float> hs(P);
  for (int i = 0; i < P; ++i) {
    ho[i] = ((long)(i * 2654435761u) % (TBL - D - 16)) & ~15L;
    hs[i] = (i & 1) ? -1.f : 1.f;
  }
  CHK(hipMemcpy(offsv, ho.data(), P * 8, hipMemcpyHostToDevice));
  CHK(hipMemcpy(signs, hs.data(), P * 4, hipMemcpyHostToDevice));
  bench("v4_fused_k4", [&] { hipLaunchKernelGGL(fwd_fused_kernel, dim3(P), dim3(256), 0, 0, theta, table, offsv, signs, obs, out, P); });
  CHK(hipDeviceSynchronize());
  return 0;
}
