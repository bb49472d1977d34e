// K6: conv_forward_pop (IMPALA-CNN) — per-member forward for the Procgen-dim
// config (BASELINE config #5).
//
// Unlike the Nature-CNN (one fused block/member kernel), IMPALA's stage-1
// activations (64x64x16 bf16 = 128 KiB) exceed LDS, so the net runs as a
// sequence of generic batched kernels over [P, C, H, W] bf16 scratch:
//   impala_conv3: 3x3 pad-1 conv, spatial 16x16 tiles, halo-staged input +
//     [k][co] transposed weight tile in LDS (~40 KiB -> 4 blocks/CU);
//     optional fused input-relu (pre-activation residual) and skip-add.
//   impala_pool:  3x3 stride-2 pad-1 max pool.
//   impala_fc_head: relu-flatten 2048 -> fc 256 (4-row-ILP bf16x8 dot2
//     stream) -> head A + argmax.
#include "common.h"
#include "gemv.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short mfma_ab;
typedef __attribute__((ext_vector_type(4))) float mfma_cd;

// 3x3 pad-1 conv as MFMA implicit GEMM per spatial tile:
//   M = TS*TS positions, N = CO, K = CI*9 (zero-padded to 32-multiples).
// Per K-step all threads build the A-slab (im2col from the halo-staged
// input tile) and B-slab (direct [co][k] rows of the member's weights),
// then each wave runs one v_mfma_f32_16x16x32_bf16 per owned 16x16 C-tile.
// Fragment maps hardware-verified by tools/probe_mfma.hip.
template <int CI, int CO, int HW, int TS, bool RELU_IN, bool ADD_SKIP>
__global__ __launch_bounds__(TS * TS) void impala_conv3_kernel(
    const uint16_t* __restrict__ in,    // [P][CI][HW][HW]
    uint16_t* __restrict__ out,         // [P][CO][HW][HW]
    const uint16_t* __restrict__ skip,  // nullable [P][CO][HW][HW]
    const uint16_t* __restrict__ popw, int64_t row_stride, int64_t w_off,
    int64_t b_off, int P) {
  constexpr int TH = TS + 2;
  constexpr int M = TS * TS;               // 256 or 64
  constexpr int K = CI * 9;
  constexpr int KSTEPS = (K + 31) / 32;
  constexpr int NWAVES = M / 64;           // 4 or 1
  constexpr int NTILES = (M / 16) * (CO / 16);
  constexpr int MAXT = NTILES / NWAVES;    // tiles per wave
  __shared__ uint16_t a_lds[CI * TH * TH];
  __shared__ uint16_t sa[M * 32];
  __shared__ uint16_t sb[CO * 32];
  __shared__ int tbl[32];  // k -> halo-tile offset (my=mx=0); -1 = zero-pad
  const int i = blockIdx.x;
  if (i >= P) return;
  constexpr int NT = HW / TS;
  const int ty0 = (blockIdx.y / NT) * TS, tx0 = (blockIdx.y % NT) * TS;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const uint16_t* w = popw + (int64_t)i * row_stride;

  // stage input halo tile (zero-padded at image edges, optional relu)
  const uint16_t* src = in + (int64_t)i * CI * HW * HW;
  for (int e = tid; e < CI * TH * TH; e += M) {
    int ci = e / (TH * TH), r = e % (TH * TH);
    int y = ty0 + r / TH - 1, x = tx0 + r % TH - 1;
    uint16_t v = 0;
    if (y >= 0 && y < HW && x >= 0 && x < HW) {
      v = src[(ci * HW + y) * HW + x];
      if (RELU_IN && (v & 0x8000)) v = 0;
    }
    a_lds[e] = v;
  }
  __syncthreads();

  const uint16_t* wsrc = w + w_off;
  mfma_cd acc[MAXT];
#pragma unroll
  for (int u = 0; u < MAXT; ++u) acc[u] = 0.f;
  for (int ks = 0; ks < KSTEPS; ++ks) {
    // k -> offset table (separable: a_lds idx = tbl[kk] + my*TH + mx)
    if (tid < 32) {
      int k = ks * 32 + tid;
      int off = -1;
      if (k < K) {
        int ci = k / 9, rm = k - ci * 9, ky = rm / 3, kx = rm - ky * 3;
        off = (ci * TH + ky) * TH + kx;
      }
      tbl[tid] = off;
    }
    __syncthreads();
    // A-slab [M][32]: im2col gather from the halo tile
    for (int e = tid; e < M * 32 / 4; e += M) {  // 4 elems per unit
      int m = (e * 4) >> 5, kk0 = (e * 4) & 31;
      int mybase = (m / TS) * TH + (m % TS);
      const int4 tq = *reinterpret_cast<const int4*>(tbl + kk0);
      uint16_t v0 = tq.x < 0 ? 0 : a_lds[tq.x + mybase];
      uint16_t v1 = tq.y < 0 ? 0 : a_lds[tq.y + mybase];
      uint16_t v2 = tq.z < 0 ? 0 : a_lds[tq.z + mybase];
      uint16_t v3 = tq.w < 0 ? 0 : a_lds[tq.w + mybase];
      *reinterpret_cast<uint2*>(sa + m * 32 + kk0) =
          uint2{(uint32_t)v0 | ((uint32_t)v1 << 16),
                (uint32_t)v2 | ((uint32_t)v3 << 16)};
    }
    // B-slab [CO][32]: direct rows of the flat [co][k] weights
    for (int e = tid; e < CO * 32; e += M) {
      int n = e >> 5, kk = e & 31;
      int k = ks * 32 + kk;
      sb[n * 32 + kk] = k < K ? wsrc[n * K + k] : (uint16_t)0;
    }
    __syncthreads();
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
    __syncthreads();
  }

  // epilogue: the naive per-lane scatter hits a 2*HW*HW-byte stride (one
  // cache line per lane) — stage the bias+relu'd tile in LDS (sa is free
  // and exactly CO*M elements) and do a coalesced cooperative store instead
  const uint16_t* bias = w + b_off;
  uint16_t* dst = out + (int64_t)i * CO * HW * HW;
  const uint16_t* sk = ADD_SKIP ? skip + (int64_t)i * CO * HW * HW : nullptr;
  static_assert(CO * M <= M * 32, "output tile exceeds the sa staging carve");
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + NWAVES * u;
    int mt = t / (CO / 16), nt = t % (CO / 16);
    int n = nt * 16 + (lane & 15);
    int m0 = mt * 16 + ((lane >> 4) << 2);
    float b = des::bf16_to_f32(bias[n]);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      sa[n * M + m0 + r] = des::f32_to_bf16(acc[u][r] + b);
  }
  __syncthreads();
  // 8 consecutive u16 of a tile row are contiguous + 16B-aligned in dst
  // (tx0 is a multiple of TS >= 8): store uint4 per thread (ablation showed
  // the narrow store tail at ~350 GB/s was 23% of the kernel)
  for (int e8 = tid; e8 < CO * M / 8; e8 += M) {
    int e = e8 * 8;
    int n = e / M, m = e - n * M;
    int64_t gidx = ((int64_t)n * HW + ty0 + m / TS) * HW + tx0 + m % TS;
    uint4 v = *reinterpret_cast<const uint4*>(sa + e);
    if (ADD_SKIP) {
      uint4 sv = *reinterpret_cast<const uint4*>(sk + gidx);
      const uint16_t* a = reinterpret_cast<const uint16_t*>(&v);
      const uint16_t* b = reinterpret_cast<const uint16_t*>(&sv);
      uint16_t o[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = des::f32_to_bf16(des::bf16_to_f32(a[j]) + des::bf16_to_f32(b[j]));
      v = *reinterpret_cast<const uint4*>(o);
    }
    *reinterpret_cast<uint4*>(dst + gidx) = v;
  }
}

template <int C, int HW>  // output is HW/2 x HW/2 (3x3 s2 pad1)
__global__ void impala_pool_kernel(const uint16_t* __restrict__ in,
                                   uint16_t* __restrict__ out, int P) {
  constexpr int HO = HW / 2;
  int64_t n = (int64_t)P * C * HO * HO;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n) return;
  int ox = idx % HO, oy = (idx / HO) % HO;
  int64_t pc = idx / (HO * HO);  // member*C + c
  const uint16_t* src = in + pc * HW * HW;
  float m = -3.0e38f;
  for (int dy = -1; dy <= 1; ++dy) {
    int y = 2 * oy + dy;
    if (y < 0 || y >= HW) continue;
    for (int dx = -1; dx <= 1; ++dx) {
      int x = 2 * ox + dx;
      if (x < 0 || x >= HW) continue;
      m = fmaxf(m, des::bf16_to_f32(src[y * HW + x]));
    }
  }
  out[idx] = des::f32_to_bf16(m);
}

// relu-flatten 2048 -> fc 256 relu -> head A + argmax
constexpr int IFC_IN = 2048, IFC_OUT = 256, IMAX_A = 32;

__global__ __launch_bounds__(256) void impala_fc_head_kernel(
    const uint16_t* __restrict__ popw, int64_t row_stride,
    const uint16_t* __restrict__ act,  // [P][2048] (pre-relu)
    int* __restrict__ actions, float* __restrict__ logits_out, int A, int P,
    int64_t wf, int64_t bf, int64_t wh, int64_t bh) {
  __shared__ uint16_t xs[IFC_IN];
  __shared__ float fcb[IFC_OUT];
  __shared__ float hd[IMAX_A];
  const int i = blockIdx.x;
  if (i >= P) return;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const uint16_t* src = act + (int64_t)i * IFC_IN;
  for (int e = tid; e < IFC_IN; e += 256) {
    uint16_t v = src[e];
    xs[e] = (v & 0x8000) ? 0 : v;  // relu
  }
  __syncthreads();
  const uint16_t* w = popw + (int64_t)i * row_stride;
  const uint16_t* w0base = w + wf;
  const uint16_t* bfp = w + bf;
  for (int rr = 0; rr < 64; rr += 4) {
    const int row0 = wid * 64 + rr;
    const uint16_t* w0 = w0base + (int64_t)row0 * IFC_IN;
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll
    for (int it = 0; it < 4; ++it) {       // 2048 = 4 * 512 exactly
      int ib = lane * 8 + it * 512;
      uint4 xp = *reinterpret_cast<const uint4*>(xs + ib);
      uint4 wa = *reinterpret_cast<const uint4*>(w0 + ib);
      uint4 wb = *reinterpret_cast<const uint4*>(w0 + IFC_IN + ib);
      uint4 wc = *reinterpret_cast<const uint4*>(w0 + 2 * IFC_IN + ib);
      uint4 wd = *reinterpret_cast<const uint4*>(w0 + 3 * IFC_IN + ib);
      const uint32_t* xq = reinterpret_cast<const uint32_t*>(&xp);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        des::bf16x2 xj = __builtin_bit_cast(des::bf16x2, xq[j]);
        a0 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
               reinterpret_cast<const uint32_t*>(&wa)[j]), a0);
        a1 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
               reinterpret_cast<const uint32_t*>(&wb)[j]), a1);
        a2 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
               reinterpret_cast<const uint32_t*>(&wc)[j]), a2);
        a3 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
               reinterpret_cast<const uint32_t*>(&wd)[j]), a3);
      }
    }
    a0 = des::wave_reduce_sum_f(a0);
    a1 = des::wave_reduce_sum_f(a1);
    a2 = des::wave_reduce_sum_f(a2);
    a3 = des::wave_reduce_sum_f(a3);
    if (lane < 4) {
      float acc = lane == 0 ? a0 : lane == 1 ? a1 : lane == 2 ? a2 : a3;
      fcb[row0 + lane] =
          fmaxf(acc + des::bf16_to_f32(bfp[row0 + lane]), 0.f);
    }
  }
  __syncthreads();
  des::gemv_rows<uint16_t, 0>(w + wh, w + bh, fcb, hd, IFC_OUT, A, IFC_OUT,
                              lane, wid, 4);
  __syncthreads();
  if (logits_out && tid < A) logits_out[(int64_t)i * A + tid] = hd[tid];
  if (tid == 0) {
    int best = 0;
    float bv = hd[0];
    for (int a = 1; a < A; ++a)
      if (hd[a] > bv) { bv = hd[a]; best = a; }
    actions[i] = best;
  }
}

}  // namespace

extern "C" {

// shape key: 0:(3,16,64) 1:(16,16,32) 2:(16,32,32) 3:(32,32,16) 4:(32,32,8)
void des_launch_impala_conv3(const uint16_t* in, uint16_t* out,
                             const uint16_t* skip, const uint16_t* popw,
                             int64_t row_stride, int64_t w_off, int64_t b_off,
                             int shape, bool relu_in, bool add_skip, int P,
                             hipStream_t stream) {
#define CASE(S, CI, CO, HW, TS)                                               \
  if (shape == S) {                                                           \
    dim3 grid((uint32_t)P, (HW / TS) * (HW / TS));                            \
    if (relu_in && add_skip)                                                  \
      hipLaunchKernelGGL((impala_conv3_kernel<CI, CO, HW, TS, true, true>),   \
                         grid, dim3(TS * TS), 0, stream, in, out, skip, popw, \
                         row_stride, w_off, b_off, P);                        \
    else if (relu_in)                                                         \
      hipLaunchKernelGGL((impala_conv3_kernel<CI, CO, HW, TS, true, false>),  \
                         grid, dim3(TS * TS), 0, stream, in, out, skip, popw, \
                         row_stride, w_off, b_off, P);                        \
    else                                                                      \
      hipLaunchKernelGGL((impala_conv3_kernel<CI, CO, HW, TS, false, false>), \
                         grid, dim3(TS * TS), 0, stream, in, out, skip, popw, \
                         row_stride, w_off, b_off, P);                        \
    return;                                                                   \
  }
  CASE(0, 3, 16, 64, 16)
  CASE(1, 16, 16, 32, 16)
  CASE(2, 16, 32, 32, 16)
  CASE(3, 32, 32, 16, 16)
  CASE(4, 32, 32, 8, 8)
#undef CASE
}

void des_launch_impala_pool(const uint16_t* in, uint16_t* out, int C, int HW,
                            int P, hipStream_t stream) {
  int64_t n = (int64_t)P * C * (HW / 2) * (HW / 2);
  int g = (int)((n + 255) / 256);
  if (C == 16 && HW == 64)
    hipLaunchKernelGGL((impala_pool_kernel<16, 64>), dim3(g), dim3(256), 0,
                       stream, in, out, P);
  else if (C == 32 && HW == 32)
    hipLaunchKernelGGL((impala_pool_kernel<32, 32>), dim3(g), dim3(256), 0,
                       stream, in, out, P);
  else if (C == 32 && HW == 16)
    hipLaunchKernelGGL((impala_pool_kernel<32, 16>), dim3(g), dim3(256), 0,
                       stream, in, out, P);
}

void des_launch_impala_fc_head(const uint16_t* popw, int64_t row_stride,
                               const uint16_t* act, int* actions,
                               float* logits_out, int A, int P, int64_t wf,
                               int64_t bf, int64_t wh, int64_t bh,
                               hipStream_t stream) {
  hipLaunchKernelGGL(impala_fc_head_kernel, dim3((uint32_t)P), dim3(256), 0,
                     stream, popw, row_stride, act, actions, logits_out, A, P,
                     wf, bf, wh, bh);
}

}  // extern "C"
