// Wave-cooperative GEMV helpers for per-member policy forwards.
//
// Population forward at batch=1/member is a batched GEMV: memory-bound on
// the member's own weights, so the design goal is clean streaming (8-byte
// bf16x4 / 16-byte float4 loads per lane, G13) + fp32 accumulate.
// Flat-layout guarantee used here: every layer's weight offset and row
// stride are multiples of 4 elements (checked in ops/pop_models.py), and
// population rows are padded to 16B (perturb row_stride), so vector loads
// stay aligned.
#pragma once
#include "common.h"

namespace des {

typedef __attribute__((ext_vector_type(4))) unsigned int nt_u32x4;

// load 4 consecutive elements as fp32
__device__ __forceinline__ float4 load4(const uint16_t* p) {  // bf16x4
  uint2 pk = *reinterpret_cast<const uint2*>(p);
  float4 r;
  r.x = bf16_to_f32((uint16_t)(pk.x & 0xFFFF));
  r.y = bf16_to_f32((uint16_t)(pk.x >> 16));
  r.z = bf16_to_f32((uint16_t)(pk.y & 0xFFFF));
  r.w = bf16_to_f32((uint16_t)(pk.y >> 16));
  return r;
}
__device__ __forceinline__ float4 load4(const float* p) {     // fp32x4
  return *reinterpret_cast<const float4*>(p);
}
__device__ __forceinline__ float load1(const uint16_t* p) { return bf16_to_f32(*p); }
__device__ __forceinline__ float load1(const float* p) { return *p; }

// NOTE: do NOT add __builtin_nontemporal_load(ext_vector*) helpers that
// scalarize their result: on this toolchain that pattern can lower to a
// broadcast of element 0 (reproduced by tools/probe_nt_load.hip) and
// silently corrupted the perturb/wsum table streams when tried.  The nt
// loads in gemv_rows4_w8 / the CNN FC streams consume the vector whole
// and are parity-tested (incl. a bit-exact golden pin).

// y[row] over rows [wid, wid+NW, ...): act(dot(W[row], x) + b[row])
// W rows at w + row*stride (elements), x fp32 in LDS, y fp32 in LDS.
// ACT: 0 = identity, 1 = tanh, 2 = relu
template <typename WT, int ACT>
__device__ void gemv_rows(const WT* __restrict__ w, const WT* __restrict__ b,
                          const float* __restrict__ x, float* __restrict__ y,
                          int in_dim, int out_dim, int64_t stride, int lane,
                          int wid, int nwaves) {
  int full = in_dim & ~3;
  for (int row = wid; row < out_dim; row += nwaves) {
    const WT* wr = w + (int64_t)row * stride;
    float acc = 0.f;
    for (int i = lane * 4; i < full; i += 64 * 4) {
      float4 wv = load4(wr + i);
      acc += wv.x * x[i] + wv.y * x[i + 1] + wv.z * x[i + 2] + wv.w * x[i + 3];
    }
    for (int i = full + lane; i < in_dim; i += 64) acc += load1(wr + i) * x[i];
    acc = wave_reduce_sum_f(acc);
    if (lane == 0) {
      acc += load1(b + row);
      if (ACT == 1) acc = tanhf(acc);
      if (ACT == 2) acc = fmaxf(acc, 0.f);
      y[row] = acc;
    }
  }
}

// 4-rows-of-ILP variant: 4 independent weight streams per wave hide load
// latency (row index clamped so partial last blocks stay in-bounds and the
// overflow lanes simply discard their result)
template <typename WT, int ACT>
__device__ void gemv_rows4(const WT* __restrict__ w, const WT* __restrict__ b,
                           const float* __restrict__ x, float* __restrict__ y,
                           int in_dim, int out_dim, int64_t stride, int lane,
                           int wid, int nwaves) {
  int full = in_dim & ~3;
  for (int r0 = wid * 4; r0 < out_dim; r0 += nwaves * 4) {
    int rc = out_dim - 1;
    const WT* w0 = w + (int64_t)r0 * stride;
    const WT* w1 = w + (int64_t)min(r0 + 1, rc) * stride;
    const WT* w2 = w + (int64_t)min(r0 + 2, rc) * stride;
    const WT* w3 = w + (int64_t)min(r0 + 3, rc) * stride;
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    for (int i = lane * 4; i < full; i += 64 * 4) {
      float x0 = x[i], x1 = x[i + 1], x2 = x[i + 2], x3 = x[i + 3];
      float4 v0 = load4(w0 + i);
      float4 v1 = load4(w1 + i);
      float4 v2 = load4(w2 + i);
      float4 v3 = load4(w3 + i);
      a0 += v0.x * x0 + v0.y * x1 + v0.z * x2 + v0.w * x3;
      a1 += v1.x * x0 + v1.y * x1 + v1.z * x2 + v1.w * x3;
      a2 += v2.x * x0 + v2.y * x1 + v2.z * x2 + v2.w * x3;
      a3 += v3.x * x0 + v3.y * x1 + v3.z * x2 + v3.w * x3;
    }
    for (int i = full + lane; i < in_dim; i += 64) {
      float xv = x[i];
      a0 += load1(w0 + i) * xv;
      a1 += load1(w1 + i) * xv;
      a2 += load1(w2 + i) * xv;
      a3 += load1(w3 + i) * xv;
    }
    a0 = wave_reduce_sum_f(a0);
    a1 = wave_reduce_sum_f(a1);
    a2 = wave_reduce_sum_f(a2);
    a3 = wave_reduce_sum_f(a3);
    if (lane < 4 && r0 + lane < out_dim) {
      float acc = lane == 0 ? a0 : lane == 1 ? a1 : lane == 2 ? a2 : a3;
      acc += load1(b + r0 + lane);
      if (ACT == 1) acc = tanhf(acc);
      if (ACT == 2) acc = fmaxf(acc, 0.f);
      y[r0 + lane] = acc;
    }
  }
}

// bf16-weight variant with 16-B (8-element) loads: preferred whenever the
// layer offset and row stride are 8-element aligned (checked by caller)
template <int ACT>
__device__ void gemv_rows4_w8(const uint16_t* __restrict__ w,
                              const uint16_t* __restrict__ b,
                              const float* __restrict__ x,
                              float* __restrict__ y, int in_dim, int out_dim,
                              int64_t stride, int lane, int wid, int nwaves) {
  int full = in_dim & ~7;
  for (int r0 = wid * 4; r0 < out_dim; r0 += nwaves * 4) {
    int rc = out_dim - 1;
    const uint16_t* w0 = w + (int64_t)r0 * stride;
    const uint16_t* w1 = w + (int64_t)min(r0 + 1, rc) * stride;
    const uint16_t* w2 = w + (int64_t)min(r0 + 2, rc) * stride;
    const uint16_t* w3 = w + (int64_t)min(r0 + 3, rc) * stride;
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    for (int i = lane * 8; i < full; i += 64 * 8) {
      // nontemporal: population weights are read exactly once per launch
      // (probe tools/probe_mlp_gemv.hip: 584 -> 551 us vs 491 pure-stream)
      nt_u32x4 v0 = __builtin_nontemporal_load(reinterpret_cast<const nt_u32x4*>(w0 + i));
      nt_u32x4 v1 = __builtin_nontemporal_load(reinterpret_cast<const nt_u32x4*>(w1 + i));
      nt_u32x4 v2 = __builtin_nontemporal_load(reinterpret_cast<const nt_u32x4*>(w2 + i));
      nt_u32x4 v3 = __builtin_nontemporal_load(reinterpret_cast<const nt_u32x4*>(w3 + i));
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xv = x[i + j];
        a0 += xv * bf16_to_f32(reinterpret_cast<const uint16_t*>(&v0)[j]);
        a1 += xv * bf16_to_f32(reinterpret_cast<const uint16_t*>(&v1)[j]);
        a2 += xv * bf16_to_f32(reinterpret_cast<const uint16_t*>(&v2)[j]);
        a3 += xv * bf16_to_f32(reinterpret_cast<const uint16_t*>(&v3)[j]);
      }
    }
    for (int i = full + lane; i < in_dim; i += 64) {
      float xv = x[i];
      a0 += load1(w0 + i) * xv;
      a1 += load1(w1 + i) * xv;
      a2 += load1(w2 + i) * xv;
      a3 += load1(w3 + i) * xv;
    }
    a0 = wave_reduce_sum_f(a0);
    a1 = wave_reduce_sum_f(a1);
    a2 = wave_reduce_sum_f(a2);
    a3 = wave_reduce_sum_f(a3);
    if (lane < 4 && r0 + lane < out_dim) {
      float acc = lane == 0 ? a0 : lane == 1 ? a1 : lane == 2 ? a2 : a3;
      acc += load1(b + r0 + lane);
      if (ACT == 1) acc = tanhf(acc);
      if (ACT == 2) acc = fmaxf(acc, 0.f);
      y[r0 + lane] = acc;
    }
  }
}

}  // namespace des
