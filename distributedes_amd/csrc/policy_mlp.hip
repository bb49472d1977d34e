// K3: mlp_forward_pop — per-member MLP policy forward.
//
// One 256-thread block (4 waves) per member.  Input normalization (running
// obs normalizer, K12 consumer side) is fused into the input load; action
// selection (argmax / tanh) fused into the epilogue — the rollout loop makes
// exactly one launch per {env step x model}.
// Layer layout mirrors models/mlp.py: fc1[,fc2],head flat offsets.
#include <algorithm>

#include "common.h"
#include "gemv.h"

namespace {

constexpr int MAX_WIDTH = 512;  // max(in_dim, hidden, A)

// NOTE: a compile-time-dims specialization (template <.., int TIN, TH1,
// TH2>) for the Humanoid shape was A/B-measured ~6% SLOWER than this
// runtime-dims kernel (15.3M vs 16.2M env steps/s, same box, 2x each):
// fully unrolling the 16-iteration row loops hurts more than the folded
// bounds help.  Keep runtime dims.
template <typename WT, bool CONTINUOUS>
__global__ __launch_bounds__(256, 6) void mlp_pop_forward_kernel(
    const WT* __restrict__ popw, int64_t row_stride,      // [P, >=D]
    const float* __restrict__ obs, int in_dim_rt,         // [P, in_dim] raw
    const float* __restrict__ nmean,                      // nullable [in_dim]
    const float* __restrict__ ninv, float clip,
    int* __restrict__ act_i, float* __restrict__ act_f,   // one of them
    float* __restrict__ logits_out,                       // nullable [P, A]
    int h1, int h2, int A,
    int64_t ow1, int64_t ob1, int64_t ow2, int64_t ob2, int64_t owh,
    int64_t obh, int P) {
  __shared__ float xbuf[MAX_WIDTH];
  __shared__ float ybuf[MAX_WIDTH];
  const int in_dim = in_dim_rt;
  int i = blockIdx.x;
  if (i >= P) return;
  const WT* w = popw + (int64_t)i * row_stride;
  int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;

  // input load + normalize
  const float* ob = obs + (int64_t)i * in_dim;
  for (int d = tid; d < in_dim; d += 256) {
    float v = ob[d];
    if (nmean) {
      v = (v - nmean[d]) * ninv[d];
      v = fminf(fmaxf(v, -clip), clip);
    }
    xbuf[d] = v;
  }
  __syncthreads();

  // bf16 weights with 8-elem-aligned layer offset + row stride -> 16-B loads
  const bool w8_1 = sizeof(WT) == 2 && (ow1 & 7) == 0 && (in_dim & 7) == 0;
  if (w8_1)
    des::gemv_rows4_w8<1>((const uint16_t*)(w + ow1),
                          (const uint16_t*)(w + ob1), xbuf, ybuf, in_dim, h1,
                          in_dim, lane, wid, 4);
  else
    des::gemv_rows4<WT, 1>(w + ow1, w + ob1, xbuf, ybuf, in_dim, h1, in_dim,
                           lane, wid, 4);
  __syncthreads();
  const float* x2 = ybuf;
  float* y2 = xbuf;
  int last_dim = h1;
  if (h2 > 0) {
    if (sizeof(WT) == 2 && (ow2 & 7) == 0 && (h1 & 7) == 0)
      des::gemv_rows4_w8<1>((const uint16_t*)(w + ow2),
                            (const uint16_t*)(w + ob2), ybuf, xbuf, h1, h2,
                            h1, lane, wid, 4);
    else
      des::gemv_rows4<WT, 1>(w + ow2, w + ob2, ybuf, xbuf, h1, h2, h1, lane,
                             wid, 4);
    __syncthreads();
    x2 = xbuf; y2 = ybuf; last_dim = h2;
  }
  if (sizeof(WT) == 2 && (owh & 7) == 0 && (last_dim & 7) == 0)
    des::gemv_rows4_w8<0>((const uint16_t*)(w + owh),
                          (const uint16_t*)(w + obh), x2, y2, last_dim, A,
                          last_dim, lane, wid, 4);
  else
    des::gemv_rows4<WT, 0>(w + owh, w + obh, x2, y2, last_dim, A, last_dim,
                           lane, wid, 4);
  __syncthreads();

  if (logits_out)
    for (int a = tid; a < A; a += 256) logits_out[(int64_t)i * A + a] = y2[a];
  if (CONTINUOUS) {
    for (int a = tid; a < A; a += 256) act_f[(int64_t)i * A + a] = tanhf(y2[a]);
  } else if (tid == 0) {
    int best = 0;
    float bv = y2[0];
    for (int a = 1; a < A; ++a)
      if (y2[a] > bv) { bv = y2[a]; best = a; }
    act_i[i] = best;
  }
}

}  // namespace

extern "C" void des_launch_mlp_forward(
    const void* popw, bool bf16, int64_t row_stride, const float* obs,
    int in_dim, const float* nmean, const float* ninv, float clip, int* act_i,
    float* act_f, float* logits_out, int h1, int h2, int A,
    const int64_t* offs6, int P, hipStream_t stream) {
  dim3 grid((uint32_t)P), block(256);
  bool cont = act_f != nullptr;
  if (bf16) {
    if (cont)
      hipLaunchKernelGGL((mlp_pop_forward_kernel<uint16_t, true>), grid, block,
                         0, stream, (const uint16_t*)popw, row_stride, obs,
                         in_dim, nmean, ninv, clip, act_i, act_f, logits_out,
                         h1, h2, A, offs6[0], offs6[1], offs6[2], offs6[3],
                         offs6[4], offs6[5], P);
    else
      hipLaunchKernelGGL((mlp_pop_forward_kernel<uint16_t, false>), grid,
                         block, 0, stream, (const uint16_t*)popw, row_stride,
                         obs, in_dim, nmean, ninv, clip, act_i, act_f,
                         logits_out, h1, h2, A, offs6[0], offs6[1], offs6[2],
                         offs6[3], offs6[4], offs6[5], P);
  } else {
    if (cont)
      hipLaunchKernelGGL((mlp_pop_forward_kernel<float, true>), grid, block, 0,
                         stream, (const float*)popw, row_stride, obs, in_dim,
                         nmean, ninv, clip, act_i, act_f, logits_out, h1, h2,
                         A, offs6[0], offs6[1], offs6[2], offs6[3], offs6[4],
                         offs6[5], P);
    else
      hipLaunchKernelGGL((mlp_pop_forward_kernel<float, false>), grid, block,
                         0, stream, (const float*)popw, row_stride, obs,
                         in_dim, nmean, ninv, clip, act_i, act_f, logits_out,
                         h1, h2, A, offs6[0], offs6[1], offs6[2], offs6[3],
                         offs6[4], offs6[5], P);
  }
}
