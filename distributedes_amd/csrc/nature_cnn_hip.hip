#include "hip/hip_runtime.h"
// K5: conv_forward_pop (Nature-CNN) — per-member policy forward as three
// kernels sized for their regimes:
//
// ONE block-per-member kernel (512 thr, ~77 KiB LDS -> 2 blocks/CU):
//   convs: 8x8s4x32 -> 4x4s2x64 -> 3x3s1x64 on packed-pair v_dot2c_f32_bf16,
//     weight tiles staged in 32-oc halves to conflict-free [k/4][oc][4] bf16
//     LDS layouts; activations live in LDS only;
//   fc 3136->512: the HBM-bound stage (94% of weights) streamed with 4-row
//     ILP bf16x8 (16 B) loads per wave;
//   head 512->A + argmax fused.
// Because 2 blocks co-reside per CU, one block's VALU-bound conv phase
// overlaps the other block's HBM-bound FC stream: step time ~ max(conv, fc)
// rather than the sum.  FC floor at pop=8192/GPU:
// 8192 * 3.2 MiB / 6.3 TBps ~= 4.3 ms.
#include "common.h"
#include "gemv.h"

namespace {

// geometry (fixed, models/nature_cnn.py)
constexpr int C_IN = 4, HW = 84;
constexpr int C1 = 32, O1 = 20, K1 = 8, S1 = 4;   // conv1
constexpr int C2 = 64, O2 = 9, K2 = 4, S2 = 2;    // conv2
constexpr int C3 = 64, O3 = 7;                    // conv3 (3x3 s1)
constexpr int FC_IN = C3 * O3 * O3;               // 3136
constexpr int FC_OUT = 512;
constexpr int MAX_A = 32;

// ---------------- kernel 1: convs ----------------
// LDS carve (bytes, 16-aligned).  conv2/conv3 weights staged in 32-oc
// halves so the whole carve stays < 80 KiB -> 2 blocks/CU (latency hiding).
constexpr int W_ELEMS = 32 * 576;                      // 18432 (conv3 half)
constexpr int OFF_W = 0;                               // bf16 [W_ELEMS]
constexpr int OFF_A1 = OFF_W + W_ELEMS * 2;            // bf16 [32*20*20]
constexpr int OFF_A2 = OFF_A1 + C1 * O1 * O1 * 2;      // bf16 [64*9*9]
constexpr int OFF_T2 = OFF_A2 + ((C2 * O2 * O2 * 2 + 15) & ~15);  // int [512]
constexpr int OFF_T3 = OFF_T2 + 512 * 4;               // int [576]
constexpr int CONV_LDS = OFF_T3 + 576 * 4;
// fc input (3136 bf16), fc output (512 f32) and head logits overlay the a1
// carve (dead after conv2): 6272 + 2048 + 128 B < 25600 B
constexpr int OFF_A3 = OFF_A1;
constexpr int OFF_FC = OFF_A3 + FC_IN * 2;
constexpr int OFF_HD = OFF_FC + FC_OUT * 4;
static_assert(OFF_HD + MAX_A * 4 <= OFF_A2, "fc overlay exceeds a1 carve");

template <typename IT>
__device__ __forceinline__ float obs_val(IT v);
template <>
__device__ __forceinline__ float obs_val<uint8_t>(uint8_t v) { return (float)v; }
template <>
__device__ __forceinline__ float obs_val<uint16_t>(uint16_t v) {
  return des::bf16_to_f32(v);
}

template <typename IT>
__global__ __launch_bounds__(512, 2) void nature_conv_kernel(
    const uint16_t* __restrict__ popw, int64_t row_stride,
    const IT* __restrict__ frames,     // [P][4][84][84]
    const int* __restrict__ slot_map,  // [4] logical->physical slot
    int* __restrict__ actions, float* __restrict__ logits_out, int A,
    int P, const int64_t* __restrict__ offs) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* w_lds = reinterpret_cast<uint16_t*>(smem + OFF_W);
  uint16_t* a1 = reinterpret_cast<uint16_t*>(smem + OFF_A1);
  uint16_t* a2 = reinterpret_cast<uint16_t*>(smem + OFF_A2);
  int* t2 = reinterpret_cast<int*>(smem + OFF_T2);
  int* t3 = reinterpret_cast<int*>(smem + OFF_T3);
  uint16_t* a3 = reinterpret_cast<uint16_t*>(smem + OFF_A3);   // fc input
  float* fcb = reinterpret_cast<float*>(smem + OFF_FC);
  float* hd = reinterpret_cast<float*>(smem + OFF_HD);

  const int i = blockIdx.x;
  if (i >= P) return;
  const uint16_t* w = popw + (int64_t)i * row_stride;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const IT* obs = frames + (int64_t)i * (4 * HW * HW);
  const int s0 = slot_map[0], s1m = slot_map[1], s2m = slot_map[2],
            s3 = slot_map[3];

  // separable activation-offset tables: a1 index = base(pos) + t2[icc]
  for (int e = tid; e < 512; e += 512) {
    int ic = e >> 4, ky = (e >> 2) & 3, kx = e & 3;
    t2[e] = ic * (O1 * O1) + ky * O1 + kx;
  }
  for (int e = tid; e < 576; e += 512) {
    int ic = e / 9, r = e % 9;
    t3[e] = ic * (O2 * O2) + (r / 3) * O2 + (r % 3);
  }
  // stage conv1 weights as [ic][ky][oc][kx8]: per-oc b128 gives 4 packed
  // kx-pairs feeding v_dot2c_f32_bf16 directly
  {
    const uint16_t* w1 = w + offs[0];
    for (int e = tid; e < C1 * 256; e += 512) {
      int oc = e >> 8, rem = e & 255, icky = rem >> 3, kx = e & 7;
      w_lds[(icky * C1 + oc) * K1 + kx] = w1[e];
    }
  }
  __syncthreads();

  // ---- conv1: thread = position (400 of 512); packed-pair dot2 ---------
  {
    const int pos = tid;
    const int oy = pos / O1, ox = pos % O1;
    const uint16_t* b1 = w + offs[1];
    float acc[C1];
    if (pos < O1 * O1) {
#pragma unroll
      for (int k = 0; k < C1; ++k) acc[k] = 0.f;
      for (int ic = 0; ic < C_IN; ++ic) {
        const int slot = (ic == 0 ? s0 : ic == 1 ? s1m : ic == 2 ? s2m : s3);
        const IT* plane = obs + slot * (HW * HW);
        for (int ky = 0; ky < K1; ++ky) {
          const IT* row = plane + (oy * S1 + ky) * HW + ox * S1;
          // 8 pixels -> 4 packed bf16 pairs
          des::bf16x2 pxp[4];
          if constexpr (sizeof(IT) == 1) {           // u8 frames (dword loads)
            uint32_t d0 = *reinterpret_cast<const uint32_t*>(row);
            uint32_t d1 = *reinterpret_cast<const uint32_t*>(row + 4);
#pragma unroll
            for (int j = 0; j < 2; ++j) {
              uint32_t d = j ? d1 : d0;
              pxp[2 * j] = des::pack_bf16((float)(d & 0xFF),
                                          (float)((d >> 8) & 0xFF));
              pxp[2 * j + 1] = des::pack_bf16((float)((d >> 16) & 0xFF),
                                              (float)(d >> 24));
            }
          } else {                                   // bf16 obs: already pairs
            uint2 d0 = *reinterpret_cast<const uint2*>(row);
            uint2 d1 = *reinterpret_cast<const uint2*>(row + 4);
            pxp[0] = __builtin_bit_cast(des::bf16x2, d0.x);
            pxp[1] = __builtin_bit_cast(des::bf16x2, d0.y);
            pxp[2] = __builtin_bit_cast(des::bf16x2, d1.x);
            pxp[3] = __builtin_bit_cast(des::bf16x2, d1.y);
          }
          const uint16_t* wrow = w_lds + ((ic * K1 + ky) * C1) * K1;
#pragma unroll
          for (int oc = 0; oc < C1; ++oc) {
            uint4 wb = *reinterpret_cast<const uint4*>(wrow + oc * K1);
            acc[oc] = des::dot2(pxp[0], __builtin_bit_cast(des::bf16x2, wb.x),
                                acc[oc]);
            acc[oc] = des::dot2(pxp[1], __builtin_bit_cast(des::bf16x2, wb.y),
                                acc[oc]);
            acc[oc] = des::dot2(pxp[2], __builtin_bit_cast(des::bf16x2, wb.z),
                                acc[oc]);
            acc[oc] = des::dot2(pxp[3], __builtin_bit_cast(des::bf16x2, wb.w),
                                acc[oc]);
          }
        }
      }
#pragma unroll
      for (int oc = 0; oc < C1; ++oc) {
        float v = fmaxf(acc[oc] + des::load1(b1 + oc), 0.f);
        a1[oc * (O1 * O1) + pos] = des::f32_to_bf16(v);
      }
    }
  }
  __syncthreads();

  // ---- conv2 in two 32-oc halves: lane&31 = oc, lane>>5 = pos stream ----
  // q = (ic, ky); the [q][oc][kx4] tile makes w a uint2 of 2 packed
  // kx-pairs, and a1's kx run is contiguous (even index -> aligned b32
  // pair reads) -> 2 LDS dwords + 2 dot2 per 4 MACs per position
  for (int half = 0; half < 2; ++half) {
    const uint16_t* w2 = w + offs[2];
    for (int e = tid; e < 32 * 512; e += 512) {
      int oc = e >> 9, icc = e & 511;
      w_lds[(icc >> 2) * 128 + oc * 4 + (icc & 3)] =
          w2[(half * 32 + oc) * 512 + icc];
    }
    __syncthreads();
    {
      const int oc = lane & 31;
      const uint16_t* b2 = w + offs[3];
      const float bias = des::load1(b2 + half * 32 + oc);
      const int pb = (wid * 2 + (lane >> 5)) * 8;  // 16 streams x 8 = 128 >= 81
      float acc[8];
      int pbase[8];
#pragma unroll
      for (int p = 0; p < 8; ++p) {
        acc[p] = 0.f;
        int pos = pb + p;
        if (pos >= O2 * O2) pos = O2 * O2 - 1;  // clamp; result discarded
        pbase[p] = (pos / O2) * S2 * O1 + (pos % O2) * S2;
      }
      for (int q = 0; q < 128; ++q) {
        uint2 wp = *reinterpret_cast<const uint2*>(w_lds + q * 128 + oc * 4);
        const int ab = t2[q << 2];        // ic*400 + ky*20 (kx = 0), even
#pragma unroll
        for (int p = 0; p < 8; ++p) {
          const uint16_t* ap = a1 + pbase[p] + ab;
          uint32_t a01 = *reinterpret_cast<const uint32_t*>(ap);
          uint32_t a23 = *reinterpret_cast<const uint32_t*>(ap + 2);
          acc[p] = des::dot2(__builtin_bit_cast(des::bf16x2, a01),
                             __builtin_bit_cast(des::bf16x2, wp.x), acc[p]);
          acc[p] = des::dot2(__builtin_bit_cast(des::bf16x2, a23),
                             __builtin_bit_cast(des::bf16x2, wp.y), acc[p]);
        }
      }
#pragma unroll
      for (int p = 0; p < 8; ++p) {
        int pos = pb + p;
        if (pos < O2 * O2)
          a2[(half * 32 + oc) * (O2 * O2) + pos] =
              des::f32_to_bf16(fmaxf(acc[p] + bias, 0.f));
      }
    }
    __syncthreads();
  }

  // ---- conv3: two oc-halves; output = fc input tile in LDS --------------
  {
    const uint16_t* w3 = w + offs[4];
    const uint16_t* b3 = w + offs[5];
    for (int half = 0; half < 2; ++half) {
      // stage [icc/4][oc32][4] for ocs [half*32, half*32+32)
      for (int e = tid; e < 32 * 576; e += 512) {
        int oc = e / 576, icc = e % 576;
        w_lds[(icc >> 2) * 128 + oc * 4 + (icc & 3)] =
            w3[(half * 32 + oc) * 576 + icc];
      }
      __syncthreads();
      // lane&31 = oc-in-half; lane>>5 selects position stream (2 streams)
      const int oc = (lane & 31);
      const float bias = des::load1(b3 + half * 32 + oc);
      for (int pb = (wid * 2 + (lane >> 5)) * 4; pb < O3 * O3; pb += 64) {
        float acc[4];
        int pbase[4];
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          acc[p] = 0.f;
          int pos = pb + p;
          if (pos >= O3 * O3) pos = O3 * O3 - 1;
          pbase[p] = (pos / O3) * O2 + (pos % O3);
        }
        for (int icc = 0; icc < 576; icc += 4) {
          float4 w4 = des::load4(w_lds + (icc >> 2) * 128 + oc * 4);
          const int4 off = *reinterpret_cast<const int4*>(t3 + icc);
#pragma unroll
          for (int p = 0; p < 4; ++p) {
            float a0 = des::bf16_to_f32(a2[pbase[p] + off.x]);
            float b0 = des::bf16_to_f32(a2[pbase[p] + off.y]);
            float c0 = des::bf16_to_f32(a2[pbase[p] + off.z]);
            float d0 = des::bf16_to_f32(a2[pbase[p] + off.w]);
            acc[p] += a0 * w4.x + b0 * w4.y + c0 * w4.z + d0 * w4.w;
          }
        }
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          int pos = pb + p;
          if (pos < O3 * O3)
            a3[(half * 32 + oc) * (O3 * O3) + pos] =
                des::f32_to_bf16(fmaxf(acc[p] + bias, 0.f));
        }
      }
      __syncthreads();
    }
  }

  // ---- fc 3136 -> 512 (relu): HBM-bound, 4 rows of ILP per wave ---------
  {
    const uint16_t* wf = w + offs[6];
    const uint16_t* bf = w + offs[7];
    for (int rr = 0; rr < 64; rr += 4) {
      const int row0 = wid * 64 + rr;
      const uint16_t* w0 = wf + (int64_t)row0 * FC_IN;
      float acc0 = 0.f, acc1 = 0.f, acc2 = 0.f, acc3 = 0.f;
      for (int ib = lane * 8; ib < FC_IN; ib += 64 * 8) {
        uint4 xp = *reinterpret_cast<const uint4*>(a3 + ib);
        uint4 wa = *reinterpret_cast<const uint4*>(w0 + ib);
        uint4 wb = *reinterpret_cast<const uint4*>(w0 + FC_IN + ib);
        uint4 wc = *reinterpret_cast<const uint4*>(w0 + 2 * FC_IN + ib);
        uint4 wd = *reinterpret_cast<const uint4*>(w0 + 3 * FC_IN + ib);
        const uint32_t* xq = reinterpret_cast<const uint32_t*>(&xp);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          des::bf16x2 xj = __builtin_bit_cast(des::bf16x2, xq[j]);
          acc0 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
                   reinterpret_cast<const uint32_t*>(&wa)[j]), acc0);
          acc1 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
                   reinterpret_cast<const uint32_t*>(&wb)[j]), acc1);
          acc2 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
                   reinterpret_cast<const uint32_t*>(&wc)[j]), acc2);
          acc3 = des::dot2(xj, __builtin_bit_cast(des::bf16x2,
                   reinterpret_cast<const uint32_t*>(&wd)[j]), acc3);
        }
      }
      acc0 = des::wave_reduce_sum_f(acc0);
      acc1 = des::wave_reduce_sum_f(acc1);
      acc2 = des::wave_reduce_sum_f(acc2);
      acc3 = des::wave_reduce_sum_f(acc3);
      if (lane < 4) {
        float acc = lane == 0 ? acc0 : lane == 1 ? acc1
                    : lane == 2 ? acc2 : acc3;
        float bias = des::bf16_to_f32(bf[row0 + lane]);
        fcb[row0 + lane] = fmaxf(acc + bias, 0.f);
      }
    }
  }
  __syncthreads();

  // ---- head 512 -> A + argmax ------------------------------------------
  des::gemv_rows<uint16_t, 0>(w + offs[8], w + offs[9], fcb, hd, FC_OUT, A,
                              FC_OUT, lane, wid, 8);
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

extern "C" void des_launch_nature_cnn_forward(
    const uint16_t* popw, int64_t row_stride, const void* frames, bool obs_u8,
    const int* slot_map_dev, int* actions, float* logits_out, int A, int P,
    const int64_t* offs_dev, hipStream_t stream) {
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)nature_conv_kernel<uint8_t>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, CONV_LDS);
    hipFuncSetAttribute((const void*)nature_conv_kernel<uint16_t>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, CONV_LDS);
    attr_set = true;
  }
  if (obs_u8)
    hipLaunchKernelGGL(nature_conv_kernel<uint8_t>, dim3((uint32_t)P),
                       dim3(512), CONV_LDS, stream, popw, row_stride,
                       (const uint8_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev);
  else
    hipLaunchKernelGGL(nature_conv_kernel<uint16_t>, dim3((uint32_t)P),
                       dim3(512), CONV_LDS, stream, popw, row_stride,
                       (const uint16_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev);
}
