#include "hip/hip_runtime.h"
// K5: conv_forward_pop (Nature-CNN) — whole per-member policy forward in ONE
// block-per-member kernel.
//
// Per member (batch=1): conv 8x8s4x32 -> 4x4s2x64 -> 3x3s1x64 -> fc 3136->512
// -> head 512->A -> argmax.  Design for MI355X:
//   * one 512-thread block (8 waves) per member; activations live entirely in
//     LDS (~123 KiB carve, 1 block/CU) and never touch HBM;
//   * per-stage weight tiles are staged global->LDS with coalesced reads and
//     a conflict-free [k/4][oc][4] layout so the inner loop is ds_read_b64 of
//     4 consecutive bf16 per lane (G13/G4);
//   * conv inner loops: activation value broadcast (uniform LDS address),
//     per-lane weight vector, fp32 accumulate;
//   * the FC layer (94% of weights -> the HBM-bound stage) streams rows with
//     the wave-GEMV (gemv.h): bf16x4 loads, 8 waves in flight per CU;
//   * action argmax fused into the epilogue (one kernel per env step).
// Step time at pop=8192/GPU is bounded by FC weight traffic:
// 8192 * 3.2 MiB / 6.3 TBps ~= 4.3 ms; conv stages are VALU work (~0.8 ms).
#include "common.h"
#include "gemv.h"

namespace {

// geometry (fixed, models/nature_cnn.py)
constexpr int C_IN = 4, HW = 84;
constexpr int C1 = 32, O1 = 20, K1 = 8, S1 = 4;   // conv1
constexpr int C2 = 64, O2 = 9, K2 = 4, S2 = 2;    // conv2
constexpr int C3 = 64, O3 = 7, K3 = 3;            // conv3 (s1)
constexpr int FC_IN = C3 * O3 * O3;               // 3136
constexpr int FC_OUT = 512;
constexpr int MAX_A = 32;

// LDS carve (bytes, 16-aligned)
constexpr int W_ELEMS = 36864;                         // max transposed tile
constexpr int OFF_W = 0;                               // bf16 [W_ELEMS]
constexpr int OFF_A1 = OFF_W + W_ELEMS * 2;            // bf16 [32*20*20]
constexpr int OFF_A2 = OFF_A1 + C1 * O1 * O1 * 2;      // bf16 [64*9*9]
constexpr int OFF_A3 = OFF_A2 + ((C2 * O2 * O2 * 2 + 15) & ~15);  // f32 [3136]
constexpr int OFF_FC = OFF_A3 + FC_IN * 4;             // f32 [512]
constexpr int OFF_HD = OFF_FC + FC_OUT * 4;            // f32 [MAX_A]
constexpr int LDS_BYTES = OFF_HD + MAX_A * 4;

template <typename IT>
__device__ __forceinline__ float obs_val(IT v);
template <>
__device__ __forceinline__ float obs_val<uint8_t>(uint8_t v) { return (float)v; }
template <>
__device__ __forceinline__ float obs_val<uint16_t>(uint16_t v) {
  return des::bf16_to_f32(v);
}

template <typename IT>
__global__ __launch_bounds__(512, 1) void nature_cnn_forward_kernel(
    const uint16_t* __restrict__ popw, int64_t row_stride,
    const IT* __restrict__ frames,  // [P][4][84][84], channel c = slot[c]
    const int* __restrict__ slot_map,  // [4] logical->physical slot
    int* __restrict__ actions, float* __restrict__ logits_out, int A, int P,
    const int64_t* __restrict__ offs) {  // ow1,ob1,ow2,ob2,ow3,ob3,owf,obf,owh,obh
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* w_lds = reinterpret_cast<uint16_t*>(smem + OFF_W);
  uint16_t* a1 = reinterpret_cast<uint16_t*>(smem + OFF_A1);
  uint16_t* a2 = reinterpret_cast<uint16_t*>(smem + OFF_A2);
  float* a3 = reinterpret_cast<float*>(smem + OFF_A3);
  float* fcb = reinterpret_cast<float*>(smem + OFF_FC);
  float* hd = reinterpret_cast<float*>(smem + OFF_HD);

  const int i = blockIdx.x;
  if (i >= P) return;
  const uint16_t* w = popw + (int64_t)i * row_stride;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const IT* obs = frames + (int64_t)i * (4 * HW * HW);
  const int s0 = slot_map[0], s1m = slot_map[1], s2m = slot_map[2],
            s3 = slot_map[3];

  // ---- stage W1: transpose conv1 weights [oc][icc] -> [icc][oc] ----------
  {
    const uint16_t* w1 = w + offs[0];
    for (int e = tid; e < C1 * 256; e += 512) {
      int oc = e >> 8, icc = e & 255;
      w_lds[icc * C1 + oc] = w1[e];
    }
  }
  __syncthreads();

  // ---- conv1: thread = output position (400), 32 oc accumulators ---------
  {
    float acc[C1];
    const int pos = tid;  // 0..399 active
    if (pos < O1 * O1) {
#pragma unroll
      for (int oc = 0; oc < C1; ++oc) acc[oc] = 0.f;
      const int oy = pos / O1, ox = pos % O1;
      for (int ic = 0; ic < C_IN; ++ic) {
        const int slot = (ic == 0 ? s0 : ic == 1 ? s1m : ic == 2 ? s2m : s3);
        const IT* plane = obs + slot * (HW * HW);
        for (int ky = 0; ky < K1; ++ky) {
          const IT* row = plane + (oy * S1 + ky) * HW + ox * S1;
          float px[K1];
#pragma unroll
          for (int kx = 0; kx < K1; ++kx) px[kx] = obs_val<IT>(row[kx]);
          const uint16_t* wrow = w_lds + ((ic * K1 + ky) * K1) * C1;
#pragma unroll
          for (int kx = 0; kx < K1; ++kx) {
            const uint16_t* wv = wrow + kx * C1;
#pragma unroll  // static acc[] indices (rule 20: runtime-indexed regs spill)
            for (int oc = 0; oc < C1; oc += 4) {
              float4 w4 = des::load4(wv + oc);
              acc[oc] += px[kx] * w4.x;
              acc[oc + 1] += px[kx] * w4.y;
              acc[oc + 2] += px[kx] * w4.z;
              acc[oc + 3] += px[kx] * w4.w;
            }
          }
        }
      }
      const uint16_t* b1 = w + offs[1];
#pragma unroll
      for (int oc = 0; oc < C1; ++oc) {
        float v = fmaxf(acc[oc] + des::load1(b1 + oc), 0.f);
        a1[oc * (O1 * O1) + pos] = des::f32_to_bf16(v);
      }
    }
  }
  __syncthreads();

  // ---- stage W2: [oc][icc](K=512) -> [icc/4][oc][4] -----------------------
  {
    const uint16_t* w2 = w + offs[2];
    for (int e = tid; e < C2 * 512; e += 512) {
      int oc = e >> 9, icc = e & 511;
      w_lds[(icc >> 2) * (C2 * 4) + oc * 4 + (icc & 3)] = w2[e];
    }
  }
  __syncthreads();

  // ---- conv2: lane = oc (64), wave = position slice (81 positions) --------
  {
    const uint16_t* b2 = w + offs[3];
    const float bias = des::load1(b2 + lane);
    for (int pos = wid; pos < O2 * O2; pos += 8) {
      const int oy = pos / O2, ox = pos % O2;
      float acc = 0.f;
      for (int icc = 0; icc < 512; icc += 4) {
        float4 w4 = des::load4(w_lds + (icc >> 2) * (C2 * 4) + lane * 4);
        // icc -> (ic, ky, kx): ic = icc/16, ky = (icc%16)/4, kx = icc%4
        float av[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int cc = icc + j;
          int ic = cc >> 4, ky = (cc >> 2) & 3, kx = cc & 3;
          av[j] = des::bf16_to_f32(
              a1[ic * (O1 * O1) + (oy * S2 + ky) * O1 + (ox * S2 + kx)]);
        }
        acc += av[0] * w4.x + av[1] * w4.y + av[2] * w4.z + av[3] * w4.w;
      }
      a2[lane * (O2 * O2) + pos] = des::f32_to_bf16(fmaxf(acc + bias, 0.f));
    }
  }
  __syncthreads();

  // ---- stage W3: [oc][icc](K=576) -> [icc/4][oc][4] -----------------------
  {
    const uint16_t* w3 = w + offs[4];
    for (int e = tid; e < C3 * 576; e += 512) {
      int oc = e / 576, icc = e % 576;
      w_lds[(icc >> 2) * (C3 * 4) + oc * 4 + (icc & 3)] = w3[e];
    }
  }
  __syncthreads();

  // ---- conv3: lane = oc (64), 49 positions --------------------------------
  {
    const uint16_t* b3 = w + offs[5];
    const float bias = des::load1(b3 + lane);
    for (int pos = wid; pos < O3 * O3; pos += 8) {
      const int oy = pos / O3, ox = pos % O3;
      float acc = 0.f;
      for (int icc = 0; icc < 576; icc += 4) {
        float4 w4 = des::load4(w_lds + (icc >> 2) * (C3 * 4) + lane * 4);
        float av[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int cc = icc + j;
          int ic = cc / 9, r = cc % 9, ky = r / 3, kx = r % 3;
          av[j] = des::bf16_to_f32(
              a2[ic * (O2 * O2) + (oy + ky) * O2 + (ox + kx)]);
        }
        acc += av[0] * w4.x + av[1] * w4.y + av[2] * w4.z + av[3] * w4.w;
      }
      a3[lane * (O3 * O3) + pos] = fmaxf(acc + bias, 0.f);
    }
  }
  __syncthreads();

  // ---- fc 3136 -> 512 (relu): the HBM-bound stage -------------------------
  des::gemv_rows<uint16_t, 2>(w + offs[6], w + offs[7], a3, fcb, FC_IN, FC_OUT,
                              FC_IN, lane, wid, 8);
  __syncthreads();

  // ---- head 512 -> A + argmax --------------------------------------------
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
  // dynamic-LDS carve is 121.5 KiB (> the 64 KiB default cap): opt in once
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)nature_cnn_forward_kernel<uint8_t>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_BYTES);
    hipFuncSetAttribute((const void*)nature_cnn_forward_kernel<uint16_t>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, LDS_BYTES);
    attr_set = true;
  }
  dim3 grid((uint32_t)P), block(512);
  if (obs_u8)
    hipLaunchKernelGGL(nature_cnn_forward_kernel<uint8_t>, grid, block,
                       LDS_BYTES, stream, popw, row_stride,
                       (const uint8_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev);
  else
    hipLaunchKernelGGL(nature_cnn_forward_kernel<uint16_t>, grid, block,
                       LDS_BYTES, stream, popw, row_stride,
                       (const uint16_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev);
}
