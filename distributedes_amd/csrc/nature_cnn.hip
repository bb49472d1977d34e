// K5: conv_forward_pop (Nature-CNN) — per-member policy forward, one fused
// block-per-member kernel (BASELINE configs #3/#4).
//
// ONE 512-thread (8-wave) block per member, ~64 KiB LDS -> 2 blocks/CU:
//
//   convs as MFMA implicit GEMM (v_mfma_f32_16x16x32_bf16):
//     conv1  M=400(25 tiles) x N=32(2) x K=256   (im2col of the obs frames)
//     conv2  M=96(6, 81 real) x N=64(4) x K=512  (im2col of a1, LDS)
//     conv3  M=64(4, 49 real) x N=64(4) x K=576  (im2col of a2, LDS)
//   Per K-step (32): all 512 threads cooperatively build the A-slab
//   (im2col gather -> [M][32] bf16, 64-B rows) and B-slab ([N][32] from the
//   member's flat [oc][k] weights: each lane's b128 fragment read is 8
//   consecutive k of its oc row -> no transpose pass at all), then each
//   wave issues one MFMA per owned 16x16 C-tile.  Fragment maps verified on
//   hardware by tools/probe_mfma.hip:
//     A: lane l -> A[l&15][(l>>4)*8+j]   B: lane l -> B[(l>>4)*8+j][l&15]
//     D: lane l, reg r -> D[(l>>4)*4+r][l&15]
//   Activations never leave LDS; relu+bias fused in the tile epilogues.
//
//   fc 3136->512 (94% of weights = the HBM floor): 4 rows of ILP per wave,
//   16-B bf16x8 loads on v_dot2c_f32_bf16; head 512->A + argmax fused.
//
// With 2 blocks/CU one block's (now tiny) conv phase and the other block's
// HBM-bound FC stream overlap; the step floor is the FC traffic:
// pop-8192 x 3.2 MiB / 6.3 TB/s ~= 4.3 ms.
#include <cstdlib>

#include "common.h"
#include "gemv.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short mfma_ab;
typedef __attribute__((ext_vector_type(4))) float mfma_cd;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;  // nt-loadable

// geometry (fixed, models/nature_cnn.py)
constexpr int C_IN = 4, HW = 84;
constexpr int O1 = 20, K1 = 8, S1 = 4;            // conv1 8x8 s4, out 32ch
constexpr int O2 = 9, S2 = 2;                     // conv2 4x4 s2, out 64ch
constexpr int O3 = 7;                             // conv3 3x3 s1, out 64ch
constexpr int FC_IN = 3136, FC_OUT = 512, MAX_A = 32;

// LDS carve (bytes, 16-aligned): ONLY the activations live in LDS.
// conv1 is DIRECT-FRAGMENT from the L2-warm frames; every conv's
// B-fragments read DIRECTLY from the member's weights (8 consecutive k of
// an oc row = one aligned 16-B global load, L1-cached across the m-tiles
// that share it) — no B-slab; conv2's A-slab overlays the not-yet-written
// a2 region and conv3's the dead a1 region.  36 KB/block -> 4 blocks/CU,
// i.e. 32 waves streaming during the HBM-bound FC phase (round 1: 16).
constexpr int OFF_A1 = 0;                         // [32][400] bf16 = 25600
constexpr int OFF_A2 = OFF_A1 + 32 * 400 * 2;     // [64][81] bf16 = 10368
constexpr int CONV_LDS = (OFF_A2 + 64 * 81 * 2 + 15) & ~15;
constexpr int OFF_A3 = OFF_A1;                    // bf16 fc input [3136]
constexpr int OFF_FC = OFF_A3 + FC_IN * 2;        // f32 [512]
constexpr int OFF_HD = OFF_FC + FC_OUT * 4;       // f32 [MAX_A]
static_assert(OFF_HD + MAX_A * 4 <= OFF_A2, "fc overlay exceeds a1 carve");

// one MFMA step over every tile a wave owns (MAXT static -> regs, rule 20)
template <int MAXT>
__device__ __forceinline__ void mfma_tiles(const uint16_t* sa,
                                           const uint16_t* sb,
                                           mfma_cd (&acc)[MAXT], int wid,
                                           int lane, int ntiles, int ncols) {
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + 8 * u;
    if (t < ntiles) {
      int mt = t / ncols, nt = t % ncols;
      mfma_ab af = *reinterpret_cast<const mfma_ab*>(
          sa + (mt * 16 + (lane & 15)) * 32 + (lane >> 4) * 8);
      mfma_ab bf = *reinterpret_cast<const mfma_ab*>(
          sb + (nt * 16 + (lane & 15)) * 32 + (lane >> 4) * 8);
      acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[u], 0, 0, 0);
    }
  }
}

// mfma step with the B-fragment read directly from the member's global
// weight rows (ldb = K of the layer); the same fragment is L1-hot for the
// m-tiles sharing an nt
template <int MAXT>
__device__ __forceinline__ void mfma_tiles_gb(const uint16_t* sa,
                                              const uint16_t* gb, int ldb,
                                              int k0, mfma_cd (&acc)[MAXT],
                                              int wid, int lane, int ntiles,
                                              int ncols) {
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + 8 * u;
    if (t < ntiles) {
      int mt = t / ncols, nt = t % ncols;
      mfma_ab af = *reinterpret_cast<const mfma_ab*>(
          sa + (mt * 16 + (lane & 15)) * 32 + (lane >> 4) * 8);
      mfma_ab bf = *reinterpret_cast<const mfma_ab*>(
          gb + (int64_t)(nt * 16 + (lane & 15)) * ldb + k0 + (lane >> 4) * 8);
      acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[u], 0, 0, 0);
    }
  }
}

// tile epilogue: bias + relu + scatter into [oc][M_real] LDS layout
template <int MAXT>
__device__ __forceinline__ void tile_epilogue(const mfma_cd (&acc)[MAXT],
                                              const uint16_t* bias,
                                              uint16_t* out, int m_real,
                                              int wid, int lane, int ntiles,
                                              int ncols) {
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + 8 * u;
    if (t < ntiles) {
      int mt = t / ncols, nt = t % ncols;
      int n = nt * 16 + (lane & 15);
      int m0 = mt * 16 + ((lane >> 4) << 2);
      float b = des::bf16_to_f32(bias[n]);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = m0 + r;
        if (m < m_real)
          out[n * m_real + m] = des::f32_to_bf16(fmaxf(acc[u][r] + b, 0.f));
      }
    }
  }
}

template <typename IT, int MINW = 6>
__global__ __launch_bounds__(512, MINW) void nature_conv_kernel(
    const uint16_t* __restrict__ popw, int64_t row_stride,
    const IT* __restrict__ frames,     // [P][4][84][84]
    const int* __restrict__ slot_map,  // [4] logical->physical slot
    int* __restrict__ actions, float* __restrict__ logits_out, int A, int P,
    const int64_t* __restrict__ offs, unsigned stagger) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* a1 = reinterpret_cast<uint16_t*>(smem + OFF_A1);
  uint16_t* a2 = reinterpret_cast<uint16_t*>(smem + OFF_A2);
  uint16_t* sa2 = a2;   // conv2 A-slab: a2 not written until conv2 epilogue
  uint16_t* sa3 = a1;   // conv3 A-slab: a1 dead once conv2 is done
  uint16_t* a3 = reinterpret_cast<uint16_t*>(smem + OFF_A3);
  float* fcb = reinterpret_cast<float*>(smem + OFF_FC);
  float* hd = reinterpret_cast<float*>(smem + OFF_HD);

  const int i = blockIdx.x;
  if (i >= P) return;
  // phase stagger: the two co-resident blocks of a CU otherwise run their
  // conv (compute) and FC (HBM) phases in lockstep, so conv time never
  // hides under the FC stream.  Delaying the odd block of the FIRST
  // resident round by ~one conv phase anti-phases the whole kernel; block
  // retirement then sustains the offset (speed heuristic only — dispatch
  // order is not a correctness assumption here).
  if (stagger && (blockIdx.x & 1) && blockIdx.x < 1024) {
    unsigned long long t0 = __builtin_amdgcn_s_memtime();
    while (__builtin_amdgcn_s_memtime() - t0 < (unsigned long long)stagger)
      __builtin_amdgcn_s_sleep(16);
  }
  const uint16_t* w = popw + (int64_t)i * row_stride;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const IT* obs = frames + (int64_t)i * (4 * HW * HW);
  const int s0 = slot_map[0], s1m = slot_map[1], s2m = slot_map[2],
            s3 = slot_map[3];

  // ======================= conv1: 400 x 32 x 256 =========================
  // DIRECT-FRAGMENT: each MFMA A-fragment (8 consecutive k of one K-step)
  // is 8 contiguous pixels of one frame row — loaded straight from the
  // L2-warm frames (28 KB/member) and packed in registers.  No A-slab, no
  // staging pass; only the tiny B-slab stays in LDS (the freed 25.6 KB
  // lifts whole-kernel residency to 3 blocks/CU for the FC stream).
  {
    const uint16_t* w1 = w + offs[0];
    mfma_cd acc[7];
#pragma unroll
    for (int u = 0; u < 7; ++u) acc[u] = 0.f;  // ext-vector splat
    for (int ks = 0; ks < 8; ++ks) {   // k-step 32 = one ic, 4 ky rows
      const int ic = ks >> 1, ky0 = (ks & 1) << 2;
      const int slot = (ic == 0 ? s0 : ic == 1 ? s1m : ic == 2 ? s2m : s3);
      const IT* plane = obs + slot * (HW * HW);
      const int kyl = lane >> 4;       // fragment chunk = one ky row
#pragma unroll
      for (int u = 0; u < 7; ++u) {
        int t = wid + 8 * u;
        if (t < 50) {
          int mt = t / 2, nt = t % 2;
          int m = mt * 16 + (lane & 15);
          int oy = m / O1, ox = m % O1;
          const IT* row = plane + (oy * S1 + ky0 + kyl) * HW + ox * S1;
          mfma_ab af;
          if constexpr (sizeof(IT) == 1) {
            uint32_t d0 = *reinterpret_cast<const uint32_t*>(row);
            uint32_t d1 = *reinterpret_cast<const uint32_t*>(row + 4);
            uint4 pk;
            pk.x = __builtin_bit_cast(uint32_t,
                   des::pack_bf16((float)(d0 & 0xFF), (float)((d0 >> 8) & 0xFF)));
            pk.y = __builtin_bit_cast(uint32_t,
                   des::pack_bf16((float)((d0 >> 16) & 0xFF), (float)(d0 >> 24)));
            pk.z = __builtin_bit_cast(uint32_t,
                   des::pack_bf16((float)(d1 & 0xFF), (float)((d1 >> 8) & 0xFF)));
            pk.w = __builtin_bit_cast(uint32_t,
                   des::pack_bf16((float)((d1 >> 16) & 0xFF), (float)(d1 >> 24)));
            af = __builtin_bit_cast(mfma_ab, pk);
          } else {
            uint2 d0 = *reinterpret_cast<const uint2*>(row);
            uint2 d1 = *reinterpret_cast<const uint2*>(row + 4);
            af = __builtin_bit_cast(mfma_ab, uint4{d0.x, d0.y, d1.x, d1.y});
          }
          mfma_ab bf = *reinterpret_cast<const mfma_ab*>(
              w1 + (nt * 16 + (lane & 15)) * 256 + ks * 32 + (lane >> 4) * 8);
          acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[u],
                                                           0, 0, 0);
        }
      }
    }
    tile_epilogue<7>(acc, w + offs[1], a1, 400, wid, lane, 50, 2);
  }
  __syncthreads();

  // ======================= conv2: 96 x 64 x 512 ==========================
  {
    const uint16_t* w2 = w + offs[2];
    mfma_cd acc[3];
#pragma unroll
    for (int u = 0; u < 3; ++u) acc[u] = 0.f;  // ext-vector splat
    for (int ks = 0; ks < 16; ++ks) {  // k-step 32 = 2 ic x 4 ky x 4 kx
      // A-slab: unit = (m, icloc, ky) -> 4 contiguous a1 x-values (8 B)
      for (int e = tid; e < 96 * 8; e += 512) {
        int m = e >> 3, icl = (e >> 2) & 1, ky = e & 3;
        uint2 v = {0u, 0u};
        if (m < O2 * O2) {
          int oy = m / O2, ox = m % O2;
          const uint16_t* ap = a1 + (ks * 2 + icl) * 400 +
                               (oy * S2 + ky) * O1 + ox * S2;
          v.x = *reinterpret_cast<const uint32_t*>(ap);       // even index
          v.y = *reinterpret_cast<const uint32_t*>(ap + 2);
        }
        *reinterpret_cast<uint2*>(sa2 + m * 32 + icl * 16 + ky * 4) = v;
      }
      __syncthreads();
      mfma_tiles_gb<3>(sa2, w2, 512, ks * 32, acc, wid, lane, 24, 4);
      __syncthreads();
    }
    tile_epilogue<3>(acc, w + offs[3], a2, 81, wid, lane, 24, 4);
  }
  __syncthreads();

  // ======================= conv3: 64 x 64 x 576 ==========================
  {
    const uint16_t* w3 = w + offs[4];
    mfma_cd acc[2];
#pragma unroll
    for (int u = 0; u < 2; ++u) acc[u] = 0.f;  // ext-vector splat
    for (int ks = 0; ks < 18; ++ks) {  // 576 = 18 x 32 exactly
      for (int e = tid; e < 64 * 32; e += 512) {
        int m = e >> 5, kk = e & 31;
        int icc = ks * 32 + kk;
        int ic = icc / 9, rm = icc - ic * 9, ky = rm / 3, kx = rm - ky * 3;
        uint16_t v = 0;
        if (m < O3 * O3) {
          int oy = m / O3, ox = m % O3;
          v = a2[ic * 81 + (oy + ky) * O2 + ox + kx];
        }
        sa3[m * 32 + kk] = v;
      }
      __syncthreads();
      mfma_tiles_gb<2>(sa3, w3, 576, ks * 32, acc, wid, lane, 16, 4);
      __syncthreads();
    }
    // a2 fully consumed above; a3 overlays the (dead) a1 region
    tile_epilogue<2>(acc, w + offs[5], a3, 49, wid, lane, 16, 4);
  }
  __syncthreads();

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
        // nt: weights are read exactly once per step by one CU -> bypass
        // caches (microarch nt-weights row: decode-like stream)
        u32x4 wa = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4*>(w0 + ib));
        u32x4 wb = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4*>(w0 + FC_IN + ib));
        u32x4 wc = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4*>(w0 + 2 * FC_IN + ib));
        u32x4 wd = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4*>(w0 + 3 * FC_IN + ib));
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
  static unsigned stagger = [] {
    const char* e = getenv("DES_NATURE_STAGGER");
    return e ? (unsigned)atoi(e) : 0u;
  }();
  // 6 waves/SIMD (3 blocks/CU, zero spills) A/B-measured ahead of the
  // 8-wave 4-block variant (1.762M vs 1.664M env steps/s: its 8 spilled
  // VGPRs cost more than the extra streaming waves buy)
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)nature_conv_kernel<uint8_t, 6>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, CONV_LDS);
    hipFuncSetAttribute((const void*)nature_conv_kernel<uint16_t, 6>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, CONV_LDS);
    attr_set = true;
  }
  if (obs_u8)
    hipLaunchKernelGGL((nature_conv_kernel<uint8_t, 6>), dim3((uint32_t)P),
                       dim3(512), CONV_LDS, stream, popw, row_stride,
                       (const uint8_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev, stagger);
  else
    hipLaunchKernelGGL((nature_conv_kernel<uint16_t, 6>), dim3((uint32_t)P),
                       dim3(512), CONV_LDS, stream, popw, row_stride,
                       (const uint16_t*)frames, slot_map_dev, actions,
                       logits_out, A, P, offs_dev, stagger);
}
