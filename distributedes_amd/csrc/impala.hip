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

// 3x3 pad-1 conv as MFMA implicit GEMM per spatial tile, DIRECT-FRAGMENT
// design (round-2 rewrite; the round-1 version staged an im2col A-slab per
// K-step, and the committed ablation attributed 0.54 ms of the 0.81 ms
// res-conv to that staging/barrier skeleton and 0.26 ms to the gather):
//
//   * the halo tile is staged TRANSPOSED: a_lds[y][x][ci] with ci
//     contiguous (stride CIS, bank-spread).  The GEMM K order is then
//     k' = tap*CIP + ci, so every MFMA A-fragment (8 consecutive k' for one
//     tap) is ONE 16-byte ds_read_b128 at a shift-computed offset — no
//     im2col pass, no A-slab, no re-read.
//   * the member's weights are reordered once per block into the same
//     k'-order (sb[co][k'], all K resident), so the k-loop contains ZERO
//     barriers and zero staging: per K-step each wave issues one A-read +
//     one B-read per owned 16x16 C-tile and the MFMA.
//   M = TS*TS positions, N = CO, K' = CIP*9 zero-padded to 32-multiples.
// Fragment maps hardware-verified by tools/probe_mfma.hip.
template <int CI, int CO, int HW, int TSY, int TSX, bool RELU_IN,
          bool ADD_SKIP>
__global__ __launch_bounds__(TSY * TSX) void impala_conv3_kernel(
    const uint16_t* __restrict__ in,    // [P][CI][HW][HW]
    uint16_t* __restrict__ out,         // [P][CO][HW][HW]
    const uint16_t* __restrict__ skip,  // nullable [P][CO][HW][HW]
    const uint16_t* __restrict__ popw, int64_t row_stride, int64_t w_off,
    int64_t b_off, int P) {
  constexpr int THY = TSY + 2, THX = TSX + 2;
  constexpr int M = TSY * TSX;             // 64..512 (rectangular tiles:
                                           // wider tiles amortize the weight
                                           // reorder and cut the halo ratio)
  constexpr int CIP = CI == 3 ? 8 : CI;    // ci zero-padded to an 8-multiple
  constexpr int CIS = CIP == 8 ? 8 : (CIP == 16 ? 24 : 40);
                                           // LDS ci stride: 16B-aligned
                                           // b128 reads, 16-way bank spread
  constexpr int K = CIP * 9;               // k' = tap*CIP + ci
  constexpr int KPAD = ((K + 31) / 32) * 32;
  constexpr int KSTEPS = KPAD / 32;
  constexpr int CPT = CIP / 8;             // 8-chunks per tap
  constexpr int SBS = KPAD + 8;            // sb row stride (bank spread)
  constexpr int NWAVES = M / 64;           // 4 or 1
  constexpr int NTILES = (M / 16) * (CO / 16);
  constexpr int MAXT = NTILES / NWAVES;    // tiles per wave
  constexpr int MP8 = M + 8;               // epilogue row stride (bank spread)
  constexpr int ALDS = THY * THX * CIS;
  constexpr int U0 = ALDS > CO * MP8 ? ALDS : CO * MP8;
  __shared__ uint16_t u_lds[U0];           // halo, then epilogue staging
  __shared__ uint16_t sb[CO * SBS];        // [co][k'] reordered weights
  const int i = blockIdx.x;
  if (i >= P) return;
  constexpr int NTX = HW / TSX;
  const int ty0 = (blockIdx.y / NTX) * TSY, tx0 = (blockIdx.y % NTX) * TSX;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const uint16_t* w = popw + (int64_t)i * row_stride;

  // stage the input halo tile transposed (zero-padded at image edges and in
  // the ci pad lanes, optional fused relu); consecutive tids read
  // consecutive x -> coalesced HBM
  // the CIP pad lanes must be ZEROED, not left as garbage: their sb rows
  // are zero, but uninitialized LDS can hold NaN bit patterns and
  // NaN * 0 = NaN would poison the MFMA accumulator
  if constexpr (CIP > CI) {
    for (int e = tid; e < (CIP - CI) * THY * THX; e += M) {
      int ci = CI + e / (THY * THX), r = e % (THY * THX);
      u_lds[r * CIS + ci] = 0;
    }
  }
  const uint16_t* src = in + (int64_t)i * CI * HW * HW;
  for (int e = tid; e < CI * THY * THX; e += M) {
    int ci = e / (THY * THX), r = e % (THY * THX);
    int yl = r / THX, xl = r % THX;
    int y = ty0 + yl - 1, x = tx0 + xl - 1;
    uint16_t v = 0;
    if (y >= 0 && y < HW && x >= 0 && x < HW) {
      v = src[(ci * HW + y) * HW + x];
      if (RELU_IN && (v & 0x8000)) v = 0;
    }
    u_lds[(yl * THX + xl) * CIS + ci] = v;
  }
  // reorder weights [co][ci][tap] -> sb[co][tap*CIP + ci]: uint4 reads of
  // the contiguous [co][k] source (coalesced HBM; w_off is 8-elem aligned,
  // checked in bindings) scattered into the tap-major LDS layout
  const uint16_t* wsrc = w + w_off;
  constexpr int KSRC = CI * 9;
  if constexpr (KPAD > KSRC || CIP > CI) {   // zero the pad slots
    for (int e = tid; e < CO * KPAD; e += M) {
      int kp = e % KPAD;
      int tap = kp / CIP, ci = kp % CIP;
      if (tap >= 9 || ci >= CI) sb[(e / KPAD) * SBS + kp] = 0;
    }
  }
  for (int e8 = tid; e8 < CO * KSRC / 8; e8 += M) {
    uint4 q = *reinterpret_cast<const uint4*>(wsrc + e8 * 8);
    const uint16_t* h = reinterpret_cast<const uint16_t*>(&q);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int idx = e8 * 8 + j;
      int n = idx / KSRC, k = idx - n * KSRC;
      int ci = k / 9, tap = k - 9 * ci;
      sb[n * SBS + tap * CIP + ci] = h[j];
    }
  }
  __syncthreads();

  mfma_cd acc[MAXT];
#pragma unroll
  for (int u = 0; u < MAXT; ++u) acc[u] = 0.f;
#pragma unroll 2
  for (int ks = 0; ks < KSTEPS; ++ks) {
    const int kc = ks * 4 + (lane >> 4);   // global 8-chunk index
    int tap = kc / CPT;
    if (tap > 8) tap = 8;                  // zero-pad taps: B rows are zero,
    const int ci0 = (kc % CPT) * 8;        // A may read any in-bounds slot
    const int ky = tap / 3, kx = tap - 3 * (tap / 3);
    mfma_ab bfr[CO / 16];
#pragma unroll
    for (int nn = 0; nn < CO / 16; ++nn)
      bfr[nn] = *reinterpret_cast<const mfma_ab*>(
          sb + (nn * 16 + (lane & 15)) * SBS + ks * 32 + (lane >> 4) * 8);
#pragma unroll
    for (int u = 0; u < MAXT; ++u) {
      int t = wid + NWAVES * u;
      int mt = t / (CO / 16), nt = t % (CO / 16);
      int m = mt * 16 + (lane & 15);
      int my = m / TSX + ky, mx = m % TSX + kx;
      mfma_ab af = *reinterpret_cast<const mfma_ab*>(
          u_lds + (my * THX + mx) * CIS + ci0);
      acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr[nt], acc[u],
                                                       0, 0, 0);
    }
  }
  __syncthreads();   // u_lds is re-carved as the epilogue staging buffer

  // epilogue: the naive per-lane scatter hits a 2*HW*HW-byte stride (one
  // cache line per lane) — stage the bias'd tile in LDS (row stride M+8 to
  // spread banks) and do a coalesced cooperative uint4 store instead
  const uint16_t* bias = w + b_off;
  uint16_t* dst = out + (int64_t)i * CO * HW * HW;
  const uint16_t* sk = ADD_SKIP ? skip + (int64_t)i * CO * HW * HW : nullptr;
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + NWAVES * u;
    int mt = t / (CO / 16), nt = t % (CO / 16);
    int n = nt * 16 + (lane & 15);
    int m0 = mt * 16 + ((lane >> 4) << 2);
    float b = des::bf16_to_f32(bias[n]);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      u_lds[n * MP8 + m0 + r] = des::f32_to_bf16(acc[u][r] + b);
  }
  __syncthreads();
  // 8 consecutive u16 of a tile row are contiguous + 16B-aligned in dst
  // (tx0 is a multiple of TS >= 8): store uint4 per thread (ablation showed
  // the narrow store tail at ~350 GB/s was 23% of the kernel)
  for (int e8 = tid; e8 < CO * M / 8; e8 += M) {
    int n = e8 / (M / 8), m = (e8 % (M / 8)) * 8;
    int64_t gidx = ((int64_t)n * HW + ty0 + m / TSX) * HW + tx0 + m % TSX;
    uint4 v = *reinterpret_cast<const uint4*>(u_lds + n * MP8 + m);
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

// Fused residual block for whole-image tiles (HW <= 16, C = 32):
//   x <- x + conv2(relu(conv1(relu(x))))        (in place)
// With the tile covering the whole image, conv1's output ring IS conv2's
// zero padding, so y1 never touches HBM: after conv1's k-loop (barrier) the
// relu'd result is written back into the SAME transposed LDS tile the
// relu'd input occupied, and conv2 runs from it.  vs two conv3 launches
// this removes the intermediate tensor's HBM write+read, one halo stage,
// one weight-reorder barrier sequence and one kernel launch per pair.
template <int C, int HW>
__global__ __launch_bounds__(HW * HW) void impala_resblock_kernel(
    uint16_t* __restrict__ x,            // [P][C][HW][HW], in & out
    const uint16_t* __restrict__ popw, int64_t row_stride, int64_t w1_off,
    int64_t b1_off, int64_t w2_off, int64_t b2_off, int P) {
  constexpr int TH = HW + 2;
  constexpr int M = HW * HW;        // 256 or 64
  constexpr int CIS = 40;           // C == 32: aligned b128, bank-spread
  constexpr int K = C * 9;          // 288 (= KPAD: no pad taps)
  constexpr int KSTEPS = K / 32;    // 9; one tap per K-step
  constexpr int SBS = K + 8;
  constexpr int NWAVES = M / 64;
  constexpr int NTILES = (M / 16) * (C / 16);
  constexpr int MAXT = NTILES / NWAVES;
  constexpr int MP8 = M + 8;
  constexpr int RING = 4 * TH - 4;  // zero-padding ring positions
  constexpr int ALDS = TH * TH * CIS;
  static_assert(ALDS >= C * MP8, "epilogue staging must fit the tile buf");
  __shared__ uint16_t ax[ALDS];     // relu(x) -> relu(y1) -> epilogue stage
  __shared__ uint16_t sb[C * SBS];  // w1 then w2, tap-major k' order
  const int i = blockIdx.x;
  if (i >= P) return;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const uint16_t* w = popw + (int64_t)i * row_stride;
  uint16_t* xme = x + (int64_t)i * C * M;

  // zero the ring once (both convs consume it as zero padding)
  for (int e = tid; e < RING * C; e += M) {
    int rp = e / C, ci = e % C;
    int yl, xl;
    if (rp < TH) { yl = 0; xl = rp; }
    else if (rp < 2 * TH) { yl = TH - 1; xl = rp - TH; }
    else { int rr = rp - 2 * TH; yl = 1 + (rr >> 1); xl = (rr & 1) ? TH - 1 : 0; }
    ax[(yl * TH + xl) * CIS + ci] = 0;
  }
  // stage relu(x) interior, transposed [y][x][ci]
  for (int e = tid; e < C * M; e += M) {
    int ci = e / M, r = e - ci * M;
    uint16_t v = xme[e];
    if (v & 0x8000) v = 0;
    ax[((r / HW + 1) * TH + (r % HW + 1)) * CIS + ci] = v;
  }
  auto stage_sb = [&](const uint16_t* wsrc) {
    for (int e8 = tid; e8 < C * K / 8; e8 += M) {
      uint4 q = *reinterpret_cast<const uint4*>(wsrc + e8 * 8);
      const uint16_t* h = reinterpret_cast<const uint16_t*>(&q);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int idx = e8 * 8 + j;
        int n = idx / K, k = idx - n * K;
        int ci = k / 9, tap = k - 9 * ci;
        sb[n * SBS + tap * C + ci] = h[j];
      }
    }
  };
  auto run_conv = [&](mfma_cd* acc) {
#pragma unroll 2
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const int ky = ks / 3, kx = ks - 3 * (ks / 3);  // tap == ks for C=32
      const int ci0 = (lane >> 4) * 8;
      mfma_ab bfr[C / 16];
#pragma unroll
      for (int nn = 0; nn < C / 16; ++nn)
        bfr[nn] = *reinterpret_cast<const mfma_ab*>(
            sb + (nn * 16 + (lane & 15)) * SBS + ks * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int u = 0; u < MAXT; ++u) {
        int t = wid + NWAVES * u;
        int mt = t / (C / 16), nt = t % (C / 16);
        int m = mt * 16 + (lane & 15);
        int my = m / HW + ky, mx = m % HW + kx;
        mfma_ab af = *reinterpret_cast<const mfma_ab*>(
            ax + (my * TH + mx) * CIS + ci0);
        acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr[nt], acc[u],
                                                         0, 0, 0);
      }
    }
  };
  stage_sb(w + w1_off);
  __syncthreads();

  mfma_cd acc[MAXT];
#pragma unroll
  for (int u = 0; u < MAXT; ++u) acc[u] = 0.f;
  run_conv(acc);
  __syncthreads();  // conv1 A/B reads complete

  // y1 = relu(acc + b1) back into ax interior; w2 into sb
  const uint16_t* b1 = w + b1_off;
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + NWAVES * u;
    int mt = t / (C / 16), nt = t % (C / 16);
    int n = nt * 16 + (lane & 15);
    int m0 = mt * 16 + ((lane >> 4) << 2);
    float b = des::bf16_to_f32(b1[n]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float v = acc[u][r] + b;
      int m = m0 + r;
      ax[((m / HW + 1) * TH + (m % HW + 1)) * CIS + n] =
          des::f32_to_bf16(v > 0.f ? v : 0.f);
    }
  }
  stage_sb(w + w2_off);
  __syncthreads();

#pragma unroll
  for (int u = 0; u < MAXT; ++u) acc[u] = 0.f;
  run_conv(acc);
  __syncthreads();  // conv2 reads complete; ax becomes the epilogue stage

  const uint16_t* b2 = w + b2_off;
#pragma unroll
  for (int u = 0; u < MAXT; ++u) {
    int t = wid + NWAVES * u;
    int mt = t / (C / 16), nt = t % (C / 16);
    int n = nt * 16 + (lane & 15);
    int m0 = mt * 16 + ((lane >> 4) << 2);
    float b = des::bf16_to_f32(b2[n]);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      ax[n * MP8 + m0 + r] = des::f32_to_bf16(acc[u][r] + b);
  }
  __syncthreads();
  // in-place skip-add store: each thread reads its x chunk then overwrites
  // it (whole-image tile: this block is the only reader of member i's x)
  for (int e8 = tid; e8 < C * M / 8; e8 += M) {
    int n = e8 / (M / 8), m = (e8 - n * (M / 8)) * 8;
    int64_t gidx = (int64_t)n * M + m;
    uint4 v = *reinterpret_cast<const uint4*>(ax + n * MP8 + m);
    uint4 sv = *reinterpret_cast<const uint4*>(xme + gidx);
    const uint16_t* a = reinterpret_cast<const uint16_t*>(&v);
    const uint16_t* b = reinterpret_cast<const uint16_t*>(&sv);
    uint16_t o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = des::f32_to_bf16(des::bf16_to_f32(a[j]) + des::bf16_to_f32(b[j]));
    *reinterpret_cast<uint4*>(xme + gidx) = *reinterpret_cast<const uint4*>(o);
  }
}

template <int C, int HW>  // output is HW/2 x HW/2 (3x3 s2 pad1)
__global__ void impala_pool_kernel(const uint16_t* __restrict__ in,
                                   uint16_t* __restrict__ out, int P) {
  // 8 contiguous outputs per thread via uint4 row loads (the round-1
  // one-scalar-per-output version issued 9 scalar loads/output and ran
  // ~15x over its HBM floor)
  constexpr int HO = HW / 2;
  constexpr int GRP = HO / 8;              // 8-output groups per out row
  int64_t nunits = (int64_t)P * C * HO * GRP;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= nunits) return;
  int g = (int)(idx % GRP);
  int oy = (int)((idx / GRP) % HO);
  int64_t pc = idx / ((int64_t)GRP * HO);  // member*C + c
  const uint16_t* src = in + pc * HW * HW;
  const int xb = 2 * (g * 8);              // 16B-aligned row base
  float best[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) best[j] = -3.0e38f;
  for (int dy = -1; dy <= 1; ++dy) {
    int y = 2 * oy + dy;
    if (y < 0 || y >= HW) continue;
    const uint16_t* row = src + y * HW;
    uint4 q0 = *reinterpret_cast<const uint4*>(row + xb);
    uint4 q1 = *reinterpret_cast<const uint4*>(row + xb + 8);
    float e[17];
    e[0] = xb > 0 ? des::bf16_to_f32(row[xb - 1]) : -3.0e38f;
    const uint16_t* h0 = reinterpret_cast<const uint16_t*>(&q0);
    const uint16_t* h1 = reinterpret_cast<const uint16_t*>(&q1);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      e[1 + j] = des::bf16_to_f32(h0[j]);
      e[9 + j] = des::bf16_to_f32(h1[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j)
      best[j] = fmaxf(best[j],
                      fmaxf(e[2 * j], fmaxf(e[2 * j + 1], e[2 * j + 2])));
  }
  uint16_t o[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) o[j] = des::f32_to_bf16(best[j]);
  *reinterpret_cast<uint4*>(out + pc * HO * HO + oy * HO + g * 8) =
      *reinterpret_cast<const uint4*>(o);
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
      // nt: fc weights are read exactly once per launch (same rationale
      // as the Nature FC / MLP streams)
      typedef __attribute__((ext_vector_type(4))) unsigned int ntv;
      uint4 wa = __builtin_bit_cast(uint4, __builtin_nontemporal_load(
          reinterpret_cast<const ntv*>(w0 + ib)));
      uint4 wb = __builtin_bit_cast(uint4, __builtin_nontemporal_load(
          reinterpret_cast<const ntv*>(w0 + IFC_IN + ib)));
      uint4 wc = __builtin_bit_cast(uint4, __builtin_nontemporal_load(
          reinterpret_cast<const ntv*>(w0 + 2 * IFC_IN + ib)));
      uint4 wd = __builtin_bit_cast(uint4, __builtin_nontemporal_load(
          reinterpret_cast<const ntv*>(w0 + 3 * IFC_IN + ib)));
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
#define CASE(S, CI, CO, HW, TSY, TSX)                                         \
  if (shape == S) {                                                           \
    dim3 grid((uint32_t)P, (HW / TSY) * (HW / TSX));                          \
    if (relu_in && add_skip)                                                  \
      hipLaunchKernelGGL(                                                     \
          (impala_conv3_kernel<CI, CO, HW, TSY, TSX, true, true>), grid,      \
          dim3(TSY * TSX), 0, stream, in, out, skip, popw, row_stride, w_off, \
          b_off, P);                                                          \
    else if (relu_in)                                                         \
      hipLaunchKernelGGL(                                                     \
          (impala_conv3_kernel<CI, CO, HW, TSY, TSX, true, false>), grid,     \
          dim3(TSY * TSX), 0, stream, in, out, skip, popw, row_stride, w_off, \
          b_off, P);                                                          \
    else                                                                      \
      hipLaunchKernelGGL(                                                     \
          (impala_conv3_kernel<CI, CO, HW, TSY, TSX, false, false>), grid,    \
          dim3(TSY * TSX), 0, stream, in, out, skip, popw, row_stride, w_off, \
          b_off, P);                                                          \
    return;                                                                   \
  }
  CASE(0, 3, 16, 64, 32, 32)
  CASE(1, 16, 16, 32, 32, 32)
  CASE(2, 16, 32, 32, 16, 32)
  CASE(3, 32, 32, 16, 16, 16)
  CASE(4, 32, 32, 8, 8, 8)
#undef CASE
}

void des_launch_impala_resblock(uint16_t* x, const uint16_t* popw,
                                int64_t row_stride, int64_t w1_off,
                                int64_t b1_off, int64_t w2_off, int64_t b2_off,
                                int HW, int P, hipStream_t stream) {
  if (HW == 16)
    hipLaunchKernelGGL((impala_resblock_kernel<32, 16>), dim3((uint32_t)P),
                       dim3(256), 0, stream, x, popw, row_stride, w1_off,
                       b1_off, w2_off, b2_off, P);
  else if (HW == 8)
    hipLaunchKernelGGL((impala_resblock_kernel<32, 8>), dim3((uint32_t)P),
                       dim3(64), 0, stream, x, popw, row_stride, w1_off,
                       b1_off, w2_off, b2_off, P);
}

void des_launch_impala_pool(const uint16_t* in, uint16_t* out, int C, int HW,
                            int P, hipStream_t stream) {
  int64_t n = (int64_t)P * C * (HW / 2) * ((HW / 2) / 8);  // 8 outputs/thread
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
