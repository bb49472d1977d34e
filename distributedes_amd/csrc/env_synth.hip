// K7: dimension-faithful synthetic steppers, on-GPU.
//
//   latent-image (Atari57/Procgen dims): z[8] chaotic map -> procedural
//     pattern rendered to [C][HW][HW] bf16 in [0,1]; one wave per env.
//   humanoid-dim: 376-f32 latent recurrence; one wave per env via LDS stage.
//
// Mirrors envs/synthetic.py; __sinf (native) used for the 28k-pixel render,
// libm sinf/cosf for the 8-wide latent update (parity with torch).
#include "common.h"

namespace {

constexpr int ZDIM = 8;

// ---------------- latent-image family ----------------
__global__ void latent_reset_kernel(float* __restrict__ z,
                                    int* __restrict__ steps,
                                    float* __restrict__ fitness,
                                    uint8_t* __restrict__ alive,
                                    float* __restrict__ ep_len,
                                    const int64_t* __restrict__ env_ids,
                                    int64_t seed, int P) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P) return;
  int64_t id = env_ids[i];
  for (int j = 0; j < ZDIM; ++j)
    z[i * ZDIM + j] =
        des::hash_uniform(id * ZDIM + j + seed * 0x51ED, -1.0f, 1.0f);
  steps[i] = 0; fitness[i] = 0.f; alive[i] = 1; ep_len[i] = 0.f;
}

// coef: [P, 6] per-env game coefficients (A, B, C, RF, RW, F0) computed on
// the host by envs/synthetic.py game_coefs (single source of truth — the
// Atari-57 multi-task parameterization; single-game envs get the classics).
template <int C, int HW>
__global__ void latent_render_kernel(const float* __restrict__ z,
                                     const float* __restrict__ coef,
                                     uint16_t* __restrict__ obs, int P) {
  // 8 contiguous pixels per unit -> uint4 stores (the per-pixel scalar
  // store version ran ~5x over the write floor); HW 84/64 are 4-multiples
  // so units never straddle rows beyond the handled tail
  int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= P) return;
  float zl[ZDIM];
#pragma unroll
  for (int j = 0; j < ZDIM; ++j) zl[j] = z[i * ZDIM + j];
  float f0 = coef[i * 6 + 5];
  uint16_t* out = obs + (int64_t)i * C * HW * HW;
  constexpr int NPIX = C * HW * HW;
  for (int p8 = lane; p8 < NPIX / 8; p8 += 64) {
    int pix0 = p8 * 8;
    int c = pix0 / (HW * HW), r = pix0 % (HW * HW);
    int h = r / HW, w0 = r % HW;
    float zc0 = zl[c % ZDIM], zc1 = zl[(c + 1) % ZDIM], zc2 = zl[(c + 2) % ZDIM];
    float base = f0 * zc0 + 0.07f * h * (1.0f + zc1);
    float dx = 0.05f * (1.0f + zc2);
    uint16_t o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int w = w0 + j;
      float ph = base, xw = (float)w;
      if (w >= HW) {  // unit crosses into the next row (HW % 8 != 0)
        xw = (float)(w - HW);
        ph = f0 * zc0 + 0.07f * (h + 1) * (1.0f + zc1);
      }
      o[j] = des::f32_to_bf16(0.5f + 0.5f * __sinf(ph + dx * xw));
    }
    *reinterpret_cast<uint4*>(out + pix0) = *reinterpret_cast<const uint4*>(o);
  }
}

template <int NA>
__global__ void latent_step_kernel(float* __restrict__ z,
                                   const float* __restrict__ coef,
                                   int* __restrict__ steps,
                                   const int* __restrict__ actions,
                                   float* __restrict__ fitness,
                                   uint8_t* __restrict__ alive,
                                   float* __restrict__ ep_len, int horizon,
                                   int P) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P) return;
  float a = (float)actions[i];
  float phase = a * (6.2831853071795864f / NA);
  float cA = coef[i * 6 + 0], cB = coef[i * 6 + 1], cC = coef[i * 6 + 2];
  float cRF = coef[i * 6 + 3], cRW = coef[i * 6 + 4];
  float zo[ZDIM], zn[ZDIM];
#pragma unroll
  for (int j = 0; j < ZDIM; ++j) zo[j] = z[i * ZDIM + j];
#pragma unroll
  for (int j = 0; j < ZDIM; ++j) {
    float zr = zo[(j + ZDIM - 1) % ZDIM];  // torch.roll(z, 1)
    zn[j] = tanhf(cA * zo[j] + cB * zr * cosf(phase) +
                  cC * sinf(phase + zr));
    z[i * ZDIM + j] = zn[j];
  }
  int st = steps[i] + 1;
  float reward = sinf(cRF * zn[0] + cRW * a);
  if (alive[i]) { fitness[i] += reward; ep_len[i] += 1.0f; }
  if (st >= horizon) st = 0;  // horizon handled by rollout; keep counter sane
  steps[i] = st;
}

// ---------------- humanoid-dim family ----------------
constexpr int HDIM = 376, HACT = 17;

__global__ void humanoid_reset_kernel(float* __restrict__ s,
                                      int* __restrict__ steps,
                                      float* __restrict__ fitness,
                                      uint8_t* __restrict__ alive,
                                      float* __restrict__ ep_len,
                                      const int64_t* __restrict__ env_ids,
                                      int64_t seed, int P) {
  int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= P) return;
  float base = des::hash_uniform(env_ids[i] + seed, -0.1f, 0.1f);
  for (int j = lane; j < HDIM; j += 64)
    s[(int64_t)i * HDIM + j] =
        base * cosf(0.1f * j) + 0.05f * sinf(0.37f * j + base);
  if (lane == 0) { steps[i] = 0; fitness[i] = 0.f; alive[i] = 1; ep_len[i] = 0.f; }
}

__global__ void humanoid_step_kernel(float* __restrict__ s,
                                     int* __restrict__ steps,
                                     const float* __restrict__ actions,
                                     float* __restrict__ fitness,
                                     uint8_t* __restrict__ alive,
                                     float* __restrict__ ep_len, int horizon,
                                     int P) {
  __shared__ float sold[HDIM];
  __shared__ float areg[HACT];
  int i = blockIdx.x;  // one 64-thread block per env
  int lane = threadIdx.x;
  if (i >= P) return;
  float* srow = s + (int64_t)i * HDIM;
  const float* arow = actions + (int64_t)i * HACT;
  // float4 staging (HDIM = 94 * 4)
  for (int j4 = lane; j4 < HDIM / 4; j4 += 64)
    *reinterpret_cast<float4*>(sold + 4 * j4) =
        *reinterpret_cast<const float4*>(srow + 4 * j4);
  if (lane < HACT) areg[lane] = arow[lane];
  __syncthreads();

  // target-tracking reward (== envs/synthetic.py HumanoidSyntheticEnv.step):
  // the first 17 dims chase a time-varying reference trajectory, so ES
  // learning curves rise over hundreds of generations (no trivial ceiling)
  float t = (float)(steps[i] + 1);
  float rpart = 0.f, apart = 0.f;
  for (int j4 = lane; j4 < HDIM / 4; j4 += 64) {
    int j = 4 * j4;
    float4 so = *reinterpret_cast<const float4*>(sold + j);
    float sm1 = sold[(j + HDIM - 1) % HDIM];
    float4 sn;
    sn.x = tanhf(1.2f * so.x + 0.6f * sm1 + 0.3f * areg[j % HACT]);
    sn.y = tanhf(1.2f * so.y + 0.6f * so.x + 0.3f * areg[(j + 1) % HACT]);
    sn.z = tanhf(1.2f * so.z + 0.6f * so.y + 0.3f * areg[(j + 2) % HACT]);
    sn.w = tanhf(1.2f * so.w + 0.6f * so.z + 0.3f * areg[(j + 3) % HACT]);
    *reinterpret_cast<float4*>(srow + j) = sn;
    if (j < HACT) {  // j in {0,4,8,12,16}: dims < 17 contribute tracking err
      const float snv[4] = {sn.x, sn.y, sn.z, sn.w};
#pragma unroll
      for (int q = 0; q < 4; ++q)
        if (j + q < HACT) {
          float target = 0.6f * sinf(0.05f * t + 0.37f * (float)(j + q));
          float e = snv[q] - target;
          rpart += e * e;
        }
    }
  }
  if (lane < HACT) apart = areg[lane] * areg[lane];
  rpart = des::wave_reduce_sum_f(rpart);
  apart = des::wave_reduce_sum_f(apart);
  if (lane == 0) {
    float reward = 1.0f - rpart / HACT - 0.05f * (apart / HACT);
    if (alive[i]) { fitness[i] += reward; ep_len[i] += 1.0f; }
    int st = steps[i] + 1;
    if (st >= horizon) st = 0;
    steps[i] = st;
  }
}

}  // namespace

extern "C" {
void des_launch_latent_reset(float* z, int* steps, float* fitness,
                             uint8_t* alive, float* ep_len,
                             const int64_t* env_ids, int64_t seed, int P,
                             hipStream_t stream) {
  int t = 256, g = (P + t - 1) / t;
  hipLaunchKernelGGL(latent_reset_kernel, dim3(g), dim3(t), 0, stream, z,
                     steps, fitness, alive, ep_len, env_ids, seed, P);
}
void des_launch_latent_render(const float* z, const float* coef,
                              uint16_t* obs, int C, int HW, int P,
                              hipStream_t stream) {
  int wpb = 4, g = (P + wpb - 1) / wpb;
  if (C == 4 && HW == 84)
    hipLaunchKernelGGL((latent_render_kernel<4, 84>), dim3(g), dim3(64 * wpb),
                       0, stream, z, coef, obs, P);
  else if (C == 3 && HW == 64)
    hipLaunchKernelGGL((latent_render_kernel<3, 64>), dim3(g), dim3(64 * wpb),
                       0, stream, z, coef, obs, P);
}
void des_launch_latent_step(float* z, const float* coef, int* steps,
                            const int* actions, float* fitness, uint8_t* alive,
                            float* ep_len, int num_actions, int horizon, int P,
                            hipStream_t stream) {
  int t = 256, g = (P + t - 1) / t;
  if (num_actions == 18)
    hipLaunchKernelGGL(latent_step_kernel<18>, dim3(g), dim3(t), 0, stream, z,
                       coef, steps, actions, fitness, alive, ep_len, horizon,
                       P);
  else
    hipLaunchKernelGGL(latent_step_kernel<15>, dim3(g), dim3(t), 0, stream, z,
                       coef, steps, actions, fitness, alive, ep_len, horizon,
                       P);
}
void des_launch_humanoid_reset(float* s, int* steps, float* fitness,
                               uint8_t* alive, float* ep_len,
                               const int64_t* env_ids, int64_t seed, int P,
                               hipStream_t stream) {
  int wpb = 4, g = (P + wpb - 1) / wpb;
  hipLaunchKernelGGL(humanoid_reset_kernel, dim3(g), dim3(64 * wpb), 0, stream,
                     s, steps, fitness, alive, ep_len, env_ids, seed, P);
}
void des_launch_humanoid_step(float* s, int* steps, const float* actions,
                              float* fitness, uint8_t* alive, float* ep_len,
                              int horizon, int P, hipStream_t stream) {
  hipLaunchKernelGGL(humanoid_step_kernel, dim3(P), dim3(64), 0, stream, s,
                     steps, actions, fitness, alive, ep_len, horizon, P);
}
}
