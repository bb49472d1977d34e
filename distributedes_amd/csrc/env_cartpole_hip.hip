#include "hip/hip_runtime.h"
// K7/K8: batched CartPole env step + fitness accumulation, on-GPU.
//
// Exact classic dynamics, auto-reset, single-episode fitness masking —
// mirrors envs/cartpole.py (the torch reference twin) including the
// deterministic per-env hash init (common.h hash_uniform).
// One thread per env; trivially memory-light.
#include "common.h"

namespace {

constexpr float GRAV = 9.8f, MC = 1.0f, MP = 0.1f, LEN = 0.5f, FORCE = 10.0f,
                TAU = 0.02f, X_LIM = 2.4f;
constexpr float TH_LIM = 12.0f * 3.14159265358979323846f / 180.0f;
constexpr int MAX_STEPS = 500;
constexpr float MT = MC + MP, PML = MP * LEN;

// counter formula matches envs/cartpole.py: (id*1000 + reset_ctr)*4 + j
// + seed*0x10001 (seed_mix precomputed host-side)
__device__ __forceinline__ void init_state(float* s, int64_t counter,
                                           int64_t seed_mix) {
#pragma unroll
  for (int j = 0; j < 4; ++j)
    s[j] = des::hash_uniform(counter * 4 + j + seed_mix, -0.05f, 0.05f);
}

__global__ void cartpole_reset_kernel(float* __restrict__ state,
                                      int* __restrict__ steps,
                                      int64_t* __restrict__ reset_ctr,
                                      float* __restrict__ fitness,
                                      uint8_t* __restrict__ alive,
                                      float* __restrict__ ep_len,
                                      const int64_t* __restrict__ env_ids,
                                      int64_t seed_term, int P) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P) return;
  reset_ctr[i] = 0;
  steps[i] = 0;
  fitness[i] = 0.f;
  alive[i] = 1;
  ep_len[i] = 0.f;
  float s[4];
  init_state(s, env_ids[i] * 1000, seed_term);
#pragma unroll
  for (int j = 0; j < 4; ++j) state[i * 4 + j] = s[j];
}

__global__ void cartpole_step_kernel(float* __restrict__ state,
                                     int* __restrict__ steps,
                                     int64_t* __restrict__ reset_ctr,
                                     const int* __restrict__ actions,
                                     float* __restrict__ fitness,
                                     uint8_t* __restrict__ alive,
                                     float* __restrict__ ep_len,
                                     const int64_t* __restrict__ env_ids,
                                     const int64_t* __restrict__ seedp,
                                     int P) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= P) return;
  const int64_t seed_term = *seedp;  // device-read: hipGraph-replayable
  float x = state[i * 4], xd = state[i * 4 + 1], th = state[i * 4 + 2],
        td = state[i * 4 + 3];
  float force = actions[i] ? FORCE : -FORCE;
  float costh = cosf(th), sinth = sinf(th);  // libm-accurate: CPU-parity tests
  float temp = (force + PML * td * td * sinth) / MT;
  float th_acc = (GRAV * sinth - costh * temp) /
                 (LEN * (4.0f / 3.0f - MP * costh * costh / MT));
  float x_acc = temp - PML * th_acc * costh / MT;
  x += TAU * xd; xd += TAU * x_acc;
  th += TAU * td; td += TAU * th_acc;
  int st = steps[i] + 1;

  // K8: reward 1/step while alive (pre-step alive mask, == torch ref)
  if (alive[i]) { fitness[i] += 1.0f; ep_len[i] += 1.0f; }

  bool fell = fabsf(x) > X_LIM || fabsf(th) > TH_LIM;
  bool done = fell || st >= MAX_STEPS;
  if (done) {
    if (alive[i]) alive[i] = 0;
    int64_t rc = reset_ctr[i] + 1;
    reset_ctr[i] = rc;
    float s[4];
    init_state(s, env_ids[i] * 1000 + rc, seed_term);
    x = s[0]; xd = s[1]; th = s[2]; td = s[3];
    st = 0;
  }
  state[i * 4] = x; state[i * 4 + 1] = xd;
  state[i * 4 + 2] = th; state[i * 4 + 3] = td;
  steps[i] = st;
}

}  // namespace

extern "C" {
void des_launch_cartpole_reset(float* state, int* steps, int64_t* reset_ctr,
                               float* fitness, uint8_t* alive, float* ep_len,
                               const int64_t* env_ids, int64_t seed_term,
                               int P, hipStream_t stream) {
  int t = 256, g = (P + t - 1) / t;
  hipLaunchKernelGGL(cartpole_reset_kernel, dim3(g), dim3(t), 0, stream, state,
                     steps, reset_ctr, fitness, alive, ep_len, env_ids,
                     seed_term, P);
}
void des_launch_cartpole_step(float* state, int* steps, int64_t* reset_ctr,
                              const int* actions, float* fitness,
                              uint8_t* alive, float* ep_len,
                              const int64_t* env_ids, const int64_t* seed_term,
                              int P, hipStream_t stream) {
  int t = 256, g = (P + t - 1) / t;
  hipLaunchKernelGGL(cartpole_step_kernel, dim3(g), dim3(t), 0, stream, state,
                     steps, reset_ctr, actions, fitness, alive, ep_len,
                     env_ids, seed_term, P);
}
}
