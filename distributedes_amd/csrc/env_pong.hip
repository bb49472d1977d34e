// K7: PongSynthetic batched env — physics + 84x84 rasterizer + K8 fitness,
// mirroring envs/pong.py step-for-step (same hash serves, same raster rules).
//
// One wave per env: physics computed redundantly by all lanes (deterministic,
// no divergence), frame rasterized cooperatively (consecutive lanes ->
// consecutive bytes, coalesced u8 stores), state written by lane 0.
// Frames live in a 4-slot ring [P][4][84][84] u8 {0,1}; the consumer
// (nature_cnn kernel) maps logical channel -> physical slot, so no frame
// copies ever happen (the torch ref's torch.cat is replaced by the ring).
#include "common.h"

namespace {

constexpr int HW = 84;
constexpr float PAD_H = 8.0f, PLAYER_X = 80.0f, OPP_X = 2.0f;
constexpr int WIN_SCORE = 5;

struct PongState {
  float* bx; float* by; float* vx; float* vy;
  float* py; float* oy;
  int* score; int* points;
  int64_t* serve_ctr;
};

__device__ __forceinline__ void serve(float& bx, float& by, float& vx,
                                      float& vy, int64_t env_id,
                                      int64_t serve_ctr, int64_t seed) {
  int64_t c = env_id * 977 + serve_ctr * 131 + seed;
  bx = 42.0f;
  by = des::hash_uniform(c, 20.0f, 64.0f);
  float mag = des::hash_uniform(c + 1, 1.2f, 2.0f);
  float sgn = des::hash_uniform(c + 2, 0.0f, 1.0f) < 0.5f ? -1.0f : 1.0f;
  vx = mag * sgn;
  vy = des::hash_uniform(c + 3, -1.5f, 1.5f);
}

__device__ __forceinline__ void raster(uint8_t* frame, float bx, float by,
                                       float py, float oy, int lane) {
  int bxi = (int)fminf(fmaxf(rintf(bx), 0.f), (float)(HW - 2));
  int byi = (int)fminf(fmaxf(rintf(by), 0.f), (float)(HW - 2));
  for (int pix = lane; pix < HW * HW; pix += 64) {
    int y = pix / HW, x = pix % HW;
    uint8_t v = 0;
    if ((x == 80 || x == 81) && fabsf((float)y - py) <= PAD_H / 2) v = 1;
    if ((x == 2 || x == 3) && fabsf((float)y - oy) <= PAD_H / 2) v = 1;
    if ((y == byi || y == byi + 1) && (x == bxi || x == bxi + 1)) v = 1;
    frame[pix] = v;
  }
}

__global__ void pong_reset_kernel(PongState st, uint8_t* __restrict__ frames,
                                  float* __restrict__ fitness,
                                  uint8_t* __restrict__ alive,
                                  float* __restrict__ ep_len,
                                  const int64_t* __restrict__ env_ids,
                                  int64_t seed, int P) {
  int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= P) return;
  int64_t id = env_ids[i];
  float bx, by, vx, vy;
  serve(bx, by, vx, vy, id, 0, seed);
  if (lane == 0) {
    st.bx[i] = bx; st.by[i] = by; st.vx[i] = vx; st.vy[i] = vy;
    st.py[i] = 42.0f; st.oy[i] = 42.0f;
    st.score[i] = 0; st.points[i] = 0; st.serve_ctr[i] = 0;
    fitness[i] = 0.f; alive[i] = 1; ep_len[i] = 0.f;
  }
  uint8_t* base = frames + (int64_t)i * 4 * HW * HW;
  for (int s = 0; s < 4; ++s)  // all 4 ring slots = initial frame
    raster(base + s * HW * HW, bx, by, 42.0f, 42.0f, lane);
}

__global__ void pong_step_kernel(PongState st, uint8_t* __restrict__ frames,
                                 int slot,  // ring slot receiving this frame
                                 const int* __restrict__ actions,
                                 float* __restrict__ fitness,
                                 uint8_t* __restrict__ alive,
                                 float* __restrict__ ep_len,
                                 const int64_t* __restrict__ env_ids,
                                 const int64_t* __restrict__ seedp, int P) {
  int i = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= P) return;
  const int64_t seed = *seedp;  // device-read: hipGraph-replayable
  int64_t id = env_ids[i];
  float bx = st.bx[i], by = st.by[i], vx = st.vx[i], vy = st.vy[i];
  float py = st.py[i], oy = st.oy[i];
  int score = st.score[i], points = st.points[i];
  int64_t sc = st.serve_ctr[i];

  // paddle moves
  int a = actions[i] % 3;
  float dy = a == 1 ? -2.0f : (a == 2 ? 2.0f : 0.0f);
  py = fminf(fmaxf(py + dy, PAD_H / 2), HW - PAD_H / 2);
  float delta = fminf(fmaxf(by - oy, -1.5f), 1.5f);
  oy = fminf(fmaxf(oy + delta, PAD_H / 2), HW - PAD_H / 2);

  // ball
  bx += vx; by += vy;
  bool low = by < 1.0f, high = by > HW - 1.0f;
  if (low) by = 2.0f - by;
  if (high) by = 2.0f * (HW - 1.0f) - by;
  if (low || high) vy = -vy;

  float reward = 0.f;
  bool at_p = bx >= PLAYER_X;
  bool hit_p = at_p && fabsf(by - py) <= PAD_H / 2;
  bool miss_p = at_p && !hit_p;
  if (hit_p) { vx = -fabsf(vx); vy += 0.25f * (by - py); bx = 2.f * PLAYER_X - bx; }
  bool at_o = bx <= OPP_X;
  bool hit_o = at_o && fabsf(by - oy) <= PAD_H / 2;
  bool miss_o = at_o && !hit_o;
  if (hit_o) { vx = fabsf(vx); vy += 0.25f * (by - oy); bx = 2.f * OPP_X - bx; }
  vy = fminf(fmaxf(vy, -2.5f), 2.5f);

  if (miss_o) reward += 1.0f;
  if (miss_p) reward -= 1.0f;
  bool scored = miss_p || miss_o;
  score += (miss_o ? 1 : 0) - (miss_p ? 1 : 0);
  points += scored ? 1 : 0;
  if (scored) { sc += 1; serve(bx, by, vx, vy, id, sc, seed); }

  bool done = (score >= WIN_SCORE || score <= -WIN_SCORE) ||
              (points >= 2 * WIN_SCORE + 5);
  if (done) {
    sc += 7;
    py = 42.0f; oy = 42.0f; score = 0; points = 0;
    serve(bx, by, vx, vy, id, sc, seed);
  }

  if (lane == 0) {
    if (alive[i]) { fitness[i] += reward; ep_len[i] += 1.0f; }
    if (done && alive[i]) alive[i] = 0;
    st.bx[i] = bx; st.by[i] = by; st.vx[i] = vx; st.vy[i] = vy;
    st.py[i] = py; st.oy[i] = oy;
    st.score[i] = score; st.points[i] = points; st.serve_ctr[i] = sc;
  }

  uint8_t* base = frames + (int64_t)i * 4 * HW * HW;
  raster(base + slot * HW * HW, bx, by, py, oy, lane);
  if (done) {  // torch ref: terminal reset floods the whole stack
    for (int s = 0; s < 4; ++s)
      if (s != slot) raster(base + s * HW * HW, bx, by, py, oy, lane);
  }
}

}  // namespace

extern "C" {
void des_launch_pong_reset(float* bx, float* by, float* vx, float* vy,
                           float* py, float* oy, int* score, int* points,
                           int64_t* serve_ctr, uint8_t* frames, float* fitness,
                           uint8_t* alive, float* ep_len,
                           const int64_t* env_ids, int64_t seed, int P,
                           hipStream_t stream) {
  PongState st{bx, by, vx, vy, py, oy, score, points, serve_ctr};
  int waves_per_block = 4;
  int g = (P + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(pong_reset_kernel, dim3(g), dim3(64 * waves_per_block), 0,
                     stream, st, frames, fitness, alive, ep_len, env_ids, seed,
                     P);
}
void des_launch_pong_step(float* bx, float* by, float* vx, float* vy,
                          float* py, float* oy, int* score, int* points,
                          int64_t* serve_ctr, uint8_t* frames, int slot,
                          const int* actions, float* fitness, uint8_t* alive,
                          float* ep_len, const int64_t* env_ids,
                          const int64_t* seed, int P, hipStream_t stream) {
  PongState st{bx, by, vx, vy, py, oy, score, points, serve_ctr};
  int waves_per_block = 4;
  int g = (P + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(pong_step_kernel, dim3(g), dim3(64 * waves_per_block), 0,
                     stream, st, frames, slot, actions, fitness, alive, ep_len,
                     env_ids, seed, P);
}
}
