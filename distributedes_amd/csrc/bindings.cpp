// pybind11 bindings for the distributedes_amd gfx950 HIP kernels.
//
// This is the only TU that includes torch headers (keeps hipcc compile time
// down); kernels live in *.hip TUs exposed through extern "C" launchers.
#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include <cstdint>
#include <vector>

extern "C" {
void des_launch_noise_fill(float*, int64_t, uint64_t, hipStream_t);
void des_launch_noise_fill_bf16(uint16_t*, int64_t, uint64_t, hipStream_t);
void des_launch_perturb(void*, bool, const float*, const void*, bool,
                        const int64_t*, const float*, float, const float*,
                        int64_t, int64_t, int64_t, hipStream_t);
void des_launch_wsum(float*, float*, const void*, bool, const int64_t*,
                     const float*, const float*, int64_t, int64_t, bool,
                     hipStream_t);
void des_launch_mlp_forward(const void*, bool, int64_t, const float*, int,
                            const float*, const float*, float, int*, float*,
                            float*, int, int, int, const int64_t*, int,
                            hipStream_t);
void des_launch_nature_cnn_forward(const uint16_t*, int64_t, const void*, bool,
                                   const int*, int*, float*, int, int,
                                   const int64_t*, hipStream_t);
void des_launch_cartpole_reset(float*, int*, int64_t*, float*, uint8_t*,
                               float*, const int64_t*, int64_t, int,
                               hipStream_t);
void des_launch_cartpole_step(float*, int*, int64_t*, const int*, float*,
                              uint8_t*, float*, const int64_t*,
                              const int64_t*, int, hipStream_t);
void des_launch_pong_reset(float*, float*, float*, float*, float*, float*,
                           int*, int*, int64_t*, uint8_t*, float*, uint8_t*,
                           float*, const int64_t*, int64_t, int, hipStream_t);
void des_launch_pong_step(float*, float*, float*, float*, float*, float*,
                          int*, int*, int64_t*, uint8_t*, int, const int*,
                          float*, uint8_t*, float*, const int64_t*,
                          const int64_t*, int, hipStream_t);
void des_launch_latent_reset(float*, int*, float*, uint8_t*, float*,
                             const int64_t*, int64_t, int, hipStream_t);
void des_launch_latent_render(const float*, const float*, uint16_t*, int, int,
                              int, hipStream_t);
void des_launch_latent_step(float*, const float*, int*, const int*, float*,
                            uint8_t*, float*, int, int, int, hipStream_t);
void des_launch_humanoid_reset(float*, int*, float*, uint8_t*, float*,
                               const int64_t*, int64_t, int, hipStream_t);
void des_launch_humanoid_step(float*, int*, const float*, float*, uint8_t*,
                              float*, int, int, hipStream_t);
void des_launch_normalizer_observe(const float*, float*, int, int,
                                   hipStream_t);
void des_launch_normalizer_observe_flush(const float*, double*, double*, int,
                                         int, hipStream_t);
void des_launch_centered_rank(const float*, float*, int, hipStream_t);
void des_launch_impala_conv3(const uint16_t*, uint16_t*, const uint16_t*,
                             const uint16_t*, int64_t, int64_t, int64_t, int,
                             bool, bool, int, hipStream_t);
void des_launch_impala_resblock(uint16_t*, const uint16_t*, int64_t, int64_t,
                                int64_t, int64_t, int64_t, int, int,
                                hipStream_t);
void des_launch_impala_pool(const uint16_t*, uint16_t*, int, int, int,
                            hipStream_t);
void des_launch_impala_fc_head(const uint16_t*, int64_t, const uint16_t*,
                               int*, float*, int, int, int64_t, int64_t,
                               int64_t, int64_t, hipStream_t);
}

namespace {

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

float* fptr(torch::Tensor& t) { return t.data_ptr<float>(); }
const float* cfptr(const torch::Tensor& t) { return t.data_ptr<float>(); }
int* iptr(torch::Tensor& t) { return t.data_ptr<int>(); }
int64_t* lptr(torch::Tensor& t) { return t.data_ptr<int64_t>(); }
uint8_t* bptr(torch::Tensor& t) { return t.data_ptr<uint8_t>(); }

void noise_fill(torch::Tensor table, int64_t seed) {
  CHECK_DEV(table); CHECK_CONT(table);
  if (table.dtype() == torch::kFloat32)
    des_launch_noise_fill(table.data_ptr<float>(), table.numel(),
                          (uint64_t)seed, cur_stream());
  else if (table.dtype() == torch::kBFloat16)
    des_launch_noise_fill_bf16(
        reinterpret_cast<uint16_t*>(table.data_ptr()), table.numel(),
        (uint64_t)seed, cur_stream());
  else
    TORCH_CHECK(false, "table must be fp32 or bf16");
}

void perturb(torch::Tensor out, torch::Tensor theta, torch::Tensor table,
             torch::Tensor offsets, torch::Tensor signs, double sigma,
             torch::Tensor sigma_vec) {
  CHECK_DEV(out); CHECK_CONT(out); CHECK_DEV(theta); CHECK_CONT(theta);
  CHECK_DEV(table); CHECK_CONT(table);
  CHECK_DEV(offsets); CHECK_CONT(offsets); CHECK_DEV(signs); CHECK_CONT(signs);
  int64_t P = out.size(0), stride = out.size(1), D = theta.numel();
  TORCH_CHECK(stride >= D && offsets.numel() == P && signs.numel() == P);
  TORCH_CHECK(offsets.dtype() == torch::kInt64 && signs.dtype() == torch::kFloat32);
  const float* sv = nullptr;
  if (sigma_vec.numel() > 0) {
    CHECK_DEV(sigma_vec); CHECK_CONT(sigma_vec);
    TORCH_CHECK(sigma_vec.numel() == D, "sigma_vec must be [D]");
    sv = sigma_vec.data_ptr<float>();
  }
  bool out_bf16 = out.dtype() == torch::kBFloat16;
  bool tab_bf16 = table.dtype() == torch::kBFloat16;
  TORCH_CHECK(out_bf16 || out.dtype() == torch::kFloat32,
              "perturb: out must be bf16 or fp32");
  TORCH_CHECK(tab_bf16 || table.dtype() == torch::kFloat32,
              "perturb: table must be bf16 or fp32");
  des_launch_perturb(out.data_ptr(), out_bf16, theta.data_ptr<float>(),
                     table.data_ptr(), tab_bf16, offsets.data_ptr<int64_t>(),
                     signs.data_ptr<float>(), (float)sigma, sv, P, D, stride,
                     cur_stream());
}

void weighted_noise_sum(torch::Tensor acc, torch::Tensor acc_sq,
                        torch::Tensor table, torch::Tensor offsets,
                        torch::Tensor signs, torch::Tensor weights,
                        bool want_sq) {
  CHECK_DEV(acc); CHECK_CONT(acc); CHECK_DEV(table); CHECK_CONT(table);
  CHECK_DEV(offsets); CHECK_DEV(signs); CHECK_DEV(weights);
  int64_t D = acc.numel(), P = offsets.numel();
  if (want_sq) TORCH_CHECK(acc_sq.numel() == D);
  des_launch_wsum(acc.data_ptr<float>(),
                  want_sq ? acc_sq.data_ptr<float>() : nullptr,
                  table.data_ptr(), table.dtype() == torch::kBFloat16,
                  offsets.data_ptr<int64_t>(), signs.data_ptr<float>(),
                  weights.data_ptr<float>(), P, D, want_sq, cur_stream());
}

// ---------------- policy forwards ----------------
void mlp_forward(torch::Tensor popw, torch::Tensor obs, torch::Tensor nmean,
                 torch::Tensor ninv, double clip, torch::Tensor actions,
                 torch::Tensor logits, int64_t h1, int64_t h2, int64_t A,
                 std::vector<int64_t> offs) {
  CHECK_DEV(popw); CHECK_CONT(popw); CHECK_DEV(obs); CHECK_CONT(obs);
  CHECK_DEV(actions); CHECK_CONT(actions);
  TORCH_CHECK(offs.size() == 6);
  int64_t P = popw.size(0);
  int in_dim = obs.size(1);
  TORCH_CHECK(obs.size(0) == P && obs.dtype() == torch::kFloat32);
  const float* nm = nullptr;
  const float* ni = nullptr;
  if (nmean.numel() > 0) {
    TORCH_CHECK(nmean.numel() == in_dim && ninv.numel() == in_dim);
    nm = cfptr(nmean); ni = cfptr(ninv);
  }
  bool bf16 = popw.dtype() == torch::kBFloat16;
  TORCH_CHECK(bf16 || popw.dtype() == torch::kFloat32);
  int* ai = nullptr;
  float* af = nullptr;
  if (actions.dtype() == torch::kInt32) {
    TORCH_CHECK(actions.numel() == P);
    ai = iptr(actions);
  } else {
    TORCH_CHECK(actions.dtype() == torch::kFloat32 && actions.numel() == P * A);
    af = fptr(actions);
  }
  float* lo = logits.numel() > 0 ? fptr(logits) : nullptr;
  des_launch_mlp_forward(popw.data_ptr(), bf16, popw.size(1), cfptr(obs),
                         in_dim, nm, ni, (float)clip, ai, af, lo, (int)h1,
                         (int)h2, (int)A, offs.data(), (int)P, cur_stream());
}

void nature_cnn_forward(torch::Tensor popw, torch::Tensor frames,
                        torch::Tensor slot_map, torch::Tensor actions,
                        torch::Tensor logits, int64_t A, torch::Tensor offs) {
  CHECK_DEV(popw); CHECK_CONT(popw); CHECK_DEV(frames); CHECK_CONT(frames);
  CHECK_DEV(slot_map); CHECK_DEV(actions); CHECK_DEV(offs);
  TORCH_CHECK(popw.dtype() == torch::kBFloat16, "pop weights must be bf16");
  TORCH_CHECK(offs.dtype() == torch::kInt64 && offs.numel() == 10);
  TORCH_CHECK(slot_map.dtype() == torch::kInt32 && slot_map.numel() == 4);
  TORCH_CHECK(actions.dtype() == torch::kInt32);
  int64_t P = popw.size(0);
  bool u8 = frames.dtype() == torch::kUInt8;
  TORCH_CHECK(u8 || frames.dtype() == torch::kBFloat16);
  float* lo = logits.numel() > 0 ? fptr(logits) : nullptr;
  des_launch_nature_cnn_forward(
      reinterpret_cast<const uint16_t*>(popw.data_ptr()), popw.size(1),
      frames.data_ptr(), u8, slot_map.data_ptr<int>(), iptr(actions), lo,
      (int)A, (int)P, offs.data_ptr<int64_t>(), cur_stream());
}

// ---------------- envs ----------------
void cartpole_reset(torch::Tensor state, torch::Tensor steps,
                    torch::Tensor reset_ctr, torch::Tensor fitness,
                    torch::Tensor alive, torch::Tensor ep_len,
                    torch::Tensor env_ids, int64_t seed_mix) {
  int P = state.size(0);
  des_launch_cartpole_reset(fptr(state), iptr(steps), lptr(reset_ctr),
                            fptr(fitness), bptr(alive), fptr(ep_len),
                            lptr(env_ids), seed_mix, P, cur_stream());
}

void cartpole_step(torch::Tensor state, torch::Tensor steps,
                   torch::Tensor reset_ctr, torch::Tensor actions,
                   torch::Tensor fitness, torch::Tensor alive,
                   torch::Tensor ep_len, torch::Tensor env_ids,
                   torch::Tensor seed_buf) {
  int P = state.size(0);
  TORCH_CHECK(seed_buf.is_cuda() && seed_buf.dtype() == torch::kInt64);
  des_launch_cartpole_step(fptr(state), iptr(steps), lptr(reset_ctr),
                           iptr(actions), fptr(fitness), bptr(alive),
                           fptr(ep_len), lptr(env_ids), lptr(seed_buf), P,
                           cur_stream());
}

void pong_reset(torch::Tensor bx, torch::Tensor by, torch::Tensor vx,
                torch::Tensor vy, torch::Tensor py, torch::Tensor oy,
                torch::Tensor score, torch::Tensor points,
                torch::Tensor serve_ctr, torch::Tensor frames,
                torch::Tensor fitness, torch::Tensor alive,
                torch::Tensor ep_len, torch::Tensor env_ids, int64_t seed) {
  int P = bx.size(0);
  des_launch_pong_reset(fptr(bx), fptr(by), fptr(vx), fptr(vy), fptr(py),
                        fptr(oy), iptr(score), iptr(points), lptr(serve_ctr),
                        bptr(frames), fptr(fitness), bptr(alive), fptr(ep_len),
                        lptr(env_ids), seed, P, cur_stream());
}

void pong_step(torch::Tensor bx, torch::Tensor by, torch::Tensor vx,
               torch::Tensor vy, torch::Tensor py, torch::Tensor oy,
               torch::Tensor score, torch::Tensor points,
               torch::Tensor serve_ctr, torch::Tensor frames, int64_t slot,
               torch::Tensor actions, torch::Tensor fitness,
               torch::Tensor alive, torch::Tensor ep_len,
               torch::Tensor env_ids, torch::Tensor seed_buf) {
  int P = bx.size(0);
  TORCH_CHECK(seed_buf.is_cuda() && seed_buf.dtype() == torch::kInt64);
  des_launch_pong_step(fptr(bx), fptr(by), fptr(vx), fptr(vy), fptr(py),
                       fptr(oy), iptr(score), iptr(points), lptr(serve_ctr),
                       bptr(frames), (int)slot, iptr(actions), fptr(fitness),
                       bptr(alive), fptr(ep_len), lptr(env_ids),
                       lptr(seed_buf), P, cur_stream());
}

void latent_reset(torch::Tensor z, torch::Tensor steps, torch::Tensor fitness,
                  torch::Tensor alive, torch::Tensor ep_len,
                  torch::Tensor env_ids, int64_t seed) {
  des_launch_latent_reset(fptr(z), iptr(steps), fptr(fitness), bptr(alive),
                          fptr(ep_len), lptr(env_ids), seed, z.size(0),
                          cur_stream());
}

void latent_render(torch::Tensor z, torch::Tensor coef, torch::Tensor obs) {
  TORCH_CHECK(obs.dtype() == torch::kBFloat16);
  TORCH_CHECK(coef.size(0) == z.size(0) && coef.size(1) == 6);
  int C = obs.size(1), HW = obs.size(2);
  des_launch_latent_render(cfptr(z), cfptr(coef),
                           reinterpret_cast<uint16_t*>(obs.data_ptr()), C, HW,
                           z.size(0), cur_stream());
}

void latent_step(torch::Tensor z, torch::Tensor coef, torch::Tensor steps,
                 torch::Tensor actions, torch::Tensor fitness,
                 torch::Tensor alive, torch::Tensor ep_len,
                 int64_t num_actions, int64_t horizon) {
  TORCH_CHECK(coef.size(0) == z.size(0) && coef.size(1) == 6);
  des_launch_latent_step(fptr(z), cfptr(coef), iptr(steps), iptr(actions),
                         fptr(fitness), bptr(alive), fptr(ep_len),
                         (int)num_actions, (int)horizon, z.size(0),
                         cur_stream());
}

void humanoid_reset(torch::Tensor s, torch::Tensor steps,
                    torch::Tensor fitness, torch::Tensor alive,
                    torch::Tensor ep_len, torch::Tensor env_ids,
                    int64_t seed) {
  des_launch_humanoid_reset(fptr(s), iptr(steps), fptr(fitness), bptr(alive),
                            fptr(ep_len), lptr(env_ids), seed, s.size(0),
                            cur_stream());
}

void humanoid_step(torch::Tensor s, torch::Tensor steps, torch::Tensor actions,
                   torch::Tensor fitness, torch::Tensor alive,
                   torch::Tensor ep_len, int64_t horizon) {
  des_launch_humanoid_step(fptr(s), iptr(steps), cfptr(actions), fptr(fitness),
                           bptr(alive), fptr(ep_len), (int)horizon, s.size(0),
                           cur_stream());
}

const uint16_t* u16p(const torch::Tensor& t) {
  return reinterpret_cast<const uint16_t*>(t.data_ptr());
}

void impala_conv(torch::Tensor in, torch::Tensor out, torch::Tensor skip,
                 torch::Tensor popw, int64_t w_off, int64_t b_off,
                 int64_t shape, bool relu_in, bool add_skip) {
  CHECK_DEV(in); CHECK_CONT(in); CHECK_DEV(out); CHECK_CONT(out);
  CHECK_DEV(popw); CHECK_CONT(popw);
  TORCH_CHECK(in.dtype() == torch::kBFloat16 && out.dtype() == torch::kBFloat16);
  // the kernel's vectorized weight reorder does uint4 (8-elem) loads from
  // popw+row*stride+w_off: both strides must be 8-elem aligned
  TORCH_CHECK(w_off % 8 == 0 && popw.size(1) % 8 == 0,
              "impala_conv: weight offset/stride must be 8-elem aligned");
  int P = popw.size(0);
  const uint16_t* sk = add_skip ? u16p(skip) : nullptr;
  des_launch_impala_conv3(u16p(in), (uint16_t*)out.data_ptr(), sk, u16p(popw),
                          popw.size(1), w_off, b_off, (int)shape, relu_in,
                          add_skip, P, cur_stream());
}

void impala_resblock(torch::Tensor x, torch::Tensor popw, int64_t w1_off,
                     int64_t b1_off, int64_t w2_off, int64_t b2_off) {
  CHECK_DEV(x); CHECK_CONT(x); CHECK_DEV(popw); CHECK_CONT(popw);
  TORCH_CHECK(x.dtype() == torch::kBFloat16);
  TORCH_CHECK(x.size(1) == 32, "resblock kernel is C=32 only");
  TORCH_CHECK(w1_off % 8 == 0 && w2_off % 8 == 0 && popw.size(1) % 8 == 0,
              "impala_resblock: weight offsets/stride must be 8-elem aligned");
  int HW = x.size(2);
  TORCH_CHECK(HW == 16 || HW == 8, "resblock kernel needs whole-image tiles");
  des_launch_impala_resblock((uint16_t*)x.data_ptr(), u16p(popw),
                             popw.size(1), w1_off, b1_off, w2_off, b2_off, HW,
                             popw.size(0), cur_stream());
}

void impala_pool(torch::Tensor in, torch::Tensor out, int64_t C, int64_t HW) {
  CHECK_DEV(in); CHECK_CONT(in); CHECK_DEV(out); CHECK_CONT(out);
  int P = in.size(0);
  des_launch_impala_pool(u16p(in), (uint16_t*)out.data_ptr(), (int)C, (int)HW,
                         P, cur_stream());
}

void impala_fc_head(torch::Tensor popw, torch::Tensor act,
                    torch::Tensor actions, torch::Tensor logits, int64_t A,
                    int64_t wf, int64_t bf, int64_t wh, int64_t bh) {
  CHECK_DEV(popw); CHECK_CONT(popw); CHECK_DEV(act); CHECK_CONT(act);
  CHECK_DEV(actions);
  int P = popw.size(0);
  float* lo = logits.numel() > 0 ? fptr(logits) : nullptr;
  des_launch_impala_fc_head(u16p(popw), popw.size(1), u16p(act),
                            iptr(actions), lo, (int)A, P, wf, bf, wh, bh,
                            cur_stream());
}

void centered_rank(torch::Tensor fitness, torch::Tensor u) {
  CHECK_DEV(fitness); CHECK_CONT(fitness); CHECK_DEV(u); CHECK_CONT(u);
  TORCH_CHECK(fitness.dtype() == torch::kFloat32 && u.numel() == fitness.numel());
  TORCH_CHECK(fitness.numel() >= 2);
  des_launch_centered_rank(cfptr(fitness), fptr(u), (int)fitness.numel(),
                           cur_stream());
}

void normalizer_observe(torch::Tensor obs, torch::Tensor scratch) {
  CHECK_DEV(obs); CHECK_CONT(obs); CHECK_DEV(scratch); CHECK_CONT(scratch);
  TORCH_CHECK(scratch.dtype() == torch::kFloat32 &&
              scratch.numel() >= 256 * 2 * obs.size(1),
              "observe scratch must be [256][2][dim] fp32");
  des_launch_normalizer_observe(cfptr(obs), fptr(scratch), obs.size(0),
                                obs.size(1), cur_stream());
}

void normalizer_observe_flush(torch::Tensor scratch, int64_t P,
                              torch::Tensor sum, torch::Tensor sumsq) {
  CHECK_DEV(scratch); CHECK_CONT(scratch);
  TORCH_CHECK(sum.dtype() == torch::kFloat64 && sumsq.dtype() == torch::kFloat64);
  int dim = sum.numel();
  des_launch_normalizer_observe_flush(cfptr(scratch), sum.data_ptr<double>(),
                                      sumsq.data_ptr<double>(), (int)P, dim,
                                      cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("noise_fill", &noise_fill, "K1: Philox N(0,1) table fill");
  m.def("perturb", &perturb, "K2: theta + sigma*sign*noise-slice");
  m.def("weighted_noise_sum", &weighted_noise_sum,
        "K10: sum_i w_i s_i eps_i (+ optional sum w_i eps_i^2)");
  m.def("mlp_forward", &mlp_forward, "K3: per-member MLP forward + action");
  m.def("nature_cnn_forward", &nature_cnn_forward,
        "K5: per-member Nature-CNN forward + argmax");
  m.def("cartpole_reset", &cartpole_reset);
  m.def("cartpole_step", &cartpole_step);
  m.def("pong_reset", &pong_reset);
  m.def("pong_step", &pong_step);
  m.def("latent_reset", &latent_reset);
  m.def("latent_render", &latent_render);
  m.def("latent_step", &latent_step);
  m.def("humanoid_reset", &humanoid_reset);
  m.def("humanoid_step", &humanoid_step);
  m.def("normalizer_observe", &normalizer_observe);
  m.def("normalizer_observe_flush", &normalizer_observe_flush);
  m.def("centered_rank", &centered_rank, "K9: stable centered-rank weights");
  m.def("impala_conv", &impala_conv, "K6: 3x3 pad-1 conv (pop, tiled)");
  m.def("impala_resblock", &impala_resblock,
        "K6: fused residual block x += conv2(relu(conv1(relu(x)))), HW<=16");
  m.def("impala_pool", &impala_pool, "K6: 3x3 s2 pad-1 max pool");
  m.def("impala_fc_head", &impala_fc_head, "K6: fc 2048->256 + head + argmax");
}
