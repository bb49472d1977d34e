// pybind11 bindings for the distributedes_amd gfx950 HIP kernels.
//
// This is the only TU that includes torch headers (keeps hipcc compile time
// down); kernels live in *.hip TUs exposed through extern "C" launchers.
#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include <cstdint>

extern "C" {
void des_launch_noise_fill(float*, int64_t, uint64_t, hipStream_t);
void des_launch_perturb_f32(float*, const float*, const float*, const int64_t*,
                            const float*, float, const float*, int64_t,
                            int64_t, hipStream_t);
void des_launch_perturb_bf16(uint16_t*, const float*, const float*,
                             const int64_t*, const float*, float, const float*,
                             int64_t, int64_t, hipStream_t);
void des_launch_wsum(float*, float*, const float*, const int64_t*,
                     const float*, const float*, int64_t, int64_t, bool,
                     hipStream_t);
}

namespace {

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

void noise_fill(torch::Tensor table, int64_t seed) {
  CHECK_DEV(table); CHECK_CONT(table);
  TORCH_CHECK(table.dtype() == torch::kFloat32, "table must be fp32");
  des_launch_noise_fill(table.data_ptr<float>(), table.numel(),
                        (uint64_t)seed, cur_stream());
}

void perturb(torch::Tensor out, torch::Tensor theta, torch::Tensor table,
             torch::Tensor offsets, torch::Tensor signs, double sigma,
             torch::Tensor sigma_vec) {
  CHECK_DEV(out); CHECK_CONT(out); CHECK_DEV(theta); CHECK_CONT(theta);
  CHECK_DEV(table); CHECK_CONT(table);
  CHECK_DEV(offsets); CHECK_CONT(offsets); CHECK_DEV(signs); CHECK_CONT(signs);
  int64_t P = out.size(0), D = out.size(1);
  TORCH_CHECK(theta.numel() == D && offsets.numel() == P && signs.numel() == P);
  TORCH_CHECK(offsets.dtype() == torch::kInt64 && signs.dtype() == torch::kFloat32);
  const float* sv = nullptr;
  if (sigma_vec.numel() > 0) {
    CHECK_DEV(sigma_vec); CHECK_CONT(sigma_vec);
    TORCH_CHECK(sigma_vec.numel() == D, "sigma_vec must be [D]");
    sv = sigma_vec.data_ptr<float>();
  }
  if (out.dtype() == torch::kBFloat16) {
    des_launch_perturb_bf16(
        reinterpret_cast<uint16_t*>(out.data_ptr()), theta.data_ptr<float>(),
        table.data_ptr<float>(), offsets.data_ptr<int64_t>(),
        signs.data_ptr<float>(), (float)sigma, sv, P, D, cur_stream());
  } else if (out.dtype() == torch::kFloat32) {
    des_launch_perturb_f32(out.data_ptr<float>(), theta.data_ptr<float>(),
                           table.data_ptr<float>(),
                           offsets.data_ptr<int64_t>(), signs.data_ptr<float>(),
                           (float)sigma, sv, P, D, cur_stream());
  } else {
    TORCH_CHECK(false, "perturb: out must be bf16 or fp32");
  }
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
                  table.data_ptr<float>(), offsets.data_ptr<int64_t>(),
                  signs.data_ptr<float>(), weights.data_ptr<float>(), P, D,
                  want_sq, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("noise_fill", &noise_fill, "K1: Philox N(0,1) table fill");
  m.def("perturb", &perturb, "K2: theta + sigma*sign*noise-slice");
  m.def("weighted_noise_sum", &weighted_noise_sum,
        "K10: sum_i w_i s_i eps_i (+ optional sum w_i eps_i^2)");
}
