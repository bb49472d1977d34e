#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) unsigned int nt_u32x4;
__global__ void k(const float* src, float* out, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i * 4 + 4 > n) return;
  nt_u32x4 v = __builtin_nontemporal_load(
      reinterpret_cast<const nt_u32x4*>(src + i * 4));
  out[i * 4 + 0] = __builtin_bit_cast(float, v.x);
  out[i * 4 + 1] = __builtin_bit_cast(float, v.y);
  out[i * 4 + 2] = __builtin_bit_cast(float, v.z);
  out[i * 4 + 3] = __builtin_bit_cast(float, v.w);
}
int main() {
  const int N = 4096;
  float *s, *o;
  hipMalloc(&s, N * 4); hipMalloc(&o, N * 4);
  float h[N], r[N];
  for (int i = 0; i < N; ++i) h[i] = (float)i * 0.5f;
  hipMemcpy(s, h, N * 4, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k, dim3(N / 4 / 64), dim3(64), 0, 0, s, o, N);
  hipMemcpy(r, o, N * 4, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < N; ++i) if (r[i] != h[i]) { if (bad < 5) printf("i=%d want %f got %f\n", i, h[i], r[i]); bad++; }
  printf("bad=%d/%d\n", bad, N);
  return 0;
}
// RESULT (MI355X, ROCm 7.2, -O3): bad=3072/4096 — v.y/z/w come back as a
// broadcast of v.x when the loaded ext_vector is scalarized component-wise.
// This is why csrc keeps nontemporal loads ONLY where the vector is
// consumed whole (MFMA fragments / dot2 streams, all parity-tested).
