// Standalone probe: verify the gfx950 v_mfma_f32_16x16x32_bf16 fragment
// layout with asymmetric operands (guide §3: A=I-check catches transposes).
//
//   hipcc --offload-arch=gfx950 -O2 tools/probe_mfma.hip -o /tmp/pm && /tmp/pm
//
// Hypothesis H (LLVM/AMD lab-notes layout):
//   A 16x32: lane l holds A[l&15][(l>>4)*8 + j], j = 0..7
//   B 32x16: lane l holds B[(l>>4)*8 + j][l&15]
//   D 16x16: lane l, reg r -> D[(l>>4)*4 + r][l&15]
// The host checks D == A@B under H; if it mismatches, it prints which
// transposed interpretation fits instead.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

static uint16_t f2bf(float f) {
  union { float f; uint32_t u; } v{f};
  uint32_t r = v.u + 0x7FFF + ((v.u >> 16) & 1);
  return (uint16_t)(r >> 16);
}
static float bf2f(uint16_t h) {
  union { uint32_t u; float f; } v{(uint32_t)h << 16};
  return v.f;
}

__global__ void probe(const uint16_t* A, const uint16_t* B, float* D) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[(l & 15) * 32 + ((l >> 4) * 8 + j)];   // A[m][k]
    b[j] = (short)B[((l >> 4) * 8 + j) * 16 + (l & 15)];   // B[k][n]
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r) D[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

int main() {
  uint16_t hA[16 * 32], hB[32 * 16];
  float ref[16 * 16] = {0};
  // asymmetric, exactly representable small ints
  for (int m = 0; m < 16; ++m)
    for (int k = 0; k < 32; ++k) hA[m * 32 + k] = f2bf((float)((m + 2 * k) % 5 - 2));
  for (int k = 0; k < 32; ++k)
    for (int n = 0; n < 16; ++n) hB[k * 16 + n] = f2bf((float)((3 * k + n) % 7 - 3));
  for (int m = 0; m < 16; ++m)
    for (int n = 0; n < 16; ++n) {
      float s = 0;
      for (int k = 0; k < 32; ++k)
        s += bf2f(hA[m * 32 + k]) * bf2f(hB[k * 16 + n]);
      ref[m * 16 + n] = s;
    }
  uint16_t *dA, *dB; float* dD;
  hipMalloc(&dA, sizeof(hA)); hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dD, 16 * 16 * 4);
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  float out[256];
  hipMemcpy(out, dD, sizeof(out), hipMemcpyDeviceToHost);
  int ok = 1, okT = 1;
  for (int m = 0; m < 16 && (ok || okT); ++m)
    for (int n = 0; n < 16; ++n) {
      if (fabsf(out[m * 16 + n] - ref[m * 16 + n]) > 1e-3) ok = 0;
      if (fabsf(out[n * 16 + m] - ref[m * 16 + n]) > 1e-3) okT = 0;
    }
  printf("H (D[m][n] direct): %s\n", ok ? "MATCH" : "mismatch");
  printf("H^T (D transposed): %s\n", okT ? "MATCH" : "mismatch");
  if (!ok && !okT) {
    printf("sample row0 out: ");
    for (int n = 0; n < 8; ++n) printf("%.0f ", out[n]);
    printf("\nsample row0 ref: ");
    for (int n = 0; n < 8; ++n) printf("%.0f ", ref[n]);
    printf("\n");
  }
  return 0;
}
