// Standalone probe: verify the v_mfma_f32_32x32x16_bf16 A/B/C lane mappings
// on gfx950 before building the wrw conv kernel on them.
// Hypothesis (CDNA3 32x32x8 convention scaled to 2xK):
//   A[32(m) x 16(k)]: lane l holds m = l&31, k = 8*(l>>5) + i, i in 0..8
//   B[16(k) x 32(n)]: lane l holds n = l&31, k = 8*(l>>5) + i
//   C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in 0..16
// Build: hipcc --offload-arch=gfx950 -o /tmp/mfma_probe mfma_probe.hip
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <vector>

typedef __bf16 bf16_t;
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__global__ void probe_kernel(const bf16_t* A, const bf16_t* B, float* C) {
  const int l = threadIdx.x;  // one wave
  bf16x8 a, b;
  const int m = l & 31, half = l >> 5;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int k = half * 8 + i;
    a[i] = A[m * 16 + k];   // A row-major [32][16]
    b[i] = B[k * 32 + m];   // B row-major [16][32], n = l&31
  }
  f32x16 c;
#pragma unroll
  for (int i = 0; i < 16; ++i) c[i] = 0.0f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * half;
    int col = l & 31;
    C[row * 32 + col] = c[r];
  }
}

int main() {
  std::vector<float> Ah(32 * 16), Bh(16 * 32);
  // asymmetric patterns (the guide warns symmetric B passes row/col swaps)
  for (int m = 0; m < 32; ++m)
    for (int k = 0; k < 16; ++k) Ah[m * 16 + k] = (float)((m * 7 + k * 3) % 11) - 5.0f;
  for (int k = 0; k < 16; ++k)
    for (int n = 0; n < 32; ++n) Bh[k * 32 + n] = (float)((k * 5 + n * 2) % 13) - 6.0f;

  std::vector<bf16_t> Abf(32 * 16), Bbf(16 * 32);
  for (int i = 0; i < 32 * 16; ++i) Abf[i] = (bf16_t)Ah[i];
  for (int i = 0; i < 16 * 32; ++i) Bbf[i] = (bf16_t)Bh[i];

  bf16_t *dA, *dB;
  float* dC;
  (void)hipMalloc(&dA, sizeof(bf16_t) * 32 * 16);
  (void)hipMalloc(&dB, sizeof(bf16_t) * 16 * 32);
  (void)hipMalloc(&dC, sizeof(float) * 32 * 32);
  (void)hipMemcpy(dA, Abf.data(), sizeof(bf16_t) * 32 * 16, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, Bbf.data(), sizeof(bf16_t) * 16 * 32, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dC);
  (void)hipDeviceSynchronize();
  std::vector<float> Ch(32 * 32);
  (void)hipMemcpy(Ch.data(), dC, sizeof(float) * 32 * 32, hipMemcpyDeviceToHost);

  // CPU reference in bf16-rounded inputs
  double max_err = 0;
  int bad = 0;
  for (int m2 = 0; m2 < 32; ++m2)
    for (int n = 0; n < 32; ++n) {
      float ref = 0;
      for (int k = 0; k < 16; ++k)
        ref += (float)Abf[m2 * 16 + k] * (float)Bbf[k * 32 + n];
      double e = std::fabs(ref - Ch[m2 * 32 + n]);
      max_err = std::max(max_err, e);
      if (e > 1e-3 && bad < 5) {
        printf("MISMATCH m=%d n=%d ref=%f got=%f\n", m2, n, ref, Ch[m2 * 32 + n]);
        ++bad;
      }
    }
  printf("max_err=%g %s\n", max_err, max_err < 1e-3 ? "MAPPING-OK" : "MAPPING-WRONG");
  return max_err < 1e-3 ? 0 : 1;
}
