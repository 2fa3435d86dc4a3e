// Fragment-layout verification probe for gfx950 v_mfma_f32_32x32x16_bf16
// (and v_mfma_f32_16x16x32_bf16).
//
// No ISA manual is available in this environment, so the A/B operand
// lane->element maps are verified empirically before the MFMA margins kernel
// relies on them. Hypothesis (CDNA3 32x32x8 doubled-K pattern):
//   A[32r x 16k]: lane l holds A[l&31][8*(l>>5) + j], j = 0..7 (contiguous k)
//   B[16k x 32c]: lane l holds B[8*(l>>5) + j][l&31]
//   C/D        : col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// For 16x16x32:
//   A[16r x 32k]: lane l holds A[l&15][8*(l>>4) + j]
//   B[32k x 16c]: lane l holds B[8*(l>>4) + j][l&15]
//   C/D        : col = lane&15, row = (lane>>4)*4 + reg
// Asymmetric random inputs (transpose-detecting, guide §5.4 rule 16) vs a
// host fp32 reference. Build:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 benchmarks/mfma_probe.hip -o gpurun_out/mfma_probe
// Run on an MI355X; prints PASS/FAIL per shape.

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <vector>

typedef unsigned short ubf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;

static ubf16 f2bf(float f) {
  union { float f; unsigned u; } v{f};
  unsigned r = (v.u + 0x7FFF + ((v.u >> 16) & 1)) >> 16;  // RNE
  return (ubf16)r;
}
static float bf2f_h(ubf16 u) {
  union { unsigned u; float f; } v{(unsigned)u << 16};
  return v.f;
}

__global__ void probe_32x32x16(const ubf16* A, const ubf16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = *(const __bf16*)&A[(l & 31) * 16 + 8 * (l >> 5) + j];
    b[j] = *(const __bf16*)&B[(8 * (l >> 5) + j) * 32 + (l & 31)];
  }
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    D[row * 32 + (l & 31)] = c[r];
  }
}

__global__ void probe_16x16x32(const ubf16* A, const ubf16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = *(const __bf16*)&A[(l & 15) * 32 + 8 * (l >> 4) + j];
    b[j] = *(const __bf16*)&B[(8 * (l >> 4) + j) * 16 + (l & 15)];
  }
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r) {
    const int row = (l >> 4) * 4 + r;
    D[row * 16 + (l & 15)] = c[r];
  }
}

static int check(const char* name, int M, int N, int K,
                 void (*kern)(const ubf16*, const ubf16*, float*)) {
  std::vector<ubf16> A(M * K), B(K * N);
  unsigned s = 12345;
  auto rnd = [&]() { s = s * 1664525u + 1013904223u; return ((s >> 8) % 2000 - 1000) / 500.0f; };
  for (auto& v : A) v = f2bf(rnd());
  for (auto& v : B) v = f2bf(rnd());
  std::vector<float> ref(M * N, 0.f);
  for (int i = 0; i < M; ++i)
    for (int j = 0; j < N; ++j) {
      float acc = 0.f;
      for (int k = 0; k < K; ++k) acc += bf2f_h(A[i * K + k]) * bf2f_h(B[k * N + j]);
      ref[i * N + j] = acc;
    }
  ubf16 *dA, *dB; float* dD;
  hipMalloc(&dA, A.size() * 2); hipMalloc(&dB, B.size() * 2); hipMalloc(&dD, M * N * 4);
  hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice);
  hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(kern, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  std::vector<float> D(M * N);
  hipMemcpy(D.data(), dD, M * N * 4, hipMemcpyDeviceToHost);
  hipDeviceSynchronize();
  int bad = 0;
  float worst = 0.f;
  for (int i = 0; i < M * N; ++i) {
    float e = fabsf(D[i] - ref[i]);
    if (e > worst) worst = e;
    if (e > 1e-3f + 1e-3f * fabsf(ref[i])) ++bad;
  }
  printf("%s: %s (bad=%d/%d, worst abs err %.3g)\n", name,
         bad == 0 ? "PASS" : "FAIL", bad, M * N, worst);
  hipFree(dA); hipFree(dB); hipFree(dD);
  return bad == 0 ? 0 : 1;
}

int main() {
  int rc = 0;
  rc |= check("mfma_f32_32x32x16_bf16 hypothesized layout", 32, 32, 16, probe_32x32x16);
  rc |= check("mfma_f32_16x16x32_bf16 hypothesized layout", 16, 16, 32, probe_16x16x32);
  return rc;
}
