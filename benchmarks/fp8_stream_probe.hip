// fp8 margins-pattern stream probe (diagnostic, not a production kernel).
//
// Question (profiles/r01_fp8_margins_investigation.txt): why does the fp8
// dense-margins kernel stream A at only ~4.5 TB/s when its bf16 twin and the
// fp8 grad kernel reach 5.9-6.6 TB/s, with PMC showing zero overfetch?
//
// Reproduces the exact k_dense_margins access pattern (R rows per wave,
// 16 B per lane per row, nt loads, one f32 w-vector load per chunk) at three
// arithmetic levels to attribute the deficit:
//   MODE 0: integer checksum only (pure stream limit of the pattern)
//   MODE 1: + cvt_pk_f32_fp8 conversions (adds the 8 cvt per 16 B)
//   MODE 2: + w loads and fma (the full production inner loop)
// swept over R in {4, 8, 16}.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 benchmarks/fp8_stream_probe.hip -o /tmp/fp8probe
// Run:   /tmp/fp8probe [rows] [d]   (defaults 16384 x 1e6 = 16.4 GB)

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define WAVE 64
#define BLOCK 256
#define WPB (BLOCK / WAVE)
typedef long long ll;

#define HIP_CHECK(x)                                                   \
  do {                                                                 \
    hipError_t e = (x);                                                \
    if (e != hipSuccess) {                                             \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), \
              __FILE__, __LINE__);                                     \
      exit(1);                                                         \
    }                                                                  \
  } while (0)

using i32x4 = __attribute__((ext_vector_type(4))) int;
using f32x2 = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;

template <int R, int MODE>
__global__ __launch_bounds__(BLOCK) void k_probe(
    const unsigned char* __restrict__ A, const float* __restrict__ w, ll n,
    ll d, float* __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WPB + wid;
  const ll n_waves = (ll)gridDim.x * WPB;
  const ll n_rg = (n + R - 1) / R;
  float facc[R];
  int iacc = 0;
#pragma unroll
  for (int j = 0; j < R; ++j) facc[j] = 0.f;
  for (ll rg = wave_gid; rg < n_rg; rg += n_waves) {
    const ll r0 = rg * R;
    if (r0 + R > n) break;  // probe: skip the ragged tail row group
    const unsigned char* __restrict__ row0 = A + r0 * d;
    for (ll c = (ll)lane * 16; c + 16 <= d; c += (ll)WAVE * 16) {
      float wv[16];
      if constexpr (MODE == 2) {
#pragma unroll
        for (int ch = 0; ch < 4; ++ch) {
          f32x4 v = *(const f32x4*)(w + c + ch * 4);
#pragma unroll
          for (int k = 0; k < 4; ++k) wv[ch * 4 + k] = v[k];
        }
      }
#pragma unroll
      for (int j = 0; j < R; ++j) {
        i32x4 v = __builtin_nontemporal_load((const i32x4*)(row0 + (ll)j * d + c));
        if constexpr (MODE == 0) {
          iacc += v[0] + v[1] + v[2] + v[3];
        } else {
#pragma unroll
          for (int ch = 0; ch < 4; ++ch) {
            const f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(v[ch], false);
            const f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(v[ch], true);
            if constexpr (MODE == 1) {
              facc[j] += lo[0] + lo[1] + hi[0] + hi[1];
            } else {
              facc[j] += lo[0] * wv[ch * 4 + 0] + lo[1] * wv[ch * 4 + 1] +
                         hi[0] * wv[ch * 4 + 2] + hi[1] * wv[ch * 4 + 3];
            }
          }
        }
      }
    }
  }
  float s = (MODE == 0) ? (float)iacc : 0.f;
#pragma unroll
  for (int j = 0; j < R; ++j) s += facc[j];
  if (s == 12345.678f) out[threadIdx.x] = s;  // defeat DCE, never taken
}

template <int R, int MODE>
static void run_case(const unsigned char* dA, const float* dw, ll n, ll d,
                     float* dout, const char* name) {
  const ll n_rg = n / R;
  int grid = (int)((n_rg + WPB - 1) / WPB);
  if (grid > 65535 * 16) grid = 65535 * 16;
  // warmup
  hipLaunchKernelGGL((k_probe<R, MODE>), dim3(grid), dim3(BLOCK), 0, 0, dA, dw,
                     n, d, dout);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  const int REPS = 5;
  HIP_CHECK(hipEventRecord(e0, 0));
  for (int i = 0; i < REPS; ++i)
    hipLaunchKernelGGL((k_probe<R, MODE>), dim3(grid), dim3(BLOCK), 0, 0, dA,
                       dw, n, d, dout);
  HIP_CHECK(hipEventRecord(e1, 0));
  HIP_CHECK(hipEventSynchronize(e1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
  ms /= REPS;
  const double gb = (double)n * d / 1e9;
  printf("%-28s R=%-3d %8.3f ms  %7.2f GB/s\n", name, R, ms, gb / (ms / 1e3));
  HIP_CHECK(hipEventDestroy(e0));
  HIP_CHECK(hipEventDestroy(e1));
}

int main(int argc, char** argv) {
  ll n = argc > 1 ? atoll(argv[1]) : 16384;
  ll d = argc > 2 ? atoll(argv[2]) : 1000000;
  unsigned char* dA;
  float *dw, *dout;
  HIP_CHECK(hipMalloc(&dA, n * d));
  HIP_CHECK(hipMemset(dA, 0x35, n * d));  // some valid fp8 bit pattern
  HIP_CHECK(hipMalloc(&dw, d * sizeof(float)));
  HIP_CHECK(hipMemset(dw, 0, d * sizeof(float)));
  HIP_CHECK(hipMalloc(&dout, BLOCK * sizeof(float)));
  printf("fp8 margins-pattern probe: n=%lld d=%lld (%.2f GB)\n", n, d,
         (double)n * d / 1e9);
  run_case<4, 0>(dA, dw, n, d, dout, "int-stream (no cvt)");
  run_case<8, 0>(dA, dw, n, d, dout, "int-stream (no cvt)");
  run_case<16, 0>(dA, dw, n, d, dout, "int-stream (no cvt)");
  run_case<4, 1>(dA, dw, n, d, dout, "cvt_pk, no w/fma");
  run_case<8, 1>(dA, dw, n, d, dout, "cvt_pk, no w/fma");
  run_case<16, 1>(dA, dw, n, d, dout, "cvt_pk, no w/fma");
  run_case<4, 2>(dA, dw, n, d, dout, "full margins inner");
  run_case<8, 2>(dA, dw, n, d, dout, "full margins inner");
  run_case<16, 2>(dA, dw, n, d, dout, "full margins inner");
  return 0;
}
