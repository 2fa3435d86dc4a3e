// sparkagd_amd CDNA4 (gfx950 / MI355X) kernel library.
//
// Hand-written HIP replacements for everything the reference delegates to
// external JVM jars:
//   K1  dense fused gradient pass  (margins = A·w ; multiplier ; grad = A^T·m ;
//       loss reduction)             <- MLlib Gradient.compute per-example loop,
//                                      reference call site AGD.scala:197-200
//   K2  CSR sparse gradient pass    <- MLlib sparse-Vector path
//   K4  fused prox/update kernels   <- MLlib Simple/L1/SquaredL2 Updater,
//                                      reference call site AGD.scala:215-220
//   K6  axpby affine combination    <- Breeze vector ops, AGD.scala:249,255
//   K7  fused multi-reduction of the 5 per-iteration scalars
//                                   <- Breeze norm/dot, AGD.scala:263-327
//   +   multinomial softmax family (beyond the reference, binary-only in
//       MLlib 1.3): KC-templated VALU margins/multiplier/grad kernels,
//       hipBLASLt bf16 NT/TN GEMM entries for the dense margins/grad
//       (skinny GEMMs on the MFMA matrix cores), and CSR gather kernels
//       with a deterministic CSC-gather transpose.
//
// Design notes (MI355X):
//  * Every data-pass kernel is HBM-bandwidth-bound (arithmetic intensity of a
//    GEMV pair is ~1 FLOP/byte vs the chip's ~400 at bf16 MFMA peak), so the
//    kernels are built wave-64-first for the memory system: 16-byte loads per
//    lane, one wave per row (dense A·w) / contiguous 256-thread column slabs
//    (dense A^T·m), grid-stride everywhere, fp32 accumulation (fp64 for fp64
//    shards), fp64 block reductions for the loss and the iteration scalars.
//  * Deterministic by construction: dense A^T·m writes private partial
//    slabs reduced in fixed order; CSR A^T·m gathers over a CSC copy by
//    default (the fp32 atomic scatter remains an option); all fp64 scalar
//    reductions are two-stage (block partials + one-block final) — no
//    same-word atomics anywhere on the default paths.
//  * No Triton, no CUDA-compat headers, no torch headers: plain HIP + a C ABI
//    taking raw device pointers and a hipStream_t, loaded via ctypes.
//
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC agd_kernels.hip -o libagd_hip.so

#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>

// ---------------------------------------------------------------------------
// Error plumbing
// ---------------------------------------------------------------------------

static char g_err[1024] = {0};

extern "C" const char* agd_last_error() { return g_err; }

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      snprintf(g_err, sizeof(g_err), "%s:%d %s: %s", __FILE__, __LINE__,       \
               #expr, hipGetErrorString(_e));                                  \
      return 1;                                                                \
    }                                                                          \
  } while (0)

// ---------------------------------------------------------------------------
// Common device helpers
// ---------------------------------------------------------------------------

#define WAVE 64
#define BLOCK 256
#define WAVES_PER_BLOCK (BLOCK / WAVE)

typedef long long ll;
typedef unsigned short ubf16;  // raw bf16 bits

__device__ __forceinline__ float bf2f(ubf16 u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

template <typename T>
__device__ __forceinline__ T wave_reduce_sum(T v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-level fp64 reduction of NACC accumulators into a per-block partial
// slot; a one-block k_reduce_partials finishes the sum. Two stages instead
// of leader atomics: 8192 same-word fp64 atomics serialize to ~0.5 ms
// (measured, d=1e7 fused-scalars) and are order-nondeterministic; the
// partial buffer costs ~30 us and is bitwise reproducible.
template <int NACC>
__device__ __forceinline__ void block_reduce_partial(double (&acc)[NACC],
                                                     double* part) {
  __shared__ double lds[WAVES_PER_BLOCK][NACC];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int k = 0; k < NACC; ++k) {
    acc[k] = wave_reduce_sum(acc[k]);
    if (lane == 0) lds[wid][k] = acc[k];
  }
  __syncthreads();
  if (wid == 0 && lane < WAVES_PER_BLOCK) {
#pragma unroll
    for (int k = 0; k < NACC; ++k) {
      double v = lds[lane][k];
#pragma unroll
      for (int off = WAVES_PER_BLOCK / 2; off > 0; off >>= 1)
        v += __shfl_xor(v, off, WAVE);
      if (lane == 0) part[(ll)blockIdx.x * NACC + k] = v;
    }
  }
}

// Final stage: one block sums the per-block partials into out[NACC]
// (ACCUMULATES into out so callers may chain; callers zero out first).
template <int NACC>
__global__ __launch_bounds__(BLOCK) void k_reduce_partials(
    const double* __restrict__ part, int nblocks, double* __restrict__ out) {
  double acc[NACC];
#pragma unroll
  for (int k = 0; k < NACC; ++k) acc[k] = 0.0;
  for (int i = threadIdx.x; i < nblocks; i += BLOCK)
#pragma unroll
    for (int k = 0; k < NACC; ++k) acc[k] += part[(ll)i * NACC + k];
  __shared__ double lds[WAVES_PER_BLOCK][NACC];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int k = 0; k < NACC; ++k) {
    acc[k] = wave_reduce_sum(acc[k]);
    if (lane == 0) lds[wid][k] = acc[k];
  }
  __syncthreads();
  if (wid == 0 && lane < WAVES_PER_BLOCK) {
#pragma unroll
    for (int k = 0; k < NACC; ++k) {
      double v = lds[lane][k];
#pragma unroll
      for (int off = WAVES_PER_BLOCK / 2; off > 0; off >>= 1)
        v += __shfl_xor(v, off, WAVE);
      if (lane == 0) out[k] += v;
    }
  }
}

// Vector load of W elements of TA starting at p (16-byte pattern for W>1),
// converted to the accumulator type TACC.
template <typename TA, typename TACC, int W, bool NT = false>
__device__ __forceinline__ void loadW(const TA* __restrict__ p, TACC (&out)[W]) {
  if constexpr (W == 1) {
    if constexpr (sizeof(TA) == 1)
      out[0] = __builtin_amdgcn_cvt_f32_fp8((int)*(const unsigned char*)p, 0);
    else if constexpr (sizeof(TA) == 2) out[0] = bf2f(*(const ubf16*)p);
    else out[0] = (TACC)p[0];
  } else if constexpr (sizeof(TA) == 1) {  // fp8 e4m3fn, W == 16 (16 B)
    using i32x4 = __attribute__((ext_vector_type(4))) int;
    using f32x2 = __attribute__((ext_vector_type(2))) float;
    i32x4 v = NT ? __builtin_nontemporal_load((const i32x4*)p) : *(const i32x4*)p;
#pragma unroll
    for (int ch = 0; ch < 4; ++ch) {
      const f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(v[ch], false);
      const f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(v[ch], true);
      out[ch * 4 + 0] = lo[0];
      out[ch * 4 + 1] = lo[1];
      out[ch * 4 + 2] = hi[0];
      out[ch * 4 + 3] = hi[1];
    }
  } else if constexpr (sizeof(TA) == 2) {  // bf16, W == 8 (16 B)
    using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;
    u16x8 v = NT ? __builtin_nontemporal_load((const u16x8*)p) : *(const u16x8*)p;
#pragma unroll
    for (int k = 0; k < 8; ++k) out[k] = bf2f((ubf16)v[k]);
  } else if constexpr (sizeof(TA) == 4) {  // f32, W == 4 (16 B)
    using f32x4 = __attribute__((ext_vector_type(4))) float;
    f32x4 v = NT ? __builtin_nontemporal_load((const f32x4*)p) : *(const f32x4*)p;
#pragma unroll
    for (int k = 0; k < 4; ++k) out[k] = (TACC)v[k];
  } else {  // f64, W == 2 (16 B)
    using f64x2 = __attribute__((ext_vector_type(2))) double;
    f64x2 v = NT ? __builtin_nontemporal_load((const f64x2*)p) : *(const f64x2*)p;
#pragma unroll
    for (int k = 0; k < 2; ++k) out[k] = (TACC)v[k];
  }
}

// Vector load of W accumulator-typed elements (16-B chunks; the address is
// W-element aligned by construction in the margins kernel).
template <typename TACC, int W>
__device__ __forceinline__ void loadAcc(const TACC* __restrict__ p, TACC (&out)[W]) {
  constexpr int EPC = 16 / sizeof(TACC);
  if constexpr (W >= EPC) {
#pragma unroll
    for (int ch = 0; ch < W / EPC; ++ch) {
      if constexpr (sizeof(TACC) == 4) {
        using f32x4 = __attribute__((ext_vector_type(4))) float;
        f32x4 v = *(const f32x4*)(p + ch * 4);
#pragma unroll
        for (int k = 0; k < 4; ++k) out[ch * 4 + k] = v[k];
      } else {
        using f64x2 = __attribute__((ext_vector_type(2))) double;
        f64x2 v = *(const f64x2*)(p + ch * 2);
#pragma unroll
        for (int k = 0; k < 2; ++k) out[ch * 2 + k] = v[k];
      }
    }
  } else {
#pragma unroll
    for (int k = 0; k < W; ++k) out[k] = p[k];
  }
}

// Vector store, mirror of loadAcc.
template <typename TACC, int W>
__device__ __forceinline__ void storeAcc(TACC* __restrict__ p, const TACC (&v)[W]) {
  constexpr int EPC = 16 / sizeof(TACC);
  if constexpr (W >= EPC) {
#pragma unroll
    for (int ch = 0; ch < W / EPC; ++ch) {
      if constexpr (sizeof(TACC) == 4) {
        using f32x4 = __attribute__((ext_vector_type(4))) float;
        f32x4 o;
#pragma unroll
        for (int k = 0; k < 4; ++k) o[k] = v[ch * 4 + k];
        *(f32x4*)(p + ch * 4) = o;
      } else {
        using f64x2 = __attribute__((ext_vector_type(2))) double;
        f64x2 o;
#pragma unroll
        for (int k = 0; k < 2; ++k) o[k] = v[ch * 2 + k];
        *(f64x2*)(p + ch * 2) = o;
      }
    }
  } else {
#pragma unroll
    for (int k = 0; k < W; ++k) p[k] = v[k];
  }
}

// ---------------------------------------------------------------------------
// K1a: dense margins  z[r] = <A[r,:], w>
//
// One wave per (row-group, column-slab) task: R rows share each 16-B-per-lane
// load of w, so at fat d (where w does not fit L1/L2 and would otherwise be
// re-streamed per row, 2x the A-traffic at bf16) the w bytes per A byte drop
// by R. Wave-shuffle reduction per row. n_slabs > 1 (thin-n / fat-d shards)
// writes per-slab partial dots into part[s*n + r]; the multiplier kernel sums
// the slab axis => deterministic (no atomics anywhere on the dense path).
// ---------------------------------------------------------------------------

#define MARGIN_ROWS 4   // R: rows per wave (w-load amortization)
#define MARGIN_ROWS_FP8 8  // 1-byte elements: R=4 would make w-bytes ~ A-bytes

template <typename TA, typename TACC, int W, bool NT>
__global__ __launch_bounds__(BLOCK) void k_dense_margins(
    const TA* __restrict__ A, const TACC* __restrict__ w, ll n, ll d,
    ll slab_w, int n_slabs, TACC* __restrict__ part) {
  constexpr int R = (sizeof(TA) == 1) ? MARGIN_ROWS_FP8 : MARGIN_ROWS;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  const ll n_rg = (n + R - 1) / R;
  const ll n_tasks = n_rg * n_slabs;
  for (ll t = wave_gid; t < n_tasks; t += n_waves) {
    const ll rg = t / n_slabs;
    const ll s = t - rg * n_slabs;
    const ll r0 = rg * R;
    const int nr = (int)((r0 + R <= n) ? R : (n - r0));
    const ll c_lo = s * slab_w;
    const ll c_hi = (c_lo + slab_w < d) ? c_lo + slab_w : d;
    const TA* __restrict__ row0 = A + r0 * d;
    TACC acc[R];
#pragma unroll
    for (int j = 0; j < R; ++j) acc[j] = (TACC)0;
    ll c = c_lo + (ll)lane * W;
    if (nr == R) {  // full row group (hot path)
      for (; c + W <= c_hi; c += (ll)WAVE * W) {
        TACC wv[W];
        loadAcc<TACC, W>(w + c, wv);
#pragma unroll
        for (int j = 0; j < R; ++j) {
          TACC v[W];
          loadW<TA, TACC, W, NT>(row0 + (ll)j * d + c, v);
#pragma unroll
          for (int k = 0; k < W; ++k) acc[j] += v[k] * wv[k];
        }
      }
      if (c < c_hi) {
#pragma unroll
        for (int k = 0; k < W; ++k)
          if (c + k < c_hi) {
            const TACC wk = w[c + k];
#pragma unroll
            for (int j = 0; j < R; ++j) {
              TACC v[1];
              loadW<TA, TACC, 1>(row0 + (ll)j * d + c + k, v);
              acc[j] += v[0] * wk;
            }
          }
      }
    } else {  // tail row group (at most one per slab; plain W stride)
      for (c = c_lo + (ll)lane * W; c < c_hi; c += (ll)WAVE * W) {
#pragma unroll
        for (int k = 0; k < W; ++k)
          if (c + k < c_hi) {
            const TACC wk = w[c + k];
            for (int j = 0; j < nr; ++j) {
              TACC v[1];
              loadW<TA, TACC, 1>(row0 + (ll)j * d + c + k, v);
              acc[j] += v[0] * wk;
            }
          }
      }
    }
#pragma unroll
    for (int j = 0; j < R; ++j) {
      acc[j] = wave_reduce_sum(acc[j]);
      if (lane == 0 && j < nr) part[s * n + r0 + j] = acc[j];
    }
  }
}

// ---------------------------------------------------------------------------
// K1a-MFMA: dense margins on the matrix cores (bf16 shards).
//
// D[32x32] = A_frag[32x16] · B_frag[16x32] with v_mfma_f32_32x32x16_bf16:
// each wave owns a 32-row group; the A fragment IS the natural coalesced
// 16-B-per-lane stream of the shard (lane l reads A[r0 + (l&31)][k + 8*(l>>5)
// .. +7]); the weight vector is staged through LDS in 16 KiB bf16 tiles
// (converted once per block, broadcast-read by all 4 waves) and replicated
// across the 32 B-columns, so column 0 of the accumulator is the margin.
// Fragment layouts verified empirically by benchmarks/mfma_probe.hip (no ISA
// doc in this environment). Requires d % 16 == 0; bf16 only. Numerics note:
// the MFMA path rounds w to bf16 (hardware input format); the VALU path keeps
// w fp32 — both pass the parity tests, and both run at the HBM roofline
// (arithmetic intensity ~1 FLOP/byte), so the selector is a measurement
// question, not a throughput one.
// ---------------------------------------------------------------------------

#define MFMA_KCHUNK 8192  // w bf16 elements staged per LDS tile (16 KiB)
#define MFMA_ROWS 128     // rows per 4-wave block (32 per wave)

__device__ __forceinline__ ubf16 f2bf_rne(float f) {
  union { float f; unsigned u; } v{f};
  return (ubf16)((v.u + 0x7FFF + ((v.u >> 16) & 1)) >> 16);
}

using bf16x8_t = __attribute__((ext_vector_type(8))) __bf16;
using f32x16_t = __attribute__((ext_vector_type(16))) float;

__global__ __launch_bounds__(BLOCK) void k_dense_margins_mfma(
    const ubf16* __restrict__ A, const float* __restrict__ w, ll n, ll d,
    ll slab_w, int n_slabs, float* __restrict__ part) {
  __shared__ ubf16 wlds[MFMA_KCHUNK];
  const int l = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll n_rb = (n + MFMA_ROWS - 1) / MFMA_ROWS;
  const ll n_tasks = n_rb * n_slabs;
  for (ll t = blockIdx.x; t < n_tasks; t += gridDim.x) {
    const ll rb = t / n_slabs;
    const ll s = t - rb * n_slabs;
    const ll c_lo = s * slab_w;
    const ll c_hi = (c_lo + slab_w < d) ? c_lo + slab_w : d;
    const ll r0 = rb * MFMA_ROWS + (ll)wid * 32;
    // lane's row (clamped on the ragged tail block: its garbage products land
    // in accumulator entries whose rows are never stored)
    const ll row = (r0 + (l & 31) < n) ? r0 + (l & 31) : n - 1;
    const ubf16* __restrict__ arow = A + row * d + 8 * (l >> 5);
    f32x16_t acc = {};
    for (ll kc = c_lo; kc < c_hi; kc += MFMA_KCHUNK) {
      const ll kcn = (kc + MFMA_KCHUNK <= c_hi) ? MFMA_KCHUNK : (c_hi - kc);
      __syncthreads();
      for (ll k = threadIdx.x; k < kcn; k += BLOCK) wlds[k] = f2bf_rne(w[kc + k]);
      __syncthreads();
      const int bbase = 8 * (l >> 5);
      for (ll kk = 0; kk + 16 <= kcn; kk += 16) {
        bf16x8_t a = *(const bf16x8_t*)(arow + kc + kk);
        bf16x8_t b = *(const bf16x8_t*)&wlds[kk + bbase];
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
      }
    }
    // column 0 of D holds the margins; lanes 0 and 32 carry all 32 rows.
    if ((l & 31) == 0) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const ll rg = r0 + (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
        if (rg < n) part[s * n + rg] = acc[r];
      }
    }
  }
}

// Final margins from per-slab partials (margins-only mode): in-place onto
// the s=0 region of part.
template <typename TACC>
__global__ __launch_bounds__(BLOCK) void k_slab_reduce(TACC* __restrict__ part,
                                                       ll n, int n_slabs) {
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll r = (ll)blockIdx.x * BLOCK + threadIdx.x; r < n; r += stride) {
    TACC v = part[r];
    for (int ss = 1; ss < n_slabs; ++ss) v += part[(ll)ss * n + r];
    part[r] = v;
  }
}

// ---------------------------------------------------------------------------
// K1b: elementwise multiplier + loss/count reduction
//
// loss conventions (z = <w, x_i>; identical algebra to MLlib 1.3's
// LogisticGradient/LeastSquaresGradient/HingeGradient with margin = -z):
//   logistic: m = sigmoid(z) - y ; loss = y>0 ? softplus(-z) : softplus(z)
//   lsq:      m = 2(z-y)         ; loss = (z-y)^2
//   hinge:    s = 2y-1 ; m = (s z < 1) ? -s : 0 ; loss = max(0, 1-s z)
// ---------------------------------------------------------------------------

#define LOSS_LOGISTIC 0
#define LOSS_LSQ 1
#define LOSS_HINGE 2
#define LOSS_SMOOTH_HINGE 3  // quadratically smoothed (Rennie) hinge

template <typename TACC>
__device__ __forceinline__ TACC softplus(TACC t) {
  // log(1 + e^t), overflow-stable
  if (t > (TACC)0) return t + log1p(exp(-t));
  return log1p(exp(t));
}
template <>
__device__ __forceinline__ float softplus<float>(float t) {
  if (t > 0.f) return t + log1pf(__expf(-t));
  return log1pf(__expf(t));
}

// Per-example multiplier + loss for the binary losses (shared by the
// margins-array and affine-combination kernels below).
template <typename TACC>
__device__ __forceinline__ void loss_mult(int loss_type, TACC z, TACC y,
                                          TACC& m, TACC& l) {
  if (loss_type == LOSS_LOGISTIC) {
    m = (TACC)1 / ((TACC)1 + exp(-z)) - y;
    l = (y > (TACC)0) ? softplus<TACC>(-z) : softplus<TACC>(z);
  } else if (loss_type == LOSS_LSQ) {
    const TACC diff = z - y;
    m = (TACC)2 * diff;
    l = diff * diff;
  } else if (loss_type == LOSS_HINGE) {
    const TACC s = (TACC)2 * y - (TACC)1;
    const TACC sz = s * z;
    m = (sz < (TACC)1) ? -s : (TACC)0;
    l = (sz < (TACC)1) ? (TACC)1 - sz : (TACC)0;
  } else {  // LOSS_SMOOTH_HINGE: 0 if sz>=1; (1-sz)^2/2 if 0<sz<1; 0.5-sz else
    const TACC s = (TACC)2 * y - (TACC)1;
    const TACC sz = s * z;
    if (sz >= (TACC)1) {
      m = (TACC)0;
      l = (TACC)0;
    } else if (sz > (TACC)0) {
      m = -s * ((TACC)1 - sz);
      l = (TACC)0.5 * ((TACC)1 - sz) * ((TACC)1 - sz);
    } else {
      m = -s;
      l = (TACC)0.5 - sz;
    }
  }
}

template <typename TACC>
__global__ __launch_bounds__(BLOCK) void k_multiplier(
    const TACC* __restrict__ margins, const float* __restrict__ labels,
    const unsigned char* __restrict__ mask,
    const float* __restrict__ sample_weight, int loss_type, ll n, int n_slabs,
    TACC* __restrict__ mult, double* __restrict__ red_part) {
  double lsum = 0.0, cnt = 0.0;
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
    if (mask && !mask[i]) {
      mult[i] = (TACC)0;
      continue;
    }
    TACC z = margins[i];
    for (int s = 1; s < n_slabs; ++s) z += margins[(ll)s * n + i];
    const TACC y = (TACC)labels[i];
    TACC m, l;
    loss_mult<TACC>(loss_type, z, y, m, l);
    if (sample_weight) {
      const TACC sw = (TACC)sample_weight[i];
      mult[i] = m * sw;
      lsum += (double)l * (double)sw;
      cnt += (double)sw;
    } else {
      mult[i] = m;
      lsum += (double)l;
      cnt += 1.0;
    }
  }
  double acc[2] = {lsum, cnt};
  block_reduce_partial<2>(acc, red_part);
}

// ---------------------------------------------------------------------------
// Gram-solver fused trial kernels (round 2): the per-trial cost after the
// bf16 K·m stream is ~12 small n-space launches + host algebra; these two
// kernels collapse them. Binary losses, full batch (no masks), affine prox.
// ---------------------------------------------------------------------------

// m_y/loss at the AT interpolation y WITHOUT materializing ym:
// z_i = a*xm[i] + b*zm[i] (a = 1-theta, b = theta).
__global__ __launch_bounds__(BLOCK) void k_multiplier_affine(
    const float* __restrict__ xm, const float* __restrict__ zm, double a,
    double b, const float* __restrict__ labels,
    const float* __restrict__ sample_weight, int loss_type, ll n,
    float* __restrict__ mult, double* __restrict__ red_part) {
  double lsum = 0.0, cnt = 0.0;
  const float fa = (float)a, fb = (float)b;
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
    const float z = fa * xm[i] + fb * zm[i];
    const float y = labels[i];
    float m, l;
    loss_mult<float>(loss_type, z, y, m, l);
    if (sample_weight) {
      const float sw = sample_weight[i];
      mult[i] = m * sw;
      lsum += (double)l * (double)sw;
      cnt += (double)sw;
    } else {
      mult[i] = m;
      lsum += (double)l;
      cnt += 1.0;
    }
  }
  double acc[2] = {lsum, cnt};
  block_reduce_partial<2>(acc, red_part);
}

// One pass over n finishing a basis registration + the AT margin updates:
//   g        = gm_raw * inv_c          (gm_raw = K·m_global, unscaled)
//   XB_t     = (f64) g                 (basis-margin row for the G dgemv)
//   md       = (f64) m_y               (dgemv right-hand side)
//   Mstore_T = m_y
//   zm_new   = pz*zm_old + pg*g        (affine prox on margins)
//   xm_new   = (1-theta)*xm_old + theta*zm_new
__global__ __launch_bounds__(BLOCK) void k_gram_state_update(
    const float* __restrict__ gm_raw, const float* __restrict__ m_y,
    const float* __restrict__ xm_old, const float* __restrict__ zm_old,
    double inv_c, double theta, double pz, double pg, ll n,
    double* __restrict__ xb_t, double* __restrict__ md,
    float* __restrict__ mstore_t, float* __restrict__ zm_new,
    float* __restrict__ xm_new) {
  const float fc = (float)inv_c, ft = (float)theta, fz = (float)pz,
              fg = (float)pg;
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
    const float g = gm_raw[i] * fc;
    xb_t[i] = (double)g;
    const float m = m_y[i];
    md[i] = (double)m;
    mstore_t[i] = m;
    const float z = fz * zm_old[i] + fg * g;
    zm_new[i] = z;
    xm_new[i] = (1.0f - ft) * xm_old[i] + ft * z;
  }
}

// ---------------------------------------------------------------------------
// K1c: dense transpose gradient  part[rb, c] = sum_{r in rb} m[r] * A[r, c]
//
// Grid covers (row-block x column-slab). Each 256-thread block owns one
// contiguous slab of BLOCK*W columns for its row block, keeps W accumulators
// per lane in VGPRs, streams its rows (4 KiB contiguous per row for bf16),
// and writes its private partial slab once. Rows with m == 0 (mini-batch
// mask, inactive hinge examples) are skipped wave-uniformly, saving their
// HBM traffic. A second kernel reduces over row blocks => deterministic.
// ---------------------------------------------------------------------------

template <typename TA, typename TACC, int W, bool NT>
__global__ __launch_bounds__(BLOCK) void k_dense_grad(
    const TA* __restrict__ A, const TACC* __restrict__ mult, ll n, ll d,
    ll n_rb, TACC* __restrict__ part) {
  const ll cols_per_block = (ll)BLOCK * W;
  const ll n_cs = (d + cols_per_block - 1) / cols_per_block;
  for (ll b = blockIdx.x; b < n_rb * n_cs; b += gridDim.x) {
    const ll rb = b / n_cs;
    const ll cs = b - rb * n_cs;
    const ll r_lo = rb * n / n_rb;
    const ll r_hi = (rb + 1) * n / n_rb;
    const ll c0 = cs * cols_per_block + (ll)threadIdx.x * W;
    TACC acc[W];
#pragma unroll
    for (int k = 0; k < W; ++k) acc[k] = (TACC)0;
    const bool full = (c0 + W <= d);
    for (ll r = r_lo; r < r_hi; ++r) {
      const TACC m = mult[r];
      if (m == (TACC)0) continue;  // wave-uniform skip
      const TA* __restrict__ p = A + r * d + c0;
      if (full) {
        TACC v[W];
        loadW<TA, TACC, W, NT>(p, v);
#pragma unroll
        for (int k = 0; k < W; ++k) acc[k] += m * v[k];
      } else {
#pragma unroll
        for (int k = 0; k < W; ++k)
          if (c0 + k < d) {
            TACC v[1];
            loadW<TA, TACC, 1>(p + k, v);
            acc[k] += m * v[0];
          }
      }
    }
    TACC* __restrict__ dst = part + rb * d + c0;
    if (full) {
#pragma unroll
      for (int k = 0; k < W; ++k) dst[k] = acc[k];
    } else {
#pragma unroll
      for (int k = 0; k < W; ++k)
        if (c0 + k < d) dst[k] = acc[k];
    }
  }
}

template <typename TACC>
__global__ __launch_bounds__(BLOCK) void k_grad_reduce(
    const TACC* __restrict__ part, ll n_rb, ll d, TACC* __restrict__ grad) {
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll c = (ll)blockIdx.x * BLOCK + threadIdx.x; c < d; c += stride) {
    TACC s = (TACC)0;
    for (ll rb = 0; rb < n_rb; ++rb) s += part[rb * d + c];
    grad[c] = s;
  }
}

// ---------------------------------------------------------------------------
// Multinomial (softmax) logistic regression kernels.
//
// Weights W [d, KC] f32 (feature-major, class-padded to KC = ceil(K/4)*4 for
// 16-B class rows); margins Z and multipliers M are [n, KC] f32 with padded
// columns exactly zero. KC is a compile-time template (4/8/16/32) so the
// per-row class accumulators stay in registers (runtime-indexed arrays go to
// scratch — guide §5.4 rule 20).
// ---------------------------------------------------------------------------

// Z[r, :] = A[r, :] @ W — one wave per row, lane-strided features; per
// element the KC class weights are one/two 16-B loads from the L2/L3-cached
// W; KC wave reductions finish the row.
template <typename TA, int W, int KC>
__global__ __launch_bounds__(BLOCK) void k_margins_multi(
    const TA* __restrict__ A, const float* __restrict__ Wm, ll n, ll d,
    float* __restrict__ Z) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll r = wave_gid; r < n; r += n_waves) {
    const TA* __restrict__ row = A + r * d;
    float acc[KC];
#pragma unroll
    for (int k = 0; k < KC; ++k) acc[k] = 0.f;
    ll c = (ll)lane * W;
    for (; c + W <= d; c += (ll)WAVE * W) {
      float v[W];
      loadW<TA, float, W, true>(row + c, v);
#pragma unroll
      for (int e = 0; e < W; ++e) {
        const float* wr = Wm + (c + e) * KC;
        float wv[KC];
        loadAcc<float, KC>(wr, wv);
#pragma unroll
        for (int k = 0; k < KC; ++k) acc[k] += v[e] * wv[k];
      }
    }
    if (c < d) {
#pragma unroll
      for (int e = 0; e < W; ++e)
        if (c + e < d) {
          float v[1];
          loadW<TA, float, 1>(row + c + e, v);
          const float* wr = Wm + (c + e) * KC;
#pragma unroll
          for (int k = 0; k < KC; ++k) acc[k] += v[0] * wr[k];
        }
    }
#pragma unroll
    for (int k = 0; k < KC; ++k) {
      acc[k] = wave_reduce_sum(acc[k]);
      if (lane == 0) Z[r * KC + k] = acc[k];
    }
  }
}

// M[r, :] = softmax(Z[r, :K]) - onehot(y_r) (times mask/weight); per-row
// loss = logsumexp - z_y. One thread per row; padded classes stay zero.
template <int KC>
__global__ __launch_bounds__(BLOCK) void k_multiplier_multi(
    const float* __restrict__ Z, const float* __restrict__ labels,
    const unsigned char* __restrict__ mask,
    const float* __restrict__ sample_weight, ll n, int K,
    float* __restrict__ M, double* __restrict__ red_part) {
  double lsum = 0.0, cnt = 0.0;
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
    float* __restrict__ mrow = M + i * KC;
    if (mask && !mask[i]) {
#pragma unroll
      for (int k = 0; k < KC; ++k) mrow[k] = 0.f;
      continue;
    }
    float z[KC];
    loadAcc<float, KC>(Z + i * KC, z);
    const int y = (int)labels[i];
    float zmax = -3.0e38f;
#pragma unroll
    for (int k = 0; k < KC; ++k)
      if (k < K && z[k] > zmax) zmax = z[k];
    float esum = 0.f;
    float p[KC];
#pragma unroll
    for (int k = 0; k < KC; ++k) {
      p[k] = (k < K) ? __expf(z[k] - zmax) : 0.f;
      esum += p[k];
    }
    const float inv = 1.0f / esum;
    float scale = 1.0f;
    if (sample_weight) scale = sample_weight[i];
#pragma unroll
    for (int k = 0; k < KC; ++k) {
      float m = p[k] * inv - ((k == y) ? 1.0f : 0.0f);
      mrow[k] = (k < K) ? m * scale : 0.f;
    }
    const float loss = (logf(esum) + zmax) - ((y >= 0 && y < K) ? z[y] : 0.f);
    lsum += (double)loss * (double)scale;
    cnt += (double)scale;
  }
  double acc[2] = {lsum, cnt};
  block_reduce_partial<2>(acc, red_part);
}

// Generic-K multiplier (K > 32, any 4-aligned KC): one WAVE per row, lanes
// stride the class axis, so no per-thread K-sized register array is needed.
// z is streamed twice (max+expsum pass, then the write pass recomputes the
// exps) — still ONE launch and 3 n*KC streams, vs the ~6-kernel torch stage
// this replaces (logsumexp/gather/softmax/scatter_add/mul/sum); the n*K
// space is tiny next to the A streams either way, the win is launch/driver
// overhead off the hot loop and exact-zero pad columns for the GEMM grad.
__global__ __launch_bounds__(BLOCK) void k_multiplier_multi_anyk(
    const float* __restrict__ Z, const float* __restrict__ labels,
    const unsigned char* __restrict__ mask,
    const float* __restrict__ sample_weight, ll n, int K, int KC,
    float* __restrict__ M, double* __restrict__ red_part) {
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  double lsum = 0.0, cnt = 0.0;
  // KC is 4-aligned (padded_k), so 16-B lane loads cover the row exactly;
  // per-element k < K predication handles the logical/padded boundary.
  for (ll i = wave_gid; i < n; i += n_waves) {
    const float* __restrict__ zrow = Z + i * KC;
    float* __restrict__ mrow = M + i * KC;
    if (mask && !mask[i]) {
      for (int k4 = lane * 4; k4 < KC; k4 += WAVE * 4)
        *(f32x4*)&mrow[k4] = f32x4{0.f, 0.f, 0.f, 0.f};
      continue;
    }
    const int y = (int)labels[i];
    float zmax = -3.0e38f;
    for (int k4 = lane * 4; k4 < KC; k4 += WAVE * 4) {
      const f32x4 zv = *(const f32x4*)&zrow[k4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (k4 + j < K) zmax = fmaxf(zmax, zv[j]);
    }
    zmax = wave_reduce_max(zmax);
    float esum = 0.f;
    for (int k4 = lane * 4; k4 < KC; k4 += WAVE * 4) {
      const f32x4 zv = *(const f32x4*)&zrow[k4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (k4 + j < K) esum += __expf(zv[j] - zmax);
    }
    esum = wave_reduce_sum(esum);
    const float inv = 1.0f / esum;
    const float scale = sample_weight ? sample_weight[i] : 1.0f;
    float zy = 0.f;  // only the lane iteration with k == y contributes
    for (int k4 = lane * 4; k4 < KC; k4 += WAVE * 4) {
      const f32x4 zv = *(const f32x4*)&zrow[k4];
      f32x4 mv;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int k = k4 + j;
        if (k < K) {
          mv[j] = (__expf(zv[j] - zmax) * inv - ((k == y) ? 1.0f : 0.0f))
                  * scale;
          if (k == y) zy = zv[j];
        } else {
          mv[j] = 0.f;
        }
      }
      *(f32x4*)&mrow[k4] = mv;
    }
    zy = wave_reduce_sum(zy);
    if (lane == 0) {
      const float loss = (logf(esum) + zmax) - ((y >= 0 && y < K) ? zy : 0.f);
      lsum += (double)loss * (double)scale;
      cnt += (double)scale;
    }
  }
  double acc[2] = {lsum, cnt};
  block_reduce_partial<2>(acc, red_part);
}

// part[rb, c, :] = sum_{r in rb} A[r, c] * M[r, :] — one thread per feature
// column (256 features per block, coalesced 512 B row segments for bf16),
// KC accumulators in registers; a reduce over rb finishes (deterministic,
// reuses k_grad_reduce on length d*KC).
template <typename TA, int KC>
__global__ __launch_bounds__(BLOCK) void k_grad_multi(
    const TA* __restrict__ A, const float* __restrict__ M, ll n, ll d,
    ll n_rb, float* __restrict__ part) {
  const ll n_cs = (d + BLOCK - 1) / BLOCK;
  for (ll b = blockIdx.x; b < n_rb * n_cs; b += gridDim.x) {
    const ll rb = b / n_cs;
    const ll cs = b - rb * n_cs;
    const ll c = cs * BLOCK + threadIdx.x;
    if (c >= d) continue;
    const ll r_lo = rb * n / n_rb;
    const ll r_hi = (rb + 1) * n / n_rb;
    float acc[KC];
#pragma unroll
    for (int k = 0; k < KC; ++k) acc[k] = 0.f;
    for (ll r = r_lo; r < r_hi; ++r) {
      float a[1];
      loadW<TA, float, 1>(A + r * d + c, a);
      const float* mr = M + r * KC;
      float mv[KC];
      loadAcc<float, KC>(mr, mv);
#pragma unroll
      for (int k = 0; k < KC; ++k) acc[k] += a[0] * mv[k];
    }
    float* dst = part + rb * d * KC + c * KC;
#pragma unroll
    for (int k = 0; k < KC; ++k) dst[k] = acc[k];
  }
}

// ---------------------------------------------------------------------------
// K2: CSR margins + transpose gradient (fp32 values)
//
// Wave per row; lanes stride the row's nnz; gathered w reads ride L2/LLC
// (w is <=40 MB fp32 even at d=1e7 and far smaller than the 256 MiB L3).
// A^T·m scatters with fp32 atomics (low contention at large d); the
// deterministic alternative is documented in ops/hiplib.py.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(BLOCK) void k_csr_margins(
    const int* __restrict__ rowptr, const int* __restrict__ col,
    const float* __restrict__ val, const float* __restrict__ w, ll n,
    float* __restrict__ margins) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  using i32x4 = __attribute__((ext_vector_type(4))) int;
  for (ll r = wave_gid; r < n; r += n_waves) {
    const int k_lo = rowptr[r], k_hi = rowptr[r + 1];
    float acc = 0.f;
    // 16-B val/col loads (4 nnz per lane); scalar head to reach 16-B
    // alignment, scalar tail for the remainder. The w gather stays 4 B
    // (random within w — L2/L3-served).
    const int head_end = (k_lo + 3) & ~3;
    const int body_end = k_hi & ~3;
    if (head_end + 4 * WAVE <= body_end) {
      for (int k = k_lo + lane; k < head_end; k += WAVE) acc += val[k] * w[col[k]];
      for (int k = head_end + 4 * lane; k + 4 <= body_end; k += 4 * WAVE) {
        const f32x4 v = *(const f32x4*)&val[k];
        const i32x4 c = *(const i32x4*)&col[k];
#pragma unroll
        for (int j = 0; j < 4; ++j) acc += v[j] * w[c[j]];
      }
      // every full 4-group in [head_end, body_end) is covered by the strided
      // group loop above; the tail is only [body_end, k_hi) (< 4 elements)
      for (int k = body_end + lane; k < k_hi; k += WAVE)
        acc += val[k] * w[col[k]];
    } else {
      for (int k = k_lo + lane; k < k_hi; k += WAVE) acc += val[k] * w[col[k]];
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) margins[r] = acc;
  }
}

// Deterministic CSR transpose gradient via a CSC copy of the shard:
// grad[c] = sum_{k in column c} val[k] * m[row[k]], one thread per column
// (mean nnz/column is small at d >= 1e6; m rides L2/LLC). No atomics =>
// bitwise reproducible; also the faster path when columns are not skewed.
__global__ __launch_bounds__(BLOCK) void k_csc_grad(
    const int* __restrict__ colptr, const int* __restrict__ row,
    const float* __restrict__ val, const float* __restrict__ mult, ll d,
    float* __restrict__ grad) {
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll c = (ll)blockIdx.x * BLOCK + threadIdx.x; c < d; c += stride) {
    const int k_lo = colptr[c], k_hi = colptr[c + 1];
    float acc = 0.f;
    for (int k = k_lo; k < k_hi; ++k) acc += val[k] * mult[row[k]];
    grad[c] = acc;
  }
}

// --- Skew-robust CSC gradient (round 2) -----------------------------------
// One thread per column serializes on power-law data: a Zipf(1.1) d=1e7
// shard puts ~9% of ALL nnz in its hottest column, and one thread grinding
// 6M entries turned the 2.26 ms uniform step into 943 ms (measured,
// profiles/r02_csr_skew_ab.txt). Fix: columns with nnz <= heavy_T keep the
// thread-per-column gather (the measured-fast path on uniform data);
// heavier columns are split into fixed S-entry tasks reduced wave-per-task,
// then combined IN TASK ORDER per column. Fixed segmentation + shuffle
// reductions + sequential combine = still bitwise deterministic.

// order (nullable): columns visited in length-sorted order so the 64
// threads of a wave carry similar-length columns — the wave executes the
// MAX of its threads, and unsorted uniform columns already waste ~2x
// (wave max ~15 vs mean 6.4 nnz at d=1e7). Each thread still owns its
// column's write, so any visit order is bitwise deterministic.
__global__ __launch_bounds__(BLOCK) void k_csc_grad_light(
    const int* __restrict__ colptr, const int* __restrict__ row,
    const float* __restrict__ val, const float* __restrict__ mult, ll d,
    int heavy_T, const int* __restrict__ order, float* __restrict__ grad) {
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < d; i += stride) {
    const ll c = order ? (ll)order[i] : i;
    const int k_lo = colptr[c], k_hi = colptr[c + 1];
    if (k_hi - k_lo > heavy_T) continue;  // heavy path owns grad[c]
    float acc = 0.f;
    for (int k = k_lo; k < k_hi; ++k) acc += val[k] * mult[row[k]];
    grad[c] = acc;
  }
}

// Wave per task: task t covers S entries of heavy column heavy_cols[i]
// starting at slot (t - taskptr[i]); lanes stride the window with 16-B
// val/row loads, wave-reduce, write partial[t].
__global__ __launch_bounds__(BLOCK) void k_csc_heavy_partial(
    const int* __restrict__ colptr, const int* __restrict__ row,
    const float* __restrict__ val, const float* __restrict__ mult,
    const int* __restrict__ heavy_cols, const int* __restrict__ taskptr,
    const int* __restrict__ task_heavy_idx, ll n_tasks, int S,
    float* __restrict__ partial) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  using i32x4 = __attribute__((ext_vector_type(4))) int;
  for (ll t = wave_gid; t < n_tasks; t += n_waves) {
    const int i = task_heavy_idx[t];
    const int c = heavy_cols[i];
    const int c_lo = colptr[c], c_hi = colptr[c + 1];
    const int k_lo = c_lo + (int)(t - taskptr[i]) * S;
    const int k_hi = min(k_lo + S, c_hi);
    float acc = 0.f;
    const int head_end = (k_lo + 3) & ~3;
    const int body_end = k_hi & ~3;
    if (head_end + 4 * WAVE <= body_end) {
      for (int k = k_lo + lane; k < head_end; k += WAVE)
        acc += val[k] * mult[row[k]];
      for (int k = head_end + 4 * lane; k + 4 <= body_end; k += 4 * WAVE) {
        const f32x4 v = *(const f32x4*)&val[k];
        const i32x4 r = *(const i32x4*)&row[k];
#pragma unroll
        for (int j = 0; j < 4; ++j) acc += v[j] * mult[r[j]];
      }
      for (int k = body_end + lane; k < k_hi; k += WAVE)
        acc += val[k] * mult[row[k]];
    } else {
      for (int k = k_lo + lane; k < k_hi; k += WAVE)
        acc += val[k] * mult[row[k]];
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) partial[t] = acc;
  }
}

// Wave per heavy column: lanes stride the task partials, shuffle-reduce.
// The reduction order is FIXED (lane-strided + the shuffle tree), so this
// stays bitwise deterministic while removing the serial tail a single
// thread would have on the hottest column (up to nnz_c/S ~ thousands of
// partials; measured 243 us -> wave-parallel).
__global__ __launch_bounds__(BLOCK) void k_csc_heavy_combine(
    const int* __restrict__ heavy_cols, const int* __restrict__ taskptr,
    const float* __restrict__ partial, ll n_heavy, float* __restrict__ grad) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll i = wave_gid; i < n_heavy; i += n_waves) {
    const int t_lo = taskptr[i], t_hi = taskptr[i + 1];
    float acc = 0.f;
    for (int t = t_lo + lane; t < t_hi; t += WAVE) acc += partial[t];
    acc = wave_reduce_sum(acc);
    if (lane == 0) grad[heavy_cols[i]] = acc;
  }
}

__global__ __launch_bounds__(BLOCK) void k_csr_grad(
    const int* __restrict__ rowptr, const int* __restrict__ col,
    const float* __restrict__ val, const float* __restrict__ mult, ll n,
    float* __restrict__ grad) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll r = wave_gid; r < n; r += n_waves) {
    const float m = mult[r];
    if (m == 0.f) continue;
    const int k_lo = rowptr[r], k_hi = rowptr[r + 1];
    for (int k = k_lo + lane; k < k_hi; k += WAVE)
      atomicAdd(&grad[col[k]], m * val[k]);
  }
}

// ---------------------------------------------------------------------------
// K6: axpby  out = a*x + b*y
// ---------------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(BLOCK) void k_axpby(double a, const T* __restrict__ x,
                                                 double b, const T* __restrict__ y,
                                                 T* __restrict__ out, ll n) {
  constexpr int VE = 16 / sizeof(T);
  const ll stride = (ll)gridDim.x * BLOCK;
  const ll gid = (ll)blockIdx.x * BLOCK + threadIdx.x;
  const T ta = (T)a, tb = (T)b;
  const ll nv = n / VE;
  for (ll i = gid; i < nv; i += stride) {
    T xv[VE], yv[VE], ov[VE];
    loadAcc<T, VE>(x + i * VE, xv);
    loadAcc<T, VE>(y + i * VE, yv);
#pragma unroll
    for (int k = 0; k < VE; ++k) ov[k] = ta * xv[k] + tb * yv[k];
    storeAcc<T, VE>(out + i * VE, ov);
  }
  const ll t = nv * VE + gid;  // tail (< VE elements)
  if (t < n) out[t] = ta * x[t] + tb * y[t];
}

// Fused AT margin update (direct tracked path): zm' = pz*zm + pg*gm and
// xm' = (1-theta)*xm + theta*zm' in ONE pass — replaces the prox-margins +
// interpolation axpby pair (3 reads + 2 writes instead of 4 + 2, one
// launch instead of two). Matters when the margin state is large
// (multiclass n*KC flats); algebra identical to the axpby composition.
template <typename T>
__global__ __launch_bounds__(BLOCK) void k_at_margin_update(
    const T* __restrict__ zm_old, const T* __restrict__ xm_old,
    const T* __restrict__ gm, double pz, double pg, double theta, ll n,
    T* __restrict__ zm_new, T* __restrict__ xm_new) {
  constexpr int VE = 16 / sizeof(T);
  const T tpz = (T)pz, tpg = (T)pg, tth = (T)theta, tom = (T)(1.0 - theta);
  const ll stride = (ll)gridDim.x * BLOCK;
  const ll gid = (ll)blockIdx.x * BLOCK + threadIdx.x;
  const ll nv = n / VE;
  for (ll i = gid; i < nv; i += stride) {
    T zo[VE], xo[VE], gv[VE], zn[VE], xn[VE];
    loadAcc<T, VE>(zm_old + i * VE, zo);
    loadAcc<T, VE>(xm_old + i * VE, xo);
    loadAcc<T, VE>(gm + i * VE, gv);
#pragma unroll
    for (int k = 0; k < VE; ++k) {
      zn[k] = tpz * zo[k] + tpg * gv[k];
      xn[k] = tom * xo[k] + tth * zn[k];
    }
    storeAcc<T, VE>(zm_new + i * VE, zn);
    storeAcc<T, VE>(xm_new + i * VE, xn);
  }
  const ll t = nv * VE + gid;  // tail (< VE elements)
  if (t < n) {
    const T z = tpz * zm_old[t] + tpg * gm[t];
    zm_new[t] = z;
    xm_new[t] = tom * xm_old[t] + tth * z;
  }
}

// ---------------------------------------------------------------------------
// K4: fused prox/update kernels (one per updater type), emitting the
// regularization value via block reduction (reference Updater semantics,
// MLlib 1.3; invoked at AGD.scala:215-220).
// ---------------------------------------------------------------------------

#define PROX_SIMPLE 0
#define PROX_L1 1
#define PROX_L2 2
#define PROX_ELASTIC 3  // l1 soft-threshold then l2 shrink (elastic net)

template <typename T>
__device__ __forceinline__ T prox_elem(int kind, T wi, T gi, T ts, T tl,
                                       T tl2, double& racc) {
  if (kind == PROX_SIMPLE) return wi - ts * gi;
  if (kind == PROX_L1) {
    const T w1 = wi - ts * gi;
    const T aw = fabs(w1) - tl * ts;
    const T wn = (aw > (T)0) ? ((w1 > (T)0) ? aw : -aw) : (T)0;
    racc += (double)tl * fabs((double)wn);
    return wn;
  }
  if (kind == PROX_L2) {
    const T wn = wi * ((T)1 - ts * tl) - ts * gi;
    racc += 0.5 * (double)tl * (double)wn * (double)wn;
    return wn;
  }
  // PROX_ELASTIC: prox of tl*|w|_1 + tl2/2*|w|^2 —
  // soft-threshold by ts*tl, then shrink by 1/(1 + ts*tl2)
  const T w1 = wi - ts * gi;
  const T aw = fabs(w1) - tl * ts;
  const T wsoft = (aw > (T)0) ? ((w1 > (T)0) ? aw : -aw) : (T)0;
  const T wn = wsoft / ((T)1 + ts * tl2);
  racc += (double)tl * fabs((double)wn) +
          0.5 * (double)tl2 * (double)wn * (double)wn;
  return wn;
}

template <typename T>
__global__ __launch_bounds__(BLOCK) void k_prox(int kind, const T* __restrict__ w,
                                                const T* __restrict__ g,
                                                double step, double lam,
                                                double lam2,
                                                T* __restrict__ out,
                                                double* __restrict__ reg_part,
                                                ll n) {
  constexpr int VE = 16 / sizeof(T);
  const ll stride = (ll)gridDim.x * BLOCK;
  const ll gid = (ll)blockIdx.x * BLOCK + threadIdx.x;
  const T ts = (T)step, tl = (T)lam, tl2 = (T)lam2;
  double racc = 0.0;
  const ll nv = n / VE;
  for (ll i = gid; i < nv; i += stride) {
    T wv[VE], gv[VE], ov[VE];
    loadAcc<T, VE>(w + i * VE, wv);
    loadAcc<T, VE>(g + i * VE, gv);
#pragma unroll
    for (int k = 0; k < VE; ++k) ov[k] = prox_elem(kind, wv[k], gv[k], ts, tl, tl2, racc);
    storeAcc<T, VE>(out + i * VE, ov);
  }
  const ll t = nv * VE + gid;
  if (t < n) out[t] = prox_elem(kind, w[t], g[t], ts, tl, tl2, racc);
  if (kind != PROX_SIMPLE) {
    double acc[1] = {racc};
    block_reduce_partial<1>(acc, reg_part);
  }
}

// ---------------------------------------------------------------------------
// K7: fused iteration scalars — ONE pass over (x, y, g_y, x_old) producing
//   out[0] = ||x-y||^2      out[1] = <x-y, g_y>   out[2] = ||x||^2
//   out[3] = ||x-x_old||^2  out[4] = <g_y, x-x_old>
// (backtracking + convergence + restart scalars of AGD.scala:263-327,
//  batched into one launch; fp64 accumulation.)
// ---------------------------------------------------------------------------

template <typename T>
__device__ __forceinline__ void fused_scalars_elem(double xi, double yi,
                                                   double gi, double oi,
                                                   double (&a)[5]) {
  const double xy = xi - yi, dx = xi - oi;
  a[0] += xy * xy;
  a[1] += xy * gi;
  a[2] += xi * xi;
  a[3] += dx * dx;
  a[4] += gi * dx;
}

template <typename T>
__global__ __launch_bounds__(BLOCK) void k_fused_scalars(
    const T* __restrict__ x, const T* __restrict__ y, const T* __restrict__ gy,
    const T* __restrict__ xold, double* __restrict__ out_part, ll n) {
  constexpr int VE = 16 / sizeof(T);
  double acc[5] = {0, 0, 0, 0, 0};
  const ll stride = (ll)gridDim.x * BLOCK;
  const ll gid = (ll)blockIdx.x * BLOCK + threadIdx.x;
  const ll nv = n / VE;
  for (ll i = gid; i < nv; i += stride) {
    T xv[VE], yv[VE], gv[VE], ov[VE];
    loadAcc<T, VE>(x + i * VE, xv);
    loadAcc<T, VE>(y + i * VE, yv);
    loadAcc<T, VE>(gy + i * VE, gv);
    loadAcc<T, VE>(xold + i * VE, ov);
#pragma unroll
    for (int k = 0; k < VE; ++k)
      fused_scalars_elem<T>(xv[k], yv[k], gv[k], ov[k], acc);
  }
  const ll t = nv * VE + gid;
  if (t < n) fused_scalars_elem<T>(x[t], y[t], gy[t], xold[t], acc);
  block_reduce_partial<5>(acc, out_part);
}

template <typename T>
__global__ __launch_bounds__(BLOCK) void k_dot_diff(
    const T* __restrict__ x, const T* __restrict__ y, const T* __restrict__ gx,
    const T* __restrict__ gy, double* __restrict__ out_part, ll n) {
  constexpr int VE = 16 / sizeof(T);
  double a = 0;
  const ll stride = (ll)gridDim.x * BLOCK;
  const ll gid = (ll)blockIdx.x * BLOCK + threadIdx.x;
  const ll nv = n / VE;
  for (ll i = gid; i < nv; i += stride) {
    T xv[VE], yv[VE], gxv[VE], gyv[VE];
    loadAcc<T, VE>(x + i * VE, xv);
    loadAcc<T, VE>(y + i * VE, yv);
    loadAcc<T, VE>(gx + i * VE, gxv);
    loadAcc<T, VE>(gy + i * VE, gyv);
#pragma unroll
    for (int k = 0; k < VE; ++k)
      a += ((double)xv[k] - (double)yv[k]) * ((double)gxv[k] - (double)gyv[k]);
  }
  const ll t = nv * VE + gid;
  if (t < n)
    a += ((double)x[t] - (double)y[t]) * ((double)gx[t] - (double)gy[t]);
  double acc[1] = {a};
  block_reduce_partial<1>(acc, out_part);
}

// ---------------------------------------------------------------------------
// Host API (C ABI). All pointers are DEVICE pointers; stream is a hipStream_t.
// dtype codes: 0 = bf16, 1 = f32, 2 = f64 (features); vector ops: 1 = f32,
// 2 = f64.
// ---------------------------------------------------------------------------

static inline int grid_for(ll work_items, ll per_block) {
  ll g = (work_items + per_block - 1) / per_block;
  if (g > 8192) g = 8192;
  if (g < 1) g = 1;
  return (int)g;
}

// W (elements per 16-B lane load) for a features dtype, or 1 if the row
// stride is not 16-B aligned.
static inline int pick_w(int dtype, ll d) {
  const int wfull = (dtype == 3) ? 16 : (dtype == 0) ? 8 : (dtype == 1) ? 4 : 2;
  return (d % wfull == 0) ? wfull : 1;
}

extern "C" int agd_version() { return 1; }

// Dense-gradient partial-slab planning: how many row blocks the A^T·m pass
// uses (callers size the partial workspace as n_rb * d accumulators).
extern "C" long long agd_dense_rowblocks(long long n, long long d, int a_dtype) {
  const int w = pick_w(a_dtype, d);
  const ll cols_per_block = (ll)BLOCK * w;
  const ll n_cs = (d + cols_per_block - 1) / cols_per_block;
  ll n_rb = 4096 / n_cs;  // target ~4096 blocks total
  if (n_rb < 1) n_rb = 1;
  if (n_rb > n) n_rb = n;
  const ll acc_bytes = (a_dtype == 2) ? 8 : 4;
  while (n_rb > 1 && n_rb * d * acc_bytes > (512LL << 20)) n_rb /= 2;
  return n_rb;
}

// margins_algo: 0 = auto, 1 = VALU row-group kernel, 2 = MFMA kernel.
// MFMA eligibility: bf16 shard with d % 16 == 0.
static inline int margins_algo_eff(int algo, int a_dtype, ll d) {
  const bool mfma_ok = (a_dtype == 0) && (d % 16 == 0);
  if (algo == 2 && !mfma_ok) return 1;
  // auto = VALU: both paths are HBM-bound, and measured round 1 the VALU
  // row-group kernel streams at 97% of the copy ceiling while the MFMA
  // variant pays for its LDS w-staging and lower row-parallelism (~2.2x on
  // the margins pass at d=1e6) AND rounds w to bf16 (matrix-core input
  // format). The MFMA kernel remains selectable (algo=2) and tested.
  if (algo == 0) return 1;
  return algo;
}

// Margin-pass column slabs (>1 only for thin-n / fat-d shards, where
// row-group parallelism alone cannot fill 256 CUs). Callers size the margins
// workspace as n_slabs * n accumulators.
extern "C" int agd_margin_slabs(long long n, long long d, int a_dtype,
                                int margins_algo) {
  const int algo = margins_algo_eff(margins_algo, a_dtype, (ll)d);
  ll n_units, target, min_slab;
  if (algo == 2) {
    n_units = (n + MFMA_ROWS - 1) / MFMA_ROWS;  // 4-wave blocks
    target = 2048;                               // ~8 blocks/CU
    min_slab = 4096;
  } else {
    const int r = (a_dtype == 3) ? MARGIN_ROWS_FP8 : MARGIN_ROWS;
    n_units = (n + r - 1) / r;  // waves
    target = 16384;
    const int w = pick_w(a_dtype, d);
    min_slab = (ll)WAVE * w * 4;
  }
  if (n_units >= target) return 1;
  ll max_slabs = (d + min_slab - 1) / min_slab;
  ll want = (target + n_units - 1) / n_units;
  ll s = want < max_slabs ? want : max_slabs;
  if (s < 1) s = 1;
  if (s > 1024) s = 1024;
  return (int)s;
}

template <typename TA, typename TACC, int W>
static int dense_eval_t(const void* A, const float* labels,
                        const unsigned char* mask, const float* sample_weight,
                        const void* w, ll n, ll d,
                        void* grad_out, double* loss_count, void* margins_ws,
                        void* mult_ws, void* part_ws, ll n_rb, int loss_type,
                        int n_slabs, int need_grad, int margins_algo,
                        int nt_loads, int mode, double* red_ws,
                        hipStream_t stream) {
  const TA* a = (const TA*)A;
  const TACC* wp = (const TACC*)w;
  TACC* margins = (TACC*)margins_ws;
  TACC* mult = (TACC*)mult_ws;
  TACC* grad = (TACC*)grad_out;
  TACC* part = (n_rb == 1) ? grad : (TACC*)part_ws;
  const int algo = margins_algo_eff(margins_algo, sizeof(TA) == 2 ? 0 : (sizeof(TA) == 4 ? 1 : 2), d);
  const bool use_mfma = (algo == 2) && (sizeof(TA) == 2) && (W == 8);

  ll slab_w = d;
  if (mode == 2 || mode == 3) n_slabs = 1;  // margins/mult provided by caller
  if (n_slabs > 1) {
    slab_w = (d + n_slabs - 1) / n_slabs;
    const ll align = (ll)WAVE * W;
    slab_w = ((slab_w + align - 1) / align) * align;
    n_slabs = (int)((d + slab_w - 1) / slab_w);
  }

  if (mode == 2 || mode == 3) {
    // skip the margins pass entirely
  } else if (use_mfma) {
    const ll tasks = ((n + MFMA_ROWS - 1) / MFMA_ROWS) * n_slabs;
    const int grid = grid_for(tasks, 1);
    hipLaunchKernelGGL(k_dense_margins_mfma, dim3(grid), dim3(BLOCK), 0,
                       stream, (const ubf16*)A, (const float*)w, n, d, slab_w,
                       n_slabs, (float*)margins);
  } else {
    const int R = (sizeof(TA) == 1) ? MARGIN_ROWS_FP8 : MARGIN_ROWS;
    const ll tasks = ((n + R - 1) / R) * n_slabs;
    const int grid = grid_for(tasks, WAVES_PER_BLOCK);
    if (nt_loads)
      hipLaunchKernelGGL((k_dense_margins<TA, TACC, W, true>), dim3(grid),
                         dim3(BLOCK), 0, stream, a, wp, n, d, slab_w, n_slabs,
                         margins);
    else
      hipLaunchKernelGGL((k_dense_margins<TA, TACC, W, false>), dim3(grid),
                         dim3(BLOCK), 0, stream, a, wp, n, d, slab_w, n_slabs,
                         margins);
  }
  if (mode == 1) {  // margins only: reduce slabs in place and return
    if (n_slabs > 1) {
      const int grid = grid_for(n, BLOCK);
      hipLaunchKernelGGL((k_slab_reduce<TACC>), dim3(grid), dim3(BLOCK), 0,
                         stream, margins, n, n_slabs);
    }
    HIP_CHECK(hipGetLastError());
    return 0;
  }
  if (mode != 3) {  // mode 3: the caller provides the multiplier vector
    const int grid = grid_for(n, BLOCK);
    hipLaunchKernelGGL((k_multiplier<TACC>), dim3(grid), dim3(BLOCK), 0, stream,
                       margins, labels, mask, sample_weight, loss_type, n,
                       n_slabs, mult, red_ws);
    hipLaunchKernelGGL((k_reduce_partials<2>), dim3(1), dim3(BLOCK), 0, stream,
                       red_ws, grid, loss_count);
  }
  if (need_grad) {
    {
      const ll cols_per_block = (ll)BLOCK * W;
      const ll n_cs = (d + cols_per_block - 1) / cols_per_block;
      const int grid = grid_for(n_rb * n_cs, 1);
      if (nt_loads)
        hipLaunchKernelGGL((k_dense_grad<TA, TACC, W, true>), dim3(grid),
                           dim3(BLOCK), 0, stream, a, mult, n, d, n_rb, part);
      else
        hipLaunchKernelGGL((k_dense_grad<TA, TACC, W, false>), dim3(grid),
                           dim3(BLOCK), 0, stream, a, mult, n, d, n_rb, part);
    }
    if (n_rb > 1) {
      const int grid = grid_for(d, BLOCK);
      hipLaunchKernelGGL((k_grad_reduce<TACC>), dim3(grid), dim3(BLOCK), 0,
                         stream, part, n_rb, d, grad);
    }
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

// The complete distributed-free part of applySmooth (AGD.scala:192-208) for a
// dense shard: margins -> multiplier/loss -> A^T·m partials -> reduce.
// loss_count[2] must be zeroed by the caller; grad_out is overwritten.
extern "C" int agd_dense_eval(const void* A, int a_dtype, const void* labels,
                              const void* mask, const void* sample_weight,
                              const void* w, long long n,
                              long long d, void* grad_out, void* loss_count,
                              void* margins_ws, void* mult_ws, void* part_ws,
                              long long n_rb, int loss_type, int n_slabs,
                              int need_grad, int margins_algo, int nt_loads,
                              int mode, void* red_ws, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const float* lab = (const float*)labels;
  const unsigned char* msk = (const unsigned char*)mask;
  const float* swt = (const float*)sample_weight;
  double* lc = (double*)loss_count;
  const int W = pick_w(a_dtype, d);
  switch (a_dtype * 10 + (W > 1 ? 1 : 0)) {
    case 1:  return dense_eval_t<ubf16, float, 8>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 0:  return dense_eval_t<ubf16, float, 1>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 11: return dense_eval_t<float, float, 4>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 10: return dense_eval_t<float, float, 1>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 21: return dense_eval_t<double, double, 2>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 20: return dense_eval_t<double, double, 1>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 31: return dense_eval_t<unsigned char, float, 16>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
    case 30: return dense_eval_t<unsigned char, float, 1>(A, lab, msk, swt, w, n, d, grad_out, lc, margins_ws, mult_ws, part_ws, n_rb, loss_type, n_slabs, need_grad, margins_algo, nt_loads, mode, (double*)red_ws, s);
  }
  snprintf(g_err, sizeof(g_err), "agd_dense_eval: bad dtype %d", a_dtype);
  return 2;
}

// CSR applySmooth: margins -> multiplier/loss -> A^T·m.
// Two transpose-gradient paths: atomic scatter over the CSR arrays
// (csc_* == nullptr), or the deterministic CSC gather when the caller
// provides the column-sorted copy (colptr [d+1], cscrow/cscval [nnz]).
// loss_count must be zeroed by the caller; grad_out additionally so on the
// atomic path (it is accumulated there, overwritten on the CSC path).
extern "C" int agd_csr_eval(const void* rowptr, const void* col, const void* val,
                            const void* labels, const void* mask,
                            const void* sample_weight, const void* w,
                            long long n, long long nnz, long long d,
                            void* grad_out, void* loss_count, void* margins_ws,
                            void* mult_ws, int loss_type,
                            const void* csc_colptr, const void* csc_row,
                            const void* csc_val, int need_grad, int mode,
                            void* red_ws, void* stream) {
  (void)nnz;
  hipStream_t s = (hipStream_t)stream;
  const int* rp = (const int*)rowptr;
  const int* ci = (const int*)col;
  const float* v = (const float*)val;
  float* margins = (float*)margins_ws;
  float* mult = (float*)mult_ws;
  if (mode != 2) {
    const int grid = grid_for(n, WAVES_PER_BLOCK);
    hipLaunchKernelGGL(k_csr_margins, dim3(grid), dim3(BLOCK), 0, s, rp, ci, v,
                       (const float*)w, n, margins);
  }
  if (mode == 1) {
    HIP_CHECK(hipGetLastError());
    return 0;
  }
  {
    const int grid = grid_for(n, BLOCK);
    hipLaunchKernelGGL((k_multiplier<float>), dim3(grid), dim3(BLOCK), 0, s,
                       margins, (const float*)labels,
                       (const unsigned char*)mask,
                       (const float*)sample_weight, loss_type, n, 1, mult,
                       (double*)red_ws);
    hipLaunchKernelGGL((k_reduce_partials<2>), dim3(1), dim3(BLOCK), 0, s,
                       (double*)red_ws, grid, (double*)loss_count);
  }
  if (!need_grad) {
    // loss-only evaluation (simple-backtracking f_x trials): skip A^T.m
  } else if (csc_colptr != nullptr) {
    const int grid = grid_for(d, BLOCK);
    hipLaunchKernelGGL(k_csc_grad, dim3(grid), dim3(BLOCK), 0, s,
                       (const int*)csc_colptr, (const int*)csc_row,
                       (const float*)csc_val, mult, d, (float*)grad_out);
  } else {
    const int grid = grid_for(n, WAVES_PER_BLOCK);
    hipLaunchKernelGGL(k_csr_grad, dim3(grid), dim3(BLOCK), 0, s, rp, ci, v,
                       mult, n, (float*)grad_out);
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_gram_mult_affine(const void* xm, const void* zm, double a,
                                    double b, const void* labels,
                                    const void* sample_weight, int loss_type,
                                    long long n, void* mult, void* loss_count,
                                    void* red_ws, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK);
  hipLaunchKernelGGL(k_multiplier_affine, dim3(grid), dim3(BLOCK), 0, s,
                     (const float*)xm, (const float*)zm, a, b,
                     (const float*)labels, (const float*)sample_weight,
                     loss_type, n, (float*)mult, (double*)red_ws);
  hipLaunchKernelGGL((k_reduce_partials<2>), dim3(1), dim3(BLOCK), 0, s,
                     (double*)red_ws, grid, (double*)loss_count);
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_gram_state_update(const void* gm_raw, const void* m_y,
                                     const void* xm_old, const void* zm_old,
                                     double inv_c, double theta, double pz,
                                     double pg, long long n, void* xb_t,
                                     void* md, void* mstore_t, void* zm_new,
                                     void* xm_new, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK);
  hipLaunchKernelGGL(k_gram_state_update, dim3(grid), dim3(BLOCK), 0, s,
                     (const float*)gm_raw, (const float*)m_y,
                     (const float*)xm_old, (const float*)zm_old, inv_c, theta,
                     pz, pg, n, (double*)xb_t, (double*)md, (float*)mstore_t,
                     (float*)zm_new, (float*)xm_new);
  HIP_CHECK(hipGetLastError());
  return 0;
}

// Skew-robust deterministic CSC gradient (see k_csc_grad_light above).
// Callers compute mult first (agd_csr_eval with need_grad=0), then invoke
// this with the heavy-column task structure built at shard construction.
extern "C" int agd_csc_grad_skew(const void* colptr, const void* row,
                                 const void* val, const void* mult,
                                 long long d, int heavy_T,
                                 const void* heavy_cols, const void* taskptr,
                                 const void* task_heavy_idx,
                                 long long n_heavy, long long n_tasks, int S,
                                 void* partial, const void* light_order,
                                 void* grad, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  {
    const int grid = grid_for(d, BLOCK);
    hipLaunchKernelGGL(k_csc_grad_light, dim3(grid), dim3(BLOCK), 0, s,
                       (const int*)colptr, (const int*)row, (const float*)val,
                       (const float*)mult, d, heavy_T,
                       (const int*)light_order, (float*)grad);
  }
  if (n_tasks > 0) {
    const int grid = grid_for(n_tasks, WAVES_PER_BLOCK);
    hipLaunchKernelGGL(k_csc_heavy_partial, dim3(grid), dim3(BLOCK), 0, s,
                       (const int*)colptr, (const int*)row, (const float*)val,
                       (const float*)mult, (const int*)heavy_cols,
                       (const int*)taskptr, (const int*)task_heavy_idx,
                       n_tasks, S, (float*)partial);
    const int grid2 = grid_for(n_heavy, WAVES_PER_BLOCK);
    hipLaunchKernelGGL(k_csc_heavy_combine, dim3(grid2), dim3(BLOCK), 0, s,
                       (const int*)heavy_cols, (const int*)taskptr,
                       (const float*)partial, n_heavy, (float*)grad);
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

// ---------------------------------------------------------------------------
// Library GEMM for the Gram-operator build (the one true GEMM in this
// workload): C[m,n] (f32, row-major) = A[m,k] (bf16 rm) · B[n,k]^T (bf16 rm),
// fp32 accumulation via hipBLASLt (bf16 in / f32 out — not expressible
// through torch.matmul, which rounds the output to bf16). Row-major is
// mapped to hipBLASLt's column-major as C'[n,m] = B'^T(k x n) · A'(k x m).
// ---------------------------------------------------------------------------

#define LT_CHECK(expr)                                                         \
  do {                                                                         \
    hipblasStatus_t _s = (expr);                                               \
    if (_s != HIPBLAS_STATUS_SUCCESS) {                                        \
      snprintf(g_err, sizeof(g_err), "%s:%d %s: hipblaslt status %d",          \
               __FILE__, __LINE__, #expr, (int)_s);                            \
      return 3;                                                                \
    }                                                                          \
  } while (0)

static hipblasLtHandle_t g_lt_handle = nullptr;
static void* g_lt_ws = nullptr;
static const size_t LT_WS_BYTES = 64u << 20;

// Measured per-shape algo autotune (round 2): the single heuristic
// candidate left the K=1000 margins GEMM at ~21% MFMA / ~1 TB/s — neither
// bound. One-time per (shape, layout-kind): request up to 16 candidates,
// time each with events on the caller's stream, cache the winner. Runs
// during warmup (first call per shape); only when beta == 0, so repeated
// tuning launches cannot corrupt an accumulating C.
struct LtAlgoEntry {
  long long m, n, k;
  int kind;
  hipblasLtMatmulAlgo_t algo;
};
static LtAlgoEntry g_lt_algos[64];
static int g_lt_algo_count = 0;

static int lt_pick_algo(int kind, hipblasLtMatmulDesc_t op,
                        hipblasLtMatrixLayout_t la, hipblasLtMatrixLayout_t lb,
                        hipblasLtMatrixLayout_t lc, const void* Amat,
                        const void* Bmat, void* Cmat, float beta, long long m,
                        long long n, long long k, hipStream_t s,
                        hipblasLtMatmulAlgo_t* out) {
  for (int i = 0; i < g_lt_algo_count; ++i)
    if (g_lt_algos[i].kind == kind && g_lt_algos[i].m == m &&
        g_lt_algos[i].n == n && g_lt_algos[i].k == k) {
      *out = g_lt_algos[i].algo;
      return 0;
    }
  hipblasLtMatmulPreference_t pref = nullptr;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &LT_WS_BYTES,
      sizeof(LT_WS_BYTES)));
  hipblasLtMatmulHeuristicResult_t heur[16];
  int found = 0;
  LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(g_lt_handle, op, la, lb, lc, lc,
                                           pref, 16, heur, &found));
  hipblasLtMatmulPreferenceDestroy(pref);
  if (found < 1) {
    snprintf(g_err, sizeof(g_err), "hipblaslt: no algo for kind %d %lldx%lldx%lld",
             kind, m, n, k);
    return 3;
  }
  int best = 0;
  // Tune only shapes whose per-call cost is small: for multi-TFLOP GEMMs
  // (e.g. the one-time Gram K build at ~1.7e13 FLOPs/chunk) the 3x16
  // tuning launches cost seconds while the heuristic is already near
  // roofline — measured 0.46 s -> 2.2 s build regression before this cap.
  // Multi-rank runs (WORLD_SIZE > 1) take the deterministic heuristic[0]
  // instead: a timing-based pick could differ across ranks, and the
  // replicated-update design requires bit-identical math on every rank.
  // SPARKAGD_GEMM_TUNE=0/1 overrides.
  static int tune_enabled = -1;
  if (tune_enabled < 0) {
    const char* e = getenv("SPARKAGD_GEMM_TUNE");
    if (e != nullptr)
      tune_enabled = (e[0] != '0');
    else {
      const char* ws = getenv("WORLD_SIZE");
      tune_enabled = !(ws != nullptr && atoll(ws) > 1);
    }
  }
  const double tune_flops = 2.0 * (double)m * (double)n * (double)k;
  if (tune_enabled && found > 1 && beta == 0.0f && tune_flops < 1.0e12) {
    const float alpha = 1.0f;
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    float best_ms = 1e30f;
    for (int a = 0; a < found; ++a) {
      // warm once, then time 2 reps
      if (hipblasLtMatmul(g_lt_handle, op, &alpha, Bmat, la, Amat, lb, &beta,
                          Cmat, lc, Cmat, lc, &heur[a].algo, g_lt_ws,
                          LT_WS_BYTES, s) != HIPBLAS_STATUS_SUCCESS)
        continue;
      HIP_CHECK(hipEventRecord(e0, s));
      for (int r = 0; r < 2; ++r)
        (void)hipblasLtMatmul(g_lt_handle, op, &alpha, Bmat, la, Amat, lb,
                              &beta, Cmat, lc, Cmat, lc, &heur[a].algo,
                              g_lt_ws, LT_WS_BYTES, s);
      HIP_CHECK(hipEventRecord(e1, s));
      HIP_CHECK(hipEventSynchronize(e1));
      float ms = 1e30f;
      HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
      if (ms < best_ms) {
        best_ms = ms;
        best = a;
      }
    }
    HIP_CHECK(hipEventDestroy(e0));
    HIP_CHECK(hipEventDestroy(e1));
  }
  *out = heur[best].algo;
  if (g_lt_algo_count < 64)
    g_lt_algos[g_lt_algo_count++] = {m, n, k, kind, heur[best].algo};
  return 0;
}

extern "C" int agd_gemm_bf16f32_nt(const void* A, const void* B, void* C,
                                   long long m, long long n, long long k,
                                   float beta, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  if (g_lt_handle == nullptr) {
    LT_CHECK(hipblasLtCreate(&g_lt_handle));
    HIP_CHECK(hipMalloc(&g_lt_ws, LT_WS_BYTES));
  }
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t la = nullptr, lb = nullptr, lc = nullptr;
  LT_CHECK(hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  const hipblasOperation_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opT, sizeof(opT)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opN, sizeof(opN)));
  // col-major views: "A" = B' (k x n, ld k) transposed; "B" = A' (k x m, ld k)
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, k, n, k));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, k, m, k));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_32F, n, m, n));
  const float alpha = 1.0f;

  hipblasLtMatmulAlgo_t algo;
  {
    const int rc = lt_pick_algo(0, op, la, lb, lc, A, B, C, beta, m, n, k, s,
                                &algo);
    if (rc != 0) return rc;
  }
  LT_CHECK(hipblasLtMatmul(g_lt_handle, op, &alpha, B, la, A, lb, &beta, C, lc,
                           C, lc, &algo, g_lt_ws, LT_WS_BYTES, s));
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatmulDescDestroy(op);
  HIP_CHECK(hipGetLastError());
  return 0;
}

// C[d,kc] (f32, row-major) = A[n,d]^T (bf16, row-major) @ M[n,kc] (bf16,
// row-major) — the multinomial gradient as a skinny TN GEMM. Col-major
// mapping: C' [kc,d] = M' [kc,n] (opN, M viewed col-major) · A'^T [n,d]
// (opT on A' = A viewed col-major [d,n]).
extern "C" int agd_gemm_bf16f32_tn(const void* A, const void* M, void* C,
                                   long long n, long long d, long long kc,
                                   void* stream) {
  hipStream_t s = (hipStream_t)stream;
  if (g_lt_handle == nullptr) {
    LT_CHECK(hipblasLtCreate(&g_lt_handle));
    HIP_CHECK(hipMalloc(&g_lt_ws, LT_WS_BYTES));
  }
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t la = nullptr, lb = nullptr, lc = nullptr;
  LT_CHECK(hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  const hipblasOperation_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opN, sizeof(opN)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opT, sizeof(opT)));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, kc, n, kc));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, d, n, d));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_32F, kc, d, kc));
  const float alpha = 1.0f, beta = 0.0f;

  hipblasLtMatmulAlgo_t algo;
  {
    const int rc = lt_pick_algo(1, op, la, lb, lc, A, M, C, beta, d, kc, n, s,
                                &algo);
    if (rc != 0) return rc;
  }
  LT_CHECK(hipblasLtMatmul(g_lt_handle, op, &alpha, M, la, A, lb, &beta, C, lc,
                           C, lc, &algo, g_lt_ws, LT_WS_BYTES, s));
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatmulDescDestroy(op);
  HIP_CHECK(hipGetLastError());
  return 0;
}

// ---------------------------------------------------------------------------
// Multinomial host entry points. KC in {4, 8, 16, 32} (K <= 32; larger K is
// rejected — BACKLOG.md). part workspace: n_rb * d * KC floats.
// ---------------------------------------------------------------------------

extern "C" long long agd_multi_rowblocks(long long n, long long d, int kc) {
  const ll n_cs = (d + BLOCK - 1) / BLOCK;
  ll n_rb = 4096 / n_cs;
  if (n_rb < 1) n_rb = 1;
  if (n_rb > n) n_rb = n;
  while (n_rb > 1 && n_rb * d * kc * 4 > (1LL << 30)) n_rb /= 2;
  return n_rb;
}

template <typename TA, int W>
static int margins_multi_t(const void* A, const float* Wm, ll n, ll d, int kc,
                           float* Z, hipStream_t s) {
  const int grid = grid_for(n, WAVES_PER_BLOCK);
#define LAUNCH_MM(KCV)                                                        \
  hipLaunchKernelGGL((k_margins_multi<TA, W, KCV>), dim3(grid), dim3(BLOCK),  \
                     0, s, (const TA*)A, Wm, n, d, Z)
  switch (kc) {
    case 4: LAUNCH_MM(4); break;
    case 8: LAUNCH_MM(8); break;
    case 16: LAUNCH_MM(16); break;
    case 32: LAUNCH_MM(32); break;
    default:
      snprintf(g_err, sizeof(g_err), "margins_multi: bad KC %d", kc);
      return 2;
  }
#undef LAUNCH_MM
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_margins_multi(const void* A, int a_dtype, const void* Wm,
                                 long long n, long long d, int kc, void* Z,
                                 void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const float* wp = (const float*)Wm;
  float* z = (float*)Z;
  const int W = pick_w(a_dtype, d);
  if (a_dtype == 0 && W == 8) return margins_multi_t<ubf16, 8>(A, wp, n, d, kc, z, s);
  if (a_dtype == 0) return margins_multi_t<ubf16, 1>(A, wp, n, d, kc, z, s);
  if (a_dtype == 1 && W == 4) return margins_multi_t<float, 4>(A, wp, n, d, kc, z, s);
  if (a_dtype == 1) return margins_multi_t<float, 1>(A, wp, n, d, kc, z, s);
  if (a_dtype == 3 && W == 16) return margins_multi_t<unsigned char, 16>(A, wp, n, d, kc, z, s);
  if (a_dtype == 3) return margins_multi_t<unsigned char, 1>(A, wp, n, d, kc, z, s);
  snprintf(g_err, sizeof(g_err), "margins_multi: dtype %d unsupported", a_dtype);
  return 2;
}

extern "C" int agd_multiplier_multi(const void* Z, const void* labels,
                                    const void* mask, const void* sample_weight,
                                    long long n, int k, int kc, void* M,
                                    void* loss_count, void* red_ws,
                                    void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK);
#define LAUNCH_MU(KCV)                                                        \
  hipLaunchKernelGGL((k_multiplier_multi<KCV>), dim3(grid), dim3(BLOCK), 0,   \
                     s, (const float*)Z, (const float*)labels,                \
                     (const unsigned char*)mask, (const float*)sample_weight, \
                     n, k, (float*)M, (double*)red_ws)
  int launched_grid = grid;
  switch (kc) {
    case 4: LAUNCH_MU(4); break;
    case 8: LAUNCH_MU(8); break;
    case 16: LAUNCH_MU(16); break;
    case 32: LAUNCH_MU(32); break;
    default: {
      if (kc < 4 || (kc & 3)) {
        snprintf(g_err, sizeof(g_err), "multiplier_multi: bad KC %d", kc);
        return 2;
      }
      // generic K > 32: wave per row, lanes stride classes
      launched_grid = grid_for(n, WAVES_PER_BLOCK);
      hipLaunchKernelGGL(k_multiplier_multi_anyk, dim3(launched_grid),
                         dim3(BLOCK), 0, s, (const float*)Z,
                         (const float*)labels, (const unsigned char*)mask,
                         (const float*)sample_weight, n, k, kc, (float*)M,
                         (double*)red_ws);
      break;
    }
  }
#undef LAUNCH_MU
  hipLaunchKernelGGL((k_reduce_partials<2>), dim3(1), dim3(BLOCK), 0, s,
                     (double*)red_ws, launched_grid, (double*)loss_count);
  HIP_CHECK(hipGetLastError());
  return 0;
}

template <typename TA>
static int grad_multi_t(const void* A, const float* M, ll n, ll d, int kc,
                        float* part, ll n_rb, float* grad, hipStream_t s) {
  const ll n_cs = (d + BLOCK - 1) / BLOCK;
  const int grid = grid_for(n_rb * n_cs, 1);
#define LAUNCH_GM(KCV)                                                        \
  hipLaunchKernelGGL((k_grad_multi<TA, KCV>), dim3(grid), dim3(BLOCK), 0, s,  \
                     (const TA*)A, M, n, d, n_rb, part)
  switch (kc) {
    case 4: LAUNCH_GM(4); break;
    case 8: LAUNCH_GM(8); break;
    case 16: LAUNCH_GM(16); break;
    case 32: LAUNCH_GM(32); break;
    default:
      snprintf(g_err, sizeof(g_err), "grad_multi: bad KC %d", kc);
      return 2;
  }
#undef LAUNCH_GM
  if (n_rb > 1) {
    const int rgrid = grid_for(d * kc, BLOCK);
    hipLaunchKernelGGL((k_grad_reduce<float>), dim3(rgrid), dim3(BLOCK), 0, s,
                       part, n_rb, d * kc, grad);
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_grad_multi(const void* A, int a_dtype, const void* M,
                              long long n, long long d, int kc, void* part,
                              long long n_rb, void* grad, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  float* p = (n_rb == 1) ? (float*)grad : (float*)part;
  if (a_dtype == 0) return grad_multi_t<ubf16>(A, (const float*)M, n, d, kc, p, n_rb, (float*)grad, s);
  if (a_dtype == 1) return grad_multi_t<float>(A, (const float*)M, n, d, kc, p, n_rb, (float*)grad, s);
  if (a_dtype == 3) return grad_multi_t<unsigned char>(A, (const float*)M, n, d, kc, p, n_rb, (float*)grad, s);
  snprintf(g_err, sizeof(g_err), "grad_multi: dtype %d unsupported", a_dtype);
  return 2;
}

// ---------------------------------------------------------------------------
// Multinomial on CSR shards. Z[n,KC] = A·W and grad[d,KC] = Aᵀ·M where the
// per-nonzero work is KC fmas on a 16-B-aligned KC-float row of W (margins)
// or M (gradient) — the gathered lines carry KC useful floats, so the
// per-line efficiency is KC/1 versus the binary kernels' single float.
// One THREAD per row/column (no cross-lane reduction; acc[KC] in registers);
// the CSC gather keeps the gradient atomics-free => bitwise deterministic.
// ---------------------------------------------------------------------------

// Gather one KC-wide W row into f32 registers. TW=float: KC/4 x 16-B loads.
// TW=ubf16: half the gathered bytes per nonzero AND double the effective
// LLC coverage of W (the margins gather is LLC-miss-bound once W [d,KC]
// exceeds the last-level cache — profiles/r01_csr_multiclass_trace.txt).
template <typename TW, int KC>
__device__ __forceinline__ void load_w_row(const TW* __restrict__ wr,
                                           float (&out)[KC]) {
  if constexpr (sizeof(TW) == 4) {
    using f32x4 = __attribute__((ext_vector_type(4))) float;
#pragma unroll
    for (int ch = 0; ch < KC / 4; ++ch) {
      const f32x4 wv = *(const f32x4*)((const float*)wr + ch * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) out[ch * 4 + j] = wv[j];
    }
  } else if constexpr (KC >= 8) {  // bf16, 16-B chunks of 8
    using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;
#pragma unroll
    for (int ch = 0; ch < KC / 8; ++ch) {
      const u16x8 wv = *(const u16x8*)((const ubf16*)wr + ch * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) out[ch * 8 + j] = bf2f((ubf16)wv[j]);
    }
  } else {  // bf16 KC == 4: one 8-B chunk
    using u16x4 = __attribute__((ext_vector_type(4))) unsigned short;
    const u16x4 wv = *(const u16x4*)wr;
#pragma unroll
    for (int j = 0; j < 4; ++j) out[j] = bf2f((ubf16)wv[j]);
  }
}

template <typename TW>
__device__ __forceinline__ float w_elem(TW v);
template <>
__device__ __forceinline__ float w_elem<float>(float v) { return v; }
template <>
__device__ __forceinline__ float w_elem<ubf16>(ubf16 v) { return bf2f(v); }

// Lane-per-class CSR multiclass margins (round 2): the wave splits into
// WAVE/KC sub-groups, one row each; within a sub-group lane j owns class j,
// so the KC lanes of a gathered W row read CONSECUTIVE floats — every 64-B
// W-row gather is a fully coalesced line and no cross-lane reduction is
// needed (each lane owns its output). The previous thread-per-row layout
// had one thread serially streaming whole W rows: measured 3.28 ms/pass at
// K=16, d=1e7, 64M nnz (~1.2 TB/s effective) vs this layout's coalesced
// line service.
template <typename TW, int KC>
__global__ __launch_bounds__(BLOCK) void k_csr_margins_multi(
    const int* __restrict__ rowptr, const int* __restrict__ col,
    const float* __restrict__ val, const TW* __restrict__ w, ll n,
    float* __restrict__ Z) {
  constexpr int RPW = WAVE / KC;  // rows (sub-groups) per wave
  const int lane = threadIdx.x & (WAVE - 1);
  const int sub = lane / KC;
  const int cls = lane - sub * KC;
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll rbase = wave_gid * RPW; rbase < n; rbase += n_waves * RPW) {
    const ll r = rbase + sub;
    if (r >= n) continue;
    const int k_lo = rowptr[r], k_hi = rowptr[r + 1];
    // 4 accumulators: independent W-line gathers in flight per sub-group
    // (the serial per-row loop is gather-latency-exposed; measured
    // 6.22 -> 5.72 ms/step at 2 accumulators already)
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    int k = k_lo;
    for (; k + 3 < k_hi; k += 4) {
      a0 += val[k] * w_elem<TW>(w[(ll)col[k] * KC + cls]);
      a1 += val[k + 1] * w_elem<TW>(w[(ll)col[k + 1] * KC + cls]);
      a2 += val[k + 2] * w_elem<TW>(w[(ll)col[k + 2] * KC + cls]);
      a3 += val[k + 3] * w_elem<TW>(w[(ll)col[k + 3] * KC + cls]);
    }
    for (; k < k_hi; ++k)
      a0 += val[k] * w_elem<TW>(w[(ll)col[k] * KC + cls]);
    Z[r * KC + cls] = (a0 + a1) + (a2 + a3);
  }
}

// Multiclass analogs of the skew-robust split (see k_csc_grad_light):
// wave per task with KC accumulators, partials combined in task order.
template <int KC>
__global__ __launch_bounds__(BLOCK) void k_csc_heavy_partial_multi(
    const int* __restrict__ colptr, const int* __restrict__ row,
    const float* __restrict__ val, const float* __restrict__ M,
    const int* __restrict__ heavy_cols, const int* __restrict__ taskptr,
    const int* __restrict__ task_heavy_idx, ll n_tasks, int S,
    float* __restrict__ partial) {
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll t = wave_gid; t < n_tasks; t += n_waves) {
    const int i = task_heavy_idx[t];
    const int c = heavy_cols[i];
    const int k_lo = colptr[c] + (int)(t - taskptr[i]) * S;
    const int k_hi = min(k_lo + S, colptr[c + 1]);
    float acc[KC];
#pragma unroll
    for (int j = 0; j < KC; ++j) acc[j] = 0.f;
    for (int k = k_lo + lane; k < k_hi; k += WAVE) {
      const float v = val[k];
      const float* __restrict__ mr = M + (ll)row[k] * KC;
#pragma unroll
      for (int ch = 0; ch < KC / 4; ++ch) {
        const f32x4 mv = *(const f32x4*)(mr + ch * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[ch * 4 + j] += v * mv[j];
      }
    }
    float* __restrict__ dst = partial + t * KC;
#pragma unroll
    for (int j = 0; j < KC; ++j) {
      const float s = wave_reduce_sum(acc[j]);
      if (lane == 0) dst[j] = s;
    }
  }
}

template <int KC>
__global__ __launch_bounds__(BLOCK) void k_csc_heavy_combine_multi(
    const int* __restrict__ heavy_cols, const int* __restrict__ taskptr,
    const float* __restrict__ partial, ll n_heavy, float* __restrict__ grad) {
  // wave per heavy column, lanes stride tasks (fixed order — deterministic)
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const ll wave_gid = (ll)blockIdx.x * WAVES_PER_BLOCK + wid;
  const ll n_waves = (ll)gridDim.x * WAVES_PER_BLOCK;
  for (ll i = wave_gid; i < n_heavy; i += n_waves) {
    const int t_lo = taskptr[i], t_hi = taskptr[i + 1];
    float acc[KC];
#pragma unroll
    for (int j = 0; j < KC; ++j) acc[j] = 0.f;
    for (int t = t_lo + lane; t < t_hi; t += WAVE)
#pragma unroll
      for (int j = 0; j < KC; ++j) acc[j] += partial[(ll)t * KC + j];
    float* __restrict__ gr = grad + (ll)heavy_cols[i] * KC;
#pragma unroll
    for (int j = 0; j < KC; ++j) {
      const float s = wave_reduce_sum(acc[j]);
      if (lane == 0) gr[j] = s;
    }
  }
}

// CSC multiclass gradient, thread-per-column with 16-B M loads/grad
// stores. A lane-per-class variant (like k_csr_margins_multi) was measured
// and REJECTED here: the random-gather target M is n*KC ~ 64 MB and
// LLC-resident, so coalescing its rows buys nothing, while thread-per-
// column keeps adjacent threads streaming ADJACENT csc val/row windows
// (1.46 vs 2.10 ms/pass at K=16, d=1e7 — profiles/r02_csr_multi notes).
template <int KC>
__global__ __launch_bounds__(BLOCK) void k_csc_grad_multi(
    const int* __restrict__ colptr, const int* __restrict__ row,
    const float* __restrict__ val, const float* __restrict__ M, ll d,
    int heavy_T, const int* __restrict__ order, float* __restrict__ grad) {
  using f32x4 = __attribute__((ext_vector_type(4))) float;
  const ll stride = (ll)gridDim.x * BLOCK;
  for (ll i = (ll)blockIdx.x * BLOCK + threadIdx.x; i < d; i += stride) {
    const ll c = order ? (ll)order[i] : i;
    const int k_lo = colptr[c], k_hi = colptr[c + 1];
    if (k_hi - k_lo > heavy_T) continue;  // heavy path owns grad[c]
    float acc[KC];
#pragma unroll
    for (int j = 0; j < KC; ++j) acc[j] = 0.f;
    for (int k = k_lo; k < k_hi; ++k) {
      const float v = val[k];
      const float* __restrict__ mr = M + (ll)row[k] * KC;
#pragma unroll
      for (int ch = 0; ch < KC / 4; ++ch) {
        const f32x4 mv = *(const f32x4*)(mr + ch * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[ch * 4 + j] += v * mv[j];
      }
    }
    float* __restrict__ gr = grad + c * KC;
#pragma unroll
    for (int ch = 0; ch < KC / 4; ++ch) {
      f32x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = acc[ch * 4 + j];
      *(f32x4*)(gr + ch * 4) = o;
    }
  }
}

// w_dtype: 1 = f32 rows, 0 = bf16 rows (half the gather bytes).
extern "C" int agd_csr_margins_multi(const void* rowptr, const void* col,
                                     const void* val, const void* w,
                                     long long n, int kc, int w_dtype, void* Z,
                                     void* stream) {
  hipStream_t s = (hipStream_t)stream;
  // lane-per-class: n rows x KC lanes of work
  const int grid = grid_for((ll)n * kc, BLOCK);
#define LAUNCH_CM(TW, KCV)                                                    \
  hipLaunchKernelGGL((k_csr_margins_multi<TW, KCV>), dim3(grid), dim3(BLOCK), \
                     0, s, (const int*)rowptr, (const int*)col,               \
                     (const float*)val, (const TW*)w, n, (float*)Z)
  if (w_dtype == 1) {
    switch (kc) {
      case 4: LAUNCH_CM(float, 4); break;
      case 8: LAUNCH_CM(float, 8); break;
      case 16: LAUNCH_CM(float, 16); break;
      case 32: LAUNCH_CM(float, 32); break;
      default:
        snprintf(g_err, sizeof(g_err), "csr_margins_multi: bad KC %d", kc);
        return 2;
    }
  } else {
    switch (kc) {
      case 4: LAUNCH_CM(ubf16, 4); break;
      case 8: LAUNCH_CM(ubf16, 8); break;
      case 16: LAUNCH_CM(ubf16, 16); break;
      case 32: LAUNCH_CM(ubf16, 32); break;
      default:
        snprintf(g_err, sizeof(g_err), "csr_margins_multi: bad KC %d", kc);
        return 2;
    }
  }
#undef LAUNCH_CM
  HIP_CHECK(hipGetLastError());
  return 0;
}

// Heavy-column args mirror agd_csc_grad_skew; n_tasks == 0 (or null heavy
// pointers) keeps the plain thread-per-column path for every column.
extern "C" int agd_csc_grad_multi(const void* colptr, const void* row,
                                  const void* val, const void* M, long long d,
                                  int kc, void* grad, int heavy_T,
                                  const void* heavy_cols, const void* taskptr,
                                  const void* task_heavy_idx,
                                  long long n_heavy, long long n_tasks, int S,
                                  void* partial, const void* light_order,
                                  void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(d, BLOCK);  // thread per column
  const int light_T = (n_tasks > 0) ? heavy_T : 0x7fffffff;
#define LAUNCH_CG(KCV)                                                         \
  do {                                                                         \
    hipLaunchKernelGGL((k_csc_grad_multi<KCV>), dim3(grid), dim3(BLOCK), 0, s, \
                       (const int*)colptr, (const int*)row, (const float*)val, \
                       (const float*)M, d, light_T,                            \
                       (const int*)light_order, (float*)grad);                 \
    if (n_tasks > 0) {                                                         \
      const int g1 = grid_for(n_tasks, WAVES_PER_BLOCK);                       \
      hipLaunchKernelGGL((k_csc_heavy_partial_multi<KCV>), dim3(g1),           \
                         dim3(BLOCK), 0, s, (const int*)colptr,                \
                         (const int*)row, (const float*)val, (const float*)M,  \
                         (const int*)heavy_cols, (const int*)taskptr,          \
                         (const int*)task_heavy_idx, n_tasks, S,               \
                         (float*)partial);                                     \
      const int g2 = grid_for(n_heavy, WAVES_PER_BLOCK);                       \
      hipLaunchKernelGGL((k_csc_heavy_combine_multi<KCV>), dim3(g2),           \
                         dim3(BLOCK), 0, s, (const int*)heavy_cols,            \
                         (const int*)taskptr, (const float*)partial, n_heavy,  \
                         (float*)grad);                                        \
    }                                                                          \
  } while (0)
  switch (kc) {
    case 4: LAUNCH_CG(4); break;
    case 8: LAUNCH_CG(8); break;
    case 16: LAUNCH_CG(16); break;
    case 32: LAUNCH_CG(32); break;
    default:
      snprintf(g_err, sizeof(g_err), "csc_grad_multi: bad KC %d", kc);
      return 2;
  }
#undef LAUNCH_CG
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_at_margin_update(const void* zm_old, const void* xm_old,
                                    const void* gm, double pz, double pg,
                                    double theta, long long n, int dtype,
                                    void* zm_new, void* xm_new, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n / (dtype == 2 ? 2 : 4) + 1, BLOCK);
  if (dtype == 1)
    hipLaunchKernelGGL((k_at_margin_update<float>), dim3(grid), dim3(BLOCK),
                       0, s, (const float*)zm_old, (const float*)xm_old,
                       (const float*)gm, pz, pg, theta, n, (float*)zm_new,
                       (float*)xm_new);
  else if (dtype == 2)
    hipLaunchKernelGGL((k_at_margin_update<double>), dim3(grid), dim3(BLOCK),
                       0, s, (const double*)zm_old, (const double*)xm_old,
                       (const double*)gm, pz, pg, theta, n, (double*)zm_new,
                       (double*)xm_new);
  else {
    snprintf(g_err, sizeof(g_err), "at_margin_update: bad dtype %d", dtype);
    return 2;
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

extern "C" int agd_axpby(double a, const void* x, double b, const void* y,
                         void* out, long long n, int dtype, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK * 4);
  if (dtype == 1)
    hipLaunchKernelGGL((k_axpby<float>), dim3(grid), dim3(BLOCK), 0, s, a,
                       (const float*)x, b, (const float*)y, (float*)out, n);
  else if (dtype == 2)
    hipLaunchKernelGGL((k_axpby<double>), dim3(grid), dim3(BLOCK), 0, s, a,
                       (const double*)x, b, (const double*)y, (double*)out, n);
  else {
    snprintf(g_err, sizeof(g_err), "agd_axpby: bad dtype %d", dtype);
    return 2;
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

// reg (double[1]) must be zeroed by the caller.
extern "C" int agd_prox(int kind, const void* w, const void* g, double step,
                        double lam, double lam2, void* out, void* reg,
                        long long n, int dtype, void* red_ws, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK * 4);
  if (dtype == 1)
    hipLaunchKernelGGL((k_prox<float>), dim3(grid), dim3(BLOCK), 0, s, kind,
                       (const float*)w, (const float*)g, step, lam, lam2,
                       (float*)out, (double*)red_ws, n);
  else if (dtype == 2)
    hipLaunchKernelGGL((k_prox<double>), dim3(grid), dim3(BLOCK), 0, s, kind,
                       (const double*)w, (const double*)g, step, lam, lam2,
                       (double*)out, (double*)red_ws, n);
  else {
    snprintf(g_err, sizeof(g_err), "agd_prox: bad dtype %d", dtype);
    return 2;
  }
  if (kind != PROX_SIMPLE)
    hipLaunchKernelGGL((k_reduce_partials<1>), dim3(1), dim3(BLOCK), 0, s,
                       (double*)red_ws, grid, (double*)reg);
  HIP_CHECK(hipGetLastError());
  return 0;
}

// out (double[5]) must be zeroed by the caller.
extern "C" int agd_fused_scalars(const void* x, const void* y, const void* gy,
                                 const void* xold, void* out, long long n,
                                 int dtype, void* red_ws, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK * 4);
  if (dtype == 1)
    hipLaunchKernelGGL((k_fused_scalars<float>), dim3(grid), dim3(BLOCK), 0, s,
                       (const float*)x, (const float*)y, (const float*)gy,
                       (const float*)xold, (double*)red_ws, n);
  else if (dtype == 2)
    hipLaunchKernelGGL((k_fused_scalars<double>), dim3(grid), dim3(BLOCK), 0, s,
                       (const double*)x, (const double*)y, (const double*)gy,
                       (const double*)xold, (double*)red_ws, n);
  else {
    snprintf(g_err, sizeof(g_err), "agd_fused_scalars: bad dtype %d", dtype);
    return 2;
  }
  hipLaunchKernelGGL((k_reduce_partials<5>), dim3(1), dim3(BLOCK), 0, s,
                     (double*)red_ws, grid, (double*)out);
  HIP_CHECK(hipGetLastError());
  return 0;
}

// out (double[1]) must be zeroed by the caller.
extern "C" int agd_dot_diff(const void* x, const void* y, const void* gx,
                            const void* gy, void* out, long long n, int dtype,
                            void* red_ws, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = grid_for(n, BLOCK * 4);
  if (dtype == 1)
    hipLaunchKernelGGL((k_dot_diff<float>), dim3(grid), dim3(BLOCK), 0, s,
                       (const float*)x, (const float*)y, (const float*)gx,
                       (const float*)gy, (double*)red_ws, n);
  else if (dtype == 2)
    hipLaunchKernelGGL((k_dot_diff<double>), dim3(grid), dim3(BLOCK), 0, s,
                       (const double*)x, (const double*)y, (const double*)gx,
                       (const double*)gy, (double*)red_ws, n);
  else {
    snprintf(g_err, sizeof(g_err), "agd_dot_diff: bad dtype %d", dtype);
    return 2;
  }
  hipLaunchKernelGGL((k_reduce_partials<1>), dim3(1), dim3(BLOCK), 0, s,
                     (double*)red_ws, grid, (double*)out);
  HIP_CHECK(hipGetLastError());
  return 0;
}
