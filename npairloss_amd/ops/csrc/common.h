// Shared device helpers for the npairloss_amd gfx950 kernels.
//
// CDNA4 notes: wavefront is 64 lanes; block size 256 = 4 waves everywhere;
// cross-lane reductions use __shfl_xor over the full wave then one LDS
// round across the 4 waves.
#pragma once

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#define WAVE 64
#define NPAIR_BLOCK 256
#define DEVINL __device__ __forceinline__

// mining method codes — must match config.params.MiningMethod
#define M_HARD 0
#define M_EASY 1
#define M_RAND 2
#define M_RELATIVE_HARD 3
#define M_RELATIVE_EASY 4

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEVINL double wave_reduce_sum(double v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEVINL float wave_reduce_min(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fminf(v, __shfl_xor(v, off, WAVE));
  return v;
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

DEVINL int wave_reduce_sum(int v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// Block-level reductions for 256-thread blocks; `scratch` is >= 4 elements
// of LDS (one slot per wave).
template <typename T, typename Op>
DEVINL T block_reduce(T v, Op op, T init, T* scratch) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  T r = (lane < NPAIR_BLOCK / WAVE && wid == 0) ? scratch[lane] : init;
  if (wid == 0) {
#pragma unroll
    for (int off = 2; off > 0; off >>= 1) r = op(r, __shfl_xor(r, off, WAVE));
  }
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  T out = scratch[0];
  __syncthreads();
  return out;
}

struct OpAddF { DEVINL float operator()(float a, float b) const { return a + b; } };
struct OpAddD { DEVINL double operator()(double a, double b) const { return a + b; } };
struct OpAddI { DEVINL int operator()(int a, int b) const { return a + b; } };
struct OpMinF { DEVINL float operator()(float a, float b) const { return fminf(a, b); } };
struct OpMaxF { DEVINL float operator()(float a, float b) const { return fmaxf(a, b); } };
// dtype-generic variants (fp64 path — reference Dtype dispatch, .cu:31-42)
struct OpMinT { template <typename T> DEVINL T operator()(T a, T b) const { return a < b ? a : b; } };
struct OpMaxT { template <typename T> DEVINL T operator()(T a, T b) const { return a > b ? a : b; } };

template <typename T> struct DtMax;
template <> struct DtMax<float> { static constexpr float v = FLT_MAX; };
template <> struct DtMax<double> { static constexpr double v = DBL_MAX; };

// exp in the storage dtype: fast hardware exp for fp32, libm for fp64
DEVINL float kexp(float x) { return __expf(x); }
DEVINL double kexp(double x) { return exp(x); }

// Pair classification: self pair (rank*B + i == j) belongs to NEITHER set
// (reference GetLabelDiffMtx, .cu:54).
DEVINL bool pair_is_self(int i, int j, int rank, int B) { return rank * B + i == j; }

// float -> uint32 key preserving order (ascending float == ascending key)
DEVINL uint32_t float_to_key(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

DEVINL float key_to_float(uint32_t k) {
  uint32_t u = (k & 0x80000000u) ? (k ^ 0x80000000u) : ~k;
  return __uint_as_float(u);
}

// Pair-selection rule (reference GetSampledPairMtx, .cu:69-122).
// thr already includes the margin.  Templated over float/double (the
// reference's Dtype dispatch).
template <typename T>
DEVINL bool select_pos(T s, T thr, int method) {
  switch (method) {
    case M_HARD: return s < thr;
    case M_EASY: return s >= thr;
    case M_RAND: return true;
    case M_RELATIVE_HARD: return s <= thr;
    default: return s >= thr;  // RELATIVE_EASY
  }
}

template <typename T>
DEVINL bool select_neg(T s, T thr, int method) {
  switch (method) {
    case M_HARD: return s > thr;
    case M_EASY: return s <= thr;
    case M_RAND: return true;
    case M_RELATIVE_HARD: return s >= thr;
    default: return s <= thr;  // RELATIVE_EASY
  }
}

// Relative-order-statistic index (reference .cu:285-287).  The reference's
// C expression `size_t - 1 + float_sn * size_t` promotes to FLOAT32, so we
// must compute in float32 too (double diverges when sn*size is near an
// integer).  Clamped to [0, size-1] (reference is UB out of range).
// Returns -1 for an empty list.
DEVINL long long relative_index(float sn, long long size) {
  if (size <= 0) return -1;
  long long pos;
  if (sn >= 0.0f) {
    pos = size - 1 - (long long)sn;
  } else {
    pos = (long long)((float)(size - 1) + sn * (float)size);
  }
  if (pos < 0) pos = 0;
  if (pos > size - 1) pos = size - 1;
  return pos;
}

// Fast x^y for x > 0: v_log_f32 + v_exp_f32 (2 hardware transcendentals).
// hipcc lowers __powf to the full-precision OCML pow (~750 VALU
// instructions of frexp/ldexp/special-case handling — measured as the
// entire LRN kernel cost); this is the CUDA-__powf-equivalent fast path.
DEVINL float fast_powf(float x, float y) {
  return __builtin_amdgcn_exp2f(y * __builtin_amdgcn_logf(x));
}

#define HIP_CHECK_LAST()                                            \
  do {                                                              \
    hipError_t _e = hipGetLastError();                              \
    if (_e != hipSuccess) {                                         \
      TORCH_CHECK(false, "HIP kernel launch failed: ",              \
                  hipGetErrorString(_e));                           \
    }                                                               \
  } while (0)
