// Mining-threshold selection kernels (gfx950), fp32 + fp64.
//
// The reference computed RELATIVE_* mining thresholds by copying the whole
// B x G similarity matrix to the HOST and running 2 global + 2B per-query
// std::sorts EVERY iteration (npair_multi_class_loss.cu:225-273, 282-305,
// 313-336).  Here:
//   local_relative_thr  — per-query LDS bitonic sort of the masked row
//                         (one workgroup per query, row staged in LDS),
//                         then the order-statistic pick + <0 clamp.
//                         For rows too long for LDS (G > 16384 fp32 /
//                         8192 fp64) a per-row MSB-first radix select
//                         takes over — any G, no abort (round-1 hard-
//                         failed above 16384).
//   global_relative_thr — device-wide k-th order statistic over the masked
//                         B x G values via MSB-first radix select over
//                         order-preserving integer keys; no host round
//                         trip, ~9 tiny launches.
// Both are templated over float/double like the reference's Dtype dispatch
// (.cu:31-42): fp64 uses 64-bit keys (8 digit passes).
// Semantics per common.h relative_index + the value<0 -> -inf clamp
// (.cu:288,303,319,334).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// order-preserving integer keys (ascending value == ascending key)
template <typename T> struct KeyTraits;
template <> struct KeyTraits<float> {
  using K = uint32_t;
  static constexpr int BITS = 32;
  DEVINL static K key(float f) {
    uint32_t u = __float_as_uint(f);
    return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
  }
  DEVINL static float val(K k) {
    uint32_t u = (k & 0x80000000u) ? (k ^ 0x80000000u) : ~k;
    return __uint_as_float(u);
  }
  DEVINL static float neg_max() { return -FLT_MAX; }
};
template <> struct KeyTraits<double> {
  using K = uint64_t;
  static constexpr int BITS = 64;
  DEVINL static K key(double f) {
    uint64_t u = __double_as_longlong(f);
    return (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
  }
  DEVINL static double val(K k) {
    uint64_t u = (k & 0x8000000000000000ull) ? (k ^ 0x8000000000000000ull) : ~k;
    return __longlong_as_double(u);
  }
  DEVINL static double neg_max() { return -DBL_MAX; }
};

template <typename T> struct TMax;
template <> struct TMax<float> { static constexpr float v = FLT_MAX; };
template <> struct TMax<double> { static constexpr double v = DBL_MAX; };

// ---------------------------------------------------------------------------
// per-row bitonic sort + pick (LOCAL RELATIVE_HARD / RELATIVE_EASY)
// ---------------------------------------------------------------------------

// use_same selects the positive (ident) list, else the negative (diff) list.
template <typename T>
__global__ void local_rel_thr_kernel(const T* __restrict__ S,
                                     const int* __restrict__ lab_l,
                                     const int* __restrict__ lab_g,
                                     int B, int G, int rank, int npow2,
                                     int use_same, float sn,
                                     T* __restrict__ thr) {
  extern __shared__ char lr_smem[];  // npow2 * sizeof(T) bytes
  T* vals = reinterpret_cast<T*>(lr_smem);
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const T* row = S + (size_t)i * G;

  int cnt = 0;
  for (int j = threadIdx.x; j < npow2; j += blockDim.x) {
    T v = TMax<T>::v;  // pad: sorts to the end of ascending order
    if (j < G && !pair_is_self(i, j, rank, B)) {
      const bool same = (lab_g[j] == li);
      if (same == (use_same != 0)) {
        v = row[j];
        ++cnt;
      }
    }
    vals[j] = v;
  }
  cnt = block_reduce(cnt, OpAddI(), 0, scratch_i);
  __syncthreads();

  // ascending bitonic sort of vals[0..npow2)
  for (int k = 2; k <= npow2; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int t = threadIdx.x; t < npow2; t += blockDim.x) {
        const int l = t ^ j;
        if (l > t) {
          const bool up = ((t & k) == 0);
          const T a = vals[t], b = vals[l];
          if ((a > b) == up) {
            vals[t] = b;
            vals[l] = a;
          }
        }
      }
      __syncthreads();
    }
  }

  if (threadIdx.x == 0) {
    const long long pos = relative_index(sn, cnt);
    T out = -TMax<T>::v;
    if (pos >= 0) {
      const T v = vals[pos];
      out = (v >= (T)0) ? v : -TMax<T>::v;
    }
    thr[i] = out;
  }
}

// ---------------------------------------------------------------------------
// per-row radix select — any G (no LDS row staging); one block per row.
// BITS/8 digit passes over the masked row, 256-bin LDS histogram each.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void local_rel_radix_kernel(const T* __restrict__ S,
                                       const int* __restrict__ lab_l,
                                       const int* __restrict__ lab_g,
                                       int B, int G, int rank,
                                       int use_same, float sn,
                                       T* __restrict__ thr) {
  using KT = KeyTraits<T>;
  using K = typename KT::K;
  __shared__ unsigned int bins[256];
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  __shared__ K sh_prefix;
  __shared__ long long sh_k;
  __shared__ int sh_done;
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const T* row = S + (size_t)i * G;

  // pass 0: masked count -> order-statistic position
  int cnt = 0;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    if ((lab_g[j] == li) == (use_same != 0)) ++cnt;
  }
  cnt = block_reduce(cnt, OpAddI(), 0, scratch_i);
  if (threadIdx.x == 0) {
    const long long pos = relative_index(sn, cnt);
    sh_done = (pos < 0);
    sh_k = pos + 1;  // k-th smallest, 1-based
    sh_prefix = 0;
    if (pos < 0) thr[i] = -TMax<T>::v;  // empty list -> select-all
  }
  __syncthreads();
  if (sh_done) return;

  for (int shift = KT::BITS - 8; shift >= 0; shift -= 8) {
    for (int t = threadIdx.x; t < 256; t += blockDim.x) bins[t] = 0;
    __syncthreads();
    const K prefix = sh_prefix;
    for (int j = threadIdx.x; j < G; j += blockDim.x) {
      if (pair_is_self(i, j, rank, B)) continue;
      if ((lab_g[j] == li) != (use_same != 0)) continue;
      const K key = KT::key(row[j]);
      if (shift < KT::BITS - 8 && (key >> (shift + 8)) != prefix) continue;
      atomicAdd(&bins[(int)((key >> shift) & 0xFF)], 1u);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      long long k = sh_k;
      int b = 0;
      for (; b < 256; ++b) {
        const long long c = (long long)bins[b];
        if (k <= c) break;
        k -= c;
      }
      if (b == 256) b = 255;  // unreachable when counts are consistent
      sh_k = k;
      sh_prefix = (prefix << 8) | (K)b;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const T v = KT::val(sh_prefix);
    thr[i] = (v >= (T)0) ? v : -TMax<T>::v;  // the <0 clamp
  }
}

// ---------------------------------------------------------------------------
// global radix select (GLOBAL RELATIVE_HARD / RELATIVE_EASY)
// ---------------------------------------------------------------------------

// state layout: [0] n_total, [1] k_remaining, [2] prefix, [3] done flag
// out: single element (the clamped threshold)

template <typename T>
__global__ void grs_count_kernel(const T* __restrict__ S,
                                 const int* __restrict__ lab_l,
                                 const int* __restrict__ lab_g,
                                 int B, int G, int rank, int use_same,
                                 long long* __restrict__ state) {
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const size_t total = (size_t)B * G;
  int cnt = 0;
  for (size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (size_t)gridDim.x * blockDim.x) {
    const int i = idx / G;
    const int j = idx % G;
    if (pair_is_self(i, j, rank, B)) continue;
    const bool same = (lab_g[j] == lab_l[i]);
    if (same == (use_same != 0)) ++cnt;
  }
  cnt = block_reduce(cnt, OpAddI(), 0, scratch_i);
  if (threadIdx.x == 0 && cnt > 0) atomicAdd((unsigned long long*)&state[0], (unsigned long long)cnt);
}

template <typename T>
__global__ void grs_init_kernel(long long* __restrict__ state, float sn,
                                T* __restrict__ out) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  const long long n = state[0];
  const long long pos = relative_index(sn, n);
  if (pos < 0) {
    out[0] = -TMax<T>::v;  // empty list
    state[3] = 1;
  } else {
    state[1] = pos + 1;  // k-th smallest, 1-based
    state[2] = 0;
    state[3] = 0;
  }
}

template <typename T>
__global__ void grs_hist_kernel(const T* __restrict__ S,
                                const int* __restrict__ lab_l,
                                const int* __restrict__ lab_g,
                                int B, int G, int rank, int use_same,
                                int shift,
                                const long long* __restrict__ state,
                                unsigned long long* __restrict__ bins) {
  using KT = KeyTraits<T>;
  using K = typename KT::K;
  if (state[3]) return;
  __shared__ unsigned int lbins[256];
  for (int t = threadIdx.x; t < 256; t += blockDim.x) lbins[t] = 0;
  __syncthreads();
  const K prefix = (K)(unsigned long long)state[2];
  const size_t total = (size_t)B * G;
  for (size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (size_t)gridDim.x * blockDim.x) {
    const int i = idx / G;
    const int j = idx % G;
    if (pair_is_self(i, j, rank, B)) continue;
    const bool same = (lab_g[j] == lab_l[i]);
    if (same != (use_same != 0)) continue;
    const K key = KT::key(S[idx]);
    if (shift < KT::BITS - 8 && (key >> (shift + 8)) != prefix) continue;
    atomicAdd(&lbins[(int)((key >> shift) & 0xFF)], 1u);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < 256; t += blockDim.x)
    if (lbins[t]) atomicAdd(&bins[t], (unsigned long long)lbins[t]);
}

template <typename T>
__global__ void grs_pick_kernel(int shift, long long* __restrict__ state,
                                unsigned long long* __restrict__ bins,
                                T* __restrict__ out) {
  using KT = KeyTraits<T>;
  using K = typename KT::K;
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  if (state[3]) return;
  long long k = state[1];
  K prefix = (K)(unsigned long long)state[2];
  for (int b = 0; b < 256; ++b) {
    const long long c = (long long)bins[b];
    if (k <= c) {
      prefix = (prefix << 8) | (K)b;
      state[1] = k;
      state[2] = (long long)(unsigned long long)prefix;
      if (shift == 0) {
        const T v = KT::val(prefix);
        out[0] = (v >= (T)0) ? v : -TMax<T>::v;  // the <0 clamp
        state[3] = 1;
      }
      return;
    }
    k -= c;
  }
  // unreachable when counts are consistent; emit select-all as a safe value
  out[0] = -TMax<T>::v;
  state[3] = 1;
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static int next_pow2(int v) {
  int p = 1;
  while (p < v) p <<= 1;
  return p;
}

torch::Tensor local_relative_thr(torch::Tensor S, torch::Tensor lab_l,
                                 torch::Tensor lab_g, int64_t rank,
                                 bool use_same, double sn) {
  TORCH_CHECK(S.is_cuda() && S.is_contiguous() &&
              (S.dtype() == torch::kFloat32 || S.dtype() == torch::kFloat64));
  const int B = S.size(0), G = S.size(1);
  auto thr = torch::empty({B}, S.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "local_relative_thr", [&] {
    using T = scalar_t;
    const int npow2 = next_pow2(G);
    // LDS row staging budget: 64 KB of dynamic shared
    const bool lds_ok = (size_t)npow2 * sizeof(T) <= 65536;
    if (lds_ok) {
      const size_t shmem = (size_t)npow2 * sizeof(T);
      local_rel_thr_kernel<T><<<B, NPAIR_BLOCK, shmem, stream>>>(
          S.data_ptr<T>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
          (int)rank, npow2, use_same ? 1 : 0, (float)sn, thr.data_ptr<T>());
    } else {
      // G too long for an LDS sort: per-row radix select, any G
      local_rel_radix_kernel<T><<<B, NPAIR_BLOCK, 0, stream>>>(
          S.data_ptr<T>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
          (int)rank, use_same ? 1 : 0, (float)sn, thr.data_ptr<T>());
    }
  });
  HIP_CHECK_LAST();
  return thr;
}

torch::Tensor global_relative_thr(torch::Tensor S, torch::Tensor lab_l,
                                  torch::Tensor lab_g, int64_t rank,
                                  bool use_same, double sn) {
  TORCH_CHECK(S.is_cuda() && S.is_contiguous() &&
              (S.dtype() == torch::kFloat32 || S.dtype() == torch::kFloat64));
  const int B = S.size(0), G = S.size(1);
  auto state = torch::zeros({4}, S.options().dtype(torch::kInt64));
  auto bins = torch::zeros({256}, S.options().dtype(torch::kInt64));
  auto out = torch::empty({1}, S.options());
  auto stream = at::hip::getCurrentHIPStream();
  const size_t total = (size_t)B * G;
  const int blocks = (int)std::min<size_t>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 1024);
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "global_relative_thr", [&] {
    using T = scalar_t;
    auto* st = reinterpret_cast<long long*>(state.data_ptr<int64_t>());
    auto* bn = reinterpret_cast<unsigned long long*>(bins.data_ptr<int64_t>());
    grs_count_kernel<T><<<blocks, NPAIR_BLOCK, 0, stream>>>(
        S.data_ptr<T>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
        (int)rank, use_same ? 1 : 0, st);
    grs_init_kernel<T><<<1, 1, 0, stream>>>(st, (float)sn, out.data_ptr<T>());
    for (int shift = KeyTraits<T>::BITS - 8; shift >= 0; shift -= 8) {
      bins.zero_();
      grs_hist_kernel<T><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          S.data_ptr<T>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
          (int)rank, use_same ? 1 : 0, shift, st, bn);
      grs_pick_kernel<T><<<1, 1, 0, stream>>>(shift, st, bn, out.data_ptr<T>());
    }
  });
  HIP_CHECK_LAST();
  return out.squeeze(0);
}
