// Fused bias + ReLU with a fused backward (dReLU + bias gradient) for
// bias-free convolutions (gfx950).
//
// Motivation (profiles/kernels_r1.md "remaining headroom"): with bias
// inside MIOpen's conv, the backward still pays TWO full passes over each
// activation tensor — torch's threshold_backward (~0.9 ms/step) plus the
// generic NHWC bias-grad reduce (~1.4 ms/step).  Moving bias out of the
// conv and fusing relu+bias forward / drelu+bias-grad backward folds the
// backward to ONE pass.
//
// NHWC fast path (C % VEC == 0, VEC = 16B/sizeof(T)): 16-byte vector
// loads/stores, and a FIXED-CHANNEL schedule — the host picks the grid so
// the grid-stride (in VEC-element groups) is a multiple of C/VEC, which
// pins every thread to the same VEC-channel window for its whole loop.
// Bias loads hoist into registers (forward) and the bias gradient
// accumulates in registers with an ordered LDS block reduce + ordered
// cross-block finalize (backward) — fully DETERMINISTIC, no atomics.
// The NCHW / odd-C fallback uses LDS atomicAdd partials (deterministic
// only up to float addition order; nothing in the framework routes
// 4D activations through it — channels_last is the default layout).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#include <numeric>

#include "common.h"

template <typename T> DEVINL float ldf1(const T* p, long long i);
template <> DEVINL float ldf1<float>(const float* p, long long i) { return p[i]; }
template <> DEVINL float ldf1<__hip_bfloat16>(const __hip_bfloat16* p, long long i) {
  return __bfloat162float(p[i]);
}
template <typename T> DEVINL void stf1(T* p, long long i, float v);
template <> DEVINL void stf1<float>(float* p, long long i, float v) { p[i] = v; }
template <> DEVINL void stf1<__hip_bfloat16>(__hip_bfloat16* p, long long i, float v) {
  p[i] = __float2bfloat16(v);
}

template <typename T> DEVINL float cvt_to_f(T v);
template <> DEVINL float cvt_to_f<float>(float v) { return v; }
template <> DEVINL float cvt_to_f<__hip_bfloat16>(__hip_bfloat16 v) { return __bfloat162float(v); }
template <typename T> DEVINL T cvt_from_f(float v);
template <> DEVINL float cvt_from_f<float>(float v) { return v; }
template <> DEVINL __hip_bfloat16 cvt_from_f<__hip_bfloat16>(float v) { return __float2bfloat16(v); }

// 16-byte vector of VEC = 16/sizeof(T) elements
template <typename T>
struct alignas(16) Pack16 {
  static constexpr int N = 16 / sizeof(T);
  T v[N];
};

// ---------------------------------------------------------------------------
// NHWC fast path: vectorized + fixed-channel schedule
// ---------------------------------------------------------------------------

// forward: y = relu(x + b[c]); host guarantees (gridDim*blockDim) % (C/VEC)==0
template <typename T>
__global__ void biasrelu_fwd_vec_kernel(const T* __restrict__ x,
                                        const float* __restrict__ bias,
                                        T* __restrict__ y, long long groups,
                                        int C) {
  constexpr int V = Pack16<T>::N;
  const long long t0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const int c0 = (int)(((long long)V * t0) % C);  // fixed for all iterations
  float b[V];
#pragma unroll
  for (int k = 0; k < V; ++k) b[k] = bias[c0 + k];
  const Pack16<T>* xp = reinterpret_cast<const Pack16<T>*>(x);
  Pack16<T>* yp = reinterpret_cast<Pack16<T>*>(y);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long g = t0; g < groups; g += stride) {
    Pack16<T> xv = xp[g];
    Pack16<T> yv;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      const float v = cvt_to_f(xv.v[k]) + b[k];
      yv.v[k] = cvt_from_f<T>(v > 0.f ? v : 0.f);
    }
    yp[g] = yv;
  }
}

// backward: dx = dy * (y > 0); db[c] = ordered sum of dx over sites.
// Each thread owns channels [c0, c0+V); register accumulate, then an
// ordered per-block LDS reduce, then the ordered cross-block finalize.
template <typename T>
__global__ void biasrelu_bwd_vec_kernel(const T* __restrict__ y,
                                        const T* __restrict__ dy,
                                        T* __restrict__ dx,
                                        float* __restrict__ partials,
                                        long long groups, int C) {
  constexpr int V = Pack16<T>::N;
  extern __shared__ float lds[];  // blockDim * V floats
  const long long t0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const int c0 = (int)(((long long)V * t0) % C);
  float acc[V];
#pragma unroll
  for (int k = 0; k < V; ++k) acc[k] = 0.f;
  const Pack16<T>* yp = reinterpret_cast<const Pack16<T>*>(y);
  const Pack16<T>* dyp = reinterpret_cast<const Pack16<T>*>(dy);
  Pack16<T>* dxp = reinterpret_cast<Pack16<T>*>(dx);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long g = t0; g < groups; g += stride) {
    Pack16<T> yv = yp[g];
    Pack16<T> dyv = dyp[g];
    Pack16<T> dxv;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      const float gk = (cvt_to_f(yv.v[k]) > 0.f) ? cvt_to_f(dyv.v[k]) : 0.f;
      dxv.v[k] = cvt_from_f<T>(gk);
      acc[k] += gk;
    }
    dxp[g] = dxv;
  }
#pragma unroll
  for (int k = 0; k < V; ++k) lds[threadIdx.x * V + k] = acc[k];
  __syncthreads();
  // Ordered per-channel-window sum over the block's threads.  Thread t's
  // window index is (base + t) mod q with q = C/V, so the threads sharing
  // window j are t0, t0+q, t0+2q, ... (ascending — deterministic order);
  // host guarantees q <= blockDim.  No modulo in the loop body.
  const int q = C / V;
  const int base_mod = (int)(((long long)blockIdx.x * blockDim.x) % q);
  for (int j = threadIdx.x; j < q; j += blockDim.x) {
    int t0 = j - base_mod;
    if (t0 < 0) t0 += q;
    float s[V];
#pragma unroll
    for (int k = 0; k < V; ++k) s[k] = 0.f;
    for (int t = t0; t < (int)blockDim.x; t += q) {
#pragma unroll
      for (int k = 0; k < V; ++k) s[k] += lds[t * V + k];
    }
    float* out = partials + (long long)blockIdx.x * C + (long long)j * V;
#pragma unroll
    for (int k = 0; k < V; ++k) out[k] = s[k];
  }
}

// ---------------------------------------------------------------------------
// scalar fallback (NCHW or C not a multiple of VEC)
// ---------------------------------------------------------------------------

template <typename T>
__global__ void biasrelu_fwd_kernel(const T* __restrict__ x,
                                    const float* __restrict__ bias,
                                    T* __restrict__ y, long long total, int C,
                                    long long cstride) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const int c = (int)((i / cstride) % C);
    const float v = ldf1(x, i) + bias[c];
    stf1(y, i, v > 0.f ? v : 0.f);
  }
}

template <typename T>
__global__ void biasrelu_bwd_kernel(const T* __restrict__ y,
                                    const T* __restrict__ dy,
                                    T* __restrict__ dx,
                                    float* __restrict__ partials,
                                    long long total, int C, long long cstride) {
  extern __shared__ float bins[];  // C floats
  for (int c = threadIdx.x; c < C; c += blockDim.x) bins[c] = 0.f;
  __syncthreads();
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long long)gridDim.x * blockDim.x) {
    const int c = (int)((i / cstride) % C);
    const float g = (ldf1(y, i) > 0.f) ? ldf1(dy, i) : 0.f;
    stf1(dx, i, g);
    if (g != 0.f) atomicAdd(&bins[c], g);
  }
  __syncthreads();
  float* out = partials + (long long)blockIdx.x * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) out[c] = bins[c];
}

// ordered over blocks (ascending) — deterministic stage 2 for both paths.
// Split into fixed chunks along the block axis (blockIdx.y) so the serial
// depth stays short; each chunk sums ascending, then the chunk results sum
// ascending — a fixed association, bitwise-reproducible run to run.
__global__ void biasrelu_db_chunk_kernel(const float* __restrict__ partials,
                                         float* __restrict__ chunk_out, int C,
                                         int nblocks, int chunk) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int b0 = blockIdx.y * chunk;
  const int b1 = min(b0 + chunk, nblocks);
  float s = 0.f;
  for (int b = b0; b < b1; ++b) s += partials[(long long)b * C + c];
  chunk_out[(long long)blockIdx.y * C + c] = s;
}

__global__ void biasrelu_db_finalize_kernel(const float* __restrict__ partials,
                                            float* __restrict__ db, int C,
                                            int nblocks) {
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int b = 0; b < nblocks; ++b) s += partials[(long long)b * C + c];
    db[c] = s;
  }
}

// two-stage deterministic column sum of partials[nblocks][C] -> db[C]
static void br_db_reduce(const torch::Tensor& partials, torch::Tensor& db,
                         int C, int nblocks, hipStream_t stream) {
  constexpr int NCHUNK = 16;
  if (nblocks <= 64) {
    biasrelu_db_finalize_kernel<<<(C + 255) / 256, 256, 0, stream>>>(
        partials.data_ptr<float>(), db.data_ptr<float>(), C, nblocks);
    return;
  }
  const int chunk = (nblocks + NCHUNK - 1) / NCHUNK;
  const int ny = (nblocks + chunk - 1) / chunk;
  auto chunks = torch::empty({ny, C}, partials.options());
  dim3 grid((C + 255) / 256, ny);
  biasrelu_db_chunk_kernel<<<grid, 256, 0, stream>>>(
      partials.data_ptr<float>(), chunks.data_ptr<float>(), C, nblocks, chunk);
  biasrelu_db_finalize_kernel<<<(C + 255) / 256, 256, 0, stream>>>(
      chunks.data_ptr<float>(), db.data_ptr<float>(), C, ny);
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

static bool br_nhwc(const torch::Tensor& t) {
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

// grid size such that (grid * NPAIR_BLOCK) % q == 0 (q = C/VEC): pins every
// thread's channel window across grid-stride iterations
static int br_fixed_grid(long long groups, int q, int maxgrid) {
  long long desired = (groups + NPAIR_BLOCK * 4 - 1) / (NPAIR_BLOCK * 4);
  if (desired < 256) desired = 256;
  if (desired > maxgrid) desired = maxgrid;
  const int m = q / std::gcd((long long)q, (long long)NPAIR_BLOCK);
  return (int)((desired + m - 1) / m * m);
}

torch::Tensor biasrelu_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4);
  TORCH_CHECK(bias.is_cuda() && bias.dtype() == torch::kFloat32 && bias.is_contiguous());
  const bool nhwc = br_nhwc(x);
  auto xc = nhwc ? x : x.contiguous();
  auto y = torch::empty_like(xc);
  const long long C = x.size(1), S = x.size(2) * x.size(3);
  const long long total = x.size(0) * C * S;
  const long long cstride = nhwc ? 1 : S;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x.scalar_type(), "biasrelu_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    TORCH_CHECK((std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>),
                "biasrelu: bf16/fp32 only");
    constexpr int V = Pack16<T>::N;
    if (nhwc && C % V == 0) {
      const long long groups = total / V;
      const int grid = br_fixed_grid(groups, (int)(C / V), 4096);
      biasrelu_fwd_vec_kernel<T><<<grid, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), bias.data_ptr<float>(),
          reinterpret_cast<T*>(y.data_ptr()), groups, (int)C);
    } else {
      const int blocks = (int)std::min<long long>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 4096);
      biasrelu_fwd_kernel<T><<<blocks, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(xc.data_ptr()), bias.data_ptr<float>(),
          reinterpret_cast<T*>(y.data_ptr()), total, (int)C, cstride);
    }
  });
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> biasrelu_bwd(torch::Tensor y, torch::Tensor dy) {
  TORCH_CHECK(y.is_cuda() && y.dim() == 4);
  const bool nhwc = br_nhwc(y);
  auto yc = nhwc ? y : y.contiguous();
  auto dyc = nhwc ? (br_nhwc(dy) ? dy : dy.contiguous(at::MemoryFormat::ChannelsLast))
                  : dy.contiguous();
  auto dx = torch::empty_like(yc);
  const long long C = y.size(1), S = y.size(2) * y.size(3);
  const long long total = y.size(0) * C * S;
  const long long cstride = nhwc ? 1 : S;
  auto db = torch::empty({C}, y.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      y.scalar_type(), "biasrelu_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    TORCH_CHECK((std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>),
                "biasrelu: bf16/fp32 only");
    constexpr int V = Pack16<T>::N;
    if (nhwc && C % V == 0 && C / V <= NPAIR_BLOCK) {
      const long long groups = total / V;
      // smaller grid than the forward: the cross-block finalize reads
      // grid*C partials serially per channel, so fewer/larger blocks win
      const int grid = br_fixed_grid(groups, (int)(C / V), 1280);
      auto partials = torch::empty({grid, C}, y.options().dtype(torch::kFloat32));
      biasrelu_bwd_vec_kernel<T><<<grid, NPAIR_BLOCK,
                                   (size_t)NPAIR_BLOCK * V * sizeof(float), stream>>>(
          reinterpret_cast<const T*>(yc.data_ptr()),
          reinterpret_cast<const T*>(dyc.data_ptr()),
          reinterpret_cast<T*>(dx.data_ptr()), partials.data_ptr<float>(),
          groups, (int)C);
      br_db_reduce(partials, db, (int)C, grid, stream);
    } else {
      const int blocks = (int)std::min<long long>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 1024);
      auto partials = torch::empty({blocks, C}, y.options().dtype(torch::kFloat32));
      biasrelu_bwd_kernel<T><<<blocks, NPAIR_BLOCK, (size_t)C * sizeof(float), stream>>>(
          reinterpret_cast<const T*>(yc.data_ptr()),
          reinterpret_cast<const T*>(dyc.data_ptr()),
          reinterpret_cast<T*>(dx.data_ptr()), partials.data_ptr<float>(),
          total, (int)C, cstride);
      br_db_reduce(partials, db, (int)C, blocks, stream);
    }
  });
  HIP_CHECK_LAST();
  return {dx, db};
}
