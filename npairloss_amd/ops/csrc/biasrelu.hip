// Fused bias + ReLU with a fused backward (dReLU + bias gradient) for
// bias-free convolutions (gfx950).
//
// Motivation (profiles/kernels_r1.md "remaining headroom"): with bias
// inside MIOpen's conv, the backward still pays TWO full passes over each
// activation tensor — torch's threshold_backward (~0.9 ms/step) plus the
// generic NHWC bias-grad reduce (~1.4 ms/step).  Moving bias out of the
// conv and fusing relu+bias forward / drelu+bias-grad backward folds the
// backward to ONE pass with a deterministic two-stage column reduce.
//
// Status: EXPERIMENTAL this round — compile-verified + CPU-parity-tested;
// GPU numerics tests are gated behind NPAIR_EXPERIMENTAL=1
// (tests/test_gpu_experimental.py) pending GPU validation next round.
// Nothing routes through these kernels unless GoogLeNet is built with
// fused_bias_relu=True.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

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

// ---------------------------------------------------------------------------
// forward: y = relu(x + b[c]); NHWC (cstride 1) or NCHW (cstride S)
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

// ---------------------------------------------------------------------------
// backward: dx = dy * (y > 0); db[c] = sum over sites of dx
// Stage 1: per-block LDS bins -> partials[block][C]; stage 2 sums blocks.
// Deterministic (fixed block count, ordered stage-2 sum).
// ---------------------------------------------------------------------------

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

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

static bool br_nhwc(const torch::Tensor& t) {
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
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
  const int blocks = (int)std::min<long long>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 4096);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      x.scalar_type(), "biasrelu_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    TORCH_CHECK((std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>),
                "biasrelu: bf16/fp32 only");
    biasrelu_fwd_kernel<T><<<blocks, NPAIR_BLOCK, 0, stream>>>(
        reinterpret_cast<const T*>(xc.data_ptr()), bias.data_ptr<float>(),
        reinterpret_cast<T*>(y.data_ptr()), total, (int)C, cstride);
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
  const int blocks = (int)std::min<long long>((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, 1024);
  auto partials = torch::empty({blocks, C}, y.options().dtype(torch::kFloat32));
  auto db = torch::empty({C}, y.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
      y.scalar_type(), "biasrelu_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    TORCH_CHECK((std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>),
                "biasrelu: bf16/fp32 only");
    biasrelu_bwd_kernel<T><<<blocks, NPAIR_BLOCK, (size_t)C * sizeof(float), stream>>>(
        reinterpret_cast<const T*>(yc.data_ptr()),
        reinterpret_cast<const T*>(dyc.data_ptr()),
        reinterpret_cast<T*>(dx.data_ptr()), partials.data_ptr<float>(),
        total, (int)C, cstride);
  });
  biasrelu_db_finalize_kernel<<<(int)std::min<long long>((C + 255) / 256, 64), 256, 0, stream>>>(
      partials.data_ptr<float>(), db.data_ptr<float>(), (int)C, blocks);
  HIP_CHECK_LAST();
  return {dx, db};
}
