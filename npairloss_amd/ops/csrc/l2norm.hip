// Row-wise L2 normalization, fused single-pass kernels (gfx950).
//
// The reference fork's `L2Normalize` layer (usage/def.prototxt:115-120)
// feeding the loss its unit-norm embeddings.  One workgroup per row,
// float4-vectorized loads, wave+LDS reduction.
//   fwd: y = x * rsqrt(max(sum x^2, eps^2)); also emits inv_norm for bwd
//   bwd: dx = (dy - y * <y,dy>) * inv_norm

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#define L2_EPS 1e-12f

__global__ void l2norm_fwd_kernel(const float* __restrict__ x, int N, int D,
                                  float* __restrict__ y,
                                  float* __restrict__ inv_norm) {
  __shared__ float scratch[NPAIR_BLOCK / WAVE];
  const int row = blockIdx.x;
  if (row >= N) return;
  const float* xr = x + (size_t)row * D;
  float* yr = y + (size_t)row * D;
  float ss = 0.f;
  const int d4 = D / 4 * 4;
  for (int d = threadIdx.x * 4; d < d4; d += blockDim.x * 4) {
    const float4 v = *reinterpret_cast<const float4*>(xr + d);
    ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  for (int d = d4 + threadIdx.x; d < D; d += blockDim.x) ss += xr[d] * xr[d];
  ss = block_reduce(ss, OpAddF(), 0.f, scratch);
  const float norm = fmaxf(sqrtf(ss), L2_EPS);
  const float inv = 1.f / norm;
  if (threadIdx.x == 0) inv_norm[row] = inv;
  for (int d = threadIdx.x * 4; d < d4; d += blockDim.x * 4) {
    float4 v = *reinterpret_cast<const float4*>(xr + d);
    v.x *= inv; v.y *= inv; v.z *= inv; v.w *= inv;
    *reinterpret_cast<float4*>(yr + d) = v;
  }
  for (int d = d4 + threadIdx.x; d < D; d += blockDim.x) yr[d] = xr[d] * inv;
}

__global__ void l2norm_bwd_kernel(const float* __restrict__ y,
                                  const float* __restrict__ inv_norm,
                                  const float* __restrict__ dy, int N, int D,
                                  float* __restrict__ dx) {
  __shared__ float scratch[NPAIR_BLOCK / WAVE];
  const int row = blockIdx.x;
  if (row >= N) return;
  const float* yr = y + (size_t)row * D;
  const float* gr = dy + (size_t)row * D;
  float* out = dx + (size_t)row * D;
  float dot = 0.f;
  const int d4 = D / 4 * 4;
  for (int d = threadIdx.x * 4; d < d4; d += blockDim.x * 4) {
    const float4 a = *reinterpret_cast<const float4*>(yr + d);
    const float4 b = *reinterpret_cast<const float4*>(gr + d);
    dot += a.x * b.x + a.y * b.y + a.z * b.z + a.w * b.w;
  }
  for (int d = d4 + threadIdx.x; d < D; d += blockDim.x) dot += yr[d] * gr[d];
  dot = block_reduce(dot, OpAddF(), 0.f, scratch);
  const float inv = inv_norm[row];
  for (int d = threadIdx.x * 4; d < d4; d += blockDim.x * 4) {
    const float4 a = *reinterpret_cast<const float4*>(yr + d);
    const float4 b = *reinterpret_cast<const float4*>(gr + d);
    float4 o;
    o.x = (b.x - a.x * dot) * inv;
    o.y = (b.y - a.y * dot) * inv;
    o.z = (b.z - a.z * dot) * inv;
    o.w = (b.w - a.w * dot) * inv;
    *reinterpret_cast<float4*>(out + d) = o;
  }
  for (int d = d4 + threadIdx.x; d < D; d += blockDim.x)
    out[d] = (gr[d] - yr[d] * dot) * inv;
}

std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(x.dtype() == torch::kFloat32, "l2norm: fp32 input expected");
  const int N = x.size(0), D = x.size(1);
  auto y = torch::empty_like(x);
  auto inv_norm = torch::empty({N}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  l2norm_fwd_kernel<<<N, NPAIR_BLOCK, 0, stream>>>(
      x.data_ptr<float>(), N, D, y.data_ptr<float>(), inv_norm.data_ptr<float>());
  HIP_CHECK_LAST();
  return {y, inv_norm};
}

torch::Tensor l2norm_bwd(torch::Tensor y, torch::Tensor inv_norm, torch::Tensor dy) {
  TORCH_CHECK(y.is_cuda() && y.dim() == 2 && y.is_contiguous() && dy.is_contiguous());
  const int N = y.size(0), D = y.size(1);
  auto dx = torch::empty_like(y);
  auto stream = at::hip::getCurrentHIPStream();
  l2norm_bwd_kernel<<<N, NPAIR_BLOCK, 0, stream>>>(
      y.data_ptr<float>(), inv_norm.data_ptr<float>(), dy.data_ptr<float>(), N, D,
      dx.data_ptr<float>());
  HIP_CHECK_LAST();
  return dx;
}
