// Low-precision MFMA similarity GEMM: S = A @ B^T with bf16 or OCP fp8
// (e4m3) operands and fp32 accumulation (gfx950
// v_mfma_f32_16x16x32_bf16 / v_mfma_f32_16x16x32_fp8_fp8).
//
// The similarity matrix is the loss path's only O(B*G*D) op; unit-norm
// embeddings are in [-1, 1] so fp8 e4m3 needs no scaling.  BASELINE.json
// config 2 (bf16 1-GPU) and config 5 (ViT fp8) use these; the default
// fp32 path (gemm_f32.hip) stays exact for oracle-matched numerics.
//
// Both operands are K-contiguous row-major (B x D and G x D), staged
// through LDS [64][32] tiles; each wave computes a 32x32 output from
// 2x2 16x16x32 fragments.  A/B fragment map: lane l holds
// elem[row = l&15][k = (l>>4)*8 + e]; C/D: col = l&15,
// row = (l>>4)*4 + reg (verified against torch.matmul on-device with
// asymmetric operands — tests/test_gpu_lowp.py).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define LP_BM 64
#define LP_BN 64
#define LP_BK 32

template <typename ET>  // unsigned short (bf16) or unsigned char (fp8)
__launch_bounds__(NPAIR_BLOCK)
__global__ void sim_nt_lowp_kernel(const ET* __restrict__ A,
                                   const ET* __restrict__ B,
                                   float* __restrict__ C,
                                   int M, int N, int K) {
  __shared__ ET As[LP_BM][LP_BK];
  __shared__ ET Bs[LP_BN][LP_BK];
  const int m0 = blockIdx.y * LP_BM;
  const int n0 = blockIdx.x * LP_BN;
  const int t = threadIdx.x;
  const int wid = t / WAVE;
  const int lane = t % WAVE;
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += LP_BK) {
    // stage: thread t loads 8 contiguous elements of one row
    {
      const int row = t >> 2;           // 0..63
      const int col = (t & 3) * 8;      // 0,8,16,24
      ET tmp[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kg = k0 + col + e;
        tmp[e] = (ET)0;
        if (m0 + row < M && kg < K) tmp[e] = A[(size_t)(m0 + row) * K + kg];
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) As[row][col + e] = tmp[e];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kg = k0 + col + e;
        tmp[e] = (ET)0;
        if (n0 + row < N && kg < K) tmp[e] = B[(size_t)(n0 + row) * K + kg];
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) Bs[row][col + e] = tmp[e];
    }
    __syncthreads();

    const int kf = l4 * 8;  // this lane's k offset within the 32-K step
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        if constexpr (sizeof(ET) == 2) {
          const bf16x8 a = *reinterpret_cast<const bf16x8*>(&As[wm + fm * 16 + l15][kf]);
          const bf16x8 b = *reinterpret_cast<const bf16x8*>(&Bs[wn + fn * 16 + l15][kf]);
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
        } else {
          const long a = *reinterpret_cast<const long*>(&As[wm + fm * 16 + l15][kf]);
          const long b = *reinterpret_cast<const long*>(&Bs[wn + fn * 16 + l15][kf]);
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc[fm][fn], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gm = m0 + wm + fm * 16 + l4 * 4 + r;
        const int gn = n0 + wn + fn * 16 + l15;
        if (gm < M && gn < N) C[(size_t)gm * N + gn] = acc[fm][fn][r];
      }
}

__global__ void cast_fp8_kernel(const float* __restrict__ x,
                                unsigned char* __restrict__ o, long long n) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    __hip_fp8_e4m3 v(x[i]);
    o[i] = v.__x;
  }
}

// ---------------------------------------------------------------------------

torch::Tensor sim_gemm_nt_bf16(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16 && A.is_contiguous());
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kBFloat16 && B.is_contiguous());
  TORCH_CHECK(A.size(1) == B.size(1));
  const int M = A.size(0), N = B.size(0), K = A.size(1);
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  dim3 grid((N + LP_BN - 1) / LP_BN, (M + LP_BM - 1) / LP_BM);
  auto stream = at::hip::getCurrentHIPStream();
  sim_nt_lowp_kernel<unsigned short><<<grid, NPAIR_BLOCK, 0, stream>>>(
      reinterpret_cast<const unsigned short*>(A.data_ptr()),
      reinterpret_cast<const unsigned short*>(B.data_ptr()),
      C.data_ptr<float>(), M, N, K);
  HIP_CHECK_LAST();
  return C;
}

torch::Tensor cast_fp8(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kFloat32 && x.is_contiguous());
  auto o = torch::empty_like(x, x.options().dtype(torch::kUInt8));
  const long long n = x.numel();
  auto stream = at::hip::getCurrentHIPStream();
  cast_fp8_kernel<<<(int)std::min<long long>((n + 255) / 256, 2048), 256, 0, stream>>>(
      x.data_ptr<float>(), o.data_ptr<unsigned char>(), n);
  HIP_CHECK_LAST();
  return o;
}

torch::Tensor sim_gemm_nt_fp8(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kUInt8 && A.is_contiguous(),
              "fp8 operands are uint8 e4m3 bit patterns (use cast_fp8)");
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kUInt8 && B.is_contiguous());
  TORCH_CHECK(A.size(1) == B.size(1));
  const int M = A.size(0), N = B.size(0), K = A.size(1);
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  dim3 grid((N + LP_BN - 1) / LP_BN, (M + LP_BM - 1) / LP_BM);
  auto stream = at::hip::getCurrentHIPStream();
  sim_nt_lowp_kernel<unsigned char><<<grid, NPAIR_BLOCK, 0, stream>>>(
      A.data_ptr<unsigned char>(), B.data_ptr<unsigned char>(),
      C.data_ptr<float>(), M, N, K);
  HIP_CHECK_LAST();
  return C;
}
