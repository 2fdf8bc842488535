// 1x1 convolution as a fused MFMA GEMM (gfx950).
//
// In NHWC a 1x1 stride-1 conv IS a GEMM: y[M,N] = x[M,K] @ w[N,K]^T with
// M = B*H*W sites, K = Cin, N = Cout.  GoogLeNet's inception blocks are
// mostly 1x1 convs, which MIOpen runs as generic igemm kernels followed by
// our separate BiasReLU pass.  This kernel fuses the whole layer forward —
// GEMM + bias + ReLU — into ONE pass over x (v_mfma_f32_16x16x32_bf16,
// fp32 accumulation, bf16 out), eliminating the intermediate conv-output
// round trip through HBM.
//
// The same kernel without the epilogue computes the data gradient
// dx[M,K] = g[M,N] @ (w^T)[K,N]^T (caller passes w pre-transposed, a
// tiny K x N copy).  The weight gradient is a plain TN GEMM with a huge
// M-reduction — that one goes to hipBLASLt via torch.matmul (library
// GEMMs are the right tool for unfused shapes; the FUSED hot op is here).
//
// Tiling: 64x64 block tile, BK=32, 4 waves of 2x2 16x16x32 fragments
// (the sim_nt_lowp_kernel fragment map, verified on-device —
// gemm_lowp.hip:10-15); A/B tiles staged through LDS with 16B row chunks.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#define C1_BM 64
#define C1_BN 64
#define C1_BK 32

// EPI: 0 = plain store, 1 = bias + relu
template <int EPI>
__launch_bounds__(NPAIR_BLOCK)
__global__ void conv1x1_nt_kernel(const unsigned short* __restrict__ A,
                                  const unsigned short* __restrict__ B,
                                  const float* __restrict__ bias,
                                  unsigned short* __restrict__ Y,
                                  int M, int N, int K) {
  __shared__ unsigned short As[C1_BM][C1_BK];
  __shared__ unsigned short Bs[C1_BN][C1_BK];
  const int m0 = blockIdx.y * C1_BM;
  const int n0 = blockIdx.x * C1_BN;
  const int t = threadIdx.x;
  const int wid = t / WAVE;
  const int lane = t % WAVE;
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  f32x4 acc[2][2] = {};

  const bool a_full = (m0 + C1_BM <= M);
  for (int k0 = 0; k0 < K; k0 += C1_BK) {
    {
      const int row = t >> 2;
      const int col = (t & 3) * 8;
      const bool k_full = (k0 + C1_BK <= K);
      unsigned short tmp[8];
      if (a_full && k_full) {
        *reinterpret_cast<bf16x8*>(tmp) =
            *reinterpret_cast<const bf16x8*>(&A[(size_t)(m0 + row) * K + k0 + col]);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int kg = k0 + col + e;
          tmp[e] = 0;
          if (m0 + row < M && kg < K) tmp[e] = A[(size_t)(m0 + row) * K + kg];
        }
      }
      *reinterpret_cast<bf16x8*>(&As[row][col]) = *reinterpret_cast<bf16x8*>(tmp);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kg = k0 + col + e;
        tmp[e] = 0;
        if (n0 + row < N && kg < K) tmp[e] = B[(size_t)(n0 + row) * K + kg];
      }
      *reinterpret_cast<bf16x8*>(&Bs[row][col]) = *reinterpret_cast<bf16x8*>(tmp);
    }
    __syncthreads();

    const int kf = l4 * 8;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        const bf16x8 a = *reinterpret_cast<const bf16x8*>(&As[wm + fm * 16 + l15][kf]);
        const bf16x8 b = *reinterpret_cast<const bf16x8*>(&Bs[wn + fn * 16 + l15][kf]);
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
      }
    __syncthreads();
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int gn = n0 + wn + fn * 16 + l15;
      const float bv = (EPI && gn < N) ? bias[gn] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gm = m0 + wm + fm * 16 + l4 * 4 + r;
        if (gm < M && gn < N) {
          float v = acc[fm][fn][r];
          if (EPI) {
            v += bv;
            v = v > 0.f ? v : 0.f;
          }
          Y[(size_t)gm * N + gn] = __hip_bfloat16_raw(__float2bfloat16(v)).x;
        }
      }
    }
}

// ---------------------------------------------------------------------------

static torch::Tensor conv1x1_nt(torch::Tensor A, torch::Tensor B,
                                c10::optional<torch::Tensor> bias, bool epi) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16 && A.is_contiguous());
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kBFloat16 && B.is_contiguous());
  TORCH_CHECK(A.size(1) == B.size(1), "reduction dims differ");
  const int64_t M = A.size(0), N = B.size(0), K = A.size(1);
  auto Y = torch::empty({M, N}, A.options());
  dim3 grid((N + C1_BN - 1) / C1_BN, (M + C1_BM - 1) / C1_BM);
  auto stream = at::hip::getCurrentHIPStream();
  const float* bptr = nullptr;
  torch::Tensor bf;
  if (epi) {
    TORCH_CHECK(bias.has_value(), "epilogue needs bias");
    bf = bias->to(torch::kFloat32).contiguous();
    bptr = bf.data_ptr<float>();
  }
  if (epi)
    conv1x1_nt_kernel<1><<<grid, NPAIR_BLOCK, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(A.data_ptr()),
        reinterpret_cast<const unsigned short*>(B.data_ptr()), bptr,
        reinterpret_cast<unsigned short*>(Y.data_ptr()), (int)M, (int)N, (int)K);
  else
    conv1x1_nt_kernel<0><<<grid, NPAIR_BLOCK, 0, stream>>>(
        reinterpret_cast<const unsigned short*>(A.data_ptr()),
        reinterpret_cast<const unsigned short*>(B.data_ptr()), nullptr,
        reinterpret_cast<unsigned short*>(Y.data_ptr()), (int)M, (int)N, (int)K);
  HIP_CHECK_LAST();
  return Y;
}

// y[M,N] = relu(x[M,K] @ w[N,K]^T + bias[N]) — the fused 1x1-conv forward
torch::Tensor conv1x1_bias_relu_fwd(torch::Tensor x, torch::Tensor w,
                                    torch::Tensor bias) {
  return conv1x1_nt(x, w, bias, true);
}

// dx[M,K] = g[M,N] @ wt[K,N]^T — data gradient (wt = w transposed, K x N)
torch::Tensor conv1x1_dgrad(torch::Tensor g, torch::Tensor wt) {
  return conv1x1_nt(g, wt, c10::nullopt, false);
}
