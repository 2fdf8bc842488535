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

#include <cstdlib>
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
// v2: 128x64 tile, BK=64, register-prefetch double buffering, +8-half LDS
// row padding (the 64-half pitch put all 16 A-frag lanes on one bank
// quartet).  4 waves in a 2x2 grid, each computing 64x32 via 4x2
// fragments -> 16 MFMAs per K-tile between barriers (v1: 4).
// ---------------------------------------------------------------------------

#define C2_BM 128
#define C2_BN 64
#define C2_BK 64
#define C2_PAD 8  // halves

template <int EPI>
__launch_bounds__(NPAIR_BLOCK)
__global__ void conv1x1_nt_v2_kernel(const unsigned short* __restrict__ A,
                                     const unsigned short* __restrict__ B,
                                     const float* __restrict__ bias,
                                     unsigned short* __restrict__ Y,
                                     int M, int N, int K) {
  __shared__ unsigned short As[C2_BM][C2_BK + C2_PAD];
  __shared__ unsigned short Bs[C2_BN][C2_BK + C2_PAD];
  const int m0 = blockIdx.y * C2_BM;
  const int n0 = blockIdx.x * C2_BN;
  const int t = threadIdx.x;
  const int wid = t / WAVE;
  const int lane = t % WAVE;
  const int wm = (wid >> 1) * 64;  // 2x2 waves: wave tile 64 x 32
  const int wn = (wid & 1) * 32;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  // stage coordinates: 256 threads x 32 halves = one 128x64 A tile pass
  const int ar = t >> 1;             // A row 0..127
  const int ac = (t & 1) * 32;       // A col 0 or 32
  const int br = t >> 2;             // B row 0..63
  const int bc = (t & 3) * 16;       // B col 0,16,32,48

  f32x4 acc[4][2] = {};

  // prefetch registers for the NEXT K-tile (4+2 16B vectors)
  using v8 = bf16x8;
  v8 pa[4], pb[2];

  auto load_a = [&](int k0, v8* dst) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = ar;
      const int col = ac + i * 8;
      const int gm = m0 + row;
      const int gk = k0 + col;
      if (gm < M && gk + 7 < K) {
        dst[i] = *reinterpret_cast<const v8*>(&A[(size_t)gm * K + gk]);
      } else {
        unsigned short tmp[8];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          tmp[e] = (gm < M && gk + e < K) ? A[(size_t)gm * K + gk + e] : 0;
        dst[i] = *reinterpret_cast<v8*>(tmp);
      }
    }
  };
  auto load_b = [&](int k0, v8* dst) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int row = br;
      const int col = bc + i * 8;
      const int gn = n0 + row;
      const int gk = k0 + col;
      if (gn < N && gk + 7 < K) {
        dst[i] = *reinterpret_cast<const v8*>(&B[(size_t)gn * K + gk]);
      } else {
        unsigned short tmp[8];
#pragma unroll
        for (int e = 0; e < 8; ++e)
          tmp[e] = (gn < N && gk + e < K) ? B[(size_t)gn * K + gk + e] : 0;
        dst[i] = *reinterpret_cast<v8*>(tmp);
      }
    }
  };
  auto store_stage = [&]() {
#pragma unroll
    for (int i = 0; i < 4; ++i)
      *reinterpret_cast<v8*>(&As[ar][ac + i * 8]) = pa[i];
#pragma unroll
    for (int i = 0; i < 2; ++i)
      *reinterpret_cast<v8*>(&Bs[br][bc + i * 8]) = pb[i];
  };

  load_a(0, pa);
  load_b(0, pb);
  store_stage();
  __syncthreads();

  for (int k0 = 0; k0 < K; k0 += C2_BK) {
    const bool last = (k0 + C2_BK >= K);
    if (!last) {  // prefetch next tile into registers during compute
      load_a(k0 + C2_BK, pa);
      load_b(k0 + C2_BK, pb);
    }
#pragma unroll
    for (int kk = 0; kk < C2_BK; kk += 32) {
      const int kf = kk + l4 * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        const v8 a = *reinterpret_cast<const v8*>(&As[wm + fm * 16 + l15][kf]);
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          const v8 b = *reinterpret_cast<const v8*>(&Bs[wn + fn * 16 + l15][kf]);
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm][fn], 0, 0, 0);
        }
      }
    }
    if (!last) {
      __syncthreads();
      store_stage();
      __syncthreads();
    }
  }

#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
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
  auto stream = at::hip::getCurrentHIPStream();
  const float* bptr = nullptr;
  torch::Tensor bf;
  if (epi) {
    TORCH_CHECK(bias.has_value(), "epilogue needs bias");
    bf = bias->to(torch::kFloat32).contiguous();
    bptr = bf.data_ptr<float>();
  }
  const char* v1 = std::getenv("NPAIR_CONV1X1_V1");
  if (v1 && v1[0] == '1') {
    dim3 grid((N + C1_BN - 1) / C1_BN, (M + C1_BM - 1) / C1_BM);
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
  } else {
    dim3 grid((N + C2_BN - 1) / C2_BN, (M + C2_BM - 1) / C2_BM);
    if (epi)
      conv1x1_nt_v2_kernel<1><<<grid, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const unsigned short*>(A.data_ptr()),
          reinterpret_cast<const unsigned short*>(B.data_ptr()), bptr,
          reinterpret_cast<unsigned short*>(Y.data_ptr()), (int)M, (int)N, (int)K);
    else
      conv1x1_nt_v2_kernel<0><<<grid, NPAIR_BLOCK, 0, stream>>>(
          reinterpret_cast<const unsigned short*>(A.data_ptr()),
          reinterpret_cast<const unsigned short*>(B.data_ptr()), nullptr,
          reinterpret_cast<unsigned short*>(Y.data_ptr()), (int)M, (int)N, (int)K);
  }
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
