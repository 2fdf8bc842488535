// fp32 MFMA GEMM for the loss-path matmuls (gfx950).
//
// gfx950 has exact f32-in/f32-accumulate MFMA (v_mfma_f32_16x16x4_f32) at
// the 157 TF f32 vector peak — the right instrument for the reference's
// fp32 cuBLAS GEMMs (similarity .cu:218; gradient GEMMs .cu:448-460) at
// identical numerics (bit-equal to an fmaf chain).  The backward's six
// GEMMs collapse to two here because bwd_weights emits the combined
// (-p1+p2+p3) matrix.
//
// One templated kernel covers the three layouts used:
//   sim_nt : S[B,G]  = F_l[B,D]  @ F_g[G,D]^T    (TB)
//   nn     : dF_l[B,D] = W[B,G]  @ F_g[G,D]
//   tn     : dF_g[G,D] = W[B,G]^T @ F_l[B,D]     (TA)
// Tile: 64x64 block, 4 waves of 32x32 (2x2 of 16x16x4 fragments), BK=16,
// LDS-staged operands in canonical [k][m] / [k][n] layout.
// Correctness-first v1; these GEMMs are ~1 GFLOP at the production shapes
// (microseconds) — the fused B x G kernels dominate the loss-path time.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

#define GEMM_BM 64
#define GEMM_BN 64
#define GEMM_BK 16

template <bool TA, bool TB>
__launch_bounds__(NPAIR_BLOCK)
__global__ void gemm_f32_kernel(const float* __restrict__ A,
                                const float* __restrict__ B,
                                float* __restrict__ C,
                                int M, int N, int K, float alpha) {
  // +1 row padding: unpadded 64-float rows put the 4 staging writes of
  // each lane quartet on ONE bank (4-way conflict; PMC showed ~2M conflict
  // cycles vs 0.5M MFMA instructions per dispatch)
  __shared__ float As[GEMM_BK][GEMM_BM + 1];
  __shared__ float Bs[GEMM_BK][GEMM_BN + 1];
  const int m0 = blockIdx.y * GEMM_BM;
  const int n0 = blockIdx.x * GEMM_BN;
  const int t = threadIdx.x;
  const int wid = t / WAVE;
  const int lane = t % WAVE;
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += GEMM_BK) {
    // ---- stage A tile: As[k][m] = opA(m0+m, k0+k)
    if (!TA) {
      // A is M x K row-major: float4 along k, coalesced
      const int km = (t & 3) * 4;
      const int m = t >> 2;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      const int kg = k0 + km;
      if (m0 + m < M) {
        if (kg + 3 < K) {
          v = *reinterpret_cast<const float4*>(A + (size_t)(m0 + m) * K + kg);
        } else {
          float tmp[4] = {0.f, 0.f, 0.f, 0.f};
          for (int e = 0; e < 4; ++e)
            if (kg + e < K) tmp[e] = A[(size_t)(m0 + m) * K + kg + e];
          v = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
        }
      }
      As[km + 0][m] = v.x;
      As[km + 1][m] = v.y;
      As[km + 2][m] = v.z;
      As[km + 3][m] = v.w;
    } else {
      // A is K x M row-major: coalesced along m
      const int m = t & 63;
      for (int kk = t >> 6; kk < GEMM_BK; kk += 4) {
        float v = 0.f;
        if (k0 + kk < K && m0 + m < M) v = A[(size_t)(k0 + kk) * M + m0 + m];
        As[kk][m] = v;
      }
    }
    // ---- stage B tile: Bs[k][n] = opB(k0+k, n0+n)
    if (!TB) {
      // B is K x N row-major: coalesced along n
      const int n = t & 63;
      for (int kk = t >> 6; kk < GEMM_BK; kk += 4) {
        float v = 0.f;
        if (k0 + kk < K && n0 + n < N) v = B[(size_t)(k0 + kk) * N + n0 + n];
        Bs[kk][n] = v;
      }
    } else {
      // B is N x K row-major: float4 along k, coalesced
      const int km = (t & 3) * 4;
      const int n = t >> 2;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      const int kg = k0 + km;
      if (n0 + n < N) {
        if (kg + 3 < K) {
          v = *reinterpret_cast<const float4*>(B + (size_t)(n0 + n) * K + kg);
        } else {
          float tmp[4] = {0.f, 0.f, 0.f, 0.f};
          for (int e = 0; e < 4; ++e)
            if (kg + e < K) tmp[e] = B[(size_t)(n0 + n) * K + kg + e];
          v = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
        }
      }
      Bs[km + 0][n] = v.x;
      Bs[km + 1][n] = v.y;
      Bs[km + 2][n] = v.z;
      Bs[km + 3][n] = v.w;
    }
    __syncthreads();

    // ---- MFMA inner loop: lane l holds A[i=l&15][k=l>>4], B[k=l>>4][j=l&15]
#pragma unroll
    for (int k4 = 0; k4 < GEMM_BK / 4; ++k4) {
      const int kl = k4 * 4 + l4;
      const float a0 = As[kl][wm + l15];
      const float a1 = As[kl][wm + 16 + l15];
      const float b0 = Bs[kl][wn + l15];
      const float b1 = Bs[kl][wn + 16 + l15];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C/D fragment map col=lane&15, row=(lane>>4)*4+reg
#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gm = m0 + wm + fm * 16 + l4 * 4 + r;
        const int gn = n0 + wn + fn * 16 + l15;
        if (gm < M && gn < N) C[(size_t)gm * N + gn] = alpha * acc[fm][fn][r];
      }
}

static torch::Tensor gemm_dispatch(torch::Tensor A, torch::Tensor B, int M,
                                   int N, int K, bool ta, bool tb,
                                   double alpha) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32 && A.is_contiguous());
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kFloat32 && B.is_contiguous());
  auto C = torch::empty({M, N}, A.options());
  dim3 grid((N + GEMM_BN - 1) / GEMM_BN, (M + GEMM_BM - 1) / GEMM_BM);
  auto stream = at::hip::getCurrentHIPStream();
  if (!ta && !tb)
    gemm_f32_kernel<false, false><<<grid, NPAIR_BLOCK, 0, stream>>>(
        A.data_ptr<float>(), B.data_ptr<float>(), C.data_ptr<float>(), M, N, K, (float)alpha);
  else if (!ta && tb)
    gemm_f32_kernel<false, true><<<grid, NPAIR_BLOCK, 0, stream>>>(
        A.data_ptr<float>(), B.data_ptr<float>(), C.data_ptr<float>(), M, N, K, (float)alpha);
  else if (ta && !tb)
    gemm_f32_kernel<true, false><<<grid, NPAIR_BLOCK, 0, stream>>>(
        A.data_ptr<float>(), B.data_ptr<float>(), C.data_ptr<float>(), M, N, K, (float)alpha);
  else
    gemm_f32_kernel<true, true><<<grid, NPAIR_BLOCK, 0, stream>>>(
        A.data_ptr<float>(), B.data_ptr<float>(), C.data_ptr<float>(), M, N, K, (float)alpha);
  HIP_CHECK_LAST();
  return C;
}

// S = F_l @ F_g^T  (similarity, .cu:218)
torch::Tensor sim_gemm_nt(torch::Tensor F_l, torch::Tensor F_g) {
  TORCH_CHECK(F_l.size(1) == F_g.size(1), "feature dims must match");
  return gemm_dispatch(F_l, F_g, F_l.size(0), F_g.size(0), F_l.size(1),
                       /*ta=*/false, /*tb=*/true, 1.0);
}

// C = alpha * A @ B   (dF_local = W @ F_g)
torch::Tensor gemm_nn(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.size(1) == B.size(0));
  return gemm_dispatch(A, B, A.size(0), B.size(1), A.size(1), false, false, 1.0);
}

// C = alpha * A^T @ B  (dF_total = W^T @ F_l)
torch::Tensor gemm_tn(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.size(0) == B.size(0));
  return gemm_dispatch(A, B, A.size(1), B.size(1), A.size(0), true, false, 1.0);
}
