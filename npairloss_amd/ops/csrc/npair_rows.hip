// Row-wise fused kernels for the N-pair loss pipeline (gfx950).
//
// Replaces the reference's per-stage B x G kernel zoo + HOST statistics
// loops (npair_multi_class_loss.cu:45-171, 225-273, 405-419) with four
// fused single-pass kernels:
//   rowstats    — min_within / max_between / max_all per query (the
//                 reference pulled the whole B x G matrix to the host and
//                 looped, .cu:225-265)
//   fused_fwd   — mask + mining-select + pair counts + stable-LSE loss
//                 sums + per-query log term in ONE pass (reference:
//                 GetLabelDiffMtx + GetSampledPairMtx + 4 muls + 4 gemvs +
//                 Minus_Querywise_Maxval + add + ManipulateDIVandLOG)
//   bwd_weights — the combined (-part1+part2+part3) softmax weight matrix
//                 in one pass (reference: 3 Get_Query_Diff_Part launches,
//                 .cu:438-446, feeding 6 GEMMs; we feed 2)
//   recall      — per-query top-(k+1) threshold + strict-> hit test for
//                 k in {1,5,10} in one kernel (reference: host D2H + B
//                 descending std::sorts per k, .cu:173-206)
//
// Label masks are recomputed on the fly from the int32 label vectors —
// the B x G same/diff/select matrices are never materialized (the
// reference kept ~10 B x G workspace blobs, .hpp:61-78).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// ---------------------------------------------------------------------------
// rowstats
// ---------------------------------------------------------------------------

__global__ void rowstats_kernel(const float* __restrict__ S,
                                const int* __restrict__ lab_l,
                                const int* __restrict__ lab_g,
                                int B, int G, int rank,
                                float* __restrict__ min_within,
                                float* __restrict__ max_between,
                                float* __restrict__ max_all) {
  __shared__ float scratch[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const float* row = S + (size_t)i * G;
  float mnw = FLT_MAX, mxb = -FLT_MAX, mxa = -FLT_MAX;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    const float s = row[j];
    if (lab_g[j] == li) {
      mnw = fminf(mnw, s);
      mxa = fmaxf(mxa, s);
    } else {
      mxb = fmaxf(mxb, s);
      mxa = fmaxf(mxa, s);
    }
  }
  mnw = block_reduce(mnw, OpMinF(), FLT_MAX, scratch);
  mxb = block_reduce(mxb, OpMaxF(), -FLT_MAX, scratch);
  mxa = block_reduce(mxa, OpMaxF(), -FLT_MAX, scratch);
  if (threadIdx.x == 0) {
    min_within[i] = mnw;
    max_between[i] = mxb;
    max_all[i] = mxa;
  }
}

// ---------------------------------------------------------------------------
// fused forward
// ---------------------------------------------------------------------------

__global__ void fused_fwd_kernel(const float* __restrict__ S,
                                 const int* __restrict__ lab_l,
                                 const int* __restrict__ lab_g,
                                 int B, int G, int rank,
                                 const float* __restrict__ thr_p,
                                 const float* __restrict__ thr_n,
                                 const float* __restrict__ max_all,
                                 float margin_ident, float margin_diff,
                                 int ap_method, int an_method,
                                 float* __restrict__ ident_num,
                                 float* __restrict__ diff_num,
                                 float* __restrict__ loss_ident,
                                 float* __restrict__ loss_sum,
                                 float* __restrict__ log_term) {
  __shared__ double scratch_d[NPAIR_BLOCK / WAVE];
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const float tp = thr_p[i] + margin_ident;
  const float tn = thr_n[i] + margin_diff;
  const float mx = max_all[i];
  const float* row = S + (size_t)i * G;
  int cnt_p = 0, cnt_n = 0;
  double sum_p = 0.0, sum_n = 0.0;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    const float s = row[j];
    const float e = __expf(s - mx);
    if (lab_g[j] == li) {
      if (select_pos(s, tp, ap_method)) {
        ++cnt_p;
        sum_p += (double)e;
      }
    } else {
      if (select_neg(s, tn, an_method)) {
        ++cnt_n;
        sum_n += (double)e;
      }
    }
  }
  cnt_p = block_reduce(cnt_p, OpAddI(), 0, scratch_i);
  cnt_n = block_reduce(cnt_n, OpAddI(), 0, scratch_i);
  sum_p = block_reduce(sum_p, OpAddD(), 0.0, scratch_d);
  sum_n = block_reduce(sum_n, OpAddD(), 0.0, scratch_d);
  if (threadIdx.x == 0) {
    ident_num[i] = (float)cnt_p;
    diff_num[i] = (float)cnt_n;
    const double lsum = sum_p + sum_n;
    loss_ident[i] = (float)sum_p;
    loss_sum[i] = (float)lsum;
    // div + log with the reference's zero-guards (.cu:162-169)
    log_term[i] = (sum_p == 0.0 || lsum == 0.0) ? 0.f : (float)log(sum_p / lsum);
  }
}

// ---------------------------------------------------------------------------
// backward weights
// ---------------------------------------------------------------------------

__global__ void bwd_weights_kernel(const float* __restrict__ S,
                                   const int* __restrict__ lab_l,
                                   const int* __restrict__ lab_g,
                                   int B, int G, int rank,
                                   const float* __restrict__ thr_p,
                                   const float* __restrict__ thr_n,
                                   const float* __restrict__ max_all,
                                   const float* __restrict__ loss_ident,
                                   const float* __restrict__ loss_sum,
                                   float margin_ident, float margin_diff,
                                   int ap_method, int an_method,
                                   float scale,
                                   float* __restrict__ W) {
  const size_t total = (size_t)B * G;
  for (size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (size_t)gridDim.x * blockDim.x) {
    const int i = idx / G;
    const int j = idx % G;
    float w = 0.f;
    if (!pair_is_self(i, j, rank, B)) {
      const float s = S[idx];
      const float e = __expf(s - max_all[i]);
      const float li = loss_ident[i];
      const float ls = loss_sum[i];
      if (lab_g[j] == lab_l[i]) {
        if (select_pos(s, thr_p[i] + margin_ident, ap_method)) {
          // -part1 + part2 (each guarded on its OWN denominator, .cu:412-417)
          const float p1 = (li == 0.f) ? 0.f : e / li;
          const float p2 = (ls == 0.f) ? 0.f : e / ls;
          w = -p1 + p2;
        }
      } else {
        if (select_neg(s, thr_n[i] + margin_diff, an_method)) {
          w = (ls == 0.f) ? 0.f : e / ls;  // part3
        }
      }
    }
    W[idx] = w * scale;
  }
}

// ---------------------------------------------------------------------------
// retrieval recall
// ---------------------------------------------------------------------------

#define RECALL_MAX_TOPK 15  // reference top list {1,5,10,15}, .cu:390-394

__global__ void recall_kernel(const float* __restrict__ S,
                              const int* __restrict__ lab_l,
                              const int* __restrict__ lab_g,
                              int B, int G, int rank,
                              const int* __restrict__ ks, int nk, int kmax,
                              int* __restrict__ hits) {
  // one block per query row; M = min(kmax, G-2) + 1 threshold candidates
  __shared__ float cand[NPAIR_BLOCK];
  __shared__ float extracted[RECALL_MAX_TOPK + 1];
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const float* row = S + (size_t)i * G;

  // phase 1: per-thread sorted (desc) local top-M list in registers
  float loc[RECALL_MAX_TOPK + 1];
  const int M = min(kmax, G - 2) + 1;  // threshold index is min(k, len-1), len = G-1
#pragma unroll
  for (int m = 0; m <= RECALL_MAX_TOPK; ++m) loc[m] = -FLT_MAX;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    float s = row[j];
    if (s <= loc[M - 1]) continue;
    // insertion into the M-length sorted list
    int p = M - 1;
    while (p > 0 && loc[p - 1] < s) {
      loc[p] = loc[p - 1];
      --p;
    }
    loc[p] = s;
  }

  // phase 2: cooperative extraction of the global top-M of the row
  int head = 0;
  for (int m = 0; m < M; ++m) {
    cand[threadIdx.x] = (head < M) ? loc[head] : -FLT_MAX;
    __syncthreads();
    // tree argmax over 256 candidates
    for (int stride = NPAIR_BLOCK / 2; stride > 0; stride >>= 1) {
      if (threadIdx.x < stride)
        cand[threadIdx.x] = fmaxf(cand[threadIdx.x], cand[threadIdx.x + stride]);
      __syncthreads();
    }
    const float winner = cand[0];
    __syncthreads();
    if (threadIdx.x == 0) extracted[m] = winner;
    // exactly ONE thread (the lowest-id holder) pops its head
    const bool mine = (head < M) && (loc[head] == winner);
    // ballot across block via LDS: find the lowest thread id holding winner
    cand[threadIdx.x] = mine ? (float)threadIdx.x : (float)NPAIR_BLOCK;
    __syncthreads();
    for (int stride = NPAIR_BLOCK / 2; stride > 0; stride >>= 1) {
      if (threadIdx.x < stride)
        cand[threadIdx.x] = fminf(cand[threadIdx.x], cand[threadIdx.x + stride]);
      __syncthreads();
    }
    if ((int)cand[0] == (int)threadIdx.x) ++head;
    __syncthreads();
  }

  // phase 3: strict-> hit test per k (.cu:190-203)
  for (int ki = 0; ki < nk; ++ki) {
    const int k = ks[ki];
    const int ti = min(k, G - 2);
    if (ti < 0) continue;  // G < 2: no retrievable database
    const float thr = extracted[min(ti, M - 1)];
    int hit = 0;
    for (int j = threadIdx.x; j < G; j += blockDim.x) {
      if (pair_is_self(i, j, rank, B)) continue;
      if (row[j] > thr && lab_g[j] == li) hit = 1;
    }
    hit = block_reduce(hit, OpAddI(), 0, scratch_i);
    if (threadIdx.x == 0 && hit > 0) atomicAdd(&hits[ki], 1);
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static void check_sg(const torch::Tensor& S, const torch::Tensor& lab_l,
                     const torch::Tensor& lab_g) {
  TORCH_CHECK(S.is_cuda() && S.dtype() == torch::kFloat32 && S.is_contiguous(),
              "S must be contiguous fp32 on GPU");
  TORCH_CHECK(lab_l.dtype() == torch::kInt32 && lab_g.dtype() == torch::kInt32,
              "labels must be int32");
  TORCH_CHECK(S.size(0) == lab_l.numel() && S.size(1) == lab_g.numel(),
              "S is B x G with B=|lab_l|, G=|lab_g|");
}

std::vector<torch::Tensor> rowstats(torch::Tensor S, torch::Tensor lab_l,
                                    torch::Tensor lab_g, int64_t rank) {
  check_sg(S, lab_l, lab_g);
  const int B = S.size(0), G = S.size(1);
  auto opts = S.options();
  auto mnw = torch::empty({B}, opts);
  auto mxb = torch::empty({B}, opts);
  auto mxa = torch::empty({B}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  rowstats_kernel<<<B, NPAIR_BLOCK, 0, stream>>>(
      S.data_ptr<float>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
      (int)rank, mnw.data_ptr<float>(), mxb.data_ptr<float>(), mxa.data_ptr<float>());
  HIP_CHECK_LAST();
  return {mnw, mxb, mxa};
}

std::vector<torch::Tensor> fused_fwd(torch::Tensor S, torch::Tensor lab_l,
                                     torch::Tensor lab_g, int64_t rank,
                                     torch::Tensor thr_p, torch::Tensor thr_n,
                                     torch::Tensor max_all, double margin_ident,
                                     double margin_diff, int64_t ap_method,
                                     int64_t an_method) {
  check_sg(S, lab_l, lab_g);
  const int B = S.size(0), G = S.size(1);
  auto opts = S.options();
  auto ident_num = torch::empty({B}, opts);
  auto diff_num = torch::empty({B}, opts);
  auto loss_ident = torch::empty({B}, opts);
  auto loss_sum = torch::empty({B}, opts);
  auto log_term = torch::empty({B}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  fused_fwd_kernel<<<B, NPAIR_BLOCK, 0, stream>>>(
      S.data_ptr<float>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
      (int)rank, thr_p.data_ptr<float>(), thr_n.data_ptr<float>(),
      max_all.data_ptr<float>(), (float)margin_ident, (float)margin_diff,
      (int)ap_method, (int)an_method, ident_num.data_ptr<float>(),
      diff_num.data_ptr<float>(), loss_ident.data_ptr<float>(),
      loss_sum.data_ptr<float>(), log_term.data_ptr<float>());
  HIP_CHECK_LAST();
  return {ident_num, diff_num, loss_ident, loss_sum, log_term};
}

torch::Tensor bwd_weights(torch::Tensor S, torch::Tensor lab_l,
                          torch::Tensor lab_g, int64_t rank,
                          torch::Tensor thr_p, torch::Tensor thr_n,
                          torch::Tensor max_all, torch::Tensor loss_ident,
                          torch::Tensor loss_sum, double margin_ident,
                          double margin_diff, int64_t ap_method,
                          int64_t an_method, double scale) {
  check_sg(S, lab_l, lab_g);
  const int B = S.size(0), G = S.size(1);
  auto W = torch::empty_like(S);
  const size_t total = (size_t)B * G;
  const int blocks = (int)min((total + NPAIR_BLOCK - 1) / NPAIR_BLOCK, (size_t)2048);
  auto stream = at::hip::getCurrentHIPStream();
  bwd_weights_kernel<<<blocks, NPAIR_BLOCK, 0, stream>>>(
      S.data_ptr<float>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
      (int)rank, thr_p.data_ptr<float>(), thr_n.data_ptr<float>(),
      max_all.data_ptr<float>(), loss_ident.data_ptr<float>(),
      loss_sum.data_ptr<float>(), (float)margin_ident, (float)margin_diff,
      (int)ap_method, (int)an_method, (float)scale, W.data_ptr<float>());
  HIP_CHECK_LAST();
  return W;
}

torch::Tensor recall_hits(torch::Tensor S, torch::Tensor lab_l,
                          torch::Tensor lab_g, int64_t rank,
                          torch::Tensor ks_t, int64_t kmax) {
  check_sg(S, lab_l, lab_g);
  const int B = S.size(0), G = S.size(1);
  TORCH_CHECK(ks_t.is_cuda() && ks_t.dtype() == torch::kInt32 && ks_t.is_contiguous());
  const int nk = ks_t.numel();
  TORCH_CHECK(nk >= 1 && nk <= 8, "1..8 k values");
  TORCH_CHECK(kmax >= 1 && kmax <= RECALL_MAX_TOPK, "k in [1,15]");
  auto hits = torch::zeros({nk}, S.options().dtype(torch::kInt32));
  auto stream = at::hip::getCurrentHIPStream();
  recall_kernel<<<B, NPAIR_BLOCK, 0, stream>>>(
      S.data_ptr<float>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
      (int)rank, ks_t.data_ptr<int>(), nk, (int)kmax, hits.data_ptr<int>());
  HIP_CHECK_LAST();
  return hits;
}
