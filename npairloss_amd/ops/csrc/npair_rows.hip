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

template <typename T>
__global__ void rowstats_kernel(const T* __restrict__ S,
                                const int* __restrict__ lab_l,
                                const int* __restrict__ lab_g,
                                int B, int G, int rank,
                                T* __restrict__ min_within,
                                T* __restrict__ max_between,
                                T* __restrict__ max_all) {
  __shared__ T scratch[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const T* row = S + (size_t)i * G;
  T mnw = DtMax<T>::v, mxb = -DtMax<T>::v, mxa = -DtMax<T>::v;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    const T s = row[j];
    if (lab_g[j] == li) {
      mnw = mnw < s ? mnw : s;
      mxa = mxa > s ? mxa : s;
    } else {
      mxb = mxb > s ? mxb : s;
      mxa = mxa > s ? mxa : s;
    }
  }
  mnw = block_reduce(mnw, OpMinT(), DtMax<T>::v, scratch);
  mxb = block_reduce(mxb, OpMaxT(), -DtMax<T>::v, scratch);
  mxa = block_reduce(mxa, OpMaxT(), -DtMax<T>::v, scratch);
  if (threadIdx.x == 0) {
    min_within[i] = mnw;
    max_between[i] = mxb;
    max_all[i] = mxa;
  }
}

// ---------------------------------------------------------------------------
// fused forward
// ---------------------------------------------------------------------------

template <typename T>
__global__ void fused_fwd_kernel(const T* __restrict__ S,
                                 const int* __restrict__ lab_l,
                                 const int* __restrict__ lab_g,
                                 int B, int G, int rank,
                                 const T* __restrict__ thr_p,
                                 const T* __restrict__ thr_n,
                                 const T* __restrict__ max_all,
                                 T margin_ident, T margin_diff,
                                 int ap_method, int an_method,
                                 T* __restrict__ ident_num,
                                 T* __restrict__ diff_num,
                                 T* __restrict__ loss_ident,
                                 T* __restrict__ loss_sum,
                                 T* __restrict__ log_term) {
  __shared__ double scratch_d[NPAIR_BLOCK / WAVE];
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const T tp = thr_p[i] + margin_ident;
  const T tn = thr_n[i] + margin_diff;
  const T mx = max_all[i];
  const T* row = S + (size_t)i * G;
  int cnt_p = 0, cnt_n = 0;
  double sum_p = 0.0, sum_n = 0.0;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    const T s = row[j];
    const T e = kexp(s - mx);
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
    ident_num[i] = (T)cnt_p;
    diff_num[i] = (T)cnt_n;
    const double lsum = sum_p + sum_n;
    loss_ident[i] = (T)sum_p;
    loss_sum[i] = (T)lsum;
    // div + log with the reference's zero-guards (.cu:162-169)
    log_term[i] = (sum_p == 0.0 || lsum == 0.0) ? (T)0 : (T)log(sum_p / lsum);
  }
}

// ---------------------------------------------------------------------------
// backward weights
// ---------------------------------------------------------------------------

template <typename T>
__global__ void bwd_weights_kernel(const T* __restrict__ S,
                                   const int* __restrict__ lab_l,
                                   const int* __restrict__ lab_g,
                                   int B, int G, int rank,
                                   const T* __restrict__ thr_p,
                                   const T* __restrict__ thr_n,
                                   const T* __restrict__ max_all,
                                   const T* __restrict__ loss_ident,
                                   const T* __restrict__ loss_sum,
                                   T margin_ident, T margin_diff,
                                   int ap_method, int an_method,
                                   T scale,
                                   T* __restrict__ W) {
  const size_t total = (size_t)B * G;
  for (size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (size_t)gridDim.x * blockDim.x) {
    const int i = idx / G;
    const int j = idx % G;
    T w = (T)0;
    if (!pair_is_self(i, j, rank, B)) {
      const T s = S[idx];
      const T e = kexp(s - max_all[i]);
      const T li = loss_ident[i];
      const T ls = loss_sum[i];
      if (lab_g[j] == lab_l[i]) {
        if (select_pos(s, (T)(thr_p[i] + margin_ident), ap_method)) {
          // -part1 + part2 (each guarded on its OWN denominator, .cu:412-417)
          const T p1 = (li == (T)0) ? (T)0 : e / li;
          const T p2 = (ls == (T)0) ? (T)0 : e / ls;
          w = -p1 + p2;
        }
      } else {
        if (select_neg(s, (T)(thr_n[i] + margin_diff), an_method)) {
          w = (ls == (T)0) ? (T)0 : e / ls;  // part3
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

template <typename T>
__global__ void recall_kernel(const T* __restrict__ S,
                              const int* __restrict__ lab_l,
                              const int* __restrict__ lab_g,
                              int B, int G, int rank,
                              const int* __restrict__ ks, int nk, int kmax,
                              int* __restrict__ hits) {
  // one block per query row; M = min(kmax, G-2) + 1 threshold candidates
  __shared__ T cand[NPAIR_BLOCK];
  __shared__ T extracted[RECALL_MAX_TOPK + 1];
  __shared__ int scratch_i[NPAIR_BLOCK / WAVE];
  const int i = blockIdx.x;
  if (i >= B) return;
  const int li = lab_l[i];
  const T* row = S + (size_t)i * G;

  // phase 1: per-thread sorted (desc) local top-M list in registers
  T loc[RECALL_MAX_TOPK + 1];
  const int M = min(kmax, G - 2) + 1;  // threshold index is min(k, len-1), len = G-1
#pragma unroll
  for (int m = 0; m <= RECALL_MAX_TOPK; ++m) loc[m] = -DtMax<T>::v;
  for (int j = threadIdx.x; j < G; j += blockDim.x) {
    if (pair_is_self(i, j, rank, B)) continue;
    T s = row[j];
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
    cand[threadIdx.x] = (head < M) ? loc[head] : -DtMax<T>::v;
    __syncthreads();
    // tree argmax over 256 candidates
    for (int stride = NPAIR_BLOCK / 2; stride > 0; stride >>= 1) {
      if (threadIdx.x < stride)
        cand[threadIdx.x] = cand[threadIdx.x] > cand[threadIdx.x + stride]
                                ? cand[threadIdx.x] : cand[threadIdx.x + stride];
      __syncthreads();
    }
    const T winner = cand[0];
    __syncthreads();
    if (threadIdx.x == 0) extracted[m] = winner;
    // exactly ONE thread (the lowest-id holder) pops its head
    const bool mine = (head < M) && (loc[head] == winner);
    // ballot across block via LDS: find the lowest thread id holding winner
    cand[threadIdx.x] = mine ? (T)threadIdx.x : (T)NPAIR_BLOCK;
    __syncthreads();
    for (int stride = NPAIR_BLOCK / 2; stride > 0; stride >>= 1) {
      if (threadIdx.x < stride)
        cand[threadIdx.x] = cand[threadIdx.x] < cand[threadIdx.x + stride]
                                ? cand[threadIdx.x] : cand[threadIdx.x + stride];
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
    const T thr = extracted[min(ti, M - 1)];
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
  TORCH_CHECK(S.is_cuda() && S.is_contiguous() &&
                  (S.dtype() == torch::kFloat32 || S.dtype() == torch::kFloat64),
              "S must be contiguous fp32/fp64 on GPU");
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
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "rowstats", [&] {
    rowstats_kernel<scalar_t><<<B, NPAIR_BLOCK, 0, stream>>>(
        S.data_ptr<scalar_t>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
        (int)rank, mnw.data_ptr<scalar_t>(), mxb.data_ptr<scalar_t>(),
        mxa.data_ptr<scalar_t>());
  });
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
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "fused_fwd", [&] {
    fused_fwd_kernel<scalar_t><<<B, NPAIR_BLOCK, 0, stream>>>(
        S.data_ptr<scalar_t>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
        (int)rank, thr_p.data_ptr<scalar_t>(), thr_n.data_ptr<scalar_t>(),
        max_all.data_ptr<scalar_t>(), (scalar_t)margin_ident, (scalar_t)margin_diff,
        (int)ap_method, (int)an_method, ident_num.data_ptr<scalar_t>(),
        diff_num.data_ptr<scalar_t>(), loss_ident.data_ptr<scalar_t>(),
        loss_sum.data_ptr<scalar_t>(), log_term.data_ptr<scalar_t>());
  });
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
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "bwd_weights", [&] {
    bwd_weights_kernel<scalar_t><<<blocks, NPAIR_BLOCK, 0, stream>>>(
        S.data_ptr<scalar_t>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
        (int)rank, thr_p.data_ptr<scalar_t>(), thr_n.data_ptr<scalar_t>(),
        max_all.data_ptr<scalar_t>(), loss_ident.data_ptr<scalar_t>(),
        loss_sum.data_ptr<scalar_t>(), (scalar_t)margin_ident, (scalar_t)margin_diff,
        (int)ap_method, (int)an_method, (scalar_t)scale, W.data_ptr<scalar_t>());
  });
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
  AT_DISPATCH_FLOATING_TYPES(S.scalar_type(), "recall_hits", [&] {
    recall_kernel<scalar_t><<<B, NPAIR_BLOCK, 0, stream>>>(
        S.data_ptr<scalar_t>(), lab_l.data_ptr<int>(), lab_g.data_ptr<int>(), B, G,
        (int)rank, ks_t.data_ptr<int>(), nk, (int)kmax, hits.data_ptr<int>());
  });
  HIP_CHECK_LAST();
  return hits;
}
