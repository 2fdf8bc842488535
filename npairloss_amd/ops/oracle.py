"""NumPy oracle for the complete multi-class N-pair loss math.

This is the executable specification of the reference layer's semantics
(quziyan/NPairLoss npair_multi_class_loss.cu), implemented independently in
NumPy.  Every kernel and the HIP compute path are tested element-wise
against this module.

Semantics reproduced, with reference citations:
- similarity S = F_local(B x D) @ F_total(G x D)^T, alpha = 1
  (dot_normalizer = 1 forward, .cu:216-218)
- same/diff 0/1 masks with the SELF pair excluded from both via
  `local_idx + rank*B != global_idx` (.cu:45-66, esp. :54)
- per-query stats min_within / max_between / max_all and the ascending-
  sorted global + per-query positive/negative similarity lists (.cu:225-273)
- (region x method) -> threshold selection with the relative-index
  arithmetic and the value<0 -> -inf clamp (.cu:275-337)
- the 5-way select rules per (method, pos/neg) incl. RAND = select ALL
  (.cu:69-122)
- stable log-sum-exp via exp(s - rowmax_all), zero-guards for empty
  selections, uniform weighting (the /identNum divisions are commented out
  in the reference, .cu:139,149) (.cu:124-171, 355-388)
- loss = -(1/B) sum_q log( P_q / (P_q + N_q) )
- backward: the three softmax-weight parts with 0-guards (.cu:405-446),
  dF_local = (-p1+p2+p3) @ F_total * (lw/B), dF_total = (...)^T @ F_local
  * (lw/B) (dot_normalizer = B backward, .cu:427,448-460), sum over ranks +
  1/NUM_GPU scale (.cu:462-489), final 0.5*local + 0.5*total[rank slice]
  (.cu:490-498)
- retrieval Recall@k: per query, all G-1 non-self sims sorted descending,
  threshold = element at index min(k, len-1), hit iff some non-self entry
  is STRICTLY greater than the threshold and label-matched (.cu:173-206).
  The reference evaluates this on exp(s - rowmax) (its `_calPrecision`
  copy, taken BEFORE the zero-guards, .cu:132); exp is strictly monotone so
  we evaluate on S directly — identical ordering up to fp rounding of exp.
- feature_asum = L1(bottom features) / B (.cu:400-401)

Deviations (reference behavior is undefined there, documented choices):
- empty mining list for a RELATIVE threshold, or out-of-range relative
  index: reference indexes out of bounds (UB). We clamp the index to
  [0, len-1] and treat an empty list as threshold -inf.
- GLOBAL absolute thresholds read list extrema (.cu:296,327); empty list is
  likewise UB in the reference -> we use -inf (pos) / +inf (neg) making the
  comparisons vacuous in the same direction as "no information".
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np

from ..config.params import MiningMethod, MiningRegion, NPairLossConfig

NEG_INF = float("-inf")
POS_INF = float("inf")


# ---------------------------------------------------------------------------
# masks + stats
# ---------------------------------------------------------------------------

def label_masks(labels_local: np.ndarray, labels_global: np.ndarray, rank: int) -> Tuple[np.ndarray, np.ndarray]:
    """same/diff boolean masks, self-pair excluded from both (.cu:45-66)."""
    B = labels_local.shape[0]
    G = labels_global.shape[0]
    eq = labels_local[:, None] == labels_global[None, :]
    not_self = np.ones((B, G), dtype=bool)
    idx = rank * B + np.arange(B)
    valid = idx < G
    not_self[np.arange(B)[valid], idx[valid]] = False
    same = eq & not_self
    diff = (~eq) & not_self
    return same, diff


@dataclass
class MiningStats:
    min_within: np.ndarray    # (B,) min positive sim per query (+inf if none)
    max_between: np.ndarray   # (B,) max negative sim per query (-inf if none)
    max_all: np.ndarray       # (B,) max sim over same|diff per query (-inf if none)
    ident_global: np.ndarray  # ascending sorted positive sims over the whole B x G
    diff_global: np.ndarray   # ascending sorted negative sims
    ident_local: List[np.ndarray]  # per query, ascending sorted
    diff_local: List[np.ndarray]


def mining_stats(S: np.ndarray, same: np.ndarray, diff: np.ndarray) -> MiningStats:
    """Per-query and global statistics + sorted lists (.cu:225-273).

    The reference initializes min_within to +FLT_MAX, max_between/max_all to
    -FLT_MAX (.cu:230-236); we use +-inf which behaves identically in every
    comparison downstream.
    """
    B = S.shape[0]
    Sw = np.where(same, S, POS_INF)
    min_within = Sw.min(axis=1)
    Sb = np.where(diff, S, NEG_INF)
    max_between = Sb.max(axis=1)
    Sa = np.where(same | diff, S, NEG_INF)
    max_all = Sa.max(axis=1)

    ident_local = [np.sort(S[i][same[i]], kind="stable") for i in range(B)]
    diff_local = [np.sort(S[i][diff[i]], kind="stable") for i in range(B)]
    ident_global = np.sort(S[same], kind="stable")
    diff_global = np.sort(S[diff], kind="stable")
    return MiningStats(min_within, max_between, max_all, ident_global, diff_global, ident_local, diff_local)


# ---------------------------------------------------------------------------
# threshold selection
# ---------------------------------------------------------------------------

def relative_index(sn: float, size: int) -> int:
    """The reference's relative-position arithmetic (.cu:285-287 etc.):
    sn >= 0 -> size-1-int(sn); sn < 0 -> int(size-1 + sn*size).  The C
    expression is `size_t - 1 + float * size_t`, whose usual arithmetic
    conversions promote everything to FLOAT32 — e.g. sn=-0.3, size=10
    gives 9 + (-3.0000001f) = 5.9999999f -> 5, not 6.  Truncation toward
    zero; we clamp to the valid range (reference is UB out of range)."""
    if size <= 0:
        return -1
    if sn >= 0:  # note: -0.0 >= 0 is True, matching C
        pos = size - 1 - int(sn)
    else:
        pos = int(np.float32(np.float32(size - 1) + np.float32(sn) * np.float32(size)))
    return min(max(pos, 0), size - 1)


def _relative_threshold(sorted_list: np.ndarray, sn: float) -> float:
    pos = relative_index(sn, len(sorted_list))
    if pos < 0:
        return NEG_INF
    v = float(sorted_list[pos])
    # value < 0 => -FLT_MAX i.e. select-all clamp (.cu:288,303,319,334).
    # NOTE: v == 0.0 passes (>= 0).
    return v if v >= 0 else NEG_INF


def select_thresholds(stats: MiningStats, cfg: NPairLossConfig, B: int) -> Tuple[np.ndarray, np.ndarray]:
    """(region, method) -> per-query AP and AN threshold vectors (.cu:275-337)."""
    thr_p = np.empty(B, dtype=np.float64)
    thr_n = np.empty(B, dtype=np.float64)
    rel = (MiningMethod.RELATIVE_HARD, MiningMethod.RELATIVE_EASY)

    # AP (positive-pair) thresholds
    if cfg.ap_mining_region == MiningRegion.LOCAL:
        if cfg.ap_mining_method not in rel:
            thr_p[:] = stats.max_between  # per-query max negative sim (.cu:277-280)
        else:
            for i in range(B):
                thr_p[i] = _relative_threshold(stats.ident_local[i], cfg.identsn)
    else:  # GLOBAL
        if cfg.ap_mining_method not in rel:
            # global max NEGATIVE sim (.cu:293-297 reads diff_prod_global_list.back())
            thr_p[:] = stats.diff_global[-1] if len(stats.diff_global) else NEG_INF
        else:
            thr_p[:] = _relative_threshold(stats.ident_global, cfg.identsn)

    # AN (negative-pair) thresholds
    if cfg.an_mining_region == MiningRegion.LOCAL:
        if cfg.an_mining_method not in rel:
            thr_n[:] = stats.min_within  # per-query min positive sim (.cu:308-311)
        else:
            for i in range(B):
                thr_n[i] = _relative_threshold(stats.diff_local[i], cfg.diffsn)
    else:  # GLOBAL
        if cfg.an_mining_method not in rel:
            # global min POSITIVE sim (.cu:324-328 reads ident_prod_global_list[0])
            thr_n[:] = stats.ident_global[0] if len(stats.ident_global) else POS_INF
        else:
            thr_n[:] = _relative_threshold(stats.diff_global, cfg.diffsn)

    return thr_p, thr_n


# ---------------------------------------------------------------------------
# pair selection
# ---------------------------------------------------------------------------

def select_pairs(
    S: np.ndarray,
    same: np.ndarray,
    diff: np.ndarray,
    thr_p: np.ndarray,
    thr_n: np.ndarray,
    cfg: NPairLossConfig,
) -> np.ndarray:
    """The 0/1 is_select_pair matrix (.cu:69-122)."""
    tp = thr_p[:, None] + cfg.margin_ident
    tn = thr_n[:, None] + cfg.margin_diff
    m = cfg.ap_mining_method
    if m == MiningMethod.HARD:
        sel_p = S < tp
    elif m == MiningMethod.EASY:
        sel_p = S >= tp
    elif m == MiningMethod.RAND:
        sel_p = np.ones_like(same)
    elif m == MiningMethod.RELATIVE_HARD:
        sel_p = S <= tp
    else:  # RELATIVE_EASY
        sel_p = S >= tp
    m = cfg.an_mining_method
    if m == MiningMethod.HARD:
        sel_n = S > tn
    elif m == MiningMethod.EASY:
        sel_n = S <= tn
    elif m == MiningMethod.RAND:
        sel_n = np.ones_like(diff)
    elif m == MiningMethod.RELATIVE_HARD:
        sel_n = S >= tn
    else:  # RELATIVE_EASY
        sel_n = S <= tn
    return np.where(same, sel_p, np.where(diff, sel_n, False))


# ---------------------------------------------------------------------------
# forward
# ---------------------------------------------------------------------------

@dataclass
class ForwardResult:
    loss: float
    recall: dict                  # k -> recall@k over the local queries
    feature_asum: float
    # intermediates (the workspace the backward + tests need)
    S: np.ndarray                 # B x G similarity
    same: np.ndarray
    diff: np.ndarray
    sel: np.ndarray               # selected-pair mask
    thr_p: np.ndarray
    thr_n: np.ndarray
    max_all: np.ndarray
    ident_num: np.ndarray         # selected positive count per query
    diff_num: np.ndarray
    loss_ident: np.ndarray        # P_q  = sum exp over selected positives
    loss_sum: np.ndarray          # P_q + N_q
    log_term: np.ndarray          # log(P/(P+N)) with 0-guards
    stats: MiningStats


def npair_forward(
    features_local: np.ndarray,
    labels_local: np.ndarray,
    features_global: np.ndarray,
    labels_global: np.ndarray,
    cfg: NPairLossConfig,
    rank: int = 0,
    top_k: Tuple[int, ...] = (1, 5, 10),
) -> ForwardResult:
    """Full reference forward for one rank (.cu:207-402).

    features_global/labels_global are the all-gathered node batch (rank r's
    rows at [r*B, (r+1)*B)); with one rank they are just the local batch.
    """
    F_l = np.asarray(features_local, dtype=np.float64)
    F_g = np.asarray(features_global, dtype=np.float64)
    B = F_l.shape[0]

    S = F_l @ F_g.T  # .cu:218, alpha = 1
    same, diff = label_masks(np.asarray(labels_local), np.asarray(labels_global), rank)
    stats = mining_stats(S, same, diff)
    thr_p, thr_n = select_thresholds(stats, cfg, B)
    sel = select_pairs(S, same, diff, thr_p, thr_n, cfg)

    sel_p = same & sel
    sel_n = diff & sel
    ident_num = sel_p.sum(axis=1).astype(np.float64)
    diff_num = sel_n.sum(axis=1).astype(np.float64)

    # stable exp; rows with no same|diff entries have max_all = -inf: the
    # reference would exp(s + FLT_MAX) = inf there, but every such entry is
    # masked out of the loss anyway; guard to keep the oracle finite.
    finite_max = np.where(np.isfinite(stats.max_all), stats.max_all, 0.0)
    E = np.exp(S - finite_max[:, None])

    loss_ident = (E * sel_p).sum(axis=1)
    loss_diff = (E * sel_n).sum(axis=1)
    loss_sum = loss_ident + loss_diff
    with np.errstate(divide="ignore", invalid="ignore"):
        div = np.where((loss_ident == 0) | (loss_sum == 0), 0.0, loss_ident / loss_sum)
        log_term = np.where(div == 0, 0.0, np.log(np.where(div > 0, div, 1.0)))
    loss = -float(log_term.sum()) / B

    recall = {k: retrieval_recall(S, np.asarray(labels_local), np.asarray(labels_global), rank, k) for k in top_k}
    feature_asum = float(np.abs(F_l).sum()) / B

    return ForwardResult(
        loss=loss,
        recall=recall,
        feature_asum=feature_asum,
        S=S,
        same=same,
        diff=diff,
        sel=sel,
        thr_p=thr_p,
        thr_n=thr_n,
        max_all=stats.max_all,
        ident_num=ident_num,
        diff_num=diff_num,
        loss_ident=loss_ident,
        loss_sum=loss_sum,
        log_term=log_term,
        stats=stats,
    )


# ---------------------------------------------------------------------------
# retrieval metric
# ---------------------------------------------------------------------------

def retrieval_recall(
    S: np.ndarray,
    labels_local: np.ndarray,
    labels_global: np.ndarray,
    rank: int,
    top_k: int,
) -> float:
    """Recall@k exactly per GetRetrivePerformance (.cu:173-206): descending
    sort of the G-1 non-self sims, threshold = (k+1)-th largest (index
    min(k, len-1)), hit iff exists non-self j with sim STRICTLY > threshold
    and matching label."""
    B, G = S.shape
    hits = 0
    for i in range(B):
        self_idx = rank * B + i
        mask = np.ones(G, dtype=bool)
        if self_idx < G:
            mask[self_idx] = False
        vals = S[i][mask]
        if len(vals) == 0:
            continue
        svals = np.sort(vals)[::-1]
        thr = svals[min(top_k, len(svals) - 1)]
        hit = ((S[i] > thr) & (labels_local[i] == labels_global) & mask).any()
        hits += bool(hit)
    return hits / B


# ---------------------------------------------------------------------------
# backward
# ---------------------------------------------------------------------------

@dataclass
class BackwardResult:
    grad_local: np.ndarray        # the final bottom[0] gradient (B x D)
    dF_local: np.ndarray          # local-role gradient before combine
    dF_total: np.ndarray          # database-role gradient (G x D), pre-allreduce
    W: np.ndarray                 # (-p1+p2+p3) weight matrix


def npair_backward_local(
    fwd: ForwardResult,
    features_local: np.ndarray,
    features_global: np.ndarray,
    loss_weight: float = 1.0,
) -> BackwardResult:
    """Single-rank backward (.cu:420-460): produces this rank's dF_local and
    its contribution to dF_total.  Cross-rank combine is in
    `npair_backward_combine`."""
    F_l = np.asarray(features_local, dtype=np.float64)
    F_g = np.asarray(features_global, dtype=np.float64)
    B = F_l.shape[0]

    finite_max = np.where(np.isfinite(fwd.max_all), fwd.max_all, 0.0)
    E = np.exp(fwd.S - finite_max[:, None])
    P = E * (fwd.same & fwd.sel)
    N = E * (fwd.diff & fwd.sel)

    with np.errstate(divide="ignore", invalid="ignore"):
        p1 = np.where(fwd.loss_ident[:, None] == 0, 0.0, P / np.where(fwd.loss_ident[:, None] == 0, 1.0, fwd.loss_ident[:, None]))
        p2 = np.where(fwd.loss_sum[:, None] == 0, 0.0, P / np.where(fwd.loss_sum[:, None] == 0, 1.0, fwd.loss_sum[:, None]))
        p3 = np.where(fwd.loss_sum[:, None] == 0, 0.0, N / np.where(fwd.loss_sum[:, None] == 0, 1.0, fwd.loss_sum[:, None]))
    W = -p1 + p2 + p3
    scale = loss_weight / B  # dot_normalizer = B in backward (.cu:427)
    dF_local = scale * (W @ F_g)
    dF_total = scale * (W.T @ F_l)
    return BackwardResult(grad_local=np.zeros_like(F_l), dF_local=dF_local, dF_total=dF_total, W=W)


def npair_backward_combine(
    bwd: BackwardResult,
    dF_total_sum: np.ndarray,
    num_gpu: int,
    rank: int,
) -> np.ndarray:
    """All-reduced database grad -> final bottom grad (.cu:462-498):
    grad = 0.5 * dF_local + 0.5 * (sum_r dF_total_r / NUM_GPU)[rank slice]."""
    B = bwd.dF_local.shape[0]
    total = np.asarray(dF_total_sum, dtype=np.float64) / num_gpu
    sl = total[rank * B : (rank + 1) * B]
    grad = 0.5 * sl + 0.5 * bwd.dF_local
    bwd.grad_local = grad
    return grad


def npair_loss_multirank(
    features: np.ndarray,
    labels: np.ndarray,
    cfg: NPairLossConfig,
    num_gpu: int,
    loss_weight: float = 1.0,
    top_k: Tuple[int, ...] = (1, 5, 10),
):
    """Simulate the full num_gpu-rank computation on one monolithic batch:
    features (G x D) is split into num_gpu rank-local batches of B rows.
    Returns (per-rank ForwardResult list, per-rank grad list (each B x D)).

    This is the single-process multi-rank harness invariant of SURVEY.md
    section 4: rank-local outputs must equal slices of this computation.
    """
    G, D = features.shape
    assert G % num_gpu == 0
    B = G // num_gpu
    fwds = []
    bwds = []
    for r in range(num_gpu):
        fl = features[r * B : (r + 1) * B]
        ll = labels[r * B : (r + 1) * B]
        fwd = npair_forward(fl, ll, features, labels, cfg, rank=r, top_k=top_k)
        bwd = npair_backward_local(fwd, fl, features, loss_weight=loss_weight)
        fwds.append(fwd)
        bwds.append(bwd)
    dF_total_sum = np.sum([b.dF_total for b in bwds], axis=0)
    grads = [npair_backward_combine(bwds[r], dF_total_sum, num_gpu, r) for r in range(num_gpu)]
    return fwds, grads


# ---------------------------------------------------------------------------
# L2 normalize (the implied upstream layer, def.prototxt:115-120)
# ---------------------------------------------------------------------------

def l2_normalize(x: np.ndarray, eps: float = 1e-12) -> np.ndarray:
    n = np.sqrt((np.asarray(x, dtype=np.float64) ** 2).sum(axis=1, keepdims=True))
    return x / np.maximum(n, eps)


def l2_normalize_backward(x: np.ndarray, dy: np.ndarray, eps: float = 1e-12) -> np.ndarray:
    x = np.asarray(x, dtype=np.float64)
    dy = np.asarray(dy, dtype=np.float64)
    n = np.sqrt((x ** 2).sum(axis=1, keepdims=True))
    n = np.maximum(n, eps)
    y = x / n
    return (dy - y * (y * dy).sum(axis=1, keepdims=True)) / n
