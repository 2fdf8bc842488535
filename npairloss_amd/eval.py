"""Offline embedding extraction + retrieval evaluation.

The reference only reports ONLINE Recall@k over the training/TEST batch
(its retrieval tops double as eval metrics).  This module
adds the offline equivalent a metric-learning user needs: embed a whole
gallery, then Recall@k over all query-vs-gallery pairs — using the same
(k+1)-th-threshold / strict-> semantics as the online metric so numbers
are comparable, evaluated via the recall HIP kernel on GPU.
"""

from __future__ import annotations

from typing import Dict, Iterable, Sequence, Tuple

import torch

from .ops import _backend


@torch.no_grad()
def extract_embeddings(model: torch.nn.Module, loader: Iterable,
                       device=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Run the model over a loader; returns (features N x D, labels N)."""
    device = device or next(model.parameters()).device
    was_training = model.training
    model.eval()
    feats, labels = [], []
    for images, labs in loader:
        f = model(images.to(device, non_blocking=True))
        feats.append(f.float())
        labels.append(labs.to(device))
    if was_training:
        model.train()
    return torch.cat(feats), torch.cat(labels)


@torch.no_grad()
def recall_at_k(features: torch.Tensor, labels: torch.Tensor,
                ks: Sequence[int] = (1, 5, 10),
                chunk: int = 1024) -> Dict[int, float]:
    """Recall@k of every sample against the rest of the gallery (self
    excluded), inner-product similarity, reference semantics (threshold =
    (k+1)-th largest, strict >).  Chunked over queries so the full N x N
    similarity never materializes."""
    N = features.shape[0]
    labels = labels.to(features.device)
    total = torch.zeros(len(ks), dtype=torch.float64)
    use_hip = features.is_cuda and _backend.has_extension()
    lab32 = labels.to(torch.int32) if use_hip else labels
    if use_hip:
        ks_t = torch.tensor(list(ks), dtype=torch.int32, device=features.device)
    for q0 in range(0, N, chunk):
        q1 = min(q0 + chunk, N)
        # rank trick: rows [q0, q1) of the N x N matrix == "rank" q0/chunk
        # only when chunk divides q0; pass explicit rank offset by building
        # the B x N block and excluding self via rank*B + i == j with
        # rank = q0 // B, B = q1-q0 — valid because q0 is a multiple of chunk
        # and the last block's self indices are still q0 + i.
        B = q1 - q0
        S = features[q0:q1] @ features.t()
        if use_hip:
            C = _backend.ext()
            # self index for row i is q0 + i = rank*B + i requires rank*B == q0:
            # true for full chunks; for the ragged last chunk use rank s.t.
            # rank*B == q0 only if divisible — otherwise fall back to torch.
            if q0 % B == 0:
                hits = C.recall_hits(S.contiguous(), lab32[q0:q1].contiguous(),
                                     lab32, q0 // B, ks_t, max(ks))
                total += hits.double().cpu()
                continue
        r = _recall_torch_block(S, labels[q0:q1], labels, q0, ks)
        total += r
    return {k: float(total[i] / N) for i, k in enumerate(ks)}


def _recall_torch_block(S, lab_q, lab_g, q0, ks):
    B, G = S.shape
    device = S.device
    not_self = torch.ones(B, G, dtype=torch.bool, device=device)
    rows = torch.arange(B, device=device)
    not_self[rows, q0 + rows] = False
    masked = torch.where(not_self, S, torch.full_like(S, float("-inf")))
    sorted_desc, _ = masked.sort(dim=1, descending=True)
    eq = lab_q.view(-1, 1) == lab_g.view(1, -1)
    out = torch.zeros(len(ks), dtype=torch.float64)
    for i, k in enumerate(ks):
        pos = min(k, G - 2)
        thr = sorted_desc[:, pos]
        hit = ((S > thr.unsqueeze(1)) & eq & not_self).any(dim=1)
        out[i] = float(hit.sum())
    return out
