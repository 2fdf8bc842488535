"""Embedding / retrieval serving (FastAPI).

Production-facing inference for a trained metric-learning model: load a
checkpoint (.pt snapshot or .caffemodel), index a gallery of embeddings,
then serve

    POST /embed   {"images": [[...CHW floats...], ...]}          -> embeddings
    POST /search  {"embeddings": [[...D floats...], ...], "k": 5} -> top-k ids
    POST /index   {"embeddings": [...], "labels": [...]}          -> add to gallery
    GET  /healthz                                                 -> status

Retrieval uses inner-product similarity over the L2-normalized gallery
(matmul + topk on the serving device).  Start with:

    python -m npairloss_amd.serve --checkpoint snap.pt --port 8000

NOTE: no `from __future__ import annotations` here — PEP 563 string
annotations break FastAPI's resolution of the request models defined
inside build_app (they'd silently become query parameters).
"""

import argparse
import threading
from typing import List, Optional

import torch

try:
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel
    _HAVE_FASTAPI = True
except Exception:  # noqa: BLE001
    _HAVE_FASTAPI = False


class GalleryIndex:
    """Thread-safe in-memory gallery of L2-normalized embeddings."""

    def __init__(self, dim: int, device: Optional[torch.device] = None):
        self.dim = dim
        self.device = device or torch.device("cuda" if torch.cuda.is_available() else "cpu")
        self._feats = torch.empty(0, dim, device=self.device)
        self._labels: List[int] = []
        self._lock = threading.Lock()

    def __len__(self):
        return len(self._labels)

    def add(self, feats: torch.Tensor, labels: List[int]):
        if feats.shape[1] != self.dim:
            raise ValueError(f"expected dim {self.dim}, got {feats.shape[1]}")
        feats = torch.nn.functional.normalize(feats.to(self.device).float(), dim=1)
        with self._lock:
            self._feats = torch.cat([self._feats, feats])
            self._labels.extend(int(l) for l in labels)

    def search(self, queries: torch.Tensor, k: int = 5):
        q = torch.nn.functional.normalize(queries.to(self.device).float(), dim=1)
        with self._lock:
            if len(self._labels) == 0:
                return [[] for _ in range(q.shape[0])]
            sims = q @ self._feats.t()
            k_eff = min(k, sims.shape[1])
            vals, idx = sims.topk(k_eff, dim=1)
        out = []
        for r in range(q.shape[0]):
            out.append([
                {"label": self._labels[int(idx[r, j])],
                 "index": int(idx[r, j]),
                 "similarity": float(vals[r, j])}
                for j in range(k_eff)
            ])
        return out


def build_app(model: Optional[torch.nn.Module], dim: int,
              device: Optional[torch.device] = None):
    if not _HAVE_FASTAPI:
        raise RuntimeError("fastapi/pydantic not importable in this environment")
    device = device or torch.device("cuda" if torch.cuda.is_available() else "cpu")
    if model is not None:
        model = model.to(device).eval()
    index = GalleryIndex(dim, device)
    app = FastAPI(title="npairloss_amd embedding service")

    class EmbedReq(BaseModel):
        images: list  # B x C x H x W nested floats

    class IndexReq(BaseModel):
        embeddings: list
        labels: list

    class SearchReq(BaseModel):
        embeddings: list
        k: int = 5

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "device": str(device), "gallery_size": len(index),
                "dim": dim, "has_model": model is not None}

    @app.post("/embed")
    def embed(req: EmbedReq):
        if model is None:
            raise HTTPException(400, "no model loaded (started without --checkpoint)")
        with torch.no_grad():
            x = torch.tensor(req.images, dtype=torch.float32, device=device)
            if x.dim() != 4:
                raise HTTPException(400, f"images must be B x C x H x W, got {tuple(x.shape)}")
            f = model(x)
        return {"embeddings": f.cpu().tolist()}

    @app.post("/index")
    def add_to_index(req: IndexReq):
        f = torch.tensor(req.embeddings, dtype=torch.float32)
        if len(req.labels) != f.shape[0]:
            raise HTTPException(400, "labels/embeddings length mismatch")
        index.add(f, req.labels)
        return {"gallery_size": len(index)}

    @app.post("/search")
    def search(req: SearchReq):
        q = torch.tensor(req.embeddings, dtype=torch.float32)
        return {"results": index.search(q, k=req.k)}

    app.state.index = index
    return app


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint", default=None, help=".pt snapshot or .caffemodel")
    p.add_argument("--model", default="googlenet")
    p.add_argument("--dim", type=int, default=1024)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args(argv)

    model = None
    if args.checkpoint:
        from .models import build_embedding_model
        from .utils.caffemodel import load_caffemodel_into

        model = build_embedding_model(args.model)
        if args.checkpoint.endswith(".caffemodel"):
            load_caffemodel_into(model, args.checkpoint)
        else:
            ck = torch.load(args.checkpoint, map_location="cpu", weights_only=False)
            model.load_state_dict(ck["model"] if "model" in ck else ck)

    app = build_app(model, dim=args.dim)
    import uvicorn

    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
