"""Serving API tests (FastAPI TestClient, CPU)."""

import numpy as np
import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from npairloss_amd.models.embedding import EmbeddingNet
from npairloss_amd.serve import GalleryIndex, build_app


def _client(dim=16, with_model=True):
    torch.manual_seed(0)
    model = EmbeddingNet(torch.nn.Sequential(
        torch.nn.Flatten(), torch.nn.Linear(3 * 8 * 8, dim))) if with_model else None
    app = build_app(model, dim=dim, device=torch.device("cpu"))
    return TestClient(app)


def test_healthz():
    c = _client()
    r = c.get("/healthz").json()
    assert r["status"] == "ok" and r["dim"] == 16 and r["has_model"]


def test_embed_returns_unit_vectors():
    c = _client()
    imgs = torch.randn(3, 3, 8, 8).tolist()
    r = c.post("/embed", json={"images": imgs})
    assert r.status_code == 200
    e = np.array(r.json()["embeddings"])
    assert e.shape == (3, 16)
    np.testing.assert_allclose(np.linalg.norm(e, axis=1), 1.0, rtol=1e-5)


def test_index_and_search():
    c = _client(with_model=False)
    rng = np.random.default_rng(0)
    gallery = rng.standard_normal((20, 16)).astype("float32")
    labels = (np.arange(20) % 5).tolist()
    r = c.post("/index", json={"embeddings": gallery.tolist(), "labels": labels})
    assert r.json()["gallery_size"] == 20
    # query with an exact gallery row: top-1 must be itself
    r = c.post("/search", json={"embeddings": [gallery[7].tolist()], "k": 3})
    res = r.json()["results"][0]
    assert len(res) == 3
    assert res[0]["index"] == 7
    assert res[0]["similarity"] == pytest.approx(1.0, abs=1e-5)


def test_search_empty_gallery():
    c = _client(with_model=False)
    r = c.post("/search", json={"embeddings": [[0.0] * 16], "k": 5})
    assert r.json()["results"] == [[]]


def test_gallery_index_dim_check():
    gi = GalleryIndex(8, device=torch.device("cpu"))
    with pytest.raises(ValueError):
        gi.add(torch.randn(2, 9), [0, 1])


def test_embed_without_model_400():
    c = _client(with_model=False)
    r = c.post("/embed", json={"images": torch.randn(1, 3, 8, 8).tolist()})
    assert r.status_code == 400
