"""Serving layer tests (FastAPI TestClient, CPU)."""

import pytest
import torch

from genrec_amd.models.tiger import Tiger
from genrec_amd.serving import RecommendationService, create_app


@pytest.fixture(scope="module")
def service():
    torch.manual_seed(0)
    model = Tiger(embedding_dim=16, attn_dim=24, dropout=0.0, num_heads=4,
                  n_layers=2, num_item_embeddings=16, num_user_embeddings=50,
                  sem_id_dim=3)
    sem_ids = torch.randint(0, 16, (40, 3))
    return RecommendationService(model, sem_ids,
                                 device=torch.device("cpu"), top_k=5)


def test_recommend_batch_maps_items(service):
    recs = service.recommend_batch([1, 2], [[0, 1, 2], [3, 4]])
    assert len(recs) == 2
    for row in recs:
        assert len(row) >= 1
        for r in row:
            assert 0 <= r["item_id"] < 40
            assert len(r["sem_ids"]) == 3
            # sem ids must match the item table entry
            assert service.item_sem_ids[r["item_id"]].tolist() == r["sem_ids"]
        scores = [r["score"] for r in row]
        assert scores == sorted(scores, reverse=True)


def test_http_endpoints(service):
    from fastapi.testclient import TestClient

    app = create_app(service, window_ms=1.0)
    client = TestClient(app)
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["num_items"] == 40
    r = client.post("/recommend",
                    json={"user_id": 3, "history": [0, 1], "top_k": 3})
    assert r.status_code == 200
    body = r.json()
    assert len(body["recommendations"]) <= 3
    assert body["latency_ms"] > 0
    r = client.post("/batch_recommend",
                    json={"user_ids": [1, 2], "histories": [[0], [1, 2]],
                          "top_k": 4})
    assert r.status_code == 200
    assert len(r.json()["recommendations"]) == 2


def test_checkpoint_roundtrip(service, tmp_path):
    model = service.model
    ck = tmp_path / "tiger.pt"
    torch.save({"epoch": 1, "model": model.state_dict()}, str(ck))
    sem = tmp_path / "sem.pt"
    torch.save(service.item_sem_ids.cpu(), str(sem))
    svc2 = RecommendationService.from_checkpoint(
        str(ck), str(sem),
        model_kwargs=dict(embedding_dim=16, attn_dim=24, dropout=0.0,
                          num_heads=4, n_layers=2, num_item_embeddings=16,
                          num_user_embeddings=50, sem_id_dim=3),
        device=torch.device("cpu"))
    recs = svc2.recommend_batch([0], [[1, 2, 3]])
    assert len(recs) == 1


def test_micro_batcher_fuses_concurrent_requests(service):
    """Concurrent submits inside the window run as ONE generate call."""
    import asyncio

    from genrec_amd.serving.server import _MicroBatcher

    calls = []
    orig = service.recommend_batch

    def counting(users, hists, k):
        calls.append(len(users))
        return orig(users, hists, k)

    service.recommend_batch = counting
    try:
        batcher = _MicroBatcher(service, max_batch=64, window_ms=50.0)

        async def drive():
            return await asyncio.gather(*[
                batcher.submit(u, [1, 2, 3], 5) for u in range(6)])

        results = asyncio.run(drive())
    finally:
        service.recommend_batch = orig
    assert len(results) == 6
    assert all(len(r) == 5 for r in results)
    assert calls == [6], calls  # one fused call, not six


def test_metrics_endpoint(service):
    from fastapi.testclient import TestClient

    from genrec_amd.serving.server import create_app

    client = TestClient(create_app(service, window_ms=1.0))
    r = client.post("/recommend",
                    json={"user_id": 1, "history": [1, 2], "top_k": 3})
    assert r.status_code == 200
    m = client.get("/metrics")
    assert m.status_code == 200
    assert "genrec_requests_total" in m.text
    assert 'endpoint="recommend"' in m.text


def test_out_of_catalog_history_is_dropped(service):
    n = service.item_sem_ids.size(0)
    recs = service.recommend_batch([1], [[0, n + 50, -3, 1]], top_k=3)
    assert len(recs) == 1 and len(recs[0]) <= 3  # served, not crashed
