"""In-process test of the embedding service (flreid_amd/serve.py)."""

import pytest


def test_embed_endpoint():
    fastapi = pytest.importorskip("fastapi")
    from starlette.testclient import TestClient

    from flreid_amd.serve import build_model, create_app

    model = build_model("fedstil", {
        "name": "resnet18", "num_classes": 16, "last_stride": 1,
        "neck": "bnneck", "fine_tuning": ["classifier"],
        "atten_default": 0.9, "lambda_k": 8}, None, "cpu")
    app = create_app(model, "cpu")
    client = TestClient(app)

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"

    import torch
    imgs = torch.randn(2, 3, 64, 32).tolist()
    r = client.post("/embed", json={"images": imgs})
    assert r.status_code == 200
    feats = r.json()["features"]
    assert len(feats) == 2 and len(feats[0]) == 512
    # L2-normalized
    import math
    assert abs(sum(v * v for v in feats[0]) - 1.0) < 1e-3

    # malformed payload -> 400
    r = client.post("/embed", json={"images": [[1.0, 2.0]]})
    assert r.status_code == 400
