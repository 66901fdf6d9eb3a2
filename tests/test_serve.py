"""Serving daemon tests (FastAPI TestClient, CPU eager path here; the
fused HIP path activates on GPU)."""

import numpy as np
import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from factorvae_amd.serve import ScoringEngine, build_app


@pytest.fixture(scope="module")
def client(tmp_path_factory):
    from factorvae_amd.models.modules import build_factorvae

    model = build_factorvae(num_latent=12, hidden_size=8, num_portfolio=4,
                            num_factor=3)
    ckpt = tmp_path_factory.mktemp("srv") / "m.pt"
    torch.save(model.state_dict(), ckpt)
    eng = ScoringEngine(str(ckpt), num_latent=12, hidden_size=8,
                        num_portfolio=4, num_factor=3, seq_length=5,
                        device="cpu")
    return TestClient(build_app(eng))


def test_health_and_model(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    m = client.get("/model").json()
    assert m["num_factor"] == 3 and m["seq_length"] == 5


def test_score_roundtrip(client):
    x = np.random.default_rng(0).standard_normal((7, 5, 12)).tolist()
    r = client.post("/score", json={"x": x})
    assert r.status_code == 200
    scores = r.json()["scores"]
    assert len(scores) == 7
    assert all(np.isfinite(scores))


def test_score_batch_and_validation(client):
    rng = np.random.default_rng(1)
    days = [rng.standard_normal((n, 5, 12)).tolist() for n in (3, 9)]
    r = client.post("/score_batch", json={"days": days})
    assert r.status_code == 200
    out = r.json()["scores"]
    assert [len(o) for o in out] == [3, 9]
    # wrong T -> 422
    bad = rng.standard_normal((4, 6, 12)).tolist()
    r = client.post("/score", json={"x": bad})
    assert r.status_code == 422


def test_score_raw_binary_endpoint():
    """Binary fast path: float32 body in, float32 scores out."""
    import numpy as np

    from factorvae_amd.serve import ScoringEngine, build_app
    from fastapi.testclient import TestClient

    eng = ScoringEngine(None, num_latent=12, hidden_size=8,
                        num_portfolio=6, num_factor=4, seq_length=5,
                        device="cpu")
    client = TestClient(build_app(eng))
    x = np.random.randn(7, 5, 12).astype("<f4")
    r = client.post("/score_raw", content=x.tobytes())
    assert r.status_code == 200
    out = np.frombuffer(r.content, dtype="<f4")
    assert out.shape == (7,) and np.isfinite(out).all()
    # malformed length rejected
    r2 = client.post("/score_raw", content=b"abc")
    assert r2.status_code == 422


def test_daemon_real_socket_binary_roundtrip():
    """Full daemon path over a real loopback socket (uvicorn + httpx),
    not just the in-process ASGI TestClient: binary /score_raw
    roundtrip at CPU-eager speed."""
    import threading
    import time

    uvicorn = pytest.importorskip("uvicorn")
    httpx = pytest.importorskip("httpx")

    eng = ScoringEngine(None, num_latent=6, hidden_size=8, num_portfolio=4,
                        num_factor=3, seq_length=4, device="cpu")
    app = build_app(eng)
    cfg = uvicorn.Config(app, host="127.0.0.1", port=8473,
                         log_level="error")
    server = uvicorn.Server(cfg)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    try:
        for _ in range(100):
            if server.started:
                break
            time.sleep(0.05)
        else:
            pytest.skip("uvicorn did not start (port in use?)")
        x = np.random.default_rng(0).standard_normal((5, 4, 6)).astype("<f4")
        with httpx.Client(base_url="http://127.0.0.1:8473",
                          timeout=30.0) as client:
            assert client.get("/health").json()["status"] == "ok"
            r = client.post("/score_raw", content=x.tobytes())
            assert r.status_code == 200
            out = np.frombuffer(r.content, dtype="<f4")
            assert out.shape == (5,) and np.isfinite(out).all()
    finally:
        server.should_exit = True
        th.join(timeout=5)
