"""Serving endpoint tests (tosem2021_amd/serve.py) — the corpus' ray
`serve/` concern.  Runs the FastAPI app in-process via TestClient on the
rules+lexicon backend (no checkpoint, CPU)."""
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from tosem2021_amd.serve import build_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    return TestClient(build_app(ckpt_dir=None))


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    body = r.json()
    assert body["ok"] is True
    assert body["backend"] == "rules+lexicon"


def test_classify_rules_backend(client):
    r = client.post("/classify", json={
        "texts": ["assertRaises(ValueError, fit, X)",
                  "assertAlmostEqual(accuracy, 0.96, places=2)"],
        "repos": ["auto_sklearn", "auto_sklearn"],
    })
    assert r.status_code == 200
    body = r.json()
    assert body["backend"] == "rules+lexicon"
    res = body["results"]
    assert len(res) == 2
    assert "value_error" in res[0]["strategies"]
    assert "rounding_tolence" in res[1]["strategies"]
    for item in res:
        assert item["method"] in ("unit_test", "regression", "integration",
                                  "end_to_end")
        assert isinstance(item["properties"], list)


def test_classify_length_mismatch(client):
    r = client.post("/classify", json={"texts": ["a"], "repos": ["x", "y"]})
    assert r.status_code == 200
    assert "error" in r.json()


def test_metrics_endpoint(client):
    client.post("/classify", json={"texts": ["assertTrue(ok)"]})
    r = client.get("/metrics")
    assert r.status_code == 200
    assert "serve_requests" in r.text
