"""HTTP serving surface (FastAPI app over the engine, CPU tiny model)."""
import pytest

fastapi = pytest.importorskip("fastapi")
pytest.importorskip("httpx")

from fastapi.testclient import TestClient  # noqa: E402

from comfyui_parallelanything_amd.serve import create_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    app = create_app("flux", devices=["cpu", "cpu"], percents=[50, 50],
                     tiny=True, microbatches=2)
    with TestClient(app) as c:
        yield c


def test_healthz(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "ok"
    assert body["devices"] == ["cpu", "cpu"]


def test_generate_batch2(client):
    r = client.post("/generate", json={"batch": 2, "steps": 2, "seed": 7})
    assert r.status_code == 200
    body = r.json()
    assert body["finite"] is True
    assert body["shape"][0] == 2
    assert body["images_per_s"] > 0


def test_generate_deterministic_by_seed(client):
    a = client.post("/generate", json={"batch": 1, "steps": 2, "seed": 3}).json()
    b = client.post("/generate", json={"batch": 1, "steps": 2, "seed": 3}).json()
    assert a["mean"] == b["mean"] and a["std"] == b["std"]


def test_generate_validation(client):
    assert client.post("/generate", json={"batch": 0}).status_code == 400
    assert client.post(
        "/generate", json={"batch": 1, "sampler": "nope"}
    ).status_code == 400


def test_stats_after_generate(client):
    r = client.get("/stats")
    assert r.status_code == 200
    assert r.json()["steps"] >= 1
