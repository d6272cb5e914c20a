"""Inference server tests (CPU, llama_test, FastAPI TestClient)."""
import pytest

fastapi = pytest.importorskip("fastapi")


@pytest.fixture(scope="module")
def client():
    from fastapi.testclient import TestClient

    from prime_amd.models import build_model
    from prime_amd.serve import create_app

    m = build_model("llama_test")
    m.eval()
    return TestClient(create_app(m, "llama_test", tokenizer=None))


def test_health_and_models(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["model"] == "llama_test"
    r = client.get("/v1/models")
    assert r.status_code == 200
    ids = [m["id"] for m in r.json()["data"]]
    assert "llama_test" in ids and "intellect_10b" in ids


def test_completion_token_ids(client):
    r = client.post("/v1/completions", json={
        "prompt": [1, 2, 3, 4], "max_tokens": 8, "temperature": 0.0,
    })
    assert r.status_code == 200, r.text
    d = r.json()
    assert d["usage"]["completion_tokens"] == 8
    assert len(d["choices"][0]["text"]) == 8  # token ids (no tokenizer)
    # greedy decode is deterministic
    r2 = client.post("/v1/completions", json={
        "prompt": [1, 2, 3, 4], "max_tokens": 8, "temperature": 0.0,
    })
    assert r2.json()["choices"][0]["text"] == d["choices"][0]["text"]


def test_text_prompt_requires_tokenizer(client):
    r = client.post("/v1/completions", json={"prompt": "hello", "max_tokens": 4})
    assert r.status_code == 400
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 400
