"""HTTP inference server tests (fastapi TestClient, CPU)."""

import json

import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402


@pytest.fixture()
def client(small_model, tokenizer):
    from luminaai_amd.inference.server import create_app
    app = create_app(small_model.eval(), tokenizer)
    return TestClient(app)


def test_health_and_models(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"
    r = client.get("/v1/models")
    assert r.json()["data"][0]["id"] == "luminaai-amd"


def test_completions(client):
    r = client.post("/v1/completions",
                    json={"prompt": "hello", "max_tokens": 4,
                          "temperature": 0.0})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "text_completion"
    assert isinstance(body["choices"][0]["text"], str)
    assert body["usage"]["completion_tokens"] <= 4
    assert body["usage"]["total_tokens"] == (
        body["usage"]["prompt_tokens"] + body["usage"]["completion_tokens"])


def test_chat_completions(client):
    r = client.post("/v1/chat/completions",
                    json={"messages": [{"role": "user", "content": "hi"}],
                          "max_tokens": 4, "temperature": 0.0})
    assert r.status_code == 200
    body = r.json()
    assert body["choices"][0]["message"]["role"] == "assistant"
    assert isinstance(body["choices"][0]["message"]["content"], str)


def test_completions_streaming(client):
    with client.stream("POST", "/v1/completions",
                       json={"prompt": "abc", "max_tokens": 3,
                             "temperature": 0.0, "stream": True}) as r:
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/event-stream")
        events = []
        for line in r.iter_lines():
            if line.startswith("data: "):
                events.append(line[6:])
    assert events[-1] == "[DONE]"
    for e in events[:-1]:
        chunk = json.loads(e)
        assert "choices" in chunk


def test_sampling_params_reach_engine(client):
    # deterministic greedy twice -> identical output
    a = client.post("/v1/completions", json={"prompt": "xyz", "max_tokens": 5,
                                             "temperature": 0.0}).json()
    b = client.post("/v1/completions", json={"prompt": "xyz", "max_tokens": 5,
                                             "temperature": 0.0}).json()
    assert a["choices"][0]["text"] == b["choices"][0]["text"]
    # request counter advanced
    h = client.get("/health").json()
    assert h["requests"] >= 2
