"""Predictor HTTP serving (reference llm/predict/flask_server.py)."""
import json

from fastapi.testclient import TestClient

from llm.predict.flask_server import PredictorServer, ServerArgument


class EchoPredictor:
    def predict(self, texts):
        return [f"echo:{t}" for t in texts]


def make_client(**kw):
    server = PredictorServer(ServerArgument(**kw), EchoPredictor())
    return TestClient(server.app)


def test_chat_endpoint():
    c = make_client()
    r = c.post("/api/chat", json={"context": "hello"})
    assert r.status_code == 200
    assert r.json() == {"result": "echo:hello"}


def test_history_folding_and_health():
    c = make_client()
    assert c.get("/health").json() == {"status": "ok"}
    r = c.post("/api/chat", json={"context": "q2", "history": ["q1", "a1"]})
    assert r.json()["result"] == "echo:q1\na1\nq2"
    assert c.post("/api/chat", json={}).status_code == 400


def test_streaming_chunks():
    c = make_client()
    r = c.post("/api/chat", json={"context": "x" * 40, "stream": True})
    lines = [json.loads(l) for l in r.text.strip().splitlines()]
    assert lines[-1]["done"] is True
    joined = "".join(l["result"] for l in lines)
    assert joined == "echo:" + "x" * 40
