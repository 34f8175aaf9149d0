"""SimpleServer routes (health, taskflow, predictor, OpenAI-compat) and the
typer CLI surface.

Reference behavior: paddlenlp/server SimpleServer; cli/main.py commands.
"""
import pytest


def test_simple_server_routes():
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from paddlenlp_amd.server import SimpleServer

    srv = SimpleServer()

    class FakePredictor:
        def predict(self, texts):
            return [t.upper() for t in texts]

    srv.register_predictor("/predict", FakePredictor())
    srv.register_taskflow("/task", lambda text: {"echo": text})
    srv.register_openai_compat(FakePredictor())
    client = TestClient(srv.app)

    assert client.get("/health").json() == {"status": "ok"}
    r = client.post("/predict", json={"data": ["hello", "world"]})
    assert r.json() == {"result": ["HELLO", "WORLD"]}
    r = client.post("/task", json={"text": "hi"})
    assert r.json() == {"result": {"echo": "hi"}}
    r = client.post("/v1/completions", json={"prompt": "abc"})
    body = r.json()
    assert body["object"] == "text_completion"
    assert body["choices"][0]["text"] == "ABC"


def test_cli_surface():
    typer = pytest.importorskip("typer")
    from typer.testing import CliRunner

    from paddlenlp_amd.cli.main import _build_app

    app = _build_app()
    runner = CliRunner()
    result = runner.invoke(app, ["search", "llama"])
    assert result.exit_code == 0
    assert "llama" in result.output.lower()
    result = runner.invoke(app, ["download", "some/model"])
    # offline environment: download explains itself rather than fetching
    assert result.exit_code != 2 or "no network" in result.output.lower() \
        or result.exit_code == 0
