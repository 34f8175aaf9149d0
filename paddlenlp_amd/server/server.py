"""SimpleServer: FastAPI wrapper over a predictor/taskflow.

Reference behavior: paddlenlp/server (SimpleServer FastAPI wrapper).
"""
from __future__ import annotations

from typing import Callable

from ..utils.log import logger


class SimpleServer:
    def __init__(self):
        try:
            from fastapi import FastAPI
        except ImportError as e:
            raise RuntimeError("SimpleServer requires fastapi") from e
        self.app = FastAPI(title="paddlenlp_amd SimpleServer")
        self._register_health()

    def _register_health(self):
        @self.app.get("/health")
        def health():
            return {"status": "ok"}

    def register(self, path: str, handler: Callable, methods=("POST",)):
        """Register a prediction handler: handler(dict) -> dict."""
        # NOTE: `from __future__ import annotations` turns the Request
        # annotation into a string FastAPI resolves against module globals —
        # a request-object endpoint signature would silently degrade to a
        # query param.  Read the body via the raw starlette scope instead.
        from starlette.requests import Request as _Req

        async def endpoint(request):
            data = await request.json()
            return handler(data)

        endpoint.__annotations__ = {"request": _Req}
        self.app.add_api_route(path, endpoint, methods=list(methods))

    def register_taskflow(self, path: str, taskflow):
        def handler(data):
            text = data.get("data") or data.get("text")
            return {"result": taskflow(text)}

        self.register(path, handler)

    def register_predictor(self, path: str, predictor):
        def handler(data):
            texts = data.get("data") or [data.get("text", "")]
            if isinstance(texts, str):
                texts = [texts]
            return {"result": predictor.predict(texts)}

        self.register(path, handler)

    def run(self, host: str = "0.0.0.0", port: int = 8189, workers: int = 1):
        import uvicorn

        logger.info(f"SimpleServer listening on {host}:{port}")
        uvicorn.run(self.app, host=host, port=port, workers=workers)

    def register_openai_compat(self, predictor, model_name: str = "default"):
        """Minimal OpenAI-style /v1/completions endpoint over a predictor."""

        def handler(data):
            prompt = data.get("prompt", "")
            prompts = [prompt] if isinstance(prompt, str) else list(prompt)
            outs = predictor.predict(prompts)
            return {
                "object": "text_completion",
                "model": data.get("model", model_name),
                "choices": [
                    {"index": i, "text": o, "finish_reason": "stop"}
                    for i, o in enumerate(outs)
                ],
            }

        self.register("/v1/completions", handler)
