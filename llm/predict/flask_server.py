"""HTTP serving over the predictor (reference: llm/predict/flask_server.py).

The reference wraps its predictor in a Flask `/api/chat` endpoint with
optional streaming and a gradio UI (:67-199).  This framework serves the
same contract over FastAPI/uvicorn (the ASGI stack in the MI355X image;
Flask and gradio are not) — `POST /api/chat` with
{"context": ..., "history": [...]} returning {"result": ...}, plus
streaming via chunked responses, `/health`, and multi-rank broadcast so
TP-sharded predictors answer from rank 0 while peers execute the same
batch (reference broadcast_msg :107).
"""
import argparse
import json
from dataclasses import dataclass
from typing import List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

__all__ = ["ServerArgument", "PredictorServer", "build_app"]


@dataclass
class ServerArgument:
    port: int = 8011
    flask_port: Optional[int] = None          # reference arg name, kept
    title: str = "paddlenlp-amd serving"
    stream: bool = False


class PredictorServer:
    """Wraps any predictor exposing `.predict(list[str]) -> list[str]`."""

    def __init__(self, args: ServerArgument, predictor):
        self.args = args
        self.predictor = predictor
        self.app = build_app(self)

    def predict(self, texts: List[str]) -> List[str]:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            # rank 0 owns the HTTP socket; broadcast the batch so every
            # TP rank runs the same forward (reference broadcast_msg)
            obj = [texts]
            dist.broadcast_object_list(obj, src=0)
            texts = obj[0]
        return self.predictor.predict(texts)

    def serve_forever(self):
        import uvicorn

        uvicorn.run(self.app, host="0.0.0.0",
                    port=self.args.flask_port or self.args.port)


def build_app(server: "PredictorServer"):
    app = FastAPI(title=server.args.title)

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.post("/api/chat")
    async def chat(request: Request):
        data = await request.json()
        context = data.get("context") or data.get("src") or ""
        history = data.get("history") or []
        if history:
            # fold history turns into the prompt (reference behavior)
            turns = [t if isinstance(t, str) else json.dumps(t)
                     for t in history]
            context = "\n".join(turns + [context])
        if not context:
            return JSONResponse({"error": "empty context"}, status_code=400)
        if data.get("stream", server.args.stream):
            def gen():
                out = server.predict([context])[0]
                # chunked emission of the final text (token-level
                # streaming needs the device scheduler's stream hook)
                for i in range(0, len(out), 16):
                    yield json.dumps(
                        {"result": out[i:i + 16], "done": False}) + "\n"
                yield json.dumps({"result": "", "done": True}) + "\n"

            return StreamingResponse(gen(), media_type="application/json")
        out = server.predict([context])[0]
        return {"result": out}

    return app


def main():
    from .predictor import PredictorArgument, create_predictor

    p = argparse.ArgumentParser()
    p.add_argument("--model_name_or_path", type=str, required=True)
    p.add_argument("--port", type=int, default=8011)
    p.add_argument("--dtype", type=str, default="bfloat16")
    p.add_argument("--max_length", type=int, default=512)
    args = p.parse_args()

    pargs = PredictorArgument(model_name_or_path=args.model_name_or_path,
                              dtype=args.dtype, max_length=args.max_length)
    predictor = create_predictor(pargs)
    PredictorServer(ServerArgument(port=args.port), predictor).serve_forever()


if __name__ == "__main__":
    main()
