"""HTTP serving layer satisfying the reference's container contract:
port 8080, 200 OK on "/" when ready, POST /v1/completions
(reference docs/container-contract.md "Server"; exercised by the
reference system test test/system.sh:70-77).
"""
from __future__ import annotations

import asyncio
import threading
import time
import uuid

from fastapi import FastAPI
from fastapi.responses import JSONResponse
from pydantic import BaseModel

from .engine import Engine
from .tokenizer import load_tokenizer


class CompletionRequest(BaseModel):
    prompt: str = ""
    max_tokens: int = 16
    temperature: float = 0.0
    model: str = ""
    stream: bool = False


def build_app(engine: Engine, tokenizer=None, model_name: str = "model") -> FastAPI:
    app = FastAPI(title="runbooks-amd-server")
    tok = tokenizer or load_tokenizer(None)
    lock = threading.Lock()

    @app.get("/")
    def ready():
        return {"status": "ok", "model": model_name}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest):
        ids = tok.encode(req.prompt)[-engine.cfg.max_seq_len + req.max_tokens + 1:]
        loop = asyncio.get_event_loop()

        def run():
            with lock:
                return engine.generate(ids, max_new_tokens=req.max_tokens,
                                       temperature=req.temperature)
        t0 = time.time()
        out_ids = await loop.run_in_executor(None, run)
        text = tok.decode(out_ids)
        return JSONResponse({
            "id": f"cmpl-{uuid.uuid4().hex[:12]}",
            "object": "text_completion",
            "created": int(t0),
            "model": req.model or model_name,
            "choices": [{"text": text, "index": 0, "logprobs": None,
                         "finish_reason": "length"}],
            "usage": {"prompt_tokens": len(ids),
                      "completion_tokens": len(out_ids),
                      "total_tokens": len(ids) + len(out_ids)},
        })

    return app


def serve_forever(engine: Engine, tokenizer=None, host: str = "0.0.0.0",
                  port: int = 8080, model_name: str = "model"):
    import uvicorn
    app = build_app(engine, tokenizer, model_name)
    uvicorn.run(app, host=host, port=port, log_level="info")
