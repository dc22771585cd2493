"""HTTP serving layer satisfying the reference's container contract:
port 8080, 200 OK on "/" when ready, POST /v1/completions with optional
SSE streaming (reference docs/container-contract.md "Server"; exercised
by the reference system test test/system.sh:70-77; the reference's
basaran image exposes the same OpenAI-style surface).

Requests are submitted to the Engine's continuous-batching scheduler; a
single background thread drives engine.step() so concurrent HTTP
requests decode together in one batch.
"""
from __future__ import annotations

import json
import logging
import queue
import threading
import time
import uuid

from fastapi import FastAPI
from fastapi.responses import JSONResponse, StreamingResponse
from pydantic import BaseModel

from .engine import Engine
from .tokenizer import load_tokenizer

# per-request latency logging (SURVEY.md §5 tracing: the reference has
# none; the serving runtime logs one structured line per request)
logger = logging.getLogger("runbooks_amd.serve")

# Prometheus metrics are process-global: created once here so a second
# build_app() in the same process (tests, uvicorn reload) cannot trip
# "Duplicated timeseries".
try:
    from prometheus_client import (
        CONTENT_TYPE_LATEST,
        Counter,
        Gauge,
        Histogram,
        generate_latest,
    )
    M_REQS = Counter("rb_requests_total", "completion requests")
    M_TOKENS = Counter("rb_generated_tokens_total", "generated tokens")
    M_LAT = Histogram("rb_request_seconds", "request latency",
                      buckets=(.05, .1, .25, .5, 1, 2.5, 5, 10, 30, 60))
    M_RUNNING = Gauge("rb_running_requests", "requests decoding")
    M_KV_FREE = Gauge("rb_kv_blocks_free", "free KV cache blocks")
    M_ENGINE = Gauge("rb_engine_stat", "engine counters", ["stat"])
except ImportError:  # pragma: no cover
    M_REQS = M_TOKENS = M_LAT = M_RUNNING = M_KV_FREE = M_ENGINE = None


class CompletionRequest(BaseModel):
    prompt: str = ""
    max_tokens: int = 16
    temperature: float = 0.0
    top_p: float = 1.0
    n: int = 1              # choices per prompt (non-stream)
    best_of: int | None = None  # sample best_of, keep the n most likely
    echo: bool = False      # prepend the prompt to each choice
    logprobs: int | None = None  # top-k logprobs per generated token
    seed: int | None = None      # reproducible sampling
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    stop: list[str] | str | None = None
    model: str = ""
    stream: bool = False

    def stop_list(self) -> list[str]:
        if self.stop is None:
            return []
        return [self.stop] if isinstance(self.stop, str) else list(self.stop)


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatRequest(BaseModel):
    messages: list[ChatMessage]
    max_tokens: int = 16
    temperature: float = 0.0
    top_p: float = 1.0
    n: int = 1
    seed: int | None = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    stop: list[str] | str | None = None
    model: str = ""
    stream: bool = False


class EngineLoop:
    """Background thread driving the continuous-batching engine."""

    def __init__(self, engine: Engine):
        self.engine = engine
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._watchers: dict[int, queue.Queue] = {}  # request_id -> token q
        self._failures = 0
        self._stop = False
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def submit(self, prompt_ids, max_new_tokens, temperature,
               top_p=1.0, logprobs=None, seed=None,
               presence_penalty=0.0, frequency_penalty=0.0,
               repetition_penalty=1.0,
               stop_token_ids=()) -> tuple[queue.Queue, "object"]:
        """Returns (queue yielding token_id | None, engine Request)."""
        q: queue.Queue = queue.Queue()
        with self._lock:
            req = self.engine.submit(prompt_ids, max_new_tokens, temperature,
                                     top_p=top_p, logprobs=logprobs,
                                     seed=seed,
                                     presence_penalty=presence_penalty,
                                     frequency_penalty=frequency_penalty,
                                     repetition_penalty=repetition_penalty,
                                     stop_token_ids=stop_token_ids)
            self._watchers[req.request_id] = q
            req._watch_sent = 0
        self._wake.set()
        return q, req

    def cancel(self, request_id: int) -> None:
        with self._lock:
            self.engine.cancel(request_id)

    def _run(self):
        while not self._stop:
            if not self.engine.has_work():
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            with self._lock:
                try:
                    finished = self.engine.step()
                    self._failures = 0
                except Exception:
                    # a rejected request is dropped from the queues —
                    # notify its watcher; on repeated failures (a request
                    # that keeps crashing the step) fail everything rather
                    # than spin
                    self._failures += 1
                    if self._failures >= 3:
                        for r in (self.engine.running +
                                  self.engine.waiting):
                            r.finished = True
                            if r.blocks:
                                self.engine.allocator.release(r.blocks)
                                r.blocks = []
                        self.engine.running.clear()
                        self.engine.waiting.clear()
                    alive = {r.request_id for r in
                             self.engine.running + self.engine.waiting}
                    for rid in list(self._watchers):
                        if rid not in alive:
                            self._watchers.pop(rid).put(None)
                    continue
                live = list(self.engine.running) + finished
                for r in live:
                    q = self._watchers.get(r.request_id)
                    if q is None:
                        continue
                    sent = getattr(r, "_watch_sent", 0)
                    for t in r.output_ids[sent:]:
                        q.put(t)
                    r._watch_sent = len(r.output_ids)
                for r in finished:
                    q = self._watchers.pop(r.request_id, None)
                    if q is not None:
                        q.put(None)

    def shutdown(self):
        self._stop = True
        self._wake.set()
        self._thread.join(timeout=2)
        from ..utils.trace import dump_global
        p = dump_global()
        if p:
            logger.info("trace written to %s", p)


def build_app(engine: Engine, tokenizer=None,
              model_name: str = "model",
              max_queue: int = 512) -> FastAPI:
    import contextlib

    engine.max_prefills_per_step = max(engine.max_prefills_per_step, 4)
    loop = EngineLoop(engine)

    @contextlib.asynccontextmanager
    async def _lifespan(app):
        yield
        loop.shutdown()

    app = FastAPI(title="runbooks-amd-server", lifespan=_lifespan)
    tok = tokenizer or load_tokenizer(None)
    # finish at the tokenizer's EOS (real checkpoints emit it; the byte
    # fallback has none). The EOS itself is filtered from the text by the
    # decoded-piece diffing, and the engine marks the request finished.
    eos = getattr(tok, "eos_id", None)
    eos_ids = (eos,) if eos is not None else ()
    app.state.engine_loop = loop

    # Prometheus metrics (the reference exposes controller metrics behind
    # kube-rbac-proxy; the serving runtime gets request/token counters and
    # latency histograms — SURVEY.md §5 observability)
    m_reqs, m_tokens, m_lat = M_REQS, M_TOKENS, M_LAT
    if M_REQS is not None:
        @app.get("/metrics")
        def metrics():
            M_RUNNING.set(len(engine.running))
            M_KV_FREE.set(len(engine.allocator.free))
            for k, v in engine.stats.items():
                M_ENGINE.labels(stat=k).set(v)
            from fastapi import Response
            return Response(generate_latest(),
                            media_type=CONTENT_TYPE_LATEST)

    app.state.metrics = (m_reqs, m_tokens, m_lat)

    @app.get("/")
    def ready():
        return {"status": "ok", "model": model_name}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model",
                          "owned_by": "runbooks-amd"}]}

    @app.get("/v1/models/{model_id}")
    def model_info(model_id: str):
        if model_id != model_name:
            return JSONResponse(status_code=404, content={"error": {
                "message": f"model '{model_id}' not found"}})
        return {"id": model_name, "object": "model",
                "owned_by": "runbooks-amd"}

    def _usage(n_prompt, n_out):
        return {"prompt_tokens": n_prompt, "completion_tokens": n_out,
                "total_tokens": n_prompt + n_out}

    def _stop_hit(text: str, stops):
        for s in stops:
            i = text.find(s)
            if i >= 0:
                return text[:i]
        return None

    def _encode(prompt: str, max_tokens: int) -> list[int]:
        # keep at most max_seq_len - max_tokens - 1 prompt tokens so
        # prompt + generation never exceeds the model's positional range
        # (keep >= 1: a naive negative slice of 0 would keep EVERYTHING)
        keep = max(1, engine.cfg.max_seq_len - max_tokens - 1)
        ids = tok.encode(prompt)[-keep:]
        # guard: a fallback tokenizer may emit ids past a small model's
        # vocab (identity for any properly paired tokenizer)
        return [i % engine.cfg.vocab_size for i in ids]

    def _pieces(q, rid, stops, state=None):
        """Incremental decoded text pieces for one request; drains the
        queue after a stop-sequence hit so cancel cleanup stays local.
        ``state`` (optional dict) gets {"finish": "stop"|"length"}."""
        out, sent = [], ""
        if state is None:
            state = {}
        state["finish"] = "length"
        while True:
            t = q.get()
            if t is None:
                return
            if t in eos_ids:
                state["finish"] = "stop"
                continue  # engine finishes on EOS; don't stream it
            out.append(t)
            text = tok.decode(out)
            trunc = _stop_hit(text, stops) if stops else None
            if trunc is not None:
                loop.cancel(rid)
                state["finish"] = "stop"
                piece = trunc[len(sent):]
                if piece:
                    yield piece
                while q.get() is not None:
                    pass
                return
            piece = text[len(sent):]
            sent += piece
            if piece:
                yield piece

    def _collect(q, rid, stops):
        out = []
        finish = "length"
        text = ""
        while True:
            t = q.get()
            if t is None:
                break
            out.append(t)
            if stops:
                trunc = _stop_hit(tok.decode(out), stops)
                if trunc is not None:
                    loop.cancel(rid)
                    text, finish = trunc, "stop"
                    while q.get() is not None:
                        pass
                    break
        if finish != "stop" and out and out[-1] in eos_ids:
            # engine stopped at the tokenizer's EOS: report "stop" and
            # keep the EOS itself out of the decoded text
            finish = "stop"
            text = tok.decode(out[:-1])
        elif finish != "stop":
            text = tok.decode(out)
        return text, finish, out

    def _fmt_logprobs(ereq, ids: list[int]):
        """OpenAI-style logprobs block for one finished choice. `ids` are
        the tokens actually delivered (stop-truncation; preemptions move
        early output into prompt_ids, so ereq.output_ids can be shorter —
        logprob_data spans the full delivered stream)."""
        data = ereq.logprob_data[:len(ids)]
        toks = [tok.decode([t]) for t in ids]
        offs, pos = [], 0
        for t in toks:
            offs.append(pos)
            pos += len(t)
        def _tok_str(i):
            t = tok.decode([i])
            # undecodable bytes all render as U+FFFD and would collide
            # as dict keys; use an explicit byte form instead
            return t if t and "\ufffd" not in t else f"bytes:0x{i:x}"

        top = None
        if data and "top" in data[0]:
            top = [{_tok_str(i): lp for i, lp in d["top"]} for d in data]
        return {"tokens": toks,
                "token_logprobs": [d["logprob"] for d in data],
                "top_logprobs": top,
                "text_offset": offs}

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        # clamp generation to what the context window can hold
        req.max_tokens = max(1, min(req.max_tokens,
                                    engine.cfg.max_seq_len - 1))
        if len(engine.waiting) >= max_queue:
            return JSONResponse(status_code=429, content={"error": {
                "message": "server overloaded: request queue full"}})
        ids = _encode(req.prompt, req.max_tokens)
        cid = f"cmpl-{uuid.uuid4().hex[:12]}"
        t0 = int(time.time())
        stops = req.stop_list()
        if req.stream and (req.n > 1 or (req.best_of or 1) > 1):
            return JSONResponse(status_code=400, content={"error": {
                "message": "n > 1 / best_of > 1 are not supported with "
                           "stream=true"}})
        # all candidates submitted up front so they decode as one batch;
        # per-request sampling noise is row-independent (ops/sampling.py).
        # best_of > n: sample best_of candidates with per-token logprobs
        # and keep the n with the highest mean logprob (OpenAI semantics).
        n = max(1, min(req.n, 16))
        n_sample = max(n, min(req.best_of or n, 16))
        want_lp = req.logprobs if req.logprobs is not None else (
            0 if n_sample > n else None)
        subs = [loop.submit(list(ids), req.max_tokens, req.temperature,
                            top_p=req.top_p, logprobs=want_lp,
                            seed=(None if req.seed is None
                                  else req.seed + i),
                            presence_penalty=req.presence_penalty,
                            frequency_penalty=req.frequency_penalty,
                            repetition_penalty=req.repetition_penalty,
                            stop_token_ids=eos_ids)
                for i in range(n_sample)]
        q, ereq = subs[0]

        if req.stream:
            def gen():
                def chunk(text):
                    c = {"id": cid, "object": "text_completion",
                         "created": t0, "model": req.model or model_name,
                         "choices": [{"text": text, "index": 0,
                                      "logprobs": None,
                                      "finish_reason": None}]}
                    return f"data: {json.dumps(c)}\n\n"
                try:
                    if req.echo and req.prompt:
                        yield chunk(req.prompt)
                    state = {}
                    for piece in _pieces(q, ereq.request_id, stops, state):
                        yield chunk(piece)
                    fin = {"id": cid, "object": "text_completion",
                           "created": t0,
                           "model": req.model or model_name,
                           "choices": [{"text": "", "index": 0,
                                        "logprobs": None,
                                        "finish_reason": state["finish"]}]}
                    yield f"data: {json.dumps(fin)}\n\n"
                    yield "data: [DONE]\n\n"
                finally:
                    # client disconnect abandons the generator mid-stream:
                    # stop generating for it (no-op after normal finish)
                    loop.cancel(ereq.request_id)
            return StreamingResponse(gen(), media_type="text/event-stream")

        collected = []
        n_out = 0
        for cq, creq_ in subs:
            text, finish, out_ids = _collect(cq, creq_.request_id, stops)
            if req.echo:
                text = req.prompt + text
            lp = (_fmt_logprobs(creq_, out_ids)
                  if req.logprobs is not None else None)
            mean_lp = (sum(d["logprob"] for d in
                           creq_.logprob_data[:len(out_ids)])
                       / max(1, len(out_ids))
                       if creq_.logprob_data else 0.0)
            collected.append((mean_lp, text, finish, lp, len(out_ids)))
            n_out += len(out_ids)
        if n_sample > n:
            collected.sort(key=lambda c: c[0], reverse=True)
            collected = collected[:n]
        choices = [{"text": c[1], "index": i, "logprobs": c[3],
                    "finish_reason": c[2]}
                   for i, c in enumerate(collected)]
        if m_reqs is not None:
            m_reqs.inc()
            m_tokens.inc(n_out)
            m_lat.observe(time.time() - t0)
        logger.info(
            "completion id=%s prompt_tokens=%d completion_tokens=%d "
            "choices=%d latency_ms=%.1f", cid, len(ids), n_out, len(choices),
            (time.time() - t0) * 1000.0)
        return JSONResponse({
            "id": cid,
            "object": "text_completion",
            "created": t0,
            "model": req.model or model_name,
            "choices": choices,
            "usage": _usage(len(ids), n_out),
        })

    @app.post("/v1/chat/completions")
    def chat(req: ChatRequest):
        """Chat surface over the same engine: messages are flattened with
        role prefixes (no chat-template metadata in tokenizer.json /
        tokenizer.model; models loaded from /content/model use whatever
        plain-text convention they were tuned on). Reuses the tested
        completions path and reshapes the payload."""
        prompt = "\n".join(f"{m.role}: {m.content}" for m in req.messages)
        prompt += "\nassistant:"
        cid = f"chatcmpl-{uuid.uuid4().hex[:12]}"
        t0 = int(time.time())
        mdl = req.model or model_name

        if req.stream:
            if req.n > 1:
                return JSONResponse(status_code=400, content={"error": {
                    "message": "n > 1 is not supported with stream=true"}})
            max_tokens = max(1, min(req.max_tokens,
                                    engine.cfg.max_seq_len - 1))
            ids = _encode(prompt, max_tokens)
            stops = CompletionRequest(stop=req.stop).stop_list()
            q, ereq = loop.submit(ids, max_tokens, req.temperature,
                                  top_p=req.top_p, seed=req.seed,
                                  presence_penalty=req.presence_penalty,
                                  frequency_penalty=req.frequency_penalty,
                                  repetition_penalty=req.repetition_penalty,
                                  stop_token_ids=eos_ids)

            def gen():
                def chunk(delta):
                    c = {"id": cid, "object": "chat.completion.chunk",
                         "created": t0, "model": mdl,
                         "choices": [{"index": 0, "delta": delta,
                                      "finish_reason": None}]}
                    return f"data: {json.dumps(c)}\n\n"
                try:
                    yield chunk({"role": "assistant"})
                    state = {}
                    for piece in _pieces(q, ereq.request_id, stops, state):
                        yield chunk({"content": piece})
                    fin = {"id": cid, "object": "chat.completion.chunk",
                           "created": t0, "model": mdl,
                           "choices": [{"index": 0, "delta": {},
                                        "finish_reason": state["finish"]}]}
                    yield f"data: {json.dumps(fin)}\n\n"
                    yield "data: [DONE]\n\n"
                finally:
                    loop.cancel(ereq.request_id)
            return StreamingResponse(gen(), media_type="text/event-stream")

        creq = CompletionRequest(
            prompt=prompt, max_tokens=req.max_tokens,
            temperature=req.temperature, top_p=req.top_p, n=req.n,
            seed=req.seed, presence_penalty=req.presence_penalty,
            frequency_penalty=req.frequency_penalty,
            repetition_penalty=req.repetition_penalty,
            stop=req.stop, model=req.model, stream=False)
        resp = completions(creq)
        if resp.status_code != 200:
            return resp
        body = json.loads(bytes(resp.body))
        return JSONResponse({
            "id": cid,
            "object": "chat.completion",
            "created": t0,
            "model": mdl,
            "choices": [{"index": c["index"],
                         "message": {"role": "assistant",
                                     "content": c["text"]},
                         "finish_reason": c["finish_reason"]}
                        for c in body["choices"]],
            "usage": body["usage"],
        })

    return app


def serve_forever(engine: Engine, tokenizer=None, host: str = "0.0.0.0",
                  port: int = 8080, model_name: str = "model"):
    import uvicorn
    app = build_app(engine, tokenizer, model_name)
    uvicorn.run(app, host=host, port=port, log_level="info")
