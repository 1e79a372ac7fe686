"""OpenAI-compatible serving layer over LlamaEngine.

Mirrors the API surface the reference's serving examples expose and probe
(vllm_inference.py:139-213: /v1/chat/completions + /v1/models + health URL;
SSE streaming client at :263-345).  Tokenization is synthetic (hash-based) —
the benchmark contract is random-init weights, so token text is `t<id>`.
"""
from __future__ import annotations

import asyncio
import hashlib
import json
import threading
import time
import uuid
from typing import List, Optional

from .engine import LlamaEngine


class SyntheticTokenizer:
    """Deterministic text↔ids without checkpoint files."""

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size
        self.bos, self.eos = 1, 2

    def encode(self, text: str) -> List[int]:
        ids = [self.bos]
        for w in text.split():
            if w[:1] == "t" and w[1:].isdigit() and int(w[1:]) < self.vocab_size:
                ids.append(int(w[1:]))  # decode() round-trips: t<id> -> id
                continue
            h = int.from_bytes(hashlib.md5(w.encode()).digest()[:4], "little")
            ids.append(10 + h % (self.vocab_size - 10))
        return ids

    def decode(self, ids: List[int]) -> str:
        return " ".join(f"t{i}" for i in ids if i > 2)


class LLMServer:
    """Engine + background scheduler thread + request futures."""

    def __init__(self, engine: LlamaEngine, model_name: str = "llama-3-8b"):
        self.engine = engine
        self.model_name = model_name
        self.tok = SyntheticTokenizer(engine.cfg.vocab_size)
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._events: dict = {}
        self._stop = False
        self._t = threading.Thread(target=self._loop, daemon=True)
        self._t.start()

    def _loop(self):
        while not self._stop:
            with self._lock:
                has = self.engine.has_work
            if not has:
                self._wake.wait(timeout=0.02)
                self._wake.clear()
                continue
            with self._lock:
                done = self.engine.step()
            for r in done:
                if r.stream_cb is not None:
                    try:
                        r.stream_cb(None)  # end-of-stream sentinel
                    except Exception:
                        pass
                ev = self._events.pop(r.req_id, None)
                if ev is not None:
                    ev.set()

    def submit(self, prompt: str, max_tokens: int = 64, temperature: float = 0.0,
               stream_cb=None, presence_penalty: float = 0.0,
               frequency_penalty: float = 0.0, stop=None,
               logprobs: bool = False) -> int:
        ids = self.tok.encode(prompt)
        seqs = [stop] if isinstance(stop, str) else list(stop or [])
        stop_ids = [self.tok.encode(q)[1:] for q in seqs]  # drop BOS
        with self._lock:
            rid = self.engine.add_request(
                ids, max_tokens, temperature, stream_cb=stream_cb,
                presence_penalty=presence_penalty,
                frequency_penalty=frequency_penalty,
                stop_seqs=[q for q in stop_ids if q], logprobs=logprobs)
            self._events[rid] = threading.Event()
        self._wake.set()
        return rid

    def wait(self, rid: int, timeout: float = 300.0):
        ev = self._events.get(rid)
        if ev is not None:
            ev.wait(timeout)
        r = self.engine.finished.get(rid)
        if r is None:
            raise TimeoutError(f"request {rid} did not finish")
        if getattr(r, "error", None):
            raise ValueError(r.error)
        return r

    @staticmethod
    def apply_stop(text: str, stop) -> tuple:
        """OpenAI `stop` semantics: truncate BEFORE the first occurrence of
        any stop sequence; returns (text, finish_reason)."""
        if not stop:
            return text, "stop"
        seqs = [stop] if isinstance(stop, str) else list(stop)
        cut = min((text.find(q) for q in seqs if q and text.find(q) >= 0),
                  default=-1)
        if cut < 0:
            return text, "stop"
        return text[:cut], "stop"

    def generate(self, prompt: str, max_tokens: int = 64,
                 temperature: float = 0.0, stop=None,
                 presence_penalty: float = 0.0,
                 frequency_penalty: float = 0.0) -> str:
        rid = self.submit(prompt, max_tokens, temperature,
                          presence_penalty=presence_penalty,
                          frequency_penalty=frequency_penalty, stop=stop)
        r = self.wait(rid)
        toks = list(r.out_tokens)
        # engine-side early stop leaves the matched stop tokens at the
        # tail — strip them (OpenAI: stop sequence is not returned)
        for q in ([stop] if isinstance(stop, str) else list(stop or [])):
            qi = self.tok.encode(q)[1:]
            if qi and len(toks) >= len(qi) and toks[-len(qi):] == qi:
                toks = toks[: -len(qi)]
                break
        text, _ = self.apply_stop(self.tok.decode(toks), stop)
        return text

    def shutdown(self):
        self._stop = True
        self._wake.set()
        if self._t.is_alive():
            self._t.join(timeout=5)
        # drop the captured decode graph while the HIP runtime is healthy
        # (a graph destructor at interpreter teardown aborts the worker)
        close = getattr(self.engine, "close", None)
        if close is not None:
            close()


def create_openai_app(server: LLMServer):
    """FastAPI app with the OpenAI-compatible routes the reference clients use."""
    from fastapi import FastAPI
    from fastapi.responses import JSONResponse, StreamingResponse

    app = FastAPI(title="modal_examples_amd LLM server")

    # built-in chat UI (the reference's llm-frontend JS app role): static ES
    # modules streaming /v1/chat/completions SSE deltas into the page
    from pathlib import Path

    frontend = Path(__file__).parent / "chat_frontend"

    @app.get("/")
    async def index():
        from fastapi.responses import FileResponse

        return FileResponse(frontend / "index.html")

    @app.get("/chat.js")
    async def chat_js():
        from fastapi.responses import FileResponse

        return FileResponse(frontend / "chat.js", media_type="text/javascript")

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def metrics():
        """Prometheus exposition (vLLM serves /metrics; scrape or push)."""
        from fastapi.responses import PlainTextResponse

        from ...observability import metrics as M

        eng = server.engine
        M.observe("llm_running_requests", float(len(eng.running)))
        M.observe("llm_waiting_requests", float(len(eng.waiting)))
        M.observe("llm_free_kv_blocks", float(len(eng.free_blocks)))
        M.inc("llm_preemptions_total", 0)  # ensure series exists
        if eng.preemptions:
            M.observe("llm_preemptions", float(eng.preemptions))
        # engine-feature counters (vLLM exposes the same families)
        M.observe("llm_spec_tokens_proposed", float(eng.spec_proposed))
        M.observe("llm_spec_tokens_accepted", float(eng.spec_accepted))
        M.observe("llm_prefix_cache_hit_tokens", float(eng.prefix_hit_tokens))
        M.observe("llm_prefix_cache_lookup_tokens",
                  float(eng.prefix_lookup_tokens))
        return PlainTextResponse(M.render_prometheus())

    @app.get("/v1/models")
    async def models():
        return {"object": "list",
                "data": [{"id": server.model_name, "object": "model"}]}

    async def _run(prompt: str, max_tokens: int, temperature: float,
                   stream: bool, chat: bool, stop=None,
                   presence_penalty: float = 0.0,
                   frequency_penalty: float = 0.0):
        created = int(time.time())
        rid_str = f"cmpl-{uuid.uuid4().hex[:12]}"
        if not stream:
            loop = asyncio.get_running_loop()
            text = await loop.run_in_executor(
                None, lambda: server.generate(
                    prompt, max_tokens, temperature, stop=stop,
                    presence_penalty=presence_penalty,
                    frequency_penalty=frequency_penalty))
            usage = {"prompt_tokens": len(server.tok.encode(prompt)),
                     "completion_tokens": len(text.split()),
                     "total_tokens": len(server.tok.encode(prompt)) + len(text.split())}
            if chat:
                return JSONResponse({
                    "id": rid_str, "object": "chat.completion", "created": created,
                    "model": server.model_name,
                    "choices": [{"index": 0, "message": {"role": "assistant", "content": text},
                                 "finish_reason": "stop"}],
                    "usage": usage,
                })
            return JSONResponse({
                "id": rid_str, "object": "text_completion", "created": created,
                "model": server.model_name,
                "choices": [{"index": 0, "text": text, "finish_reason": "stop"}],
                "usage": usage,
            })

        loop = asyncio.get_running_loop()
        q: asyncio.Queue = asyncio.Queue()

        def cb(tok_id):
            loop.call_soon_threadsafe(q.put_nowait, tok_id)

        rid = server.submit(prompt, max_tokens, temperature, stream_cb=cb,
                            stop=stop)  # engine-side early stop

        async def gen():
            sent = 0
            acc = ""
            while True:
                try:
                    tok_id = await asyncio.wait_for(q.get(), timeout=120)
                except asyncio.TimeoutError:
                    break
                if tok_id is None:
                    break  # request finished (eos / stop / max_tokens)
                sent += 1
                piece = f"t{tok_id} "
                if stop:
                    probe, reason = LLMServer.apply_stop(acc + piece, stop)
                    if reason == "stop" and len(probe) < len(acc + piece):
                        tail = probe[len(acc):]
                        if tail:
                            payload = {"id": rid_str,
                                       "object": "chat.completion.chunk" if chat
                                       else "text_completion",
                                       "created": created,
                                       "model": server.model_name,
                                       "choices": [{"index": 0,
                                                    **({"delta": {"content": tail}}
                                                       if chat else {"text": tail}),
                                                    "finish_reason": None}]}
                            yield f"data: {json.dumps(payload)}\n\n"
                        break
                    acc += piece
                if chat:
                    payload = {"id": rid_str, "object": "chat.completion.chunk",
                               "created": created, "model": server.model_name,
                               "choices": [{"index": 0, "delta": {"content": piece},
                                            "finish_reason": None}]}
                else:
                    payload = {"id": rid_str, "object": "text_completion",
                               "created": created, "model": server.model_name,
                               "choices": [{"index": 0, "text": piece,
                                            "finish_reason": None}]}
                yield f"data: {json.dumps(payload)}\n\n"
                if tok_id == server.engine.eos_id or sent >= max_tokens:
                    break
            yield "data: [DONE]\n\n"

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.post("/v1/completions")
    async def completions(body: dict):
        n = int(body.get("n", 1) or 1)
        if n > 1 and not body.get("stream"):
            # OpenAI `n`: independent sampled completions (per-row gumbel
            # noise differs inside a batch, so parallel submits diverge)
            loop = asyncio.get_running_loop()

            def run_n():
                rids = [server.submit(body.get("prompt", ""),
                                      int(body.get("max_tokens", 64)),
                                      float(body.get("temperature", 1.0)))
                        for _ in range(n)]
                return [server.wait(rid) for rid in rids]

            rs = await loop.run_in_executor(None, run_n)
            return JSONResponse({
                "id": f"cmpl-{uuid.uuid4().hex[:12]}",
                "object": "text_completion", "model": server.model_name,
                "choices": [
                    {"index": i, "text": server.tok.decode(r.out_tokens),
                     "finish_reason": "stop"} for i, r in enumerate(rs)],
            })
        if body.get("logprobs") and not body.get("stream"):
            # eval-harness surface (vLLM serves the same field)
            loop = asyncio.get_running_loop()

            def run_lp():
                rid = server.submit(body.get("prompt", ""),
                                    int(body.get("max_tokens", 64)),
                                    float(body.get("temperature", 0.0)),
                                    logprobs=True)
                return server.wait(rid)

            r = await loop.run_in_executor(None, run_lp)
            toks = [f"t{t}" for t in r.out_tokens]
            return JSONResponse({
                "id": f"cmpl-{uuid.uuid4().hex[:12]}",
                "object": "text_completion", "model": server.model_name,
                "choices": [{"index": 0,
                             "text": server.tok.decode(r.out_tokens),
                             "logprobs": {"tokens": toks,
                                          "token_logprobs": r.out_logprobs},
                             "finish_reason": "stop"}],
            })
        return await _run(body.get("prompt", ""), int(body.get("max_tokens", 64)),
                          float(body.get("temperature", 0.0)),
                          bool(body.get("stream", False)), chat=False,
                          stop=body.get("stop"),
                          presence_penalty=float(
                              body.get("presence_penalty", 0) or 0),
                          frequency_penalty=float(
                              body.get("frequency_penalty", 0) or 0))

    @app.post("/v1/chat/completions")
    async def chat_completions(body: dict):
        msgs = body.get("messages", [])
        prompt = "\n".join(m.get("content", "") for m in msgs)
        return await _run(prompt, int(body.get("max_tokens", 64)),
                          float(body.get("temperature", 0.0)),
                          bool(body.get("stream", False)), chat=True,
                          stop=body.get("stop"),
                          presence_penalty=float(
                              body.get("presence_penalty", 0) or 0),
                          frequency_penalty=float(
                              body.get("frequency_penalty", 0) or 0))

    return app


def serve_openai(server: LLMServer, port: int = 8000, block: bool = True):
    import uvicorn

    app = create_openai_app(server)
    cfg = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="warning")
    srv = uvicorn.Server(cfg)
    if block:
        srv.run()
        return srv
    t = threading.Thread(target=srv.run, daemon=True)
    t.start()
    deadline = time.monotonic() + 20
    while not srv.started and time.monotonic() < deadline:
        time.sleep(0.05)
    return srv
