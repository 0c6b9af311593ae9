"""OpenAI-compatible HTTP server for the MI355X backend engine.

One server per engine (one engine per GPU; TP replicas run one server
on rank 0 of the TP group).  Serves the surface the gateway's model
sync and routing expect:

  GET  /v1/models            — model listing (sync source)
  POST /v1/chat/completions  — chat, stream + non-stream
  POST /v1/completions       — legacy completions
  GET  /api/health           — health with GPU telemetry (xLLM-style)
  GET  /metrics              — Prometheus counters

Chat messages are rendered with a trivial template and tokenized with a
deterministic hash tokenizer by default (synthetic serving: BASELINE
prescribes random-init weights + synthetic prompts; there is no network
for real tokenizer downloads).  A HF tokenizer can be plugged in when
available on disk.
"""
from __future__ import annotations

import asyncio
import hashlib
import json
import logging
import time
import uuid

from .engine import Engine
from .sampler import SamplingParams

log = logging.getLogger("llmlb.backend")


class HFTokenizer:
    """Wrap a local HuggingFace tokenizer.json (``tokenizers`` package;
    no network).  Used when the operator passes --tokenizer so real
    checkpoints serve real text; falls back to SyntheticTokenizer
    otherwise."""

    def __init__(self, path: str, vocab: int):
        from tokenizers import Tokenizer
        self.tok = Tokenizer.from_file(path)
        self.vocab = vocab

    def encode(self, text: str) -> list[int]:
        ids = [t % self.vocab for t in self.tok.encode(text).ids]
        return ids or [1]

    def decode_token(self, tok: int) -> str:
        return self.tok.decode([tok])


class SyntheticTokenizer:
    """Deterministic text<->token mapping for synthetic serving."""

    def __init__(self, vocab: int):
        self.vocab = vocab

    def encode(self, text: str) -> list[int]:
        toks = []
        for word in text.split():
            h = int.from_bytes(
                hashlib.blake2s(word.encode(), digest_size=4).digest(),
                "little")
            toks.append(h % max(self.vocab - 16, 1) + 16)
        return toks or [1]

    def decode_token(self, tok: int) -> str:
        return f"tok{tok} "


def render_chat(messages: list[dict]) -> str:
    parts = []
    for m in messages:
        content = m.get("content", "")
        if isinstance(content, list):
            content = " ".join(
                b.get("text", "") for b in content
                if isinstance(b, dict))
        parts.append(f"<|{m.get('role', 'user')}|> {content}")
    parts.append("<|assistant|>")
    return "\n".join(parts)


class BackendServer:
    def __init__(self, engine: Engine, model_name: str,
                 served_models: list[str] | None = None,
                 max_tokens_default: int = 128,
                 tokenizer_path: str | None = None):
        self.engine = engine
        self.model_name = model_name
        self.served = served_models or [model_name]
        vocab = engine.cfg.model.vocab
        self.tokenizer = (HFTokenizer(tokenizer_path, vocab)
                          if tokenizer_path
                          else SyntheticTokenizer(vocab))
        self.max_tokens_default = max_tokens_default
        self.requests_total = 0
        self.tokens_total = 0
        self.start_time = time.time()
        import threading
        self.embed_lock = threading.Lock()

    # ---------------- request handling (framework-agnostic) ----------
    def models_payload(self) -> dict:
        return {
            "object": "list",
            "data": [
                {"id": m, "object": "model", "owned_by": "llmlb_amd",
                 "max_tokens": self.engine.cfg.model.max_pos,
                 "capabilities":
                     "chat,completions,responses,embeddings"}
                for m in self.served
            ],
        }

    def health_payload(self) -> dict:
        gpu = {}
        try:
            import torch
            if torch.cuda.is_available():
                free, total = torch.cuda.mem_get_info(self.engine.device)
                gpu = {
                    "gpu_memory_total_mb": total // (1 << 20),
                    "gpu_memory_used_mb": (total - free) // (1 << 20),
                    "gpu_usage_percent": None,
                }
        except Exception:
            pass
        return {
            "status": "ok",
            "model": self.model_name,
            "active_requests": self.engine.num_running,
            "queued_requests": self.engine.waiting.qsize(),
            "tokens_per_sec": round(self.engine.tokens_per_sec_ema, 2),
            "gpu_step_ms": round(self.engine.gpu_step_ms_ema, 3),
            "spec_decode": (
                {"proposed": self.engine.spec_proposed,
                 "accepted": self.engine.spec_accepted}
                if self.engine.spec_ngram > 0 else None),
            "prefix_cache": (
                {"hit_tokens": self.engine.prefix.hit_tokens,
                 "prompt_tokens": self.engine.prefix.prompt_tokens,
                 "parked_pages": len(self.engine.prefix.lru)}
                if self.engine.prefix is not None else None),
            "uptime_s": time.time() - self.start_time,
            **gpu,
        }

    def parse_sampling(self, body: dict) -> SamplingParams:
        return SamplingParams(
            temperature=float(body.get("temperature", 0.0) or 0.0),
            top_p=float(body.get("top_p", 1.0) or 1.0),
            top_k=int(body.get("top_k", 0) or 0),
            min_p=float(body.get("min_p", 0.0) or 0.0),
            frequency_penalty=float(body.get("frequency_penalty", 0.0)
                                    or 0.0),
            presence_penalty=float(body.get("presence_penalty", 0.0)
                                   or 0.0),
            repetition_penalty=float(body.get("repetition_penalty", 1.0)
                                     or 1.0),
            max_new_tokens=int(body.get("max_tokens")
                               or body.get("max_completion_tokens")
                               or self.max_tokens_default),
            seed=body.get("seed"),
            logprobs=self._parse_logprobs(body),
            logit_bias=tuple(
                (int(k), max(-100.0, min(100.0, float(vv))))
                for k, vv in (body.get("logit_bias") or {}).items()),
        )

    @staticmethod
    def _parse_logprobs(body: dict) -> int:
        lp = body.get("logprobs")
        if lp is True:                    # chat form
            return int(body.get("top_logprobs", 0) or 0) or 1
        if isinstance(lp, int) and not isinstance(lp, bool):
            return min(lp, 20)            # completions form
        return 0

    def submit_chat(self, body: dict, sink=None):
        prompt = render_chat(body.get("messages", []))
        prompt_ids = self.tokenizer.encode(prompt)
        params = self.parse_sampling(body)
        rid, out_q = self.engine.submit(prompt_ids, params, sink=sink)
        self.requests_total += 1
        return rid, out_q, prompt_ids

    def submit_completion(self, body: dict, sink=None):
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = " ".join(map(str, prompt))
        prompt_ids = self.tokenizer.encode(str(prompt))
        params = self.parse_sampling(body)
        rid, out_q = self.engine.submit(prompt_ids, params, sink=sink)
        self.requests_total += 1
        return rid, out_q, prompt_ids


async def handle_responses(server: "BackendServer", body: dict, send):
    """Open Responses API surface (object=response, output_text items;
    streaming emits response.output_text.delta / response.completed)."""
    import uuid as _uuid
    model = body.get("model", server.model_name)
    if model not in server.served:
        data = json.dumps({"error": {"message": "model not served",
                                     "code": "model_not_found"}}).encode()
        await send({"type": "http.response.start", "status": 404,
                    "headers": [(b"content-type", b"application/json")]})
        await send({"type": "http.response.body", "body": data})
        return
    inp = body.get("input", "")
    instructions = body.get("instructions") or ""
    if isinstance(inp, list):
        # list of messages or content parts
        parts = []
        for item in inp:
            if isinstance(item, dict):
                c = item.get("content", "")
                if isinstance(c, list):
                    c = " ".join(p.get("text", "") for p in c
                                 if isinstance(p, dict))
                parts.append(str(c))
            else:
                parts.append(str(item))
        inp = "\n".join(parts)
    rendered = (f"<|system|> {instructions}\n{inp}"
                if instructions else str(inp))
    prompt_ids = server.tokenizer.encode(rendered or "hello")
    params = server.parse_sampling(body)
    import asyncio as _asyncio

    class _Sink:
        def __init__(self):
            self.loop = _asyncio.get_running_loop()
            self.q: _asyncio.Queue = _asyncio.Queue()

        def put(self, item):
            try:
                self.loop.call_soon_threadsafe(self.q.put_nowait, item)
            except RuntimeError:
                pass  # loop closed during shutdown

    sink = _Sink()
    server.engine.submit(prompt_ids, params, sink=sink)
    server.requests_total += 1
    rid = f"resp_{_uuid.uuid4().hex[:24]}"
    created = int(time.time())
    stream = bool(body.get("stream"))

    async def collect():
        toks = []
        while True:
            tok, fin, _extra = await sink.q.get()
            toks.append(tok)
            if fin:
                return toks, fin

    if not stream:
        toks, fin = await collect()
        text = "".join(server.tokenizer.decode_token(t) for t in toks)
        payload = {
            "id": rid, "object": "response", "created_at": created,
            "model": model, "status": "completed",
            "output": [{"type": "message", "role": "assistant",
                        "content": [{"type": "output_text",
                                     "text": text}]}],
            "usage": {"input_tokens": len(prompt_ids),
                      "output_tokens": len(toks),
                      "total_tokens": len(prompt_ids) + len(toks)}}
        data = json.dumps(payload).encode()
        await send({"type": "http.response.start", "status": 200,
                    "headers": [(b"content-type", b"application/json"),
                                (b"content-length",
                                 str(len(data)).encode())]})
        await send({"type": "http.response.body", "body": data})
        return

    await send({"type": "http.response.start", "status": 200,
                "headers": [(b"content-type", b"text/event-stream"),
                            (b"cache-control", b"no-cache")]})

    def ev(obj):
        return ("data: " + json.dumps(obj) + "\n\n").encode()

    await send({"type": "http.response.body",
                "body": ev({"type": "response.created",
                            "response": {"id": rid,
                                         "object": "response",
                                         "model": model,
                                         "status": "in_progress"}}),
                "more_body": True})
    ntok = 0
    while True:
        tok, fin, _extra = await sink.q.get()
        ntok += 1
        await send({"type": "http.response.body",
                    "body": ev({"type": "response.output_text.delta",
                                "delta":
                                    server.tokenizer.decode_token(tok)}),
                    "more_body": True})
        if fin:
            break
    await send({"type": "http.response.body",
                "body": ev({"type": "response.completed",
                            "response": {
                                "id": rid, "object": "response",
                                "status": "completed", "model": model,
                                "usage": {
                                    "input_tokens": len(prompt_ids),
                                    "output_tokens": ntok,
                                    "total_tokens":
                                        len(prompt_ids) + ntok}}}),
                "more_body": True})
    await send({"type": "http.response.body",
                "body": b"data: [DONE]\n\n", "more_body": False})


def build_asgi_app(server: BackendServer):
    """Raw ASGI app (no framework dependency on the hot path)."""

    async def read_body(receive) -> bytes:
        chunks = []
        while True:
            msg = await receive()
            if msg["type"] == "http.request":
                chunks.append(msg.get("body", b""))
                if not msg.get("more_body"):
                    break
            elif msg["type"] == "http.disconnect":
                break
        return b"".join(chunks)

    async def send_json(send, status: int, obj: dict):
        data = json.dumps(obj).encode()
        await send({"type": "http.response.start", "status": status,
                    "headers": [(b"content-type", b"application/json"),
                                (b"content-length",
                                 str(len(data)).encode())]})
        await send({"type": "http.response.body", "body": data})

    class _AsyncSink:
        """Engine-thread -> asyncio bridge: put() is called from the
        engine loop thread; items land in an asyncio.Queue without
        blocking any executor thread (scales past 64 streams).

        Writes COALESCE: a decode step emits one token for every
        running stream back to back, so put() buffers under a lock
        and schedules at most one loop wakeup until drained — ~64x
        fewer call_soon_threadsafe round trips per step."""

        def __init__(self):
            import threading as _th
            self.loop = asyncio.get_running_loop()
            self.q: asyncio.Queue = asyncio.Queue()
            self._buf: list = []
            self._mu = _th.Lock()
            self._scheduled = False

        def put(self, item):
            with self._mu:
                self._buf.append(item)
                if self._scheduled:
                    return
                self._scheduled = True
            try:
                self.loop.call_soon_threadsafe(self._drain)
            except RuntimeError:
                # event loop closed (server shutting down): the engine
                # thread must not die over an orphaned stream
                pass

        def _drain(self):
            with self._mu:
                items, self._buf = self._buf, []
                self._scheduled = False
            for it in items:
                self.q.put_nowait(it)

    async def drain_queue_async(sink):
        while True:
            item = await sink.q.get()
            yield item
            if item[1]:
                return

    async def app(scope, receive, send):
        if scope["type"] == "lifespan":
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await send({"type": "lifespan.shutdown.complete"})
                    return
            return
        assert scope["type"] == "http"
        path = scope["path"]
        method = scope["method"]

        if method == "GET" and path == "/v1/models":
            await send_json(send, 200, server.models_payload())
            return
        if method == "GET" and path.startswith("/v1/models/"):
            mid = path[len("/v1/models/"):]
            for m in server.models_payload()["data"]:
                if m["id"] == mid:
                    await send_json(send, 200, m)
                    return
            await send_json(send, 404, {"error": {
                "message": "model not found",
                "type": "invalid_request_error",
                "code": "model_not_found"}})
            return
        if method == "GET" and path in ("/api/health", "/health"):
            await send_json(send, 200, server.health_payload())
            return
        if method == "GET" and path == "/metrics":
            eng = server.engine
            extra = ""
            if eng.prefix is not None:
                extra += (f"backend_prefix_hit_tokens "
                          f"{eng.prefix.hit_tokens}\n"
                          f"backend_prefix_prompt_tokens "
                          f"{eng.prefix.prompt_tokens}\n")
            if eng.spec_ngram > 0:
                extra += (f"backend_spec_proposed {eng.spec_proposed}\n"
                          f"backend_spec_accepted {eng.spec_accepted}\n")
            text = (
                f"backend_requests_total {server.requests_total}\n"
                f"backend_tokens_total {eng.tokens_out}\n"
                f"backend_active_requests {eng.num_running}\n"
                f"backend_tokens_per_sec {eng.tokens_per_sec_ema:.2f}\n"
                f"backend_gpu_step_ms {eng.gpu_step_ms_ema:.3f}\n"
                f"backend_kv_free_pages {eng.kv.free_blocks()}\n"
                f"{extra}"
            ).encode()
            await send({"type": "http.response.start", "status": 200,
                        "headers": [(b"content-type", b"text/plain")]})
            await send({"type": "http.response.body", "body": text})
            return

        if method == "POST" and path == "/v1/embeddings":
            try:
                body = json.loads(await read_body(receive) or b"{}")
            except json.JSONDecodeError:
                await send_json(send, 400, {"error": {
                    "message": "invalid JSON",
                    "type": "invalid_request_error"}})
                return
            inputs = body.get("input", "")
            if isinstance(inputs, str):
                inputs = [inputs]
            inputs = [str(x) for x in inputs][:64]
            import torch as _t

            def _embed():
                data, ntok = [], 0
                with server.embed_lock:
                    for i, text in enumerate(inputs):
                        ids = server.tokenizer.encode(text)
                        ntok += len(ids)
                        t = _t.tensor([ids], dtype=_t.long,
                                      device=server.engine.device)
                        h = server.engine.model.forward_dense(t)
                        vec = h[0, -1].float()
                        vec = vec / (vec.norm() + 1e-8)
                        data.append({"object": "embedding", "index": i,
                                     "embedding": vec.cpu().tolist()})
                return data, ntok

            data, ntok = await asyncio.to_thread(_embed)
            await send_json(send, 200, {
                "object": "list", "data": data,
                "model": body.get("model", server.model_name),
                "usage": {"prompt_tokens": ntok, "total_tokens": ntok}})
            return

        if method == "POST" and path in ("/v1/chat/completions",
                                         "/v1/completions",
                                         "/v1/responses"):
            try:
                body = json.loads(await read_body(receive) or b"{}")
            except json.JSONDecodeError:
                await send_json(send, 400, {"error": {
                    "message": "invalid JSON",
                    "type": "invalid_request_error"}})
                return
            if path == "/v1/responses":
                await handle_responses(server, body, send)
                return
            is_chat = path != "/v1/completions"
            model = body.get("model", server.model_name)
            if model not in server.served:
                await send_json(send, 404, {"error": {
                    "message": f"model '{model}' not served here",
                    "type": "invalid_request_error",
                    "code": "model_not_found"}})
                return
            # context-window guard (OpenAI: context_length_exceeded)
            probe_ids = server.tokenizer.encode(
                render_chat(body.get("messages", []))
                if is_chat else str(body.get("prompt", "")))
            if len(probe_ids) >= server.engine.cfg.model.max_pos:
                await send_json(send, 400, {"error": {
                    "message": f"prompt has {len(probe_ids)} tokens; "
                               f"model context is "
                               f"{server.engine.cfg.model.max_pos}",
                    "type": "invalid_request_error",
                    "code": "context_length_exceeded"}})
                return
            n_choices = max(1, min(int(body.get("n", 1) or 1), 8))
            sinks = [_AsyncSink() for _ in range(n_choices)]
            rids = []
            try:
                for ci, sink in enumerate(sinks):
                    # distinct seeds per choice so sampled choices
                    # differ (greedy choices are identical by design)
                    cbody = dict(body)
                    if body.get("seed") is not None and ci:
                        cbody["seed"] = int(body["seed"]) + ci
                    if is_chat:
                        rid, out_q, prompt_ids = server.submit_chat(
                            cbody, sink)
                    else:
                        rid, out_q, prompt_ids = server.submit_completion(
                            cbody, sink)
                    rids.append(rid)
            except MemoryError:
                await send_json(send, 429, {"error": {
                    "message": "KV cache exhausted",
                    "type": "server_error",
                    "code": "capacity_exceeded"}})
                return
            out_q = sinks[0]
            created = int(time.time())
            cid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
            stream = bool(body.get("stream"))
            include_usage = bool(
                (body.get("stream_options") or {}).get("include_usage"))

            want_lp = server._parse_logprobs(body) > 0
            stops = body.get("stop") or []
            if isinstance(stops, str):
                stops = [stops]
            stops = [s for s in map(str, stops)][:4]

            def find_stop(text: str):
                """Earliest stop-sequence hit -> cut position."""
                best = -1
                for s in stops:
                    i = text.find(s)
                    if i >= 0 and (best < 0 or i < best):
                        best = i
                return best

            def lp_entry(tok, extra):
                return {
                    "token": server.tokenizer.decode_token(tok).strip(),
                    "logprob": extra["logprob"], "bytes": None,
                    "top_logprobs": [
                        {"token":
                             server.tokenizer.decode_token(t).strip(),
                         "logprob": v, "bytes": None}
                        for t, v in extra["top"]]}

            if not stream:
                disconnected = asyncio.Event()

                async def _watch_disconnect_ns():
                    while True:
                        msg = await receive()
                        if msg["type"] == "http.disconnect":
                            disconnected.set()
                            return

                watcher = asyncio.create_task(_watch_disconnect_ns())
                choices = []
                total_out = 0
                for ci, sink_i in enumerate(sinks):
                    if disconnected.is_set():
                        break
                    toks, finish, lp_content = [], "stop", []
                    text = ""
                    async for tok, fin, extra in drain_queue_async(
                            sink_i):
                        toks.append(tok)
                        text += server.tokenizer.decode_token(tok)
                        if extra is not None:
                            lp_content.append(lp_entry(tok, extra))
                        if fin:
                            finish = fin
                        if stops and find_stop(text) >= 0:
                            finish = "stop"
                            server.engine.abort(rids[ci])
                            break
                        if disconnected.is_set():
                            break
                    if disconnected.is_set():
                        break
                    cut = find_stop(text) if stops else -1
                    if cut >= 0:
                        text = text[:cut]
                    total_out += len(toks)
                    if is_chat:
                        choices.append(
                            {"index": ci,
                             "message": {"role": "assistant",
                                         "content": text},
                             "logprobs": ({"content": lp_content}
                                          if want_lp else None),
                             "finish_reason": finish})
                    else:
                        if body.get("echo"):
                            text = str(body.get("prompt", "")) + text
                        choices.append({"index": ci, "text": text,
                                        "finish_reason": finish})
                watcher.cancel()
                if disconnected.is_set():
                    for r in rids:
                        server.engine.abort(r)
                    return
                usage = {"prompt_tokens": len(prompt_ids),
                         "completion_tokens": total_out,
                         "total_tokens": len(prompt_ids) + total_out}
                payload = {
                    "id": cid,
                    "object": ("chat.completion" if is_chat
                               else "text_completion"),
                    "created": created, "model": model,
                    "choices": choices, "usage": usage}
                await send_json(send, 200, payload)
                return

            # streaming SSE
            await send({"type": "http.response.start", "status": 200,
                        "headers": [(b"content-type",
                                     b"text/event-stream"),
                                    (b"cache-control", b"no-cache")]})

            def chunk_payload(delta: dict, finish=None, usage=None,
                              index=0):
                obj = {"id": cid,
                       "object": ("chat.completion.chunk" if is_chat
                                  else "text_completion"),
                       "created": created, "model": model,
                       "choices": [{"index": index,
                                    ("delta" if is_chat else "text"):
                                        delta if is_chat else
                                        delta.get("content", ""),
                                    "finish_reason": finish}]}
                if usage is not None:
                    obj["usage"] = usage
                return ("data: " + json.dumps(obj) + "\n\n").encode()

            if is_chat:
                await send({"type": "http.response.body",
                            "body": chunk_payload({"role": "assistant"}),
                            "more_body": True})
            # watch for client disconnect so an abandoned stream stops
            # burning decode steps (engine.abort frees its pages)
            disconnected = asyncio.Event()

            async def _watch_disconnect():
                while True:
                    msg = await receive()
                    if msg["type"] == "http.disconnect":
                        disconnected.set()
                        return

            watcher = asyncio.create_task(_watch_disconnect())
            ntok = 0
            for ci, sink_i in enumerate(sinks):
                if disconnected.is_set():
                    break
                finish = "stop"
                acc = ""
                async for tok, fin, extra in drain_queue_async(sink_i):
                    ntok += 1
                    if fin:
                        finish = fin
                    text = server.tokenizer.decode_token(tok)
                    if stops:
                        cut = find_stop(acc + text)
                        if cut >= 0:
                            # emit only up to the stop sequence
                            text = (acc + text)[:cut][len(acc):]
                            finish = "stop"
                            server.engine.abort(rids[ci])
                            if text:
                                await send({
                                    "type": "http.response.body",
                                    "body": chunk_payload(
                                        {"content": text}, index=ci),
                                    "more_body": True})
                            break
                        acc += text
                    payload = chunk_payload({"content": text}, index=ci)
                    if extra is not None and is_chat:
                        obj = json.loads(payload[len(b"data: "):])
                        obj["choices"][0]["logprobs"] = {
                            "content": [lp_entry(tok, extra)]}
                        payload = ("data: " + json.dumps(obj)
                                   + "\n\n").encode()
                    await send({"type": "http.response.body",
                                "body": payload,
                                "more_body": True})
                    if disconnected.is_set():
                        break
                if disconnected.is_set():
                    break
                if ci + 1 < len(sinks):
                    await send({"type": "http.response.body",
                                "body": chunk_payload({}, finish=finish,
                                                      index=ci),
                                "more_body": True})
            usage = ({"prompt_tokens": len(prompt_ids),
                      "completion_tokens": ntok,
                      "total_tokens": len(prompt_ids) + ntok}
                     if include_usage else None)
            await send({"type": "http.response.body",
                        "body": chunk_payload({}, finish=finish,
                                              usage=usage,
                                              index=len(sinks) - 1),
                        "more_body": True})
            watcher.cancel()
            if disconnected.is_set():
                for r in rids:
                    server.engine.abort(r)
                return
            await send({"type": "http.response.body",
                        "body": b"data: [DONE]\n\n", "more_body": False})
            return

        await send_json(send, 404, {"error": {"message": "not found",
                                              "type":
                                                  "invalid_request_error"}})

    return app


def serve(engine: Engine, model_name: str, host: str = "0.0.0.0",
          port: int = 9200, served_models: list[str] | None = None,
          tokenizer_path: str | None = None):
    """Run the backend server (blocking).  Starts the engine loop."""
    import uvicorn

    server = BackendServer(engine, model_name, served_models,
                           tokenizer_path=tokenizer_path)
    app = build_asgi_app(server)
    engine.start()
    config = uvicorn.Config(app, host=host, port=port, log_level="warning",
                            access_log=False)
    uvicorn.Server(config).run()
