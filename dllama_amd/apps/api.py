"""`dllama-api` — OpenAI-compatible HTTP server.

Behavior parity with the reference API server (src/dllama-api.cpp):
  - POST /v1/chat/completions (stream SSE + non-stream), GET /v1/models
    (routes, dllama-api.cpp:550-561)
  - params: stream / temperature / seed / max_tokens / stop
    (dllama-api.cpp:491-520)
  - NaiveCache: longest-prefix chat-history KV reuse by message-list
    comparison (dllama-api.cpp:298-343)
  - single-slot sequential serving over a stdlib HTTP server (the reference
    hand-rolls HTTP/1.1 the same way, dllama-api.cpp:45-186)
"""

from __future__ import annotations

import json
import sys
import time
import uuid
from http.server import BaseHTTPRequestHandler, HTTPServer

from ..tokenizer import (ChatItem, ChatTemplateGenerator, EosDetector,
                         TEMPLATE_UNKNOWN, chat_stops)
from .main import build_parser, load_engine


class NaiveCache:
    """Reuse the KV cache for the longest shared chat-message prefix
    (reference NaiveCache, dllama-api.cpp:298-343)."""

    def __init__(self):
        self.tokens: list[int] = []

    def resolve(self, new_tokens: list[int]) -> int:
        """-> start_pos: length of the shared prefix with the cached run."""
        n = 0
        for a, b in zip(self.tokens, new_tokens):
            if a != b:
                break
            n += 1
        # never reuse the full prompt (need at least 1 token to evaluate)
        n = min(n, len(new_tokens) - 1)
        return max(n, 0)

    def update(self, tokens: list[int]) -> None:
        self.tokens = list(tokens)


class ApiState:
    def __init__(self, args):
        self.engine, self.m, self.comm = load_engine(args)
        self.tok = self.engine.tokenizer
        eos_piece = (self.tok.vocab[self.tok.eos_token_ids[0]]
                     .decode("utf-8", "replace") if self.tok.eos_token_ids else "")
        self.template = ChatTemplateGenerator(TEMPLATE_UNKNOWN,
                                              self.tok.chat_template, eos_piece)
        self.cache = NaiveCache()
        self.model_name = "dllama"
        # per-request sampler defaults (reference re-parses params with CLI
        # defaults each request, dllama-api.cpp:491-520)
        self.default_temp = args.temperature
        self.default_topp = args.topp
        self.default_seed = args.seed if args.seed is not None else int(time.time())

    def complete(self, body: dict, emit):
        """Run one chat completion; emit(delta_text) streams chunks."""
        items = [ChatItem(m.get("role", "user"), m.get("content", ""))
                 for m in body.get("messages", [])]
        text = self.template.generate(items, True).content
        tokens = self.tok.encode(text)
        start = self.cache.resolve(tokens)
        self.engine.reset(start)
        max_tokens = int(body.get("max_tokens") or 256)
        # reset to CLI defaults so one request's overrides don't leak into the
        # next (reference dllama-api.cpp:491-520)
        self.engine.sampler.set_temp(self.default_temp)
        self.engine.sampler.topp = self.default_topp
        self.engine.sampler.set_seed(self.default_seed)
        if body.get("temperature") is not None:
            self.engine.sampler.set_temp(float(body["temperature"]))
        if body.get("top_p") is not None:
            self.engine.sampler.topp = float(body["top_p"])
        if body.get("seed") is not None:
            self.engine.sampler.set_seed(int(body["seed"]))
        user_stop = body.get("stop") or []
        if isinstance(user_stop, str):  # OpenAI allows string or list
            user_stop = [user_stop]
        stops = chat_stops(self.tok) + user_stop
        detector = EosDetector(self.tok.eos_token_ids, stops)
        self.tok.reset_decoder()
        out_text = []

        def on_token(t):
            # streaming-decoder first: UTF-8-safe pieces into the detector
            piece = self.tok.decode(t)
            kind = detector.append(t, piece)
            if kind != 0:  # not MAYBE_EOS
                delta = detector.get_delta()
                if delta:
                    out_text.append(delta)
                    emit(delta)
                    detector.reset()

        gen_tokens, _ = self.engine.generate(
            tokens[start:], max_tokens, on_token=on_token,
            stop_check=lambda t: detector.is_eos(t) or detector.eos_pos >= 0)
        # cache only EVALUATED tokens: the last sampled token was never fed
        # through the model, so its KV row does not exist (reference caches
        # the evaluated endPos, dllama-api.cpp:470-473)
        self.cache.update(tokens + gen_tokens[:-1])
        return "".join(out_text), len(tokens), len(gen_tokens)


STATE: ApiState | None = None


class Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"

    def log_message(self, fmt, *a):  # quiet
        pass

    def _json(self, code: int, obj: dict):
        data = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_GET(self):
        if self.path == "/v1/models":
            self._json(200, {"object": "list", "data": [
                {"id": STATE.model_name, "object": "model",
                 "created": int(time.time()), "owned_by": "dllama_amd"}]})
        elif self.path == "/health":
            self._json(200, {"status": "ok"})
        else:
            self._json(404, {"error": "not found"})

    def do_POST(self):
        if self.path not in ("/v1/chat/completions", "/chat/completions"):
            self._json(404, {"error": "not found"})
            return
        length = int(self.headers.get("Content-Length", 0))
        try:
            body = json.loads(self.rfile.read(length) or b"{}")
        except json.JSONDecodeError:
            self._json(400, {"error": "invalid json"})
            return
        rid = f"chatcmpl-{uuid.uuid4().hex[:12]}"
        created = int(time.time())
        stream = bool(body.get("stream"))
        if stream:
            self.send_response(200)
            self.send_header("Content-Type", "text/event-stream")
            self.send_header("Cache-Control", "no-cache")
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()

            def emit(delta):
                chunk = {"id": rid, "object": "chat.completion.chunk",
                         "created": created, "model": STATE.model_name,
                         "choices": [{"index": 0, "delta": {"content": delta},
                                      "finish_reason": None}]}
                self._chunk(f"data: {json.dumps(chunk)}\n\n")

            try:
                STATE.complete(body, emit)
            finally:
                fin = {"id": rid, "object": "chat.completion.chunk",
                       "created": created, "model": STATE.model_name,
                       "choices": [{"index": 0, "delta": {},
                                    "finish_reason": "stop"}]}
                self._chunk(f"data: {json.dumps(fin)}\n\n")
                self._chunk("data: [DONE]\n\n")
                self.wfile.write(b"0\r\n\r\n")
        else:
            try:
                text, n_prompt, n_gen = STATE.complete(body, lambda d: None)
            except Exception as e:  # noqa: BLE001
                self._json(500, {"error": {"message": str(e), "type": "server_error"}})
                return
            self._json(200, {
                "id": rid, "object": "chat.completion", "created": created,
                "model": STATE.model_name,
                "choices": [{"index": 0, "message":
                             {"role": "assistant", "content": text},
                             "finish_reason": "stop"}],
                "usage": {"prompt_tokens": n_prompt,
                          "completion_tokens": n_gen,
                          "total_tokens": n_prompt + n_gen}})

    def _chunk(self, s: str):
        data = s.encode()
        self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
        self.wfile.flush()


def main(argv=None) -> int:
    global STATE
    parser = build_parser()
    args = parser.parse_args(["inference"] + (argv if argv is not None
                                              else sys.argv[1:]))
    STATE = ApiState(args)
    comm = STATE.comm
    if comm.world > 1:
        # TP serving: rank 0 owns HTTP + sampling; ranks > 0 replay control
        # packets (reference root/worker split, app.cpp:168-230) — without
        # this every rank would bind the port
        from ..parallel.lockstep import RootModel, follower_loop
        if comm.rank > 0:
            follower_loop(STATE.engine.model, comm)
            return 0
        STATE.engine.model = RootModel(STATE.engine.model, comm)
    server = HTTPServer((args.host, args.port), Handler)
    print(f"⭐ dllama-api listening on {args.host}:{args.port}")
    try:
        server.serve_forever()
    except KeyboardInterrupt:
        pass
    finally:
        if comm.world > 1 and comm.rank == 0:
            STATE.engine.model.stop_followers()
    return 0


if __name__ == "__main__":
    sys.exit(main())
