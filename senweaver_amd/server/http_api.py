"""OpenAI-compatible HTTP serving surface over the local engine.

The reference talks to remote providers over HTTPS; serving the LOCAL
backbone to other processes wants the same wire shape, so tools built for
the OpenAI API work against an MI355X node unchanged:

  GET  /v1/models                      — model list with capability records
  POST /v1/completions                 — prompt in, text out; stream=true
                                         emits SSE `data:` chunks with text
                                         DELTAS, then `data: [DONE]`
  POST /v1/chat/completions            — messages in (rendered through the
                                         engine's chat template)
  POST /v1/embeddings                  — pooled-representation vectors
  POST /v1/tokenize | /v1/detokenize   — budget math for clients
  GET  /v1/stats                       — per-request token usage

Generation options map 1:1 onto the engine's sampler: temperature, top_p,
seed, stop, max_tokens.  Generations serialize per backend (the engine is
single-stream); concurrent requests queue inside LLMMessageService.

Run: `python -m senweaver_amd.server.http_api [--port 8008] [--model M]`
(uvicorn; loopback by default).  Tests drive the ASGI app in-process.
"""

from __future__ import annotations

import json
import queue
import threading
import time
import uuid
from typing import Optional

from ..transport.service import LLMChatMessage, LLMMessageService


def create_app(service: LLMMessageService):
    from fastapi import FastAPI
    from fastapi.responses import JSONResponse, StreamingResponse

    app = FastAPI(title="senweaver-amd", version="0.2.0")

    def _model_name() -> str:
        models = service.list_models()
        return models[0] if models else "local"

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [dict(d, object="model", id=d["name"])
                         for d in service.list_models_detailed()]}

    @app.get("/v1/stats")
    def stats():
        return service.usage.stats()

    def _options(body: dict) -> Optional[dict]:
        opts = {}
        if body.get("temperature"):
            opts["temperature"] = float(body["temperature"])
            opts["topP"] = float(body.get("top_p", 1.0))
            if body.get("seed") is not None:
                opts["sampleSeed"] = int(body["seed"])
        stop = body.get("stop")
        if stop:
            opts["stop"] = [stop] if isinstance(stop, str) else list(stop)
        for src, dst in (("presence_penalty", "presencePenalty"),
                         ("frequency_penalty", "frequencyPenalty")):
            if body.get(src):
                opts[dst] = float(body[src])
        if body.get("logit_bias"):
            opts["logitBias"] = {int(k): float(v)
                                 for k, v in body["logit_bias"].items()}
        return opts or None

    def _generate(body: dict, raw_prompt: Optional[str],
                  messages: list, kind: str):
        rid = f"cmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())
        model = _model_name()
        max_tokens = int(body.get("max_tokens",
                             body.get("max_completion_tokens", 128)))
        stream = bool(body.get("stream", False))

        if not stream:
            n = max(1, min(int(body.get("n", 1)), 16))
            base_opts = _options(body)
            choices = []
            for i in range(n):
                opts = dict(base_opts) if base_opts else None
                if n > 1:
                    # n samples need sampling; vary the seed per choice
                    opts = opts or {"temperature": 0.8, "topP": 1.0}
                    opts["sampleSeed"] = int(opts.get("sampleSeed", 0)) + i
                state = {}
                done = threading.Event()
                service.send_llm_message(
                    messages,
                    on_text=lambda **k: None,
                    on_final_message=lambda full_text="", **k: (
                        state.update(text=full_text), done.set()),
                    on_error=lambda message="", **k: (
                        state.update(error=message), done.set()),
                    max_new_tokens=max_tokens, model_options=opts,
                    raw_prompt=raw_prompt, synchronous=True)
                done.wait(timeout=1)
                if "error" in state:
                    return JSONResponse({"error": {"message": state["error"]}},
                                        status_code=500)
                text = state.get("text", "")
                if kind == "chat":
                    choices.append({"index": i, "finish_reason": "stop",
                                    "message": {"role": "assistant",
                                                "content": text}})
                else:
                    choices.append({"index": i, "finish_reason": "stop",
                                    "text": text})
            usage = service.usage.stats()
            return {"id": rid, "object": f"{kind}.completion",
                    "created": created, "model": model, "choices": choices,
                    "usage": {"total_requests": usage["totalRequests"]}}

        q: "queue.Queue" = queue.Queue()
        sent = {"len": 0}

        def on_text(full_text="", **k):
            delta = full_text[sent["len"]:]
            if delta:
                sent["len"] = len(full_text)
                q.put(delta)

        def on_final(full_text="", **k):
            on_text(full_text=full_text)
            q.put(None)

        sid = service.send_llm_message(
            messages, on_text=on_text, on_final_message=on_final,
            on_error=lambda message="", **k: q.put(None),
            max_new_tokens=max_tokens, model_options=_options(body),
            raw_prompt=raw_prompt)

        def sse():
            try:
                while True:
                    delta = q.get()
                    if delta is None:
                        break
                    if kind == "chat":
                        choice = {"index": 0, "delta": {"content": delta}}
                    else:
                        choice = {"index": 0, "text": delta}
                    chunk = {"id": rid, "object": f"{kind}.completion.chunk",
                             "created": created, "model": model,
                             "choices": [choice]}
                    yield f"data: {json.dumps(chunk)}\n\n"
                yield "data: [DONE]\n\n"
            finally:
                # client gone (GeneratorExit) or stream finished: stop the
                # decode — a vanished consumer must not keep the GPU busy
                service.abort(sid)

        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.post("/v1/tokenize")
    def tokenize(body: dict):
        tok = getattr(service._backend, "tokenizer", None)
        if tok is None:
            return JSONResponse(
                {"error": {"message": "backend has no tokenizer"}}, 501)
        ids = tok.encode(str(body.get("text", "")))
        return {"tokens": ids, "count": len(ids)}

    @app.post("/v1/detokenize")
    def detokenize(body: dict):
        tok = getattr(service._backend, "tokenizer", None)
        if tok is None:
            return JSONResponse(
                {"error": {"message": "backend has no tokenizer"}}, 501)
        return {"text": tok.decode([int(i) for i in body.get("tokens", [])])}

    @app.post("/v1/embeddings")
    def embeddings(body: dict):
        texts = body.get("input", [])
        if isinstance(texts, str):
            texts = [texts]
        embed = getattr(service._backend, "embed", None)
        if embed is None:
            return JSONResponse(
                {"error": {"message": "backend has no embedding support"}},
                status_code=501)
        vecs = embed([str(t) for t in texts])
        return {"object": "list", "model": _model_name(),
                "data": [{"object": "embedding", "index": i, "embedding": v}
                         for i, v in enumerate(vecs)]}

    @app.post("/v1/completions")
    def completions(body: dict):
        return _generate(body, raw_prompt=str(body.get("prompt", "")),
                         messages=[], kind="text")

    @app.post("/v1/chat/completions")
    def chat_completions(body: dict):
        messages = [LLMChatMessage(m.get("role", "user"), m.get("content", ""))
                    for m in body.get("messages", [])]
        return _generate(body, raw_prompt=None, messages=messages, kind="chat")

    return app


def main(argv=None) -> int:
    import argparse

    ap = argparse.ArgumentParser(prog="senweaver-amd http server")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8008)
    ap.add_argument("--model", default=None)
    ap.add_argument("--max-seq", type=int, default=2048)
    ap.add_argument("--daemon", default=None, metavar="SOCKET",
                    help="front the native daemon at SOCKET instead of "
                         "loading a model in-process")
    args = ap.parse_args(argv)

    import uvicorn

    if args.daemon:
        from .client import DaemonBackend
        backend = DaemonBackend(args.daemon)
    else:
        import torch

        from ..engine.scorer import LlamaBackend

        model = args.model or ("llama-3-8b" if torch.cuda.is_available()
                               else "tiny-debug")
        backend = LlamaBackend(model, max_seq=args.max_seq)
    app = create_app(LLMMessageService(backend))
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
