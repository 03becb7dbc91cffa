"""Engine worker — the model-side half of the native daemon.

Spawned by senweaver_daemon with requests on stdin and events on stdout
(newline-delimited JSON, request-id tagged).  Runs the local backbone
through LLMMessageService, streaming cumulative-text events exactly like
the reference's channel (onText_/onFinalMessage_/onError_sendLLMMessage).

Request payload:
  {"method":"sendLLMMessage","requestId":"r1",
   "messages":[{"role":"user","content":"..."}],
   "chatMode":"agent"|null, "maxNewTokens":256}
"""

from __future__ import annotations

import json
import os
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))


def main() -> int:
    import torch

    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.models.config import get_config
    from senweaver_amd.transport.service import LLMChatMessage, LLMMessageService

    model_name = os.environ.get("SENWEAVER_MODEL", "tiny-debug" if not torch.cuda.is_available() else "llama-3-8b")
    max_seq = int(os.environ.get("SENWEAVER_MAX_SEQ", "2048" if torch.cuda.is_available() else "256"))
    backend = LlamaBackend(get_config(model_name), max_seq=max_seq)
    service = LLMMessageService(backend)
    out_lock = threading.Lock()
    request_map = {}  # client requestId -> service request id

    def emit(obj) -> None:
        with out_lock:
            sys.stdout.write(json.dumps(obj) + "\n")
            sys.stdout.flush()

    emit({"event": "ready", "model": model_name})

    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        try:
            req = json.loads(line)
        except ValueError:
            emit({"event": "onError", "message": "bad json"})
            continue
        method = req.get("method")
        rid = req.get("requestId", "")
        if method == "sendLLMMessage":
            messages = [LLMChatMessage(m.get("role", "user"), m.get("content", ""))
                        for m in req.get("messages", [])]

            def on_text(full_text="", full_reasoning="", tool_call=None, _rid=rid, **kw):
                emit({"event": "onText", "requestId": _rid, "fullText": full_text,
                      "fullReasoning": full_reasoning,
                      "toolCall": tool_call.__dict__ if tool_call else None})

            def on_final(full_text="", full_reasoning="", tool_call=None, _rid=rid, **kw):
                request_map.pop(_rid, None)  # terminal: free the entry
                emit({"event": "onFinalMessage", "requestId": _rid, "fullText": full_text,
                      "fullReasoning": full_reasoning,
                      "toolCall": tool_call.__dict__ if tool_call else None})

            def on_error(message="", _rid=rid, **kw):
                request_map.pop(_rid, None)
                emit({"event": "onError", "requestId": _rid, "message": message})

            sid = service.send_llm_message(
                messages, on_text, on_final, on_error,
                chat_mode=req.get("chatMode"),
                max_new_tokens=int(req.get("maxNewTokens", 256)),
                model_options=req.get("modelOptions"),
                raw_prompt=req.get("rawPrompt"))
            request_map[rid] = sid
        elif method == "abort":
            sid = request_map.pop(rid, None)
            if sid:
                service.abort(sid)
        elif method == "list":
            emit({"event": "listResult", "models": service.list_models(),
                  "detailed": service.list_models_detailed()})
        elif method == "stats":
            emit({"event": "statsResult", "usage": service.usage.stats()})
        elif method == "ping":
            emit({"event": "pong"})
        elif method == "shutdown":
            break
    return 0


if __name__ == "__main__":
    sys.exit(main())
