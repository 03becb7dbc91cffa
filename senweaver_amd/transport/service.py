"""LLMMessageService — the streaming send/abort layer over the local backbone.

Mirrors the reference's L2 transport contract without the Electron IPC hop:
the renderer-side proxy kept a request-id-keyed hook registry and posted over
channel 'senweaver-channel-llmMessage' (common/sendLLMMessageService.ts:37-140);
the main process streamed provider chunks back as request-id-tagged events
(electron-main/sendLLMMessageChannel.ts).  Here ``send_llm_message`` drives
the local MI355X backbone on a worker thread and fires the same callbacks:

  on_text(full_text=..., full_reasoning=..., tool_call=...)   # CUMULATIVE
  on_final_message(full_text=..., full_reasoning=..., tool_call=...)
  on_error(message=...) / on_abort()

with the reasoning and XML-tool grammars applied exactly as the reference's
extractGrammar wrappers do.  ``send_llm_message`` returns the request id,
which doubles as the abort token (abort resolves instantly client-side, no
round-trip — the decode loop observes the flag at its next token).
"""

from __future__ import annotations

import threading
import uuid
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .grammar import ReasoningExtractor, XMLToolExtractor


@dataclass
class LLMChatMessage:
    role: str  # 'system' | 'user' | 'assistant' | 'tool'
    content: str
    name: Optional[str] = None


@dataclass
class _Request:
    id: str
    abort_event: threading.Event = field(default_factory=threading.Event)
    thread: Optional[threading.Thread] = None


class StreamingBackendProtocol:
    """What the transport needs: an incremental token generator."""

    def stream_generate(self, prompt: str, max_new_tokens: int,
                        should_stop: Callable[[], bool],
                        on_chunk: Callable[[str], None]) -> str:
        """Greedy-decode; call on_chunk(cumulative_text) as text grows;
        honor should_stop() between tokens; return the final text."""
        raise NotImplementedError


class LLMMessageService:
    def __init__(self, backend) -> None:
        self._backend = backend
        self._requests: Dict[str, _Request] = {}
        self._lock = threading.Lock()
        # one generation at a time per backend: stream_generate resets the
        # shared paged cache (and on GPU replays one captured graph), so
        # concurrent generations would corrupt each other — requests queue
        # here and stream once started
        self._gen_lock = threading.Lock()
        from ..utils.observability import TokenUsageTracker
        self.usage = TokenUsageTracker()  # per-request in/out token log

    # -- prompt rendering (chat messages -> backbone prompt text) --

    @staticmethod
    def render_messages(messages: List[LLMChatMessage]) -> str:
        parts = []
        for m in messages:
            if isinstance(m, dict):  # duck-type the wire shape too
                role, content = m.get("role", "user"), m.get("content", "")
            else:
                role, content = m.role, m.content
            parts.append(f"<|{role}|>\n{content}")
        parts.append("<|assistant|>\n")
        return "\n".join(parts)

    def send_llm_message(
        self,
        messages: List[LLMChatMessage],
        on_text: Callable[..., None],
        on_final_message: Callable[..., None],
        on_error: Callable[..., None],
        on_abort: Optional[Callable[[], None]] = None,
        chat_mode: Optional[str] = None,
        mcp_tools: Optional[List[dict]] = None,
        max_new_tokens: int = 512,
        think_tags: tuple = ("<think>", "</think>"),
        synchronous: bool = False,
        model_options: Optional[dict] = None,
        raw_prompt: Optional[str] = None,
    ) -> str:
        """Start a streaming generation; returns the request id (abort token)."""
        request_id = str(uuid.uuid4())
        req = _Request(id=request_id)
        with self._lock:
            self._requests[request_id] = req

        reasoning = ReasoningExtractor(think_tags)
        tools = XMLToolExtractor(chat_mode, mcp_tools)

        def pump_chunk(cumulative_raw: str) -> None:
            text, rsn, emit = reasoning.feed(cumulative_raw)
            if not emit:
                return
            vis_text, tool_call = tools.feed(text)
            try:
                on_text(full_text=vis_text, full_reasoning=rsn, tool_call=tool_call)
            except Exception:
                pass

        def run() -> None:
            try:
                prompt = (raw_prompt if raw_prompt is not None
                          else self.render_messages(messages))
                with self._gen_lock:
                    if req.abort_event.is_set():
                        if on_abort:
                            on_abort()
                        return
                    return _generate(prompt)
            except Exception as e:
                try:
                    on_error(message=str(e))
                except Exception:
                    pass
            finally:
                with self._lock:
                    self._requests.pop(request_id, None)

        def _generate(prompt: str) -> None:
            # modelSelectionOptions analog: temperature/top-p pass through
            # to backends that sample (scripted test backends keep the
            # positional-only signature)
            opts = model_options or {}
            if opts:
                kwargs = {
                    "temperature": float(opts.get("temperature", 0.0)),
                    "top_p": float(opts.get("topP", opts.get("top_p", 1.0))),
                    "sample_seed": opts.get("sampleSeed"),
                    "stop": opts.get("stop"),
                }
                # optional knobs only when requested: duck-typed backends
                # that predate them keep working
                if opts.get("presencePenalty"):
                    kwargs["presence_penalty"] = float(opts["presencePenalty"])
                if opts.get("frequencyPenalty"):
                    kwargs["frequency_penalty"] = float(opts["frequencyPenalty"])
                if opts.get("logitBias"):
                    kwargs["logit_bias"] = {int(k): float(v)
                                            for k, v in opts["logitBias"].items()}
                final_raw = self._backend.stream_generate(
                    prompt, max_new_tokens, req.abort_event.is_set,
                    pump_chunk, **kwargs)
            else:
                final_raw = self._backend.stream_generate(
                    prompt, max_new_tokens, req.abort_event.is_set, pump_chunk)
            if req.abort_event.is_set():
                if on_abort:
                    on_abort()
                return
            text, rsn = reasoning.finalize(final_raw)
            vis_text, tool_call = tools.finalize(text)
            if not vis_text and not rsn and tool_call is None:
                on_error(message="Response from model was empty.")
                return
            self._record_usage(request_id, prompt, final_raw)
            on_final_message(full_text=vis_text, full_reasoning=rsn, tool_call=tool_call)

        if synchronous:
            run()
        else:
            t = threading.Thread(target=run, name=f"llm-{request_id[:8]}", daemon=True)
            req.thread = t
            t.start()
        return request_id

    def _record_usage(self, request_id: str, prompt: str, output: str) -> None:
        """Token accounting per request: exact counts when the backend has a
        tokenizer, the reference's 4-chars/token estimate otherwise."""
        tok = getattr(self._backend, "tokenizer", None)
        if tok is not None:
            tin, tout = len(tok.encode(prompt)), len(tok.encode(output))
        else:
            tin, tout = len(prompt) // 4, len(output) // 4
        model = getattr(getattr(self._backend, "config", None), "name", "local")
        self.usage.record(request_id, model, tin, tout)

    def abort(self, request_id: str) -> None:
        """Instant client-side abort (no round trip), like the reference."""
        with self._lock:
            req = self._requests.get(request_id)
        if req:
            req.abort_event.set()

    def wait(self, request_id: str, timeout: Optional[float] = None) -> None:
        with self._lock:
            req = self._requests.get(request_id)
        if req and req.thread:
            req.thread.join(timeout)

    def list_models(self) -> List[str]:
        """Local analog of ollamaList/openAICompatibleList."""
        name = getattr(getattr(self._backend, "config", None), "name", None)
        if name:
            return [name]
        # proxy backends (DaemonBackend) report the remote engine's models
        return list(getattr(self._backend, "model_names", []) or [])

    def list_models_detailed(self) -> List[dict]:
        """Model list with the capability record the reference's
        RefreshModelService attaches (modelCapabilities.ts lookup)."""
        from ..models.capabilities import get_model_capabilities
        out = []
        for name in self.list_models():
            caps = get_model_capabilities(name)
            out.append({
                "name": name,
                "contextWindow": caps.contextWindow,
                "reservedOutputTokenSpace": caps.reservedOutputTokenSpace,
                "supportsFIM": caps.supportsFIM,
                "supportsSystemMessage": caps.supportsSystemMessage,
            })
        return out
