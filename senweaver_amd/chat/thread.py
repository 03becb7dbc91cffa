"""ChatThreadService — thread state + the agent loop.

Rebuilds the reference's core loop (browser/chatThreadService.ts):
``add_user_message_and_stream_response`` (:2694) starts a trace, records the
user message and enters ``_run_chat_agent`` (:1172) — the tool-use while
loop with the layered recovery ladder:

  - context-overflow errors (string-matched, :1438-1446) -> progressive
    pruning (drop tool outputs, trim history) and retry (<=5);
  - rate limits -> reactive cooldown + retry (attempt counter decremented);
  - other errors -> CHAT_RETRIES with exponential backoff;
  - tool failures become tool_error messages fed back to the model, not
    crashes (:1111-1128);
  - every turn is traced (recordUserMessage/AssistantMessage/LLMCall/
    ToolCall/Error) and the trace ends with the 9-dim reward.

Tool approval gating (:958-996): invalid params -> invalid_params message
and the loop continues; tools in an approval class pause the loop with a
pending tool_request unless auto-approved.  Checkpoints are added before
each user message and at stream end (:1734, 2710).
"""

from __future__ import annotations

import threading

import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ..context.compress import CompressibleMessage
from ..context.pipeline import ConvertToLLMMessages
from ..trace.collector import TraceCollector
from ..transport.grammar import RawToolCall
from ..transport.service import LLMChatMessage, LLMMessageService
from ..tools.service import ToolError, ToolsService
from .limits import (
    CHAT_RETRIES,
    TPMRateLimiter,
    get_retry_delay_ms,
    is_context_length_error,
    is_rate_limit_error,
)

MAX_AGENT_LOOP_TURNS = 50  # build agent maxSteps


@dataclass
class ThreadMessage:
    role: str  # 'user' | 'assistant' | 'tool' | 'checkpoint' | 'tool_request' | 'invalid_params'
    content: str
    tool_name: Optional[str] = None
    tool_params: Optional[Dict[str, Any]] = None
    tool_success: Optional[bool] = None


@dataclass
class ChatThread:
    id: str
    messages: List[ThreadMessage] = field(default_factory=list)
    checkpoints: List[int] = field(default_factory=list)
    pending_tool: Optional[RawToolCall] = None
    streaming: bool = False


@dataclass
class GlobalSettings:
    auto_approve: Dict[str, bool] = field(default_factory=dict)  # approval class -> bool
    chat_mode: str = "agent"
    context_window: int = 8192


class ChatThreadService:
    def __init__(self, llm: LLMMessageService, tools: ToolsService,
                 trace_collector: TraceCollector,
                 converter: Optional[ConvertToLLMMessages] = None,
                 settings: Optional[GlobalSettings] = None,
                 sleep: Callable[[float], None] = time.sleep) -> None:
        self._llm = llm
        self._tools = tools
        self._traces = trace_collector
        self._converter = converter or ConvertToLLMMessages()
        self.settings = settings or GlobalSettings()
        self._threads: Dict[str, ChatThread] = {}
        self._sleep = sleep
        self._rate_limiter = TPMRateLimiter()

    # ---- thread state ----
        from ..context.compress import EnhancedContextManager
        self._ctx_mgr = EnhancedContextManager(
            context_limit=self.settings.context_window)

    def open_thread(self) -> ChatThread:
        t = ChatThread(id=str(uuid.uuid4()))
        self._threads[t.id] = t
        return t

    def get_thread(self, thread_id: str) -> Optional[ChatThread]:
        return self._threads.get(thread_id)

    def _add_checkpoint(self, thread: ChatThread) -> None:
        thread.checkpoints.append(len(thread.messages))
        thread.messages.append(ThreadMessage("checkpoint", f"checkpoint@{len(thread.messages)}"))

    # ---- entry point ----

    def add_user_message_and_stream_response(self, thread_id: str, content: str,
                                             max_new_tokens: int = 256) -> ChatThread:
        thread = self._threads[thread_id]
        self._add_checkpoint(thread)
        thread.messages.append(ThreadMessage("user", content))
        self._traces.start_trace(thread_id, {"chatMode": self.settings.chat_mode})
        self._traces.record_user_message(thread_id, len(thread.messages) - 1, content)
        try:
            self._run_chat_agent(thread, max_new_tokens=max_new_tokens)
        finally:
            self._add_checkpoint(thread)
            self._traces.end_trace_for_thread(thread_id)
        return thread

    # ---- the agent loop ----

    def _history(self, thread: ChatThread) -> List[CompressibleMessage]:
        out = []
        for m in thread.messages:
            if m.role in ("user", "assistant"):
                out.append(CompressibleMessage(m.role, m.content))
            elif m.role == "tool":
                out.append(CompressibleMessage("tool", m.content, m.tool_name))
        return out

    def _run_chat_agent(self, thread: ChatThread, max_new_tokens: int = 256,
                        preapproved_tool: Optional[RawToolCall] = None) -> None:
        thread.streaming = True
        pruned_context = 0
        turn = 0
        pending_first_tool = preapproved_tool
        try:
            while turn < MAX_AGENT_LOOP_TURNS:
                turn += 1
                if pending_first_tool is not None:
                    tool_call = pending_first_tool
                    pending_first_tool = None
                    should_continue = self._run_tool_call(thread, tool_call, approved=True)
                    if not should_continue:
                        return
                    continue

                wait_ms = self._rate_limiter.get_wait_time_ms("local")
                if wait_ms > 0:
                    self._sleep(wait_ms / 1000)

                # in-loop per-tool compaction (chatThreadService.ts:1456-1461):
                # above 55% occupancy the oldest/largest tool outputs are
                # pruned before the next send
                history = self._ctx_mgr.maybe_prune(self._history(thread))
                sys_msg, fitted = self._converter.prepare_llm_chat_messages(
                    history, self.settings.chat_mode,
                    self.settings.context_window)
                messages = [LLMChatMessage("system", sys_msg)] + [
                    LLMChatMessage(m.role, m.content) for m in fitted]

                result = self._send_with_retries(thread, messages, max_new_tokens)
                if result is None:
                    return  # unrecoverable; error recorded
                full_text, tool_call = result
                msg_idx = len(thread.messages)
                thread.messages.append(ThreadMessage("assistant", full_text))
                self._traces.record_assistant_message(thread.id, msg_idx, full_text,
                                                      model="local", provider="senweaver_amd")
                if tool_call is None:
                    return  # conversation turn complete
                # tool-call || system-message warmup overlap
                # (chatThreadService.ts:1659-1684): while the tool runs, the
                # next turn's system message is generated on a side thread so
                # the post-tool prepare hits the cache
                warm = threading.Thread(
                    target=self._warm_system_message, daemon=True)
                warm.start()
                should_continue = self._run_tool_call(thread, tool_call)
                warm.join(timeout=10)
                if not should_continue:
                    return
        finally:
            thread.streaming = False


    def _warm_system_message(self) -> None:
        """Populate the converter's system-message cache (runs beside a tool)."""
        try:
            self._converter.generate_system_message(self.settings.chat_mode)
        except Exception:
            pass

    def _send_with_retries(self, thread: ChatThread, messages: List[LLMChatMessage],
                           max_new_tokens: int):
        """The retry ladder around one LLM send.  Returns (text, tool_call) or None."""
        attempts = 0
        context_prunes = 0
        while True:
            state: Dict[str, Any] = {}

            def on_text(**kw):
                pass

            def on_final(full_text="", full_reasoning="", tool_call=None, **kw):
                state["text"] = full_text
                state["reasoning"] = full_reasoning
                state["tool_call"] = tool_call

            def on_error(message="", **kw):
                state["error"] = message

            t0 = time.time()
            self._rate_limiter.record_request("local")
            self._llm.send_llm_message(
                messages, on_text, on_final, on_error,
                chat_mode=self.settings.chat_mode, max_new_tokens=max_new_tokens,
                synchronous=True)
            duration = (time.time() - t0) * 1000

            if "error" not in state:
                self._rate_limiter.record_success("local")
                self._traces.record_llm_call(thread.id, len(thread.messages),
                                             model="local", provider="senweaver_amd",
                                             input_tokens=sum(len(m.content) for m in messages) // 4,
                                             output_tokens=len(state.get("text", "")) // 4,
                                             duration=duration)
                return state.get("text", ""), state.get("tool_call")

            err = state["error"]
            if is_context_length_error(err) and context_prunes < 5:
                context_prunes += 1
                messages = self._prune_for_context(messages, context_prunes)
                continue
            if is_rate_limit_error(err):
                cooldown = self._rate_limiter.handle_rate_limit_error("local", err, attempts)
                self._sleep(cooldown / 1000)
                attempts = max(0, attempts - 1)  # reactive: doesn't consume retries
                continue
            attempts += 1
            if attempts < CHAT_RETRIES:
                self._sleep(get_retry_delay_ms(attempts, False) / 1000)
                continue
            self._traces.record_error(thread.id, len(thread.messages), err)
            thread.messages.append(ThreadMessage("assistant", f"[error] {err}"))
            return None

    @staticmethod
    def _prune_for_context(messages: List[LLMChatMessage], attempt: int) -> List[LLMChatMessage]:
        """Progressive pruning (reference :1450-1478): first drop tool outputs,
        then trim middle history."""
        out = []
        for m in messages:
            c = m.content
            if attempt >= 1 and c.startswith("[tool "):
                c = "[tool output pruned]"
            out.append(LLMChatMessage(m.role, c, m.name))
        if attempt >= 3 and len(out) > 6:
            out = out[:2] + out[-4:]
        return out

    # ---- tool execution + approval gate ----

    def _run_tool_call(self, thread: ChatThread, tool_call: RawToolCall,
                       approved: bool = False) -> bool:
        """Returns True when the agent loop should send another message."""
        try:
            params = self._tools.validate_params(tool_call.name, tool_call.raw_params)
        except ToolError as e:
            thread.messages.append(ThreadMessage(
                "tool", f"Invalid parameters: {e}", tool_name=tool_call.name,
                tool_success=False))
            self._traces.record_tool_call(thread.id, len(thread.messages) - 1,
                                          tool_name=tool_call.name, tool_success=False,
                                          tool_result=str(e))
            return True  # invalid_params: loop continues (reference :972)

        approval = self._tools.approval_type(tool_call.name)
        if approval and not approved and not self.settings.auto_approve.get(approval, False):
            thread.pending_tool = tool_call
            thread.messages.append(ThreadMessage(
                "tool_request", f"awaiting approval ({approval}) for {tool_call.name}",
                tool_name=tool_call.name, tool_params=params))
            return False  # awaitingUserApproval

        t0 = time.time()
        try:
            result = self._tools.call_tool(tool_call.name, params)
            ok, text = True, result.text
        except ToolError as e:
            ok, text = False, f"Tool error: {e}"
        except Exception as e:  # noqa: BLE001 — tool crash becomes tool_error
            ok, text = False, f"Tool crashed: {e}"
        duration = (time.time() - t0) * 1000
        thread.messages.append(ThreadMessage("tool", text, tool_name=tool_call.name,
                                             tool_params=params, tool_success=ok))
        self._traces.record_tool_call(thread.id, len(thread.messages) - 1,
                                      tool_name=tool_call.name,
                                      tool_params=str(params), tool_result=text,
                                      tool_success=ok, duration=duration)
        return True

    # ---- approval resume (reference :841-869) ----

    def approve_latest_tool_request(self, thread_id: str) -> None:
        thread = self._threads[thread_id]
        if thread.pending_tool is None:
            return
        tool = thread.pending_tool
        thread.pending_tool = None
        self._run_chat_agent(thread, preapproved_tool=tool)
        self._traces.end_trace_for_thread(thread_id)

    def reject_latest_tool_request(self, thread_id: str) -> None:
        thread = self._threads[thread_id]
        if thread.pending_tool is None:
            return
        tool = thread.pending_tool
        thread.pending_tool = None
        thread.messages.append(ThreadMessage("tool", "Tool request rejected by user",
                                             tool_name=tool.name, tool_success=False))
