"""Subagent runner — the spawn_subagent tool backend.

Capability-compatible with the reference's SubagentToolService
(browser/subagentToolService.ts): independent-context subagents with
MAX_PARALLEL_SUBAGENTS=8, MAX_SUBAGENT_DEPTH=4, CONTEXT_LOW_THRESHOLD=0.25,
5-minute default timeout (:33-36), a single LLM call per subagent (:414-430),
the same system-prompt template (:437-458) and the 4-chars/token 128k
context estimate (:361-366).  The "LLM call" runs through the local
transport (MI355X backbone) instead of IPC+HTTPS.
"""

from __future__ import annotations

import threading
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

MAX_PARALLEL_SUBAGENTS = 8
MAX_SUBAGENT_DEPTH = 4
CONTEXT_LOW_THRESHOLD = 0.25
DEFAULT_SUBAGENT_TIMEOUT_MS = 300000
CONTEXT_WINDOW_ESTIMATE = 128000
CHARS_PER_TOKEN_ESTIMATE = 4


@dataclass
class SubagentInput:
    label: str
    task_prompt: str
    summary_prompt: str = "Summarize your findings concisely."
    context_low_prompt: str = "Context is low; summarize now."
    timeout_ms: int = DEFAULT_SUBAGENT_TIMEOUT_MS
    allowed_tools: Optional[List[str]] = None


@dataclass
class SubagentResult:
    task_id: str
    success: bool
    summary: str
    execution_time_ms: float
    context_exhausted: bool = False
    error: Optional[str] = None


def build_subagent_system_prompt(inp: SubagentInput) -> str:
    tools = ", ".join(inp.allowed_tools) if inp.allowed_tools else "All tools from parent agent"
    return f"""You are a subagent with a specific task to complete.

## Your Task
{inp.label}

## Guidelines
1. Focus on completing the assigned task efficiently
2. Be concise in your responses
3. If you encounter errors, try alternative approaches
4. When you complete the task or cannot proceed further, clearly state your findings

## Available Tools
{tools}

## Important
- You have a limited context window. Be concise in your responses.
- Always respond with actionable information that helps the parent agent.

## Summary Requirement
{inp.summary_prompt}"""


class SubagentRunner:
    """Runs subagent tasks on the local transport with depth/parallel limits."""

    def __init__(self, llm_service) -> None:
        self._llm = llm_service
        self._active = 0
        self._lock = threading.Lock()
        self.progress_events: List[Dict[str, Any]] = []

    def spawn(self, inp: SubagentInput, depth: int = 0,
              max_new_tokens: int = 512) -> SubagentResult:
        task_id = str(uuid.uuid4())
        t0 = time.time()
        if depth >= MAX_SUBAGENT_DEPTH:
            return SubagentResult(task_id, False, "", 0,
                                  error=f"Max subagent depth {MAX_SUBAGENT_DEPTH} exceeded")
        with self._lock:
            if self._active >= MAX_PARALLEL_SUBAGENTS:
                return SubagentResult(task_id, False, "", 0,
                                      error=f"Max parallel subagents {MAX_PARALLEL_SUBAGENTS} exceeded")
            self._active += 1
        try:
            from ..transport.service import LLMChatMessage

            state: Dict[str, Any] = {"text": "", "error": None, "done": threading.Event(),
                                     "context_exhausted": False}

            def on_text(full_text="", **kw):
                state["text"] = full_text
                used = -(-len(full_text) // CHARS_PER_TOKEN_ESTIMATE)  # ceil
                pct = used / CONTEXT_WINDOW_ESTIMATE
                if pct > (1 - CONTEXT_LOW_THRESHOLD):
                    state["context_exhausted"] = True
                    self.progress_events.append({"taskId": task_id, "type": "context_low",
                                                 "percentage": pct})

            def on_final(full_text="", **kw):
                state["text"] = full_text or state["text"]
                state["done"].set()

            def on_error(message="", **kw):
                state["error"] = message
                state["done"].set()

            messages = [
                LLMChatMessage("system", build_subagent_system_prompt(inp)),
                LLMChatMessage("user", inp.task_prompt),
            ]
            req = self._llm.send_llm_message(messages, on_text, on_final, on_error,
                                             max_new_tokens=max_new_tokens)
            finished = state["done"].wait(timeout=inp.timeout_ms / 1000)
            if not finished:
                self._llm.abort(req)
                return SubagentResult(task_id, False, state["text"],
                                      (time.time() - t0) * 1000, error="Subagent timed out")
            if state["error"]:
                return SubagentResult(task_id, False, "", (time.time() - t0) * 1000,
                                      error=state["error"])
            summary = state["text"] or f'[Subagent "{inp.label}"] Task completed.'
            return SubagentResult(task_id, True, summary, (time.time() - t0) * 1000,
                                  context_exhausted=state["context_exhausted"])
        finally:
            with self._lock:
                self._active -= 1
