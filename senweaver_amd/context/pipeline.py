"""System-message assembly + prepared-request pipeline.

Rebuilds the reference's ConvertToLLMMessageService surface
(browser/convertToLLMMessageService.ts): system-message generation with the
multi-agent section (:788-830) and the APO 2000-char rule injection
(:832-853, via senweaver_amd.apo.rules.inject_rules), key+timestamp caching
(:661-663), and prepare_llm_chat_messages combining history compression with
the 4-phase fitting.
"""

from __future__ import annotations

import time
from typing import List, Optional, Tuple

from ..agents.registry import get_agent_composition, get_agent_definition
from ..apo.rules import inject_rules
from .compress import CompressibleMessage, compress_old_messages
from .fitting import Msg, prepare_messages

SYSTEM_CACHE_TTL_MS = 30_000


def base_system_message(chat_mode: str, workspace_overview: str = "",
                        system_info: str = "",
                        workspace_folders: Optional[List[str]] = None,
                        active_file: Optional[str] = None,
                        open_files: Optional[List[str]] = None,
                        terminal_ids: Optional[List[str]] = None,
                        os_name: str = "",
                        now: Optional[str] = None,
                        mcp_tool_names: Optional[List[str]] = None,
                        custom_api_description: str = "",
                        supports_vision: bool = True) -> str:
    """Analog of chat_systemMessage (prompts.ts:806-1237) with the same
    information slots in the same order: mode header, <system_info> (OS,
    current date/time, workspace folders, active file, open files, agent-mode
    persistent terminal IDs), <files_overview>, then MCP/custom-API/vision
    notes.  Tool XML definitions are appended by the transport's tool
    registry when tools are enabled; the prose itself is original (the
    reference's ~1800-line wording is product copy)."""
    headers = {
        "normal": "You are SenWeaver-AMD, an expert coding assistant. Answer questions "
                  "about the user's codebase; you may read files but not modify them.",
        "agent": "You are SenWeaver-AMD in agent mode: complete the user's task end to "
                 "end using the available tools. Verify your work before finishing.",
        "designer": "You are SenWeaver-AMD in designer mode: build and refine UI "
                    "components and their backing APIs. Plan the complete set of "
                    "related pages before generating, then produce them one by one.",
        "gather": "You are SenWeaver-AMD in gather mode: collect and summarize relevant "
                  "context from the codebase. Read-only.",
    }
    parts = [headers.get(chat_mode, headers["normal"])]
    if system_info:
        parts.append(f"<system_info>\n{system_info}\n</system_info>")
    else:
        info_lines = []
        if os_name:
            info_lines.append(f"- Operating System: {os_name}")
        if now:
            info_lines.append(f"- Current date/time: {now} (reference point for "
                              "relative dates like today/tomorrow/this week)")
        info_lines.append("- Workspace folders:\n"
                          + ("\n".join(workspace_folders) if workspace_folders
                             else "NO FOLDERS OPEN"))
        info_lines.append(f"- Active file:\n{active_file or 'NO ACTIVE FILE'}")
        info_lines.append("- Open files:\n"
                          + ("\n".join(open_files) if open_files else "NO OPENED FILES"))
        if chat_mode == "agent" and terminal_ids:
            info_lines.append("- Persistent terminal IDs available for commands: "
                              + ", ".join(terminal_ids))
        parts.append("<system_info>\n" + "\n".join(info_lines) + "\n</system_info>")
    if workspace_overview:
        parts.append(f"<files_overview>\n{workspace_overview}\n</files_overview>")
    if mcp_tool_names:
        parts.append("MCP tools available in this session: " + ", ".join(mcp_tool_names))
    if custom_api_description:
        parts.append(f"Custom API: {custom_api_description}")
    if not supports_vision:
        parts.append("This model has no vision capability: use the image-description "
                     "tool when the user provides images.")
    return "\n\n".join(parts)


def multi_agent_section(chat_mode: str) -> str:
    """The multi-agent system section (:788-830)."""
    comp = get_agent_composition(chat_mode)
    if not comp.available_sub_agents:
        return ""
    parts = [f"\n## Primary Agent\n{get_agent_definition(comp.primary_agent).name}"]
    subs = []
    for a in comp.available_sub_agents:
        d = get_agent_definition(a)
        if d:
            subs.append(f"- {d.name}: {d.description}")
    if subs:
        parts.append("\n## Available Sub-Agents\n" + "\n".join(subs))
    if comp.enable_parallel:
        parts.append(
            f"\n## Parallel Execution Capability\nYou support up to {comp.max_parallel} "
            "sub-tasks running in parallel. For sub-tasks that can be completed "
            "independently, prefer parallel processing to improve efficiency.")
    return "# Multi-Agent System" + "".join(parts)


class ConvertToLLMMessages:
    def __init__(self, apo_service=None, clock=None) -> None:
        self._apo = apo_service
        self._clock = clock or (lambda: int(time.time() * 1000))
        self._cached_msg: Optional[str] = None
        self._cached_key: Optional[str] = None
        self._cached_at = 0

    def invalidate_cache(self) -> None:
        """Directory-change invalidation hook (reference :698-700, 2s debounce)."""
        self._cached_msg = None

    def generate_system_message(self, chat_mode: str, workspace_overview: str = "",
                                system_info: str = "", **env_slots) -> str:
        """``env_slots`` pass through to base_system_message (workspace
        folders, active/open files, terminal IDs, MCP tools, ...)."""
        key = (f"{chat_mode}:{hash(workspace_overview)}:{hash(system_info)}:"
               f"{hash(tuple(sorted((k, str(v)) for k, v in env_slots.items())))}")
        now = self._clock()
        if (self._cached_msg is not None and self._cached_key == key
                and now - self._cached_at < SYSTEM_CACHE_TTL_MS):
            return self._cached_msg
        msg = base_system_message(chat_mode, workspace_overview, system_info,
                                  **env_slots)
        ma = multi_agent_section(chat_mode)
        if ma:
            msg = msg + "\n\n" + ma
        if self._apo is not None:
            try:
                msg = inject_rules(msg, self._apo.get_optimized_rules())
            except Exception:
                pass  # APO failure never breaks message assembly
        self._cached_msg = msg
        self._cached_key = key
        self._cached_at = now
        return msg

    def prepare_llm_chat_messages(self, history: List[CompressibleMessage],
                                  chat_mode: str, context_window: int,
                                  workspace_overview: str = "",
                                  reserved_output: Optional[int] = None
                                  ) -> Tuple[str, List[Msg]]:
        """History compression -> 4-phase fit -> (system_message, messages)."""
        sys_msg = self.generate_system_message(chat_mode, workspace_overview)
        compressed = compress_old_messages(history)
        msgs = [Msg(m.role if m.role != "tool" else "assistant",
                    m.content if m.role != "tool" else f"[tool {m.tool_name} result]\n{m.content}")
                for m in compressed]
        return prepare_messages(msgs, sys_msg, context_window, reserved_output)
