"""Multi-agent registry: 3 primary + 7 sub + 3 system agents with per-agent
tool permissions, chatMode compositions, and keyword-based recommendation.

Capability-compatible with the reference's static registry
(common/agentService.ts:114-460 definitions, :486-522 AGENT_COMPOSITIONS,
:583-613 recommendSubAgents, :643-665 shouldUseSubAgents).
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Union


@dataclass
class AgentPermission:
    can_read: bool
    can_write: bool
    can_delete: bool
    allowed_tools: Union[str, List[str]]  # '*' or explicit list
    denied_tools: List[str]
    can_access_network: bool
    can_execute_terminal: bool
    can_use_mcp: bool


@dataclass
class AgentDefinition:
    id: str
    name: str
    description: str
    mode: str  # 'primary' | 'subagent' | 'system'
    permission: AgentPermission
    system_prompt: Optional[str] = None
    temperature: Optional[float] = None
    max_steps: Optional[int] = None
    hidden: bool = False


FULL = AgentPermission(True, True, True, "*", [], True, True, True)
READ_ONLY = AgentPermission(
    True, False, False,
    ["read_file", "ls_dir", "get_dir_tree", "search_pathnames_only",
     "search_for_files", "search_in_file", "read_lint_errors"],
    ["rewrite_file", "edit_file", "create_file_or_folder", "delete_file_or_folder", "run_command"],
    True, False, False)
EXPLORE_PERM = AgentPermission(
    True, False, False,
    ["read_file", "ls_dir", "get_dir_tree", "search_pathnames_only",
     "search_for_files", "search_in_file", "web_search", "fetch_url"],
    ["rewrite_file", "edit_file", "create_file_or_folder", "delete_file_or_folder"],
    True, False, False)
SYSTEM_PERM = AgentPermission(False, False, False, [], [], False, False, False)


def _d(**kw) -> AgentDefinition:
    return AgentDefinition(**kw)


BUILTIN_AGENTS: Dict[str, AgentDefinition] = {
    # primary
    "build": _d(id="build", name="Build Agent", mode="primary", permission=FULL,
                description="Primary build agent: full read/write/terminal/tool access",
                max_steps=50),
    "chat": _d(id="chat", name="Chat Agent", mode="primary",
               permission=AgentPermission(**{**READ_ONLY.__dict__, "can_access_network": True}),
               description="Conversation agent: reads code, never edits directly",
               max_steps=20),
    "designer": _d(id="designer", name="Designer Agent", mode="primary", permission=FULL,
                   description="Designer agent: UI/component and full-stack work",
                   max_steps=100),
    # sub-agents
    "explore": _d(id="explore", name="Explore Agent", mode="subagent", permission=EXPLORE_PERM,
                  description="Fast read-only codebase exploration",
                  system_prompt="You are a code-exploration agent. Search files and paths, "
                                "read code, map the project structure. You cannot modify files; "
                                "find the relevant code quickly and report a clear analysis.",
                  max_steps=15, temperature=0.3),
    "plan": _d(id="plan", name="Plan Agent", mode="subagent",
               permission=AgentPermission(**{**READ_ONLY.__dict__, "allowed_tools": [
                   "read_file", "ls_dir", "get_dir_tree", "search_pathnames_only", "search_for_files"]}),
               description="Breaks complex tasks into an executable step plan",
               system_prompt="You are a task-planning agent. Understand the goal, inspect the "
                             "codebase state, produce a numbered execution plan and note risks "
                             "and dependencies.",
               max_steps=10, temperature=0.2),
    "code": _d(id="code", name="Code Agent", mode="subagent",
               permission=AgentPermission(True, True, False,
                                          ["read_file", "edit_file", "rewrite_file", "create_file_or_folder",
                                           "search_for_files", "search_in_file", "read_lint_errors"],
                                          ["delete_file_or_folder", "run_command"], False, False, False),
               description="Writes and modifies code for concrete tasks",
               system_prompt="You are a coding agent. Follow the existing style, keep changes "
                             "minimal and clear, add error handling, keep comments, and check "
                             "lint errors after editing.",
               max_steps=30, temperature=0.1),
    "review": _d(id="review", name="Review Agent", mode="subagent", permission=READ_ONLY,
                 description="Reviews code for correctness, performance, security and style",
                 system_prompt="You are a code-review agent. Check correctness, performance, "
                               "security, style and best practices; list each finding with a "
                               "suggestion.",
                 max_steps=10, temperature=0.2),
    "test": _d(id="test", name="Test Agent", mode="subagent",
               permission=AgentPermission(True, True, False,
                                          ["read_file", "edit_file", "rewrite_file", "create_file_or_folder",
                                           "search_for_files", "run_command"],
                                          ["delete_file_or_folder"], False, True, False),
               description="Writes and runs tests to verify behavior",
               max_steps=20, temperature=0.1),
    "ui": _d(id="ui", name="UI Agent", mode="subagent",
             permission=AgentPermission(True, True, False,
                                        ["read_file", "edit_file", "rewrite_file", "create_file_or_folder",
                                         "search_for_files", "web_search", "fetch_url"],
                                        ["delete_file_or_folder", "run_command"], True, False, False),
             description="Interface design, component development and styling",
             system_prompt="You are a UI design/development agent: modern styling, responsive "
                           "layout, good UX, design-system consistency, accessibility.",
             max_steps=30, temperature=0.3),
    "api": _d(id="api", name="API Agent", mode="subagent",
              permission=AgentPermission(True, True, False,
                                         ["read_file", "edit_file", "rewrite_file", "create_file_or_folder",
                                          "search_for_files", "web_search"],
                                         ["delete_file_or_folder"], True, False, False),
              description="Backend API design, implementation and docs",
              max_steps=25, temperature=0.1),
    # system
    "compaction": _d(id="compaction", name="Compaction Agent", mode="system",
                     permission=SYSTEM_PERM, hidden=True, temperature=0.3,
                     description="Generates concise summaries of conversation history"),
    "summary": _d(id="summary", name="Summary Agent", mode="system",
                  permission=SYSTEM_PERM, hidden=True, temperature=0.3,
                  description="Generates task-execution summary reports"),
    "title": _d(id="title", name="Title Agent", mode="system",
                permission=SYSTEM_PERM, hidden=True, temperature=0.5,
                description="Generates short conversation titles"),
}


@dataclass
class AgentComposition:
    primary_agent: str
    available_sub_agents: List[str]
    enable_parallel: bool
    max_parallel: int
    auto_select_sub_agents: bool


AGENT_COMPOSITIONS: Dict[str, AgentComposition] = {
    "normal": AgentComposition("chat", ["explore"], False, 1, False),
    "agent": AgentComposition("build", ["explore", "plan", "code", "review", "test"], True, 3, True),
    "designer": AgentComposition("designer", ["explore", "plan", "ui", "api", "code", "review"], True, 4, True),
    "gather": AgentComposition("chat", ["explore"], False, 1, False),
}


def get_agent_composition(chat_mode: str) -> AgentComposition:
    return AGENT_COMPOSITIONS[chat_mode]


def get_agent_definition(agent_id: str) -> Optional[AgentDefinition]:
    return BUILTIN_AGENTS.get(agent_id)


def get_visible_agents() -> List[AgentDefinition]:
    return [a for a in BUILTIN_AGENTS.values() if not a.hidden]


def get_agents_by_mode(mode: str) -> List[AgentDefinition]:
    return [a for a in BUILTIN_AGENTS.values() if a.mode == mode]


def can_agent_use_tool(agent_id: str, tool_name: str) -> bool:
    agent = get_agent_definition(agent_id)
    if not agent:
        return False
    p = agent.permission
    if tool_name in p.denied_tools:
        return False
    if p.allowed_tools == "*":
        return True
    return tool_name in p.allowed_tools


# keyword rules — agentService.ts:594-603 (bilingual keywords kept)
_RECOMMEND_RULES = [
    (["搜索", "查找", "找到", "探索", "search", "find", "explore", "locate"], "explore"),
    (["计划", "规划", "设计方案", "plan", "design"], "plan"),
    (["编写", "修改", "实现", "代码", "code", "implement", "write", "modify"], "code"),
    (["审查", "检查", "优化", "review", "check", "optimize"], "review"),
    (["测试", "验证", "test", "verify"], "test"),
    (["界面", "ui", "组件", "样式", "component", "style", "layout"], "ui"),
    (["接口", "api", "后端", "backend", "endpoint"], "api"),
]

_COMPLEX_KEYWORDS = [
    "重构", "优化", "实现", "创建", "设计",
    "refactor", "optimize", "implement", "create", "design",
    "多个文件", "整个项目", "全面",
    "multiple files", "entire project", "comprehensive",
]


def recommend_sub_agents(task_description: str, chat_mode: str) -> List[str]:
    comp = get_agent_composition(chat_mode)
    if not comp.auto_select_sub_agents:
        return []
    lower = task_description.lower()
    recommended: List[str] = []
    for keywords, agent in _RECOMMEND_RULES:
        if any(kw in lower for kw in keywords) and agent in comp.available_sub_agents:
            recommended.append(agent)
    seen = []
    for a in recommended:
        if a not in seen:
            seen.append(a)
    return seen[: comp.max_parallel]


def should_use_sub_agents(task_description: str, chat_mode: str) -> bool:
    comp = get_agent_composition(chat_mode)
    if not comp.auto_select_sub_agents:
        return False
    if len(task_description) < 50:
        return False
    lower = task_description.lower()
    return any(kw in lower for kw in _COMPLEX_KEYWORDS)


def create_agent_execution_context(agent_id: str, task_description: str,
                                   parent_agent_id: Optional[str] = None,
                                   depth: int = 0, max_depth: int = 3) -> dict:
    return {
        "agentId": agent_id,
        "parentAgentId": parent_agent_id,
        "taskDescription": task_description,
        "depth": depth,
        "maxDepth": max_depth,
        "startTime": int(time.time() * 1000),
    }
