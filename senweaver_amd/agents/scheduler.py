"""Session-scoped multi-agent scheduler.

Capability-compatible with the reference's AgentScheduler
(common/agentScheduler.ts): planSubAgents via keyword recommendation (:125),
chunked parallel execution bounded by the composition's maxParallel with
per-chunk settle semantics (:203-258), and result merging (:314-330).
Parallelism here is a thread pool (the reference used Promise.allSettled
chunks on the JS event loop).
"""

from __future__ import annotations

import time
from concurrent.futures import ThreadPoolExecutor, wait
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from .registry import (
    create_agent_execution_context,
    get_agent_composition,
    get_agent_definition,
    recommend_sub_agents,
    should_use_sub_agents,
)


@dataclass
class SubAgentTask:
    agent_id: str
    task_description: str
    priority: int
    status: str = "pending"  # pending | running | completed | failed
    result: Optional[Dict[str, Any]] = None


@dataclass
class SchedulingSession:
    id: str
    chat_mode: str
    primary_agent_id: str
    sub_agent_tasks: List[SubAgentTask] = field(default_factory=list)
    start_time: float = field(default_factory=lambda: time.time() * 1000)
    end_time: Optional[float] = None
    status: str = "planning"


# per-subagent task phrasing (agentScheduler.ts:152-177)
_SUBTASK_TEMPLATES = {
    "explore": "Explore the codebase and find the files and code relevant to: {task}",
    "plan": "Analyze the following task and produce an execution plan: {task}",
    "code": "Implement the following coding task: {task}",
    "review": "Review the code related to the following task: {task}",
    "test": "Write tests for the following functionality: {task}",
    "ui": "Design and implement the following UI: {task}",
    "api": "Design and implement the following API: {task}",
}


class AgentScheduler:
    def __init__(self) -> None:
        self._session_counter = 0
        self._current: Optional[SchedulingSession] = None
        self.events: List[Dict[str, Any]] = []

    def start_session(self, chat_mode: str) -> SchedulingSession:
        comp = get_agent_composition(chat_mode)
        self._session_counter += 1
        self._current = SchedulingSession(
            id=f"session_{self._session_counter}", chat_mode=chat_mode,
            primary_agent_id=comp.primary_agent)
        self.events.append({"sessionId": self._current.id, "type": "session_start"})
        return self._current

    @property
    def current_session(self) -> Optional[SchedulingSession]:
        return self._current

    def plan_sub_agents(self, task_description: str) -> List[str]:
        if not self._current:
            return []
        mode = self._current.chat_mode
        if not should_use_sub_agents(task_description, mode):
            return []
        recommended = recommend_sub_agents(task_description, mode)
        self._current.sub_agent_tasks = [
            SubAgentTask(agent_id=a,
                         task_description=self._sub_task_description(a, task_description),
                         priority=i)
            for i, a in enumerate(recommended)
        ]
        return recommended

    @staticmethod
    def _sub_task_description(agent_id: str, original: str) -> str:
        tpl = _SUBTASK_TEMPLATES.get(agent_id)
        return tpl.format(task=original) if tpl else original

    def add_sub_agent_task(self, agent_id: str, task_description: str,
                           priority: int = 0) -> Optional[SubAgentTask]:
        if not self._current:
            return None
        agent = get_agent_definition(agent_id)
        if not agent or agent.mode != "subagent":
            return None
        task = SubAgentTask(agent_id=agent_id, task_description=task_description,
                            priority=priority)
        self._current.sub_agent_tasks.append(task)
        return task

    def execute_sub_agent_tasks(self, executor: Callable[[dict], Dict[str, Any]]
                                ) -> List[Dict[str, Any]]:
        """Run pending tasks; parallel in chunks of maxParallel like the
        reference's Promise.allSettled chunks — a failed task is skipped,
        the chunk settles before the next starts."""
        if not self._current:
            return []
        session = self._current
        session.status = "executing"
        comp = get_agent_composition(session.chat_mode)
        pending = [t for t in session.sub_agent_tasks if t.status == "pending"]
        results: List[Dict[str, Any]] = []
        if not pending:
            session.status = "completed"
            return results

        def run_one(task: SubAgentTask) -> Optional[Dict[str, Any]]:
            task.status = "running"
            ctx = create_agent_execution_context(task.agent_id, task.task_description)
            try:
                res = executor(ctx)
                task.status = "completed"
                task.result = res
                return res
            except Exception as e:
                task.status = "failed"
                task.result = {"agentId": task.agent_id, "success": False, "error": str(e)}
                return None

        if comp.enable_parallel:
            with ThreadPoolExecutor(max_workers=comp.max_parallel) as pool:
                for i in range(0, len(pending), comp.max_parallel):
                    chunk = pending[i: i + comp.max_parallel]
                    futures = [pool.submit(run_one, t) for t in chunk]
                    wait(futures)
                    for f in futures:
                        r = f.result()
                        if r is not None:
                            results.append(r)
        else:
            for t in pending:
                r = run_one(t)
                if r is not None:
                    results.append(r)
        session.status = "completed"
        session.end_time = time.time() * 1000
        self.events.append({"sessionId": session.id, "type": "session_complete"})
        return results

    @staticmethod
    def merge_sub_agent_results(results: List[Dict[str, Any]]) -> str:
        """Reference mergeSubAgentResults (:314-330): labeled sections."""
        if not results:
            return ""
        parts = []
        for r in results:
            agent = get_agent_definition(r.get("agentId", "")) if r.get("agentId") else None
            name = agent.name if agent else r.get("agentId", "unknown")
            status = "OK" if r.get("success") else "FAILED"
            parts.append(f"## {name} [{status}]\n{r.get('output', r.get('error', ''))}")
        return "\n\n".join(parts)


_global_scheduler: Optional[AgentScheduler] = None


def get_agent_scheduler() -> AgentScheduler:
    global _global_scheduler
    if _global_scheduler is None:
        _global_scheduler = AgentScheduler()
    return _global_scheduler
