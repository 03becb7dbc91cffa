from .registry import (
    AGENT_COMPOSITIONS,
    BUILTIN_AGENTS,
    AgentComposition,
    AgentDefinition,
    AgentPermission,
    can_agent_use_tool,
    create_agent_execution_context,
    get_agent_composition,
    get_agent_definition,
    get_agents_by_mode,
    get_visible_agents,
    recommend_sub_agents,
    should_use_sub_agents,
)
from .scheduler import AgentScheduler, SchedulingSession, SubAgentTask, get_agent_scheduler
from .subagents import (
    CONTEXT_LOW_THRESHOLD,
    DEFAULT_SUBAGENT_TIMEOUT_MS,
    MAX_PARALLEL_SUBAGENTS,
    MAX_SUBAGENT_DEPTH,
    SubagentInput,
    SubagentResult,
    SubagentRunner,
    build_subagent_system_prompt,
)
