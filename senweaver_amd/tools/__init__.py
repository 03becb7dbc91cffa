from .registry import (
    APPROVAL_TYPE_OF_TOOL,
    BUILTIN_TOOLS,
    TOOL_APPROVAL_TYPES,
    available_tools,
    is_builtin_tool,
)
from .service import PersistentTerminal, ToolError, ToolResult, ToolsService, parse_search_replace_blocks
from .skills import Skill, SkillService, parse_frontmatter
