"""Built-in tool registry — the 31 tools of the reference, with their param
schemas and approval classes.

Reference: common/prompt/prompts.ts:225-719 (registry),
common/toolsServiceTypes.ts:28-49 (approval classes),
prompts.ts:730-753 (availableTools filter by chatMode / vision).
"""

from __future__ import annotations

from typing import Dict, List, Optional

# name -> ordered param names (subset semantics like the reference: more can
# be parsed, these are the recognized ones)
BUILTIN_TOOLS: Dict[str, List[str]] = {
    # --- context-gathering (read/search/list) ---
    "read_file": ["uri", "start_line", "end_line", "page_number"],
    "ls_dir": ["uri", "page_number"],
    "get_dir_tree": ["uri"],
    "search_pathnames_only": ["query", "include_pattern", "page_number"],
    "search_for_files": ["query", "search_in_folder", "is_regex", "page_number"],
    "search_in_file": ["uri", "query", "is_regex"],
    "read_lint_errors": ["uri"],
    # --- editing ---
    "create_file_or_folder": ["uri"],
    "delete_file_or_folder": ["uri", "is_recursive"],
    "edit_file": ["uri", "search_replace_blocks"],
    "rewrite_file": ["uri", "new_content"],
    # --- terminal ---
    "run_command": ["command", "cwd"],
    "run_persistent_command": ["command", "persistent_terminal_id"],
    "open_persistent_terminal": ["cwd"],
    "kill_persistent_terminal": ["persistent_terminal_id"],
    # --- web / browser ---
    "open_browser": ["url", "headless"],
    "fetch_url": ["url", "method", "headers", "body", "crawl_links", "max_pages", "max_depth"],
    "web_search": ["query", "max_results"],
    # --- vision / docs ---
    "analyze_image": ["image_data", "prompt", "api_key", "model"],
    "screenshot_to_code": ["source", "image_data", "url", "stack", "custom_prompt"],
    "api_request": ["url", "method", "headers", "body", "auth", "timeout"],
    "read_document": ["uri", "start_index", "max_length"],
    "edit_document": ["uri", "content", "backup", "replacements"],
    "create_document": ["type", "file_path", "document_data", "options"],
    "pdf_operation": ["operation", "input_files", "input_file", "output_path",
                      "output_dir", "watermark_text", "options"],
    "document_convert": ["input_file", "output_path", "format", "options"],
    "document_merge": ["input_files", "output_path", "options"],
    "document_extract": ["input_file", "output_dir", "extract_type", "options"],
    # --- agents / skills ---
    "spawn_subagent": ["label", "task_prompt", "summary_prompt", "context_low_prompt",
                       "timeout_ms", "allowed_tools"],
    "edit_agent": ["uri", "mode", "description", "current_content", "selection_range"],
    "skill": ["name"],
}

# approval classes — toolsServiceTypes.ts:28-37
APPROVAL_TYPE_OF_TOOL: Dict[str, str] = {
    "create_file_or_folder": "edits",
    "delete_file_or_folder": "edits",
    "rewrite_file": "edits",
    "edit_file": "edits",
    "run_command": "terminal",
    "run_persistent_command": "terminal",
    "open_persistent_terminal": "terminal",
    "kill_persistent_terminal": "terminal",
}

TOOL_APPROVAL_TYPES = {"edits", "terminal", "MCP tools"}


def is_builtin_tool(name: str) -> bool:
    return name in BUILTIN_TOOLS


def available_tools(chat_mode: Optional[str], mcp_tools: Optional[List[dict]] = None,
                    supports_vision: bool = False) -> Optional[List[dict]]:
    """Reference availableTools (prompts.ts:730-753): 'normal' gets none,
    'gather' gets the approval-free subset, 'agent'/'designer' get all; MCP
    tools only in agent mode; analyze_image dropped when the model has
    native vision."""
    if chat_mode == "normal" or chat_mode is None:
        names = None
    elif chat_mode == "gather":
        names = [n for n in BUILTIN_TOOLS if n not in APPROVAL_TYPE_OF_TOOL]
    elif chat_mode in ("agent", "designer"):
        names = list(BUILTIN_TOOLS)
    else:
        names = None
    if supports_vision and names:
        names = [n for n in names if n != "analyze_image"]
    builtin = [{"name": n, "params": BUILTIN_TOOLS[n]} for n in names] if names else None
    mcp = mcp_tools if chat_mode == "agent" else None
    if builtin is None and mcp is None:
        return None
    return (builtin or []) + (mcp or [])
