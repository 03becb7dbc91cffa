"""ToolsService — validation + execution of the built-in tools.

Capability-compatible with the reference's ToolsService
(browser/toolsService.ts: callTool map :1693, stringOfResult map :3265,
param validation).  The filesystem/search/terminal tools are implemented
locally (sandboxed under a workspace root); web/vision/document tools —
which the reference served via localhost sidecar HTTP servers
(browser/start*.cjs, ports 3000-3008) — raise a structured offline error in
this no-network environment.
"""

from __future__ import annotations

import fnmatch
import os
import re
import shutil
import subprocess
import threading
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Tuple

from .registry import APPROVAL_TYPE_OF_TOOL, BUILTIN_TOOLS, is_builtin_tool

MAX_TERMINAL_CHARS = 100_000            # prompts.ts:29
MAX_TERMINAL_INACTIVE_TIME_S = 8        # prompts.ts:30
MAX_TERMINAL_BG_COMMAND_TIME_S = 5      # prompts.ts:31
MAX_FILE_CHARS = 200_000
PAGE_SIZE_RESULTS = 50

# search/replace block format — prompts.ts:38-40
SR_ORIGINAL = "<<<<<<< ORIGINAL"
SR_DIVIDER = "======="
SR_UPDATED = ">>>>>>> UPDATED"


class ToolError(Exception):
    pass


@dataclass
class ToolResult:
    tool_name: str
    result: Any
    text: str  # stringOfResult


def parse_search_replace_blocks(blocks: str) -> List[Tuple[str, str]]:
    """Parse '<<<<<<< ORIGINAL / ======= / >>>>>>> UPDATED' blocks."""
    out: List[Tuple[str, str]] = []
    lines = blocks.split("\n")
    i = 0
    while i < len(lines):
        if lines[i].strip() == SR_ORIGINAL:
            orig: List[str] = []
            updated: List[str] = []
            i += 1
            while i < len(lines) and lines[i].strip() != SR_DIVIDER:
                orig.append(lines[i])
                i += 1
            i += 1  # divider
            while i < len(lines) and lines[i].strip() != SR_UPDATED:
                updated.append(lines[i])
                i += 1
            out.append(("\n".join(orig), "\n".join(updated)))
        i += 1
    if not out:
        raise ToolError("No ORIGINAL/UPDATED search-replace blocks found")
    return out


class PersistentTerminal:
    def __init__(self, cwd: str) -> None:
        self.id = str(uuid.uuid4())[:8]
        self.cwd = cwd
        self.proc = subprocess.Popen(
            ["/bin/bash"], stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, cwd=cwd, text=True, bufsize=1)
        self._buffer: List[str] = []
        self._lock = threading.Lock()
        self._reader = threading.Thread(target=self._read, daemon=True)
        self._reader.start()

    def _read(self) -> None:
        try:
            for line in self.proc.stdout:
                with self._lock:
                    self._buffer.append(line)
        except Exception:
            pass

    def _respawn(self) -> None:
        self.proc = subprocess.Popen(
            ["/bin/bash"], stdin=subprocess.PIPE, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, cwd=self.cwd, text=True, bufsize=1)
        self._reader = threading.Thread(target=self._read, daemon=True)
        self._reader.start()

    def run(self, command: str, wait_s: float = MAX_TERMINAL_BG_COMMAND_TIME_S) -> str:
        if self.proc.poll() is not None:  # shell exited (e.g. `exit`)
            self._respawn()
        with self._lock:
            self._buffer.clear()
        marker = f"__DONE_{uuid.uuid4().hex[:8]}__"
        try:
            self.proc.stdin.write(command + f"\necho {marker}\n")
            self.proc.stdin.flush()
        except (BrokenPipeError, OSError):
            self._respawn()
            self.proc.stdin.write(command + f"\necho {marker}\n")
            self.proc.stdin.flush()
        deadline = time.time() + wait_s
        while time.time() < deadline:
            with self._lock:
                text = "".join(self._buffer)
            if marker in text:
                return text.split(marker)[0][:MAX_TERMINAL_CHARS]
            time.sleep(0.05)
        with self._lock:
            return ("".join(self._buffer))[:MAX_TERMINAL_CHARS] + "\n[command still running in background]"

    def kill(self) -> None:
        try:
            self.proc.kill()
        except Exception:
            pass


class ToolsService:
    def __init__(self, workspace_root: str, subagent_runner=None,
                 edit_backend=None) -> None:
        self.root = os.path.abspath(workspace_root)
        self.terminals: Dict[str, PersistentTerminal] = {}
        self._subagents = subagent_runner
        self._edit_backend = edit_backend  # PromptOptimizerBackend-like .generate

    # ---- path safety ----
    def _resolve(self, uri: Optional[str]) -> str:
        p = uri or ""
        if not os.path.isabs(p):
            p = os.path.join(self.root, p)
        p = os.path.abspath(p)
        if not (p == self.root or p.startswith(self.root + os.sep)):
            raise ToolError(f"path {uri!r} escapes the workspace root")
        return p

    # ---- validation (reference validateParams semantics) ----
    def validate_params(self, tool_name: str, raw: Dict[str, str]) -> Dict[str, Any]:
        if not is_builtin_tool(tool_name):
            raise ToolError(f"unknown tool {tool_name!r}")
        allowed = set(BUILTIN_TOOLS[tool_name])
        params = {k: v for k, v in raw.items() if k in allowed}
        required = {
            "read_file": ["uri"], "get_dir_tree": ["uri"], "search_pathnames_only": ["query"],
            "search_for_files": ["query"], "search_in_file": ["uri", "query"],
            "read_lint_errors": ["uri"], "create_file_or_folder": ["uri"],
            "delete_file_or_folder": ["uri"], "edit_file": ["uri", "search_replace_blocks"],
            "rewrite_file": ["uri", "new_content"], "run_command": ["command"],
            "run_persistent_command": ["command", "persistent_terminal_id"],
            "kill_persistent_terminal": ["persistent_terminal_id"],
            "spawn_subagent": ["label", "task_prompt"], "skill": ["name"],
        }.get(tool_name, [])
        missing = [r for r in required if not params.get(r)]
        if missing:
            raise ToolError(f"missing required param(s) {missing} for {tool_name}")
        for bkey in ("is_regex", "is_recursive", "headless", "crawl_links"):
            if bkey in params:
                params[bkey] = str(params[bkey]).strip().lower() in ("true", "1", "yes")
        for ikey in ("start_line", "end_line", "page_number", "timeout_ms", "max_results"):
            if ikey in params and params[ikey] not in (None, ""):
                try:
                    params[ikey] = int(str(params[ikey]).strip())
                except ValueError:
                    raise ToolError(f"param {ikey} must be an integer")
        return params

    def approval_type(self, tool_name: str) -> Optional[str]:
        return APPROVAL_TYPE_OF_TOOL.get(tool_name)

    # ---- execution ----
    def call_tool(self, tool_name: str, params: Dict[str, Any]) -> ToolResult:
        fn = getattr(self, f"_tool_{tool_name}", None)
        if fn is None:
            raise ToolError(f"tool {tool_name!r} is not available in this environment "
                            "(reference serves it via a localhost sidecar; offline here)")
        return fn(params)

    # --- context-gathering ---
    def _tool_read_file(self, p):
        path = self._resolve(p["uri"])
        with open(path, "r", encoding="utf-8", errors="replace") as f:
            lines = f.readlines()
        s = p.get("start_line") or 1
        e = p.get("end_line") or len(lines)
        text = "".join(lines[s - 1: e])[:MAX_FILE_CHARS]
        return ToolResult("read_file", {"uri": p["uri"], "lines": len(lines)}, text)

    def _tool_ls_dir(self, p):
        path = self._resolve(p.get("uri") or "")
        entries = sorted(os.listdir(path))
        page = (p.get("page_number") or 1) - 1
        chunk = entries[page * PAGE_SIZE_RESULTS: (page + 1) * PAGE_SIZE_RESULTS]
        text = "\n".join((e + "/" if os.path.isdir(os.path.join(path, e)) else e) for e in chunk)
        return ToolResult("ls_dir", {"entries": chunk}, text)

    def _tool_get_dir_tree(self, p):
        # caps from directoryStrService.ts: MAX_FILES_TOTAL=1000, depth 3
        root = self._resolve(p.get("uri") or "")
        out: List[str] = []
        count = 0
        for dirpath, dirnames, filenames in os.walk(root):
            depth = dirpath[len(root):].count(os.sep)
            if depth >= 3:
                dirnames[:] = []
                continue
            dirnames[:] = [d for d in sorted(dirnames) if not d.startswith(".") and d != "__pycache__"]
            indent = "  " * depth
            out.append(f"{indent}{os.path.basename(dirpath) or '.'}/")
            for fn in sorted(filenames):
                if count >= 1000:
                    out.append(f"{indent}  ...[truncated at 1000 files]")
                    return ToolResult("get_dir_tree", {}, "\n".join(out))
                out.append(f"{indent}  {fn}")
                count += 1
        return ToolResult("get_dir_tree", {}, "\n".join(out))

    def _tool_search_pathnames_only(self, p):
        q = p["query"]
        include = p.get("include_pattern")
        hits = []
        for dirpath, dirnames, filenames in os.walk(self.root):
            dirnames[:] = [d for d in dirnames if not d.startswith(".") and d not in ("__pycache__", "node_modules")]
            for fn in filenames:
                rel = os.path.relpath(os.path.join(dirpath, fn), self.root)
                if q.lower() in rel.lower() and (not include or fnmatch.fnmatch(rel, include)):
                    hits.append(rel)
        page = (p.get("page_number") or 1) - 1
        chunk = hits[page * PAGE_SIZE_RESULTS: (page + 1) * PAGE_SIZE_RESULTS]
        return ToolResult("search_pathnames_only", {"hits": chunk}, "\n".join(chunk) or "(no matches)")

    def _tool_search_for_files(self, p):
        q = p["query"]
        rx = re.compile(q) if p.get("is_regex") else None
        base = self._resolve(p.get("search_in_folder") or "")
        hits = []
        for dirpath, dirnames, filenames in os.walk(base):
            dirnames[:] = [d for d in dirnames if not d.startswith(".") and d not in ("__pycache__", "node_modules")]
            for fn in filenames:
                fp = os.path.join(dirpath, fn)
                try:
                    if os.path.getsize(fp) > 2_000_000:
                        continue
                    with open(fp, "r", encoding="utf-8", errors="ignore") as f:
                        content = f.read()
                except OSError:
                    continue
                if (rx.search(content) if rx else q in content):
                    hits.append(os.path.relpath(fp, self.root))
                if len(hits) >= PAGE_SIZE_RESULTS:
                    break
        return ToolResult("search_for_files", {"hits": hits}, "\n".join(hits) or "(no matches)")

    def _tool_search_in_file(self, p):
        path = self._resolve(p["uri"])
        q = p["query"]
        rx = re.compile(q) if p.get("is_regex") else None
        lines = open(path, "r", encoding="utf-8", errors="replace").readlines()
        nums = [i + 1 for i, ln in enumerate(lines) if (rx.search(ln) if rx else q in ln)]
        return ToolResult("search_in_file", {"lines": nums}, ", ".join(map(str, nums)) or "(no matches)")

    def _tool_read_lint_errors(self, p):
        # marker-service backed (features/markers.py = _markerCheckService.ts
        # analog): lint the file through the registered providers, publish
        # markers, and report Error-severity diagnostics
        from ..features.markers import MarkerService, python_lint
        path = self._resolve(p["uri"])
        if not hasattr(self, "marker_service"):
            self.marker_service = MarkerService()
        markers = []
        if path.endswith(".py"):
            with open(path, encoding="utf-8", errors="replace") as f:
                markers = python_lint(path, f.read())
        self.marker_service.changed(path, markers)
        errors = [f"{m.startLineNumber}: {m.message}"
                  for m in self.marker_service.read(resource=path)]
        text = "\n".join(errors) if errors else "No lint errors found."
        return ToolResult("read_lint_errors", {"errors": errors}, text)

    # --- editing ---
    def _tool_create_file_or_folder(self, p):
        uri = p["uri"]
        path = self._resolve(uri)
        if uri.endswith("/"):
            os.makedirs(path, exist_ok=True)
        else:
            os.makedirs(os.path.dirname(path), exist_ok=True)
            if not os.path.exists(path):
                open(path, "w").close()
        return ToolResult("create_file_or_folder", {"uri": uri}, f"Created {uri}")

    def _tool_delete_file_or_folder(self, p):
        path = self._resolve(p["uri"])
        if os.path.isdir(path):
            if p.get("is_recursive"):
                shutil.rmtree(path)
            else:
                os.rmdir(path)
        elif os.path.exists(path):
            os.unlink(path)
        return ToolResult("delete_file_or_folder", {}, f"Deleted {p['uri']}")

    def _tool_rewrite_file(self, p):
        path = self._resolve(p["uri"])
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w", encoding="utf-8") as f:
            f.write(p["new_content"])
        return ToolResult("rewrite_file", {}, f"Rewrote {p['uri']}")

    def _tool_edit_file(self, p):
        path = self._resolve(p["uri"])
        content = open(path, "r", encoding="utf-8", errors="replace").read()
        for orig, updated in parse_search_replace_blocks(p["search_replace_blocks"]):
            if orig not in content:
                raise ToolError(f"ORIGINAL block not found in {p['uri']}: {orig[:80]!r}")
            content = content.replace(orig, updated, 1)
        with open(path, "w", encoding="utf-8") as f:
            f.write(content)
        return ToolResult("edit_file", {}, f"Applied edits to {p['uri']}")

    # --- terminal ---
    def _tool_run_command(self, p):
        cwd = self._resolve(p.get("cwd") or "")
        try:
            r = subprocess.run(["/bin/bash", "-c", p["command"]], cwd=cwd,
                               capture_output=True, text=True,
                               timeout=MAX_TERMINAL_INACTIVE_TIME_S * 4)
            out = (r.stdout + r.stderr)[:MAX_TERMINAL_CHARS]
            text = out + (f"\n[exit code {r.returncode}]" if r.returncode else "")
        except subprocess.TimeoutExpired as e:
            text = ((e.stdout or "") + (e.stderr or ""))[:MAX_TERMINAL_CHARS] + "\n[command timed out]"
        return ToolResult("run_command", {}, text)

    def _tool_open_persistent_terminal(self, p):
        term = PersistentTerminal(self._resolve(p.get("cwd") or ""))
        self.terminals[term.id] = term
        return ToolResult("open_persistent_terminal", {"persistent_terminal_id": term.id},
                          f"Opened persistent terminal {term.id}")

    def _tool_run_persistent_command(self, p):
        term = self.terminals.get(p["persistent_terminal_id"])
        if not term:
            raise ToolError(f"no persistent terminal {p['persistent_terminal_id']!r}")
        return ToolResult("run_persistent_command", {}, term.run(p["command"]))

    def _tool_kill_persistent_terminal(self, p):
        term = self.terminals.pop(p["persistent_terminal_id"], None)
        if term:
            term.kill()
        return ToolResult("kill_persistent_terminal", {}, "Terminal killed")

    # --- agents ---
    def _tool_spawn_subagent(self, p):
        if self._subagents is None:
            raise ToolError("subagent runner not configured")
        from ..agents.subagents import SubagentInput
        inp = SubagentInput(
            label=p["label"], task_prompt=p["task_prompt"],
            summary_prompt=p.get("summary_prompt") or "Summarize your findings concisely.",
            timeout_ms=p.get("timeout_ms") or 300000,
            allowed_tools=(p.get("allowed_tools") or "").split(",") if p.get("allowed_tools") else None,
        )
        res = self._subagents.spawn(inp)
        if not res.success:
            raise ToolError(res.error or "subagent failed")
        return ToolResult("spawn_subagent", {"taskId": res.task_id}, res.summary)

    def _tool_edit_agent(self, p):
        """LLM-driven file edit (reference browser/editAgentService.ts:458):
        modes 'create' | 'edit' | 'overwrite'; the model produces the new
        content from the description (+ current content for edits)."""
        if self._edit_backend is None:
            raise ToolError("edit_agent backend not configured")
        mode = (p.get("mode") or "edit").strip().lower()
        uri = p.get("uri")
        if not uri:
            raise ToolError("edit_agent requires uri")
        current = p.get("current_content")
        if current is None and mode != "create":
            path = self._resolve(uri)
            current = (open(path, encoding="utf-8", errors="replace").read()
                       if os.path.exists(path) else "")
        prompt = (f"File: {uri}\nTask ({mode}): {p.get('description', '')}\n"
                  + (f"Current content:\n{(current or '')[:8000]}\n" if mode != "create" else "")
                  + "Return ONLY the complete new file content.")
        new_content = self._edit_backend.generate(prompt, max_new_tokens=512)
        path = self._resolve(uri)
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with open(path, "w", encoding="utf-8") as f:
            f.write(new_content)
        return ToolResult("edit_agent", {"uri": uri, "mode": mode},
                          f"{mode} applied to {uri} ({len(new_content)} chars)")

    def _tool_api_request(self, p):
        """Custom/user API invocation (customApiService.ts + api_request tool).

        Offline environment: only loopback targets are reachable — a real
        local service (the engine daemon, an MCP HTTP server, a test
        server) can be called; anything else raises the structured offline
        error the other network tools use."""
        import json as _json
        import urllib.request
        import urllib.error
        from urllib.parse import urlparse
        url = p.get("url") or ""
        host = urlparse(url).hostname or ""
        if host not in ("127.0.0.1", "localhost", "::1"):
            raise ToolError(
                "api_request can only reach loopback targets in this "
                "environment (reference proxies it via a localhost sidecar; "
                "offline here)")
        method = (p.get("method") or "GET").upper()
        body = p.get("body")
        data = None
        if body not in (None, ""):
            data = body.encode() if isinstance(body, str) else _json.dumps(body).encode()
        req = urllib.request.Request(url, data=data, method=method)
        headers = p.get("headers")
        if isinstance(headers, str) and headers.strip():
            try:
                headers = _json.loads(headers)
            except ValueError:
                raise ToolError("headers must be a JSON object")
        for k, v in (headers or {}).items():
            req.add_header(k, str(v))
        timeout = min(int(p.get("timeout") or 10000), 60000) / 1000
        try:
            with urllib.request.urlopen(req, timeout=timeout) as resp:
                text = resp.read(1_000_000).decode("utf-8", "replace")
                status = resp.status
        except urllib.error.HTTPError as e:
            text = e.read(100_000).decode("utf-8", "replace")
            status = e.code
        except (urllib.error.URLError, OSError) as e:
            raise ToolError(f"api_request failed: {e}")
        return ToolResult("api_request", {"status": status},
                          f"HTTP {status}\n{text[:MAX_FILE_CHARS]}")

    # --- document family (offline backends: tools/documents.py) ---
    def _tool_read_document(self, p):
        from . import documents as docs
        path = self._resolve(p["uri"]) if not os.path.isabs(p.get("uri", "")) \
            else p["uri"]
        try:
            text = docs.read_document(path)
        except docs.DocumentError as e:
            raise ToolError(str(e))
        start = int(p.get("start_index") or 0)
        maxlen = int(p.get("max_length") or MAX_FILE_CHARS)
        chunk = text[start: start + maxlen]
        return ToolResult("read_document",
                          {"uri": p["uri"], "total_chars": len(text),
                           "start_index": start}, chunk)

    def _tool_edit_document(self, p):
        from . import documents as docs
        path = self._resolve(p["uri"])
        reps = p.get("replacements")
        if isinstance(reps, str):
            import json as _json
            try:
                reps = _json.loads(reps)
            except ValueError:
                raise ToolError("replacements must be a JSON array")
        try:
            docs.edit_document(path, p.get("content"), reps,
                               backup=str(p.get("backup", "")).lower() in ("true", "1"))
        except docs.DocumentError as e:
            raise ToolError(str(e))
        return ToolResult("edit_document", {"uri": p["uri"]},
                          f"edited {p['uri']}")

    def _tool_create_document(self, p):
        from . import documents as docs
        path = self._resolve(p.get("file_path") or "document.docx")
        try:
            out = docs.create_document(p.get("type") or "word", path,
                                       p.get("document_data") or "")
        except docs.DocumentError as e:
            raise ToolError(str(e))
        return ToolResult("create_document", {"path": out}, f"created {out}")

    def _tool_document_convert(self, p):
        from . import documents as docs
        src = self._resolve(p["input_file"])
        dst = self._resolve(p.get("output_path") or
                            os.path.splitext(src)[0] + ".txt")
        try:
            out = docs.convert_document(src, dst, p.get("format"))
        except docs.DocumentError as e:
            raise ToolError(str(e))
        return ToolResult("document_convert", {"path": out}, f"converted to {out}")

    def _tool_document_extract(self, p):
        from . import documents as docs
        src = self._resolve(p["input_file"])
        out_dir = self._resolve(p.get("output_dir") or ".")
        try:
            written = docs.extract_document(src, out_dir,
                                            p.get("extract_type") or "text")
        except docs.DocumentError as e:
            raise ToolError(str(e))
        return ToolResult("document_extract", {"files": written},
                          "\n".join(written))

    def _tool_document_merge(self, p):
        from . import documents as docs
        files = p.get("input_files")
        if isinstance(files, str):
            files = [f.strip() for f in files.split(",") if f.strip()]
        if not files:
            raise ToolError("document_merge needs input_files")
        srcs = [self._resolve(f) for f in files]
        dst = self._resolve(p.get("output_path") or "merged.txt")
        try:
            out = docs.merge_documents(srcs, dst)
        except docs.DocumentError as e:
            raise ToolError(str(e))
        return ToolResult("document_merge", {"path": out}, f"merged to {out}")

    def _tool_skill(self, p):
        from .skills import SkillService
        svc = SkillService(self.root)
        skill = svc.get_skill(p["name"])
        if skill is None:
            raise ToolError(f"skill {p['name']!r} not found")
        return ToolResult("skill", {"name": skill.name}, skill.content)
