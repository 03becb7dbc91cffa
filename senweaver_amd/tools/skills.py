"""SkillService — SKILL.md / skills.json discovery and injection.

Capability-compatible with the reference's SkillService
(common/skillService.ts:99-143): skills live as SKILL.md files with YAML
frontmatter (name/description) or entries in a skills.json manifest, under
``.senweaver/skills/`` in the workspace; the ``skill`` tool injects a
skill's content into context.
"""

from __future__ import annotations

import json
import os
import re
from dataclasses import dataclass
from typing import Dict, List, Optional

_FRONTMATTER_RE = re.compile(r"\A---\s*\n(.*?)\n---\s*\n", re.S)


@dataclass
class Skill:
    name: str
    description: str
    content: str
    source: str


def parse_frontmatter(text: str) -> tuple:
    """Returns (meta dict, body)."""
    m = _FRONTMATTER_RE.match(text)
    if not m:
        return {}, text
    meta: Dict[str, str] = {}
    for line in m.group(1).split("\n"):
        if ":" in line:
            k, v = line.split(":", 1)
            meta[k.strip()] = v.strip().strip("\"'")
    return meta, text[m.end():]


class SkillService:
    def __init__(self, workspace_root: str) -> None:
        self.root = workspace_root
        self._skills: Optional[Dict[str, Skill]] = None

    def _discover(self) -> Dict[str, Skill]:
        if self._skills is not None:
            return self._skills
        skills: Dict[str, Skill] = {}
        base = os.path.join(self.root, ".senweaver", "skills")
        if os.path.isdir(base):
            for entry in sorted(os.listdir(base)):
                p = os.path.join(base, entry)
                md = os.path.join(p, "SKILL.md") if os.path.isdir(p) else (p if entry.endswith(".md") else None)
                if md and os.path.isfile(md):
                    try:
                        text = open(md, encoding="utf-8").read()
                    except OSError:
                        continue
                    meta, body = parse_frontmatter(text)
                    name = meta.get("name") or os.path.splitext(entry)[0]
                    skills[name] = Skill(name, meta.get("description", ""), body, md)
            manifest = os.path.join(base, "skills.json")
            if os.path.isfile(manifest):
                try:
                    for item in json.load(open(manifest, encoding="utf-8")):
                        name = item.get("name")
                        if name and name not in skills:
                            skills[name] = Skill(name, item.get("description", ""),
                                                 item.get("content", ""), manifest)
                except (OSError, ValueError):
                    pass
        self._skills = skills
        return skills

    def list_skills(self) -> List[Skill]:
        return list(self._discover().values())

    def get_skill(self, name: str) -> Optional[Skill]:
        return self._discover().get(name)
