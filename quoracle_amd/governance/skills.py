"""Skills: SKILL.md knowledge packs, loadable into agent context.

Behavior-parity with the reference (reference: lib/quoracle/skills/loader.ex:
27-133,315 and skills/creator.ex): YAML-frontmatter skill files; a grove-local
skills directory shadows the global one; metadata listing is cheap (name +
description only), content loads on demand; create_skill writes a new skill
with optional scripts/references/assets attachments.
"""

from __future__ import annotations

import os
import re
from typing import Any, Dict, List, Optional

import yaml

_FRONTMATTER_RE = re.compile(r"\A---\s*\n(.*?)\n---\s*(\n|\Z)", re.DOTALL)
_NAME_RE = re.compile(r"^[a-z0-9][a-z0-9-]{0,63}$")

ATTACHMENT_DIRS = {"script": "scripts", "reference": "references", "asset": "assets"}


class SkillError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


def parse_skill_markdown(text: str) -> Dict[str, Any]:
    match = _FRONTMATTER_RE.match(text)
    if not match:
        raise SkillError("missing_frontmatter")
    meta = yaml.safe_load(match.group(1)) or {}
    if not isinstance(meta, dict) or not meta.get("name"):
        raise SkillError("malformed_frontmatter")
    return {
        "name": str(meta["name"]),
        "description": str(meta.get("description", "")),
        "metadata": {k: v for k, v in meta.items()
                     if k not in ("name", "description")},
        "content": text[match.end():].strip(),
    }


class SkillLoader:
    """grove_dir/skills shadows global skills_dir."""

    def __init__(self, skills_dir: Optional[str] = None,
                 grove_skills_dir: Optional[str] = None):
        self.skills_dir = skills_dir
        self.grove_skills_dir = grove_skills_dir

    def _dirs(self) -> List[str]:
        return [d for d in (self.grove_skills_dir, self.skills_dir) if d]

    def _skill_file(self, name: str) -> Optional[str]:
        for base in self._dirs():
            for candidate in (os.path.join(base, name, "SKILL.md"),
                              os.path.join(base, f"{name}.md")):
                if os.path.isfile(candidate):
                    return candidate
        return None

    def list_metadata(self) -> List[Dict[str, str]]:
        seen: Dict[str, Dict[str, str]] = {}
        for base in self._dirs():
            if not os.path.isdir(base):
                continue
            for entry in sorted(os.listdir(base)):
                path = os.path.join(base, entry)
                name = entry[:-3] if entry.endswith(".md") else entry
                if name in seen:
                    continue  # grove-local shadows global
                try:
                    skill = self.load(name)
                except SkillError:
                    continue
                seen[name] = {"name": skill["name"],
                              "description": skill["description"]}
        return list(seen.values())

    def load(self, name: str) -> Dict[str, Any]:
        path = self._skill_file(name)
        if path is None:
            raise SkillError("skill_not_found")
        with open(path) as f:
            skill = parse_skill_markdown(f.read())
        skill["path"] = path
        return skill

    def create(self, name: str, description: str, content: str,
               metadata: Optional[Dict[str, Any]] = None,
               attachments: Optional[List[Dict[str, str]]] = None) -> str:
        if not _NAME_RE.match(name or ""):
            raise SkillError("invalid_name")
        if len(description or "") > 1024:
            raise SkillError("description_too_long")
        base = self.grove_skills_dir or self.skills_dir
        if base is None:
            raise SkillError("no_skills_dir")
        skill_dir = os.path.join(base, name)
        if os.path.exists(skill_dir) or self._skill_file(name):
            raise SkillError("skill_exists")
        os.makedirs(skill_dir)
        frontmatter = {"name": name, "description": description}
        frontmatter.update(metadata or {})
        text = "---\n" + yaml.safe_dump(frontmatter, sort_keys=False) + "---\n\n" + content
        skill_path = os.path.join(skill_dir, "SKILL.md")
        with open(skill_path, "w") as f:
            f.write(text)
        for att in attachments or []:
            sub = ATTACHMENT_DIRS.get(att.get("type", ""))
            filename = att.get("filename", "")
            if sub is None or not filename or "/" in filename or ".." in filename:
                raise SkillError("invalid_attachment")
            att_dir = os.path.join(skill_dir, sub)
            os.makedirs(att_dir, exist_ok=True)
            with open(os.path.join(att_dir, filename), "w") as f:
                f.write(att.get("content", ""))
        return skill_path
