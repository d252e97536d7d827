"""Team template engine (the teamrender analog, reference
internal/teamrender's Go text/template + partials pipeline).

Beyond plain ``${VAR}`` substitution the harness templates support:

* ``{{> name}}``        — include ``partials/<name>.tmpl`` from the agents
  source (recursive, depth-capped),
* ``{{#if VAR}} … {{else}} … {{/if}}`` — conditional sections on the
  truthiness of a context value ("" / missing / "false" / "0" are falsy),
* ``{{#each VAR}} … {{/each}}``     — iterate a list value; inside the
  body ``${ITEM}`` and ``${ITEM_INDEX}`` are bound per element,
* ``${VAR}``            — scalar substitution (unknown keys are left
  as-is, matching string.Template.safe_substitute semantics).

This is deliberately a small, deterministic engine (no expressions, no
method calls): team templates are configuration, not programs.
"""
from __future__ import annotations

import re
from pathlib import Path
from string import Template
from typing import Any, Dict, Optional

from kukeon_amd.api import errors

_PARTIAL_RE = re.compile(r"\{\{>\s*([\w./-]+)\s*\}\}")
_IF_RE = re.compile(
    r"\{\{#if\s+(\w+)\s*\}\}(.*?)(?:\{\{else\}\}(.*?))?\{\{/if\}\}",
    re.S)
_EACH_RE = re.compile(r"\{\{#each\s+(\w+)\s*\}\}(.*?)\{\{/each\}\}", re.S)

_FALSY = {"", "false", "0", "no", "none", "null"}


def _truthy(v: Any) -> bool:
    if isinstance(v, str):
        return v.strip().lower() not in _FALSY
    return bool(v)


class TemplateEngine:
    def __init__(self, partials_dir: Optional[Path] = None,
                 max_depth: int = 8):
        self.partials_dir = partials_dir
        self.max_depth = max_depth

    def render(self, text: str, ctx: Dict[str, Any]) -> str:
        return self._render(text, ctx, 0)

    def _render(self, text: str, ctx: Dict[str, Any], depth: int) -> str:
        if depth > self.max_depth:
            raise errors.ValidationError(
                "template partial recursion exceeds depth "
                f"{self.max_depth}")

        def sub_partial(m: "re.Match[str]") -> str:
            name = m.group(1)
            if self.partials_dir is None:
                raise errors.ValidationError(
                    f"template uses partial {name!r} but the source has "
                    "no partials directory")
            p = self.partials_dir / f"{name}.tmpl"
            if not p.exists():
                p = self.partials_dir / name
            if not p.exists():
                raise errors.ValidationError(f"partial {name!r} not found "
                                             f"under {self.partials_dir}")
            body = self._render(p.read_text(), ctx, depth + 1)
            # indentation-preserving include (YAML templates): when the
            # directive sits alone after whitespace, every line of the
            # partial inherits that column
            ls = text.rfind("\n", 0, m.start()) + 1
            lead = text[ls:m.start()]
            if lead.strip() == "" and lead:
                lines = body.splitlines()
                body = ("\n" + lead).join(lines)
                if body.endswith(lead.rstrip("\n")):
                    pass
            return body

        text = _PARTIAL_RE.sub(sub_partial, text)

        def sub_each(m: "re.Match[str]") -> str:
            items = ctx.get(m.group(1)) or []
            if isinstance(items, str):
                items = [s for s in items.split(",") if s]
            body = m.group(2)
            out = []
            for i, item in enumerate(items):
                inner = dict(ctx, ITEM=item, ITEM_INDEX=str(i))
                out.append(self._render(body, inner, depth + 1))
            return "".join(out)

        text = _EACH_RE.sub(sub_each, text)

        def sub_if(m: "re.Match[str]") -> str:
            cond, then, els = m.group(1), m.group(2), m.group(3) or ""
            body = then if _truthy(ctx.get(cond)) else els
            return self._render(body, ctx, depth + 1)

        # innermost-first: repeat while an if-block remains
        prev = None
        while prev != text:
            prev = text
            text = _IF_RE.sub(sub_if, text)

        return Template(text).safe_substitute(
            {k: str(v) for k, v in ctx.items()
             if not isinstance(v, (list, dict))})


def merge_needs(*need_lists) -> list:
    """needs-merge (reference teamrender): capability requirements from
    team defaults, role and harness union into one ordered set."""
    seen = []
    for lst in need_lists:
        for item in lst or []:
            if item not in seen:
                seen.append(item)
    return seen
