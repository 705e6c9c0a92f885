"""Prompt styles (sdwui styles.csv semantics).

A style is a named prompt template; applying it substitutes ``{prompt}``
(or appends with ", " when no placeholder is present) and concatenates
negative prompts. Styles load from ``SDWD_STYLES_FILE`` — either sdwui's
``styles.csv`` (columns: name, prompt, negative_prompt) or a JSON list of
``{"name", "prompt", "negative_prompt"}`` objects.
"""
from __future__ import annotations

import csv
import json
import os
from typing import Dict, List, Tuple

from ..utils import get_logger

log = get_logger("styles")

_styles: Dict[str, Tuple[str, str]] = {}


def styles_file() -> str:
    return os.environ.get("SDWD_STYLES_FILE", "styles.csv")


def refresh_styles(path: str | None = None) -> List[str]:
    global _styles
    p = path or styles_file()
    out: Dict[str, Tuple[str, str]] = {}
    if os.path.exists(p):
        try:
            with open(p, "r", encoding="utf-8-sig", newline="") as fh:
                if p.endswith(".json"):
                    for e in json.load(fh):
                        out[e["name"]] = (
                            e.get("prompt", ""), e.get("negative_prompt", "")
                        )
                else:
                    reader = csv.DictReader(fh)
                    for row in reader:
                        name = (row.get("name") or "").strip()
                        if name and name != "None":
                            out[name] = (
                                row.get("prompt") or "",
                                row.get("negative_prompt") or "",
                            )
        except Exception as exc:
            log.warning("failed to read styles from %s: %s", p, exc)
    _styles = out
    return sorted(out)


def all_styles() -> Dict[str, Tuple[str, str]]:
    return dict(_styles)


def _merge(base: str, template: str) -> str:
    if not template:
        return base
    if "{prompt}" in template:
        return template.replace("{prompt}", base)
    return f"{base}, {template}" if base else template


def apply_styles(
    prompt: str, negative: str, names: List[str]
) -> Tuple[str, str]:
    """Apply styles in order (sdwui apply_styles_to_prompt)."""
    for name in names or []:
        entry = _styles.get(name)
        if entry is None:
            log.warning("unknown style '%s' ignored", name)
            continue
        prompt = _merge(prompt, entry[0])
        negative = _merge(negative, entry[1])
    return prompt, negative
