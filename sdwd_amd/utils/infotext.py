"""sdwui "parameters" infotext parsing — the inverse of the infotext the
pipeline emits (pipeline.py generation-summary block) and the reference's
remotes returned per image. sdwui's own parser
(modules/infotext_utils.parse_generation_parameters) reads: prompt lines,
an optional "Negative prompt:" block, then ONE final line of
comma-separated ``Key: value`` pairs where values may be double-quoted
(quotes escape embedded commas). Keeping our emitted format round-trippable
through this grammar is what makes "send to txt2img"-style tooling work on
our PNGs.
"""
from __future__ import annotations

import re
from typing import Dict

_RE_PARAM = re.compile(
    r'\s*(?P<key>[\w \-/]+):\s*(?P<val>"(?:\\.|[^\\"])*"|[^,]*)(?:,|$)'
)


def _unquote(v: str) -> str:
    v = v.strip()
    if len(v) >= 2 and v[0] == '"' and v[-1] == '"':
        v = v[1:-1].replace('\\"', '"').replace("\\\\", "\\")
    return v


def parse_infotext(text: str) -> Dict[str, str]:
    """Parse a "parameters" infotext into a flat dict.

    Returns at least ``prompt`` and ``negative_prompt`` (possibly empty)
    plus one entry per ``Key: value`` pair from the final parameter line,
    keys as written ("Steps", "CFG scale", "Worker Label", ...).
    """
    lines = (text or "").split("\n")
    # the parameter line is the LAST line iff it parses as k:v pairs
    # (sdwui heuristic: it must contain "Steps:" or >=3 pairs)
    param_line = ""
    if lines:
        tail = lines[-1]
        pairs = _RE_PARAM.findall(tail)
        if pairs and ("Steps:" in tail or len(pairs) >= 3):
            param_line = tail
            lines = lines[:-1]
    prompt_lines, neg_lines, in_neg = [], [], False
    for ln in lines:
        if not in_neg and ln.startswith("Negative prompt:"):
            in_neg = True
            neg_lines.append(ln[len("Negative prompt:"):].lstrip())
        elif in_neg:
            neg_lines.append(ln)
        else:
            prompt_lines.append(ln)
    out: Dict[str, str] = {
        "prompt": "\n".join(prompt_lines).strip(),
        "negative_prompt": "\n".join(neg_lines).strip(),
    }
    for m in _RE_PARAM.finditer(param_line):
        key = m.group("key").strip()
        if key:
            out[key] = _unquote(m.group("val"))
    return out
