"""sdwui prompt editing / alternation.

``[from:to:when]`` swaps text at a step threshold (fraction of steps if
``when`` <= 1, absolute step otherwise), ``[to:when]`` adds text late,
``[from::when]`` drops text, ``[a|b|c]`` alternates per step. Plain
``[word]`` (no ``:`` / ``|``) stays attention down-weighting and is left
untouched. Mirrors the semantics of sdwui's prompt_parser schedules
(the reference forwarded prompts verbatim to remotes that ran that
parser, distributed.py:251-254).
"""
from __future__ import annotations

import re
from typing import List, Tuple

_inner = re.compile(r"\[([^\[\]]*)\]")


def split_and(text: str) -> List[Tuple[str, float]]:
    """sdwui composable diffusion: "a AND b:0.6" -> [("a",1.0),("b",0.6)].
    The optional trailing ``:number`` on each sub-prompt is its weight."""
    parts = []
    for sub in text.split(" AND "):
        head, sep, tail = sub.rpartition(":")
        w = 1.0
        if sep and head.strip():
            try:
                w = float(tail.strip())
                sub = head
            except ValueError:
                pass
        parts.append((sub.strip(), w))
    return parts or [(text, 1.0)]
# sentinels for brackets we must preserve (attention syntax)
_L, _R = "\x01", "\x02"


def _resolve_once(text: str, step: int, steps: int) -> Tuple[str, bool]:
    """Resolve every innermost bracket group once; returns (text, changed)."""
    changed = False

    def sub(m: re.Match) -> str:
        nonlocal changed
        body = m.group(1)
        if "|" in body:
            parts = body.split("|")
            changed = True
            return parts[step % len(parts)]
        if ":" in body:
            rest, _, when_s = body.rpartition(":")
            try:
                when = float(when_s)
            except ValueError:
                return _L + body + _R  # attention syntax like [x:1.2]? keep
            frm, sep, to = rest.partition(":")
            if not sep:
                frm, to = "", rest
            thr = when * steps if when <= 1.0 else when
            changed = True
            return frm if step < thr else to
        return _L + body + _R  # plain [word] attention group

    out = _inner.sub(sub, text)
    return out, changed


def prompt_at_step(prompt: str, step: int, steps: int) -> str:
    """Concrete prompt text for sampler step ``step`` (0-based)."""
    text = prompt
    for _ in range(8):  # nested schedules resolve inside-out
        text, changed = _resolve_once(text, step, steps)
        if not changed:
            break
    return text.replace(_L, "[").replace(_R, "]")


def prompt_schedule(prompt: str, steps: int) -> List[Tuple[int, str]]:
    """-> [(first_step, text)] segments, deduped consecutively. A single
    segment [(0, prompt)] means no editing syntax is present."""
    segs: List[Tuple[int, str]] = []
    last = None
    for i in range(steps):
        p = prompt_at_step(prompt, i, steps)
        if p != last:
            segs.append((i, p))
            last = p
    return segs or [(0, prompt)]
