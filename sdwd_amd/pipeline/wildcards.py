"""Dynamic-prompts execution (the sd-dynamic-prompts alwayson script,
the most-used weight-free extension the reference could only FORWARD to
remotes - here it runs natively).

Supported syntax (the extension's core):
  {red|green|blue}          one variant, uniformly
  {2$$a|b|c}                N distinct variants joined by ", "
  {3::heavy|1::light}       weighted variants
  __animals__               one line from <wildcard_dir>/animals.txt
  nesting of all of the above

Expansion is per-image and seeded with that image's seed, so shard
placement never changes the gallery (the C22 determinism contract).
Wildcard files live in SDWD_WILDCARDS_DIR (default ./wildcards), one
option per line, '#' comments ignored; subdirectories are addressed as
__dir/name__.
"""
from __future__ import annotations

import os
import random
import re
from functools import lru_cache
from typing import List

from ..utils import get_logger

log = get_logger("wildcards")

_WILDCARD = re.compile(r"__([A-Za-z0-9_\-/ ]+?)__")


def wildcard_dir() -> str:
    return os.environ.get("SDWD_WILDCARDS_DIR", "wildcards")


@lru_cache(maxsize=256)
def _wildcard_lines(name: str, root: str) -> tuple:
    path = os.path.join(root, *name.split("/")) + ".txt"
    try:
        with open(path, "r", encoding="utf-8") as fh:
            lines = [
                ln.strip() for ln in fh
                if ln.strip() and not ln.strip().startswith("#")
            ]
        return tuple(lines)
    except OSError:
        log.warning("wildcard file not found: %s", path)
        return ()


def _split_variants(body: str) -> List[str]:
    """Split on '|' at nesting depth 0."""
    out, depth, cur = [], 0, []
    for ch in body:
        if ch == "{":
            depth += 1
        elif ch == "}":
            depth -= 1
        if ch == "|" and depth == 0:
            out.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    out.append("".join(cur))
    return out


def _find_group(text: str):
    """Innermost-first {..} group as (start, end) or None."""
    start = None
    for i, ch in enumerate(text):
        if ch == "{":
            start = i
        elif ch == "}":
            if start is not None:
                return start, i
            return None
    return None


def _pick(body: str, rng: random.Random) -> str:
    count = 1
    m = re.match(r"\s*(\d+)\$\$(.*)", body, re.S)
    if m:
        count = int(m.group(1))
        body = m.group(2)
    opts = _split_variants(body)
    weights = []
    clean = []
    for o in opts:
        wm = re.match(r"\s*(\d+(?:\.\d+)?)::(.*)", o, re.S)
        if wm:
            weights.append(float(wm.group(1)))
            clean.append(wm.group(2))
        else:
            weights.append(1.0)
            clean.append(o)
    if count <= 1:
        return rng.choices(clean, weights=weights, k=1)[0]
    count = min(count, len(clean))
    picked = []
    pool = list(zip(clean, weights))
    for _ in range(count):
        total = sum(w for _, w in pool)
        r = rng.random() * total
        acc = 0.0
        for i, (o, w) in enumerate(pool):
            acc += w
            if r <= acc:
                picked.append(o)
                pool.pop(i)
                break
    return ", ".join(picked)


def expand(text: str, seed: int, root: str = None) -> str:
    """Expand one prompt deterministically from `seed`."""
    rng = random.Random(int(seed) & 0xFFFFFFFF)
    root = root or wildcard_dir()
    out = text
    for _ in range(64):  # nesting/wildcard-recursion bound
        g = _find_group(out)
        if g is not None:
            s, e = g
            out = out[:s] + _pick(out[s + 1:e], rng) + out[e + 1:]
            continue
        m = _WILDCARD.search(out)
        if m is not None:
            lines = _wildcard_lines(m.group(1).strip(), root)
            rep = rng.choice(lines) if lines else ""
            out = out[:m.start()] + rep + out[m.end():]
            continue
        break
    return out


def has_dynamic_syntax(text: str) -> bool:
    return "{" in text or bool(_WILDCARD.search(text))


def expand_batch(prompt: str, seeds: List[int], root: str = None) -> List[str]:
    """One expansion per image, seeded with that image's seed."""
    return [expand(prompt, s, root) for s in seeds]
