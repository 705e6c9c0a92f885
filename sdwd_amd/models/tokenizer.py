"""Deterministic prompt tokenizer.

There is no network access for the CLIP BPE vocab, and the engine runs
random-init weights, so tokenization only has to be
deterministic and well-shaped: lower-cased word pieces are hashed into the
CLIP vocab range, bracketed by BOS/EOS, padded to 77. A real BPE vocab can
be dropped in later via ``from_files`` without touching callers.
"""
from __future__ import annotations

import hashlib
import re
from typing import List

import torch

VOCAB_SIZE = 49408
BOS = 49406
EOS = 49407
MAX_LEN = 77

_word_re = re.compile(r"[a-z0-9]+|[^\sa-z0-9]")


def _hash_token(word: str) -> int:
    h = int.from_bytes(hashlib.sha1(word.encode()).digest()[:4], "little")
    return h % (VOCAB_SIZE - 2)  # keep clear of BOS/EOS


def encode(text: str, max_len: int = MAX_LEN) -> List[int]:
    words = _word_re.findall(text.lower())
    ids = [BOS] + [_hash_token(w) for w in words][: max_len - 2] + [EOS]
    ids += [EOS] * (max_len - len(ids))
    return ids


def encode_batch(texts: List[str], device="cpu") -> torch.Tensor:
    return torch.tensor([encode(t) for t in texts], dtype=torch.long, device=device)


def parse_weighted(text: str):
    """sdwui prompt-attention syntax -> [(fragment, weight)].

    ``(x)`` -> 1.1x, ``((x))`` -> 1.21x, ``[x]`` -> /1.1, ``(x:1.3)`` ->
    exactly 1.3; nesting multiplies.
    """
    out = []
    stack = [1.0]
    buf = ""
    i = 0

    def flush():
        nonlocal buf
        if buf:
            out.append((buf, stack[-1]))
            buf = ""

    while i < len(text):
        ch = text[i]
        if ch == "(":
            flush()
            # look ahead for an explicit (text:weight)
            depth, j = 1, i + 1
            while j < len(text) and depth:
                if text[j] == "(":
                    depth += 1
                elif text[j] == ")":
                    depth -= 1
                j += 1
            seg = text[i + 1 : j - 1]
            k = seg.rfind(":")
            if k != -1:
                try:
                    wt = float(seg[k + 1 :])
                    inner = seg[:k]
                    for frag, w in parse_weighted(inner):
                        out.append((frag, w * stack[-1] * wt))
                    i = j
                    continue
                except ValueError:
                    pass
            stack.append(stack[-1] * 1.1)
        elif ch == ")":
            flush()
            if len(stack) > 1:
                stack.pop()
        elif ch == "[":
            flush()
            stack.append(stack[-1] / 1.1)
        elif ch == "]":
            flush()
            if len(stack) > 1:
                stack.pop()
        else:
            buf += ch
        i += 1
    flush()
    return [(f, w) for f, w in out if f.strip()]


def encode_weighted(text: str, max_len: int = MAX_LEN):
    """-> (ids [77], weights [77] float). BOS/EOS/pad carry weight 1."""
    ids = [BOS]
    weights = [1.0]
    for frag, w in parse_weighted(text):
        for word in _word_re.findall(frag.lower()):
            if len(ids) < max_len - 1:
                ids.append(_hash_token(word))
                weights.append(w)
    ids.append(EOS)
    weights.append(1.0)
    while len(ids) < max_len:
        ids.append(EOS)
        weights.append(1.0)
    return ids, weights


def encode_batch_weighted(texts: List[str], device="cpu"):
    pairs = [encode_weighted(t) for t in texts]
    ids = torch.tensor([p[0] for p in pairs], dtype=torch.long, device=device)
    wts = torch.tensor([p[1] for p in pairs], dtype=torch.float32,
                       device=device)
    return ids, wts
