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
