"""Deterministic prompt tokenizer, with optional real CLIP BPE.

There is no network access for the CLIP BPE vocab, and the engine runs
random-init weights, so by default lower-cased word pieces
are hashed into the CLIP vocab range, bracketed by BOS/EOS, padded to 77 —
deterministic and well-shaped. When a user has the real vocab on disk,
``use_bpe(vocab.json, merges.txt)`` (or ``BPETokenizer.from_files``) swaps
in OpenAI-CLIP byte-level BPE without touching callers.
"""
from __future__ import annotations

import hashlib
import json
import re
from typing import Dict, List, Optional, Tuple

import torch

VOCAB_SIZE = 49408
BOS = 49406
EOS = 49407
MAX_LEN = 77

_word_re = re.compile(r"[a-z0-9]+|[^\sa-z0-9]")


def _hash_token(word: str) -> int:
    h = int.from_bytes(hashlib.sha1(word.encode()).digest()[:4], "little")
    return h % (VOCAB_SIZE - 2)  # keep clear of BOS/EOS


def _bytes_to_unicode() -> Dict[int, str]:
    """GPT-2/CLIP byte→printable-unicode map (reversible, no control chars)."""
    bs = (
        list(range(ord("!"), ord("~") + 1))
        + list(range(ord("¡"), ord("¬") + 1))
        + list(range(ord("®"), ord("ÿ") + 1))
    )
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


# CLIP's word-split pattern (contractions, letter runs, single digits,
# punctuation runs) — \p{L}/\p{N} rewritten for the stdlib re module.
_bpe_word_re = re.compile(
    r"'s|'t|'re|'ve|'m|'ll|'d|[^\W\d_]+|\d|[^\s\w]+", re.IGNORECASE
)


class BPETokenizer:
    """OpenAI-CLIP byte-level BPE (lowercase, ``</w>`` end-of-word marker).

    Symbols missing from the vocab fall back to the hash tokenizer so a
    truncated vocab degrades gracefully instead of raising.
    """

    def __init__(self, vocab: Dict[str, int], merges: List[Tuple[str, str]]):
        self.vocab = vocab
        self.ranks = {pair: i for i, pair in enumerate(merges)}
        self.byte_enc = _bytes_to_unicode()
        self._cache: Dict[str, List[str]] = {}

    @classmethod
    def from_files(cls, vocab_json: str, merges_txt: str) -> "BPETokenizer":
        with open(vocab_json, "r", encoding="utf-8") as fh:
            vocab = json.load(fh)
        merges: List[Tuple[str, str]] = []
        with open(merges_txt, "r", encoding="utf-8") as fh:
            for line in fh:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                a, _, b = line.partition(" ")
                if b:
                    merges.append((a, b))
        return cls(vocab, merges)

    def _bpe(self, word: str) -> List[str]:
        if word in self._cache:
            return self._cache[word]
        symbols = list(word)
        symbols[-1] = symbols[-1] + "</w>"
        while len(symbols) > 1:
            pairs = [(symbols[i], symbols[i + 1]) for i in range(len(symbols) - 1)]
            best = min(pairs, key=lambda p: self.ranks.get(p, 1 << 30))
            if best not in self.ranks:
                break
            merged: List[str] = []
            i = 0
            while i < len(symbols):
                if (
                    i < len(symbols) - 1
                    and (symbols[i], symbols[i + 1]) == best
                ):
                    merged.append(symbols[i] + symbols[i + 1])
                    i += 2
                else:
                    merged.append(symbols[i])
                    i += 1
            symbols = merged
        self._cache[word] = symbols
        return symbols

    def encode_text(self, text: str) -> List[int]:
        """Text -> token ids (no BOS/EOS/padding)."""
        text = re.sub(r"\s+", " ", text).strip().lower()
        ids: List[int] = []
        for word in _bpe_word_re.findall(text):
            encoded = "".join(self.byte_enc[b] for b in word.encode("utf-8"))
            for sym in self._bpe(encoded):
                tid = self.vocab.get(sym)
                ids.append(tid if tid is not None else _hash_token(sym))
        return ids


_active: Optional[BPETokenizer] = None


def use_bpe(vocab_json: str, merges_txt: str) -> BPETokenizer:
    """Switch the module to real CLIP BPE loaded from disk."""
    global _active
    _active = BPETokenizer.from_files(vocab_json, merges_txt)
    return _active


def use_hash() -> None:
    """Revert to the deterministic hash tokenizer (the default)."""
    global _active
    _active = None


def _fragment_ids(text: str) -> List[int]:
    from .embeddings import trigger_ids

    out: List[int] = []
    if _active is not None:
        # BPE path: check whole words for textual-inversion triggers first
        for word in text.lower().split():
            tids = trigger_ids(word)
            if tids is not None:
                out.extend(tids)
            else:
                out.extend(_active.encode_text(word))
        return out
    for w in _word_re.findall(text.lower()):
        tids = trigger_ids(w)
        out.extend(tids if tids is not None else [_hash_token(w)])
    return out


def encode(text: str, max_len: int = MAX_LEN) -> List[int]:
    ids = [BOS] + _fragment_ids(text)[: max_len - 2] + [EOS]
    ids += [EOS] * (max_len - len(ids))
    return ids


def encode_batch(texts: List[str], device="cpu") -> torch.Tensor:
    return torch.tensor([encode(t) for t in texts], dtype=torch.long, device=device)


_ESC = {"\\(": "\x03", "\\)": "\x04", "\\[": "\x05", "\\]": "\x06"}


def parse_weighted(text: str):
    """sdwui prompt-attention syntax -> [(fragment, weight)].

    ``(x)`` -> 1.1x, ``((x))`` -> 1.21x, ``[x]`` -> /1.1, ``(x:1.3)`` ->
    exactly 1.3; nesting multiplies. Escaped ``\\(`` etc. are literal.
    """
    for esc, sent in _ESC.items():
        text = text.replace(esc, sent)
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
    rev = {v: k[-1] for k, v in _ESC.items()}

    def unesc(f):
        for sent, ch in rev.items():
            f = f.replace(sent, ch)
        return f

    return [(unesc(f), w) for f, w in out if f.strip()]


def encode_weighted(text: str, max_len: int = MAX_LEN):
    """-> (ids [k*77], weights [k*77] float): sdwui "unlimited prompt
    length" — tokens beyond 75 spill into additional BOS/EOS-bracketed
    77-token chunks, each later encoded by CLIP separately; the literal
    keyword ``BREAK`` forces a chunk boundary. BOS/EOS/pad carry
    weight 1."""
    import re as _re

    body = max_len - 2
    chunk_lists: List[List[tuple]] = []
    for section in _re.split(r"\bBREAK\b", text):
        flat: List[tuple] = []
        for frag, w in parse_weighted(section):
            for tid in _fragment_ids(frag):
                flat.append((tid, w))
        n_chunks = max(1, -(-max(1, len(flat)) // body))
        for c in range(n_chunks):
            chunk_lists.append(flat[c * body : (c + 1) * body])
    ids: List[int] = []
    weights: List[float] = []
    for part in chunk_lists:
        ids.append(BOS)
        weights.append(1.0)
        for tid, w in part:
            ids.append(tid)
            weights.append(w)
        ids.append(EOS)
        weights.append(1.0)
        while len(ids) % max_len:
            ids.append(EOS)
            weights.append(1.0)
    return ids, weights


def encode_batch_weighted(texts: List[str], device="cpu"):
    """-> (ids [B, K, 77], weights [B, K*77]); K = max chunk count over
    the batch (short prompts pad with empty BOS/EOS chunks)."""
    pairs = [encode_weighted(t) for t in texts]
    k = max(len(p[0]) // MAX_LEN for p in pairs)
    pad_ids = [BOS] + [EOS] * (MAX_LEN - 1)
    ids_rows, wt_rows = [], []
    for pids, pwts in pairs:
        while len(pids) < k * MAX_LEN:
            pids = pids + pad_ids
            pwts = pwts + [1.0] * MAX_LEN
        ids_rows.append(pids)
        wt_rows.append(pwts)
    ids = torch.tensor(ids_rows, dtype=torch.long, device=device)
    wts = torch.tensor(wt_rows, dtype=torch.float32, device=device)
    return ids.reshape(len(texts), k, MAX_LEN), wts
