"""Textual-inversion embeddings.

sdwui loads trained embedding files and splices their vectors into the
CLIP token stream when the trigger word appears in a prompt
(/sdapi/v1/embeddings surface; the reference's remotes applied them
host-side). Here: files from ``SDWD_EMBEDDINGS_DIR`` (or programmatic
``register``) map a trigger word to a ``[k, d_model]`` tensor; the
tokenizer emits placeholder ids >= VOCAB_SIZE for the trigger, and the
text encoder overwrites those positions with the trained vectors after
the token-embedding lookup.

Supported file shapes (safetensors or torch.load pickles are NOT read —
safetensors only, like the rest of this repo):
- sdwui-style: a tensor under the key ``emb_params``;
- a single tensor under any sole key;
- [d] vectors are treated as [1, d].
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import torch

from ..utils import get_logger

log = get_logger("embeddings")

# placeholder token ids start here (outside the CLIP vocab; the encoder
# clamps the embedding lookup and overwrites these positions)
PLACEHOLDER_BASE = 49408

_registry: Dict[str, torch.Tensor] = {}   # trigger -> [k, d]
_id_map: Dict[int, Tuple[str, int]] = {}  # placeholder id -> (name, row)
_name_base: Dict[str, int] = {}           # trigger -> first placeholder id


def _assign_ids(name: str, vecs: torch.Tensor) -> None:
    base = PLACEHOLDER_BASE + sum(v.shape[0] for v in _registry.values())
    _registry[name] = vecs
    _name_base[name] = base
    for i in range(vecs.shape[0]):
        _id_map[base + i] = (name, i)


def register(name: str, vectors: torch.Tensor) -> None:
    """Register an embedding under a trigger word (lower-cased)."""
    if vectors.dim() == 1:
        vectors = vectors[None]
    name = name.lower()
    if name in _registry:
        # keep the existing id block if the shape matches
        if _registry[name].shape == vectors.shape:
            _registry[name] = vectors.float()
            return
        clear()  # shapes changed: rebuild the id space
    _assign_ids(name, vectors.float())


def clear() -> None:
    _registry.clear()
    _id_map.clear()
    _name_base.clear()


def embeddings_dir() -> str:
    return os.environ.get("SDWD_EMBEDDINGS_DIR", "embeddings")


def refresh_embedding_files(dirpath: Optional[str] = None) -> List[str]:
    """Scan for *.safetensors embedding files; the stem is the trigger."""
    from safetensors import safe_open

    clear()
    d = dirpath or embeddings_dir()
    if os.path.isdir(d):
        for fn in sorted(os.listdir(d)):
            if not fn.endswith(".safetensors"):
                continue
            path = os.path.join(d, fn)
            try:
                with safe_open(path, framework="pt") as f:
                    keys = list(f.keys())
                    key = "emb_params" if "emb_params" in keys else keys[0]
                    vecs = f.get_tensor(key)
                register(os.path.splitext(fn)[0], vecs)
            except Exception as exc:
                log.warning("skipping embedding %s: %s", path, exc)
    return sorted(_registry)


def trigger_ids(word: str) -> Optional[List[int]]:
    """Placeholder ids for a trigger word, or None if not an embedding."""
    name = word.lower()
    vecs = _registry.get(name)
    if vecs is None:
        return None
    base = _name_base[name]
    return list(range(base, base + vecs.shape[0]))


def loaded() -> Dict[str, int]:
    """{trigger: n_vectors} for the API surface."""
    return {n: int(v.shape[0]) for n, v in _registry.items()}


def apply_to_hidden(tokens: torch.Tensor, hidden: torch.Tensor) -> torch.Tensor:
    """Overwrite placeholder positions of ``hidden`` [B,S,D] with the
    registered vectors (called right after the token-embedding lookup)."""
    if not _id_map:
        return hidden
    mask = tokens >= PLACEHOLDER_BASE
    if not bool(mask.any()):
        return hidden
    hidden = hidden.clone()
    d = hidden.shape[-1]
    for b, s in mask.nonzero(as_tuple=False).tolist():
        entry = _id_map.get(int(tokens[b, s]))
        if entry is None:
            continue
        name, row = entry
        vec = _registry[name][row]
        if vec.shape[0] != d:
            log.warning(
                "embedding '%s' width %d != encoder width %d; skipped",
                name, vec.shape[0], d,
            )
            continue
        hidden[b, s] = vec.to(hidden.dtype).to(hidden.device)
    return hidden
