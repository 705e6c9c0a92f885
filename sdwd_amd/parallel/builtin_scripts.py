"""Native implementations of sdwui's other built-in selectable scripts
(the host's ``scripts/prompt_matrix.py`` and ``scripts/prompts_from_file.py``;
the reference's users reached them through the host API with
``script_name``/``script_args``). Like the X/Y/Z plot (parallel/xyz.py),
every constituent generation runs through the engine and is therefore
benchmark-weighted-sharded across the node's GPUs.
"""
from __future__ import annotations

import shlex
from dataclasses import replace
from typing import Any, Dict, List, Optional

import torch

from ..utils import get_logger
from ..utils.images import make_grid

log = get_logger("scripts")


# ---------------------------------------------------------------- matrix --
def prompt_matrix_prompts(
    prompt: str, delimiter: str = ", ", put_at_start: bool = False
) -> List[str]:
    """sdwui prompt-matrix combinations: the prompt splits on ``|`` into a
    base part plus N optional parts; all 2^N subsets are generated, joined
    with ``delimiter`` (optional parts before the base if put_at_start)."""
    parts = [p.strip() for p in prompt.split("|")]
    base, options = parts[0], parts[1:]
    combos = []
    for mask in range(1 << len(options)):
        picked = [o for i, o in enumerate(options) if mask & (1 << i)]
        chunks = (picked + [base]) if put_at_start else ([base] + picked)
        combos.append(delimiter.join(c for c in chunks if c))
    return combos


def run_prompt_matrix(
    engine,
    gen,
    *,
    put_at_start: bool = False,
    different_seeds: bool = False,
    prompt_type: str = "positive",
    variations_delimiter: str = "comma",
) -> Dict[str, Any]:
    """Run the 2^N prompt combinations (each a sharded engine generation
    with the request's fixed seed, unless different_seeds) and assemble
    the near-square grid sdwui produces."""
    delim = " " if str(variations_delimiter).lower().startswith("space") else ", "
    source = gen.negative_prompt if prompt_type == "negative" else gen.prompt
    combos = prompt_matrix_prompts(source, delim, put_at_start)
    if len(combos) > 256:
        raise ValueError(f"prompt matrix of {len(combos)} combinations (max 256)")
    if gen.seed == -1:
        from . import group as pg

        gen = replace(gen, seed=pg.broadcast_object(
            int(torch.randint(0, 2**31 - 1, (1,)).item())
        ))
    images: List[torch.Tensor] = []
    seeds: List[int] = []
    infos: List[str] = []
    interrupted = False
    for i, combo in enumerate(combos):
        cell = replace(
            gen,
            prompt=combo if prompt_type != "negative" else gen.prompt,
            negative_prompt=combo if prompt_type == "negative" else gen.negative_prompt,
            seed=gen.seed + i if different_seeds else gen.seed,
        )
        res = engine.generate(cell)
        if res.images.shape[0]:
            images.append(res.images[0])
            seeds.append(res.seeds[0])
            infos.append(res.infotexts[0])
        if res.interrupted:
            interrupted = True
            break
    if not images:
        raise ValueError("prompt matrix produced no images")
    grid = make_grid(torch.stack(images))
    return {
        "grid": grid,
        "images": images,
        "seeds": seeds,
        "infotexts": infos,
        "prompts": combos[: len(images)],
        "interrupted": interrupted,
    }


# ---------------------------------------------------------- prompts-file --
# "--steps 4 --prompt "a cow"" per line; the recognized keys mirror the
# host script's prompt_tags table
_LINE_TAGS = {
    "prompt": str,
    "negative_prompt": str,
    "steps": int,
    "seed": int,
    "subseed": int,
    "subseed_strength": float,
    "width": int,
    "height": int,
    "cfg_scale": float,
    "sampler_name": str,
    "scheduler": str,
    "batch_size": int,
    "denoising_strength": float,
    "clip_skip": int,
}


def parse_prompt_line(line: str) -> Dict[str, Any]:
    """One job line: either a bare prompt or ``--key value`` pairs
    (shlex-split, so quoted values work — sdwui cmdargs syntax)."""
    line = line.strip()
    if not line.startswith("--"):
        return {"prompt": line}
    out: Dict[str, Any] = {}
    toks = shlex.split(line)
    i = 0
    while i < len(toks):
        tok = toks[i]
        if not tok.startswith("--"):
            raise ValueError(f"expected --option, got {tok!r}")
        key = tok[2:].replace("-", "_")
        if key not in _LINE_TAGS:
            raise ValueError(
                f"unknown option --{key}; known: "
                + ", ".join(sorted(_LINE_TAGS))
            )
        if i + 1 >= len(toks):
            raise ValueError(f"--{key} needs a value")
        # values may span multiple tokens until the next --flag (sdwui
        # allows unquoted multi-word prompts)
        j = i + 1
        vals = []
        while j < len(toks) and not toks[j].startswith("--"):
            vals.append(toks[j])
            j += 1
        raw = " ".join(vals)
        out[key] = _LINE_TAGS[key](raw)
        i = j
    return out


def run_prompts_from_file(
    engine,
    gen,
    lines_text: str,
    *,
    checkbox_iterate: bool = False,
    checkbox_iterate_batches: bool = False,
) -> Dict[str, Any]:
    """One engine generation per non-empty line; per-line overrides via
    ``--key value``. checkbox_iterate advances the seed across lines so
    every line draws fresh-but-reproducible noise."""
    lines = [ln for ln in (lines_text or "").splitlines() if ln.strip()]
    if not lines:
        raise ValueError("prompts from file: no lines given")
    if len(lines) > 1024:
        raise ValueError(f"prompts from file: {len(lines)} lines (max 1024)")
    jobs = [parse_prompt_line(ln) for ln in lines]
    if (checkbox_iterate or checkbox_iterate_batches) and gen.seed == -1:
        from . import group as pg

        gen = replace(gen, seed=pg.broadcast_object(
            int(torch.randint(0, 2**31 - 1, (1,)).item())
        ))
    images: List[torch.Tensor] = []
    seeds: List[int] = []
    infos: List[str] = []
    prompts: List[str] = []
    interrupted = False
    seed_cursor: Optional[int] = gen.seed if gen.seed != -1 else None
    for overrides in jobs:
        cell = replace(gen, **overrides)
        if "seed" not in overrides and seed_cursor is not None and (
            checkbox_iterate or checkbox_iterate_batches
        ):
            cell = replace(cell, seed=seed_cursor)
        res = engine.generate(cell)
        for i in range(res.images.shape[0]):
            images.append(res.images[i])
            seeds.append(res.seeds[i])
            infos.append(res.infotexts[i])
            prompts.append(cell.prompt)
        if checkbox_iterate and seed_cursor is not None:
            seed_cursor += cell.batch_size
        if res.interrupted:
            interrupted = True
            break
    if not images:
        raise ValueError("prompts from file produced no images")
    uniform = all(im.shape == images[0].shape for im in images)
    return {
        "grid": (
            make_grid(torch.stack(images))
            if uniform and len(images) > 1 else None
        ),
        "images": images,
        "seeds": seeds,
        "infotexts": infos,
        "prompts": prompts,
        "interrupted": interrupted,
    }
