"""X/Y/Z plot — native implementation of sdwui's grid-sweep script.

The reference's users ran this through the host webui (the host's
``scripts/xyz_grid.py`` selectable script, invoked over the API with
``script_name``/``script_args``); each of its cells was a normal
generation the extension then sharded across workers. Here the whole
sweep runs in-process: every cell is one engine.generate() call, so each
cell is itself benchmark-weighted-sharded across the node's GPUs.

Value syntax matches sdwui's: comma-separated entries where numeric
entries may be ranges — ``1-5`` (inclusive), ``1-10 (+2)`` (step),
``1-10 [5]`` (count, evenly spaced) — and string axes are CSV-parsed so
quoted values may contain commas.

Boundary (documented in docs/usage.md): axis labels are returned in the
response metadata and per-cell infotexts, not rasterised into the grid
pixels (``draw_legend`` needs a font; none can ship offline).
"""
from __future__ import annotations

import csv
import io
import re
from dataclasses import replace
from typing import Any, Callable, Dict, List, NamedTuple, Optional

import torch

from ..utils import get_logger

log = get_logger("xyz")

_re_range = re.compile(
    r"^\s*([+-]?\s*\d+)\s*-\s*([+-]?\s*\d+)(?:\s*\(([+-]\d+)\s*\))?\s*$"
)
_re_range_float = re.compile(
    r"^\s*([+-]?\s*\d+(?:\.\d*)?)\s*-\s*([+-]?\s*\d+(?:\.\d*)?)"
    r"(?:\s*\(([+-]\d+(?:\.\d*)?)\s*\))?\s*$"
)
_re_count = re.compile(
    r"^\s*([+-]?\s*\d+)\s*-\s*([+-]?\s*\d+)\s*\[(\d+)\s*\]\s*$"
)
_re_count_float = re.compile(
    r"^\s*([+-]?\s*\d+(?:\.\d*)?)\s*-\s*([+-]?\s*\d+(?:\.\d*)?)"
    r"\s*\[(\d+)\s*\]\s*$"
)


def _parse_int_token(tok: str) -> List[int]:
    m = _re_count.match(tok)
    if m:
        a, b, n = int(m.group(1)), int(m.group(2)), int(m.group(3))
        if n <= 1:
            return [a]
        return [round(a + (b - a) * i / (n - 1)) for i in range(n)]
    m = _re_range.match(tok)
    if m:
        a, b = int(m.group(1).replace(" ", "")), int(m.group(2).replace(" ", ""))
        step = int(m.group(3).replace(" ", "")) if m.group(3) else 1
        if step == 0:
            raise ValueError(f"zero step in range {tok!r}")
        vals = list(range(a, b + (1 if step > 0 else -1), step))
        return vals or [a]
    return [int(tok.strip())]


def _parse_float_token(tok: str) -> List[float]:
    m = _re_count_float.match(tok)
    if m:
        a, b, n = float(m.group(1)), float(m.group(2)), int(m.group(3))
        if n <= 1:
            return [a]
        return [a + (b - a) * i / (n - 1) for i in range(n)]
    m = _re_range_float.match(tok)
    if m:
        a = float(m.group(1).replace(" ", ""))
        b = float(m.group(2).replace(" ", ""))
        step = float(m.group(3).replace(" ", "")) if m.group(3) else 1.0
        if step == 0:
            raise ValueError(f"zero step in range {tok!r}")
        vals, v = [], a
        while (step > 0 and v <= b + 1e-9) or (step < 0 and v >= b - 1e-9):
            vals.append(round(v, 8))
            v += step
        return vals or [a]
    return [float(tok.strip())]


def parse_axis_values(kind: str, raw: Any) -> List[Any]:
    """Parse one axis' value spec. ``raw`` may already be a list (API
    dropdown-style args) or a string in sdwui's comma/range syntax."""
    if isinstance(raw, (list, tuple)):
        items = [str(v) for v in raw]
    elif raw is None:
        items = []
    else:
        s = str(raw)
        if kind in ("int", "float"):
            items = [t for t in s.split(",") if t.strip()]
        else:
            # CSV so quoted strings may contain commas (sdwui csv_mode)
            items = [
                t.strip()
                for t in next(csv.reader(io.StringIO(s)), [])
                if t.strip()
            ]
    out: List[Any] = []
    for tok in items:
        if kind == "int":
            out.extend(_parse_int_token(tok))
        elif kind == "float":
            out.extend(_parse_float_token(tok))
        else:
            out.append(tok)
    return out


class AxisOption(NamedTuple):
    name: str
    kind: str  # int | float | str
    apply: Optional[Callable[[Any, Any], Any]]  # (gen, value) -> gen


def _prompt_sr(gen, value):
    """Prompt S/R: the FIRST axis value is the search token; every value
    (including the first) substitutes it in prompt and negative prompt."""
    raise RuntimeError("applied specially in run_xyz")  # pragma: no cover


# Index-addressable axis list (published via /sdapi/v1/script-info and
# docs/api_reference.md). The first eight match the host's dropdown order;
# string names are the version-proof way to address any of them.
AXIS_OPTIONS: List[AxisOption] = [
    AxisOption("Nothing", "str", None),
    AxisOption("Seed", "int", lambda g, v: replace(g, seed=int(v))),
    AxisOption("Var. seed", "int", lambda g, v: replace(g, subseed=int(v))),
    AxisOption(
        "Var. strength", "float",
        lambda g, v: replace(g, subseed_strength=float(v)),
    ),
    AxisOption("Steps", "int", lambda g, v: replace(g, steps=int(v))),
    AxisOption("Hires steps", "int", lambda g, v: replace(g, hr_steps=int(v))),
    AxisOption("CFG Scale", "float", lambda g, v: replace(g, cfg_scale=float(v))),
    AxisOption("Prompt S/R", "str", _prompt_sr),
    AxisOption("Sampler", "str", lambda g, v: replace(g, sampler_name=str(v))),
    AxisOption("Checkpoint name", "str", lambda g, v: replace(g, model=str(v))),
    AxisOption("Clip skip", "int", lambda g, v: replace(g, clip_skip=int(v))),
    AxisOption(
        "Denoising", "float",
        lambda g, v: replace(g, denoising_strength=float(v)),
    ),
    AxisOption("Hires upscaler", "str", lambda g, v: replace(g, hr_upscaler=str(v))),
    AxisOption("Hires scale", "float", lambda g, v: replace(g, hr_scale=float(v))),
    AxisOption("Schedule type", "str", lambda g, v: replace(g, scheduler=str(v))),
    AxisOption("Sigma Churn", "float", lambda g, v: replace(g, s_churn=float(v))),
    AxisOption("Sigma min", "float", lambda g, v: replace(g, s_tmin=float(v))),
    AxisOption("Sigma max", "float", lambda g, v: replace(g, s_tmax=float(v))),
    AxisOption("Sigma noise", "float", lambda g, v: replace(g, s_noise=float(v))),
    AxisOption(
        "ENSD", "int",
        lambda g, v: replace(g, eta_noise_seed_delta=int(v)),
    ),
    AxisOption("Eta", "float", lambda g, v: replace(g, eta=float(v))),
    AxisOption("Width", "int", lambda g, v: replace(g, width=int(v))),
    AxisOption("Height", "int", lambda g, v: replace(g, height=int(v))),
]

_BY_NAME = {o.name.lower(): i for i, o in enumerate(AXIS_OPTIONS)}
# common aliases users/other frontends send
_BY_NAME.update({
    "var seed": 2, "variation seed": 2,
    "var strength": 3, "variation strength": 3,
    "var. seed strength": 3,
    "cfg": 6, "cfg scale": 6,
    "prompt sr": 7, "prompt s/r": 7,
    "sampler name": 8, "checkpoint": 9, "model": 9,
    "clip_skip": 10, "denoising strength": 11,
    "schedule": 14, "scheduler": 14,
    "eta noise seed delta": 19,
})


def resolve_axis(spec: Any) -> AxisOption:
    """Axis selector -> AxisOption. Accepts an int index into AXIS_OPTIONS
    or a (case-insensitive) axis name / alias."""
    if isinstance(spec, bool):  # bool is an int subclass; reject it
        raise ValueError(f"invalid axis selector {spec!r}")
    if isinstance(spec, int):
        if 0 <= spec < len(AXIS_OPTIONS):
            return AXIS_OPTIONS[spec]
        raise ValueError(
            f"axis index {spec} out of range; known axes: "
            + ", ".join(f"{i}={o.name}" for i, o in enumerate(AXIS_OPTIONS))
        )
    key = str(spec).strip().lower()
    if key in _BY_NAME:
        return AXIS_OPTIONS[_BY_NAME[key]]
    raise ValueError(
        f"unknown axis {spec!r}; known axes: "
        + ", ".join(o.name for o in AXIS_OPTIONS)
    )


def _apply(axis: AxisOption, gen, value):
    if axis.apply is None:
        return gen
    if axis.apply is _prompt_sr:
        return gen  # handled with the axis' first value in run_xyz
    return axis.apply(gen, value)


def _apply_sr(gen, search: str, value: str):
    if search not in gen.prompt and search not in gen.negative_prompt:
        log.warning("Prompt S/R: %r not found in prompt", search)
    return replace(
        gen,
        prompt=gen.prompt.replace(search, value),
        negative_prompt=gen.negative_prompt.replace(search, value),
    )


def run_xyz(
    engine,
    gen,
    x_axis: Any, x_values: Any,
    y_axis: Any, y_values: Any,
    z_axis: Any = 0, z_values: Any = "",
    *,
    no_fixed_seeds: bool = False,
    include_lone_images: bool = False,
    include_sub_grids: bool = False,
) -> Dict[str, Any]:
    """Run the sweep. Returns a dict with:

    - ``grid``: [GH,GW,3] uint8 — z sub-grids stacked vertically, each
      sub-grid x-major columns × y rows (first image of each cell);
    - ``sub_grids``: list of per-z grids (len(z) > 1 and requested);
    - ``images``/``seeds``/``infotexts``: per-cell first images (all
      cell images when ``include_lone_images``);
    - ``labels``: {"x": [...], "y": [...], "z": [...], axes: names}.
    """
    ax, ay, az = resolve_axis(x_axis), resolve_axis(y_axis), resolve_axis(z_axis)
    xs = parse_axis_values(ax.kind, x_values) if ax.apply else [None]
    ys = parse_axis_values(ay.kind, y_values) if ay.apply else [None]
    zs = parse_axis_values(az.kind, z_values) if az.apply else [None]
    xs, ys, zs = xs or [None], ys or [None], zs or [None]

    # Prompt S/R: first value is the search token
    sr_search = {}
    for axis, vals in ((ax, xs), (ay, ys), (az, zs)):
        if axis.apply is _prompt_sr:
            if not vals or vals[0] is None:
                raise ValueError("Prompt S/R needs at least the search value")
            sr_search[axis] = str(vals[0])

    # sdwui fixes the seed ONCE so cells are comparable, unless the user
    # asked for free seeds or sweeps the seed axis itself. Broadcast from
    # rank 0 so DistributedEngine ranks build identical cell requests.
    if not no_fixed_seeds and gen.seed == -1:
        from . import group as pg

        seed = pg.broadcast_object(
            int(torch.randint(0, 2**31 - 1, (1,)).item())
        )
        gen = replace(gen, seed=seed)

    n_cells = len(xs) * len(ys) * len(zs)
    if n_cells > 1024:
        raise ValueError(f"X/Y/Z plot of {n_cells} cells (max 1024)")
    log.info(
        "X/Y/Z plot: %s(%d) x %s(%d) x %s(%d) = %d cells",
        ax.name, len(xs), ay.name, len(ys), az.name, len(zs), n_cells,
    )

    cell_first: List[Optional[torch.Tensor]] = []
    all_images: List[torch.Tensor] = []
    all_seeds: List[int] = []
    all_infos: List[str] = []
    interrupted = False
    for zi, zv in enumerate(zs):
        for yi, yv in enumerate(ys):
            for xi, xv in enumerate(xs):
                cell = gen
                for axis, val in ((az, zv), (ay, yv), (ax, xv)):
                    if val is None or axis.apply is None:
                        continue
                    if axis.apply is _prompt_sr:
                        cell = _apply_sr(cell, sr_search[axis], str(val))
                    else:
                        cell = _apply(axis, cell, val)
                if interrupted:
                    cell_first.append(None)
                    continue
                res = engine.generate(cell)
                tag = ", ".join(
                    f"{a.name}: {v}"
                    for a, v in ((ax, xv), (ay, yv), (az, zv))
                    if a.apply is not None and v is not None
                )
                infos = [
                    (i + f"\nXYZ: {tag}" if tag else i)
                    for i in res.infotexts
                ]
                cell_first.append(res.images[0] if res.images.shape[0] else None)
                if include_lone_images:
                    for i in range(res.images.shape[0]):
                        all_images.append(res.images[i])
                        all_seeds.append(res.seeds[i])
                        all_infos.append(infos[i])
                elif res.images.shape[0]:
                    all_images.append(res.images[0])
                    all_seeds.append(res.seeds[0])
                    all_infos.append(infos[0])
                if res.interrupted:
                    interrupted = True

    # assemble sub-grid per z (rows = y, cols = x), stack z vertically
    shapes = [c.shape for c in cell_first if c is not None]
    if not shapes:
        raise ValueError("X/Y/Z plot produced no images")
    h, w, _ = shapes[0]
    uniform = all(s == shapes[0] for s in shapes)
    sub_grids: List[torch.Tensor] = []
    if uniform:
        for zi in range(len(zs)):
            cells = torch.zeros(
                len(ys) * len(xs), h, w, 3, dtype=torch.uint8
            )
            for yi in range(len(ys)):
                for xi in range(len(xs)):
                    c = cell_first[zi * len(ys) * len(xs) + yi * len(xs) + xi]
                    if c is not None:
                        cells[yi * len(xs) + xi] = c
            sub_grids.append(
                cells.reshape(len(ys), len(xs), h, w, 3)
                .permute(0, 2, 1, 3, 4)
                .reshape(len(ys) * h, len(xs) * w, 3)
            )
        grid = torch.cat(sub_grids, dim=0) if len(sub_grids) > 1 else sub_grids[0]
    else:
        # axes that change the output size (Width/Height/Hires scale)
        # cannot tile into one grid; return cells only
        log.warning("X/Y/Z plot cells differ in size; no grid assembled")
        grid = None
    return {
        "grid": grid,
        "sub_grids": sub_grids if (include_sub_grids and len(zs) > 1) else [],
        "images": all_images,
        "seeds": all_seeds,
        "infotexts": all_infos,
        "labels": {
            "x_axis": ax.name, "x_values": [str(v) for v in xs] if ax.apply else [],
            "y_axis": ay.name, "y_values": [str(v) for v in ys] if ay.apply else [],
            "z_axis": az.name, "z_values": [str(v) for v in zs] if az.apply else [],
        },
        "interrupted": interrupted,
    }
