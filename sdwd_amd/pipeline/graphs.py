"""hipGraph capture of the denoise-loop model call.

The sampler calls the UNet (with CFG) 20+ times per generation with
identical shapes; capturing one forward into a hipGraph (torch.cuda.graphs
on ROCm) removes per-step launch overhead for the hundreds of kernels in
the UNet. Input tensors are static buffers copied into before each replay;
the captured graph is cached per (shape, cfg) and reused across requests.

Disable with SDWD_HIPGRAPH=0. A failed capture falls back to eager launches
of the SAME kernels (the graph is a launch mechanism, not a compute path, so
this is not a backend fallback).
"""
from __future__ import annotations

import gc
import os
from typing import Callable, Dict, Optional, Tuple

import torch

from ..utils import get_logger

log = get_logger("graphs")


def graphs_enabled() -> bool:
    return os.environ.get("SDWD_HIPGRAPH", "1") not in ("", "0", "false")


class _Entry:
    def __init__(self, graph, x, ts, ctx, y, out):
        self.graph = graph
        self.x = x
        self.ts = ts
        self.ctx = ctx
        self.y = y
        self.out = out


class _FnEntry:
    def __init__(self, graph, x, t, ctx, y, out):
        self.graph = graph
        self.x = x
        self.t = t
        self.ctx = ctx
        self.y = y
        self.out = out
        self.ctx_src: Optional[int] = None  # bind-token of last copy


class GraphedModelFn:
    """Whole-step capture: CFG duplication + UNet forward + guidance
    combine run as ONE hipGraph replay per step (the sampler's few
    elementwise kernels and the deterministic per-image CPU noise stay
    eager by design). The timestep rides a 0-dim device buffer so one
    graph serves every step; conditioning tensors are static buffers
    re-copied only when a new generation binds different tensors."""

    def __init__(self, core: Callable, device: torch.device):
        self.core = core  # core(x, t0d, ctx, y) -> eps
        self.device = device
        self.cache: Dict[Tuple, _FnEntry] = {}
        self.enabled = graphs_enabled() and device.type == "cuda"
        self.failed = False
        self._ctx: Optional[torch.Tensor] = None
        self._y: Optional[torch.Tensor] = None
        self._bind_gen = 0  # bumped per bind(): keys the ctx re-copy

    def bind(self, ctx: torch.Tensor, y: Optional[torch.Tensor]) -> None:
        # a monotone token, NOT id(ctx): a freed ctx object's id can be
        # reused by the next generation's tensor, which would skip the
        # static-buffer copy and replay the previous conditioning
        self._ctx, self._y = ctx, y
        self._bind_gen += 1

    def _t0d(self, t: float) -> torch.Tensor:
        return torch.tensor(float(t), device=self.device,
                            dtype=torch.float32)

    def __call__(self, x: torch.Tensor, t: float) -> torch.Tensor:
        if not self.enabled or self.failed:
            return self.core(x, self._t0d(t), self._ctx, self._y)
        key = (
            tuple(x.shape),
            tuple(self._ctx.shape),
            tuple(self._y.shape) if self._y is not None else None,
        )
        e = self.cache.get(key)
        if e is None:
            e = self._capture(x, t, key)
            if e is None:
                return self.core(x, self._t0d(t), self._ctx, self._y)
        e.x.copy_(x)
        e.t.fill_(float(t))
        if e.ctx_src != self._bind_gen:
            e.ctx.copy_(self._ctx)
            if self._y is not None:
                e.y.copy_(self._y)
            e.ctx_src = self._bind_gen
        e.graph.replay()
        # CLONE: multi-eval samplers (Heun, DPM2, UniPC...) hold the first
        # eval's eps while the second replay overwrites the static buffer
        return e.out.clone()

    def _capture(self, x, t, key) -> Optional[_FnEntry]:
        try:
            sx = x.clone()
            st = self._t0d(t)
            sctx = self._ctx.clone()
            sy = self._y.clone() if self._y is not None else None
            side = torch.cuda.Stream(self.device)
            side.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(side):
                for _ in range(2):
                    self.core(sx, st, sctx, sy)
            torch.cuda.current_stream(self.device).wait_stream(side)
            torch.cuda.synchronize(self.device)
            # a GC cycle-collect during capture frees CUDA memory inside
            # the capture (hipFree -> hard abort); collect first, then
            # hold GC off until the graph is sealed
            gc.collect()
            gc.disable()
            try:
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    out = self.core(sx, st, sctx, sy)
            finally:
                gc.enable()
            e = _FnEntry(graph, sx, st, sctx, sy, out)
            e.ctx_src = self._bind_gen  # capture cloned the current ctx
            self.cache[key] = e
            log.info("whole-step hipGraph captured for shape %s", key[0])
            return e
        except Exception as exc:  # pragma: no cover - device-specific
            log.warning(
                "whole-step hipGraph capture failed (%s); eager", exc
            )
            self.failed = True
            return None


class GraphedDenoiser:
    """Wraps fn(x, ts, ctx, y) -> eps with hipGraph capture per shape."""

    def __init__(self, fn: Callable, device: torch.device):
        self.fn = fn
        self.device = device
        self.cache: Dict[Tuple, _Entry] = {}
        self.enabled = graphs_enabled() and device.type == "cuda"
        self.failed = False

    def __call__(
        self,
        x: torch.Tensor,
        ts: torch.Tensor,
        ctx: torch.Tensor,
        y: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if not self.enabled or self.failed or not torch.is_tensor(ctx):
            # non-tensor ctx (RegionalContext) carries python-side control
            # flow the capture cannot freeze — run eager
            return self.fn(x, ts, ctx, y)
        key = (
            tuple(x.shape),
            tuple(ctx.shape),
            tuple(y.shape) if y is not None else None,
        )
        entry = self.cache.get(key)
        if entry is None:
            entry = self._capture(x, ts, ctx, y, key)
            if entry is None:
                return self.fn(x, ts, ctx, y)
        entry.x.copy_(x)
        entry.ts.copy_(ts)
        entry.ctx.copy_(ctx)
        if y is not None:
            entry.y.copy_(y)
        entry.graph.replay()
        # CLONE: see GraphedModelFn - callers may hold this across evals
        return entry.out.clone()

    def _capture(self, x, ts, ctx, y, key) -> Optional[_Entry]:
        try:
            sx = x.clone()
            sts = ts.clone()
            sctx = ctx.clone()
            sy = y.clone() if y is not None else None
            side = torch.cuda.Stream(self.device)
            side.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(side):
                for _ in range(2):  # MIOpen autotune / allocator warmup
                    self.fn(sx, sts, sctx, sy)
            torch.cuda.current_stream(self.device).wait_stream(side)
            torch.cuda.synchronize(self.device)
            gc.collect()
            gc.disable()  # see GraphedModelFn._capture
            try:
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    out = self.fn(sx, sts, sctx, sy)
            finally:
                gc.enable()
            entry = _Entry(graph, sx, sts, sctx, sy, out)
            self.cache[key] = entry
            log.info("hipGraph captured for shape %s", key[0])
            return entry
        except Exception as exc:  # pragma: no cover - device-specific
            log.warning("hipGraph capture failed (%s); running eager", exc)
            self.failed = True
            return None
