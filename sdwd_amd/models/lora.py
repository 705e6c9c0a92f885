"""LoRA: low-rank adapter loading and weight merging.

sdwui applies LoRAs named in the prompt (``<lora:name:0.8>``); the reference
forwarded such prompts to its remotes and refreshed their lora lists
(worker.py:580-581 POST /refresh-loras). Here LoRAs are first-class: a LoRA
is a set of (A [r, in], B [out, r]) pairs keyed by module path; applying
merges ``scale * B @ A`` into the target weights (fast inference, no extra
GEMMs per step) and is exactly reversible, so the pipeline swaps adapter
sets between requests without reloading the model.

Files: safetensors with keys ``<module_path>.lora_A`` / ``.lora_B`` and
metadata ``{"arch": ...}``. Without files (this environment), named LoRAs
are deterministic random per name — same contract as the model registry.
"""
from __future__ import annotations

import zlib
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

from ..utils import get_logger

log = get_logger("lora")


@dataclass
class LoRA:
    name: str
    # module_path -> (A [r, fan_in], B [fan_out, r]) fp32 CPU tensors
    tensors: Dict[str, Tuple[torch.Tensor, torch.Tensor]] = field(
        default_factory=dict
    )

    def delta(self, path: str, like: torch.Tensor) -> Optional[torch.Tensor]:
        """scale-1 weight delta for a module, shaped like its weight."""
        if path not in self.tensors:
            return None
        a, b = self.tensors[path]
        d = (b.float() @ a.float())
        if like.dim() == 4:  # conv [out, in, kh, kw]: 1x1-style adapter
            d = d.reshape(like.shape[0], like.shape[1], 1, 1).expand_as(like) \
                / (like.shape[2] * like.shape[3])
        return d.reshape(like.shape)


def _iter_targets(unet: torch.nn.Module):
    """LoRA-able modules: the attention/ff projections (sdwui convention)."""
    for path, mod in unet.named_modules():
        if isinstance(mod, torch.nn.Linear) and any(
            k in path for k in ("to_q", "to_k", "to_v", "to_out", "ff", "proj_in",
                                "proj_out")
        ):
            yield path, mod


def make_random_lora(name: str, unet: torch.nn.Module, rank: int = 4) -> LoRA:
    """Deterministic random LoRA for a model (test/demo weights)."""
    gen = torch.Generator().manual_seed(zlib.crc32(name.encode()) % (2**31))
    lora = LoRA(name=name)
    for path, mod in _iter_targets(unet):
        fan_out, fan_in = mod.weight.shape[:2]
        a = torch.randn(rank, fan_in, generator=gen) * 0.02
        b = torch.randn(fan_out, rank, generator=gen) * 0.02
        lora.tensors[path] = (a, b)
    return lora


def load_lora_file(path: str) -> LoRA:
    from safetensors import safe_open

    lora = LoRA(name=path)
    with safe_open(path, framework="pt") as f:
        keys = [k for k in f.keys() if k.endswith(".lora_A")]
        for k in keys:
            base = k[: -len(".lora_A")]
            lora.tensors[base] = (f.get_tensor(k), f.get_tensor(base + ".lora_B"))
    return lora


def save_lora_file(lora: LoRA, path: str) -> str:
    from safetensors.torch import save_file

    tensors = {}
    for base, (a, b) in lora.tensors.items():
        tensors[base + ".lora_A"] = a.contiguous()
        tensors[base + ".lora_B"] = b.contiguous()
    save_file(tensors, path)
    return path


_lora_files: Dict[str, str] = {}


def lora_dir() -> str:
    import os

    return os.environ.get("SDWD_LORA_DIR", "loras")


def refresh_lora_files(dirpath: Optional[str] = None) -> Dict[str, str]:
    """Scan the lora directory (sdwui's Lora folder + refresh-loras route);
    *.safetensors become addressable <lora:stem:scale> names."""
    import os

    global _lora_files
    d = dirpath or lora_dir()
    found: Dict[str, str] = {}
    if os.path.isdir(d):
        for fn in sorted(os.listdir(d)):
            if fn.endswith(".safetensors"):
                found[os.path.splitext(fn)[0]] = os.path.join(d, fn)
    _lora_files = found
    return dict(found)


def lora_files() -> Dict[str, str]:
    return _lora_files


class LoraManager:
    """Tracks the adapter set merged into one UNet; swaps sets reversibly."""

    def __init__(self, unet: torch.nn.Module):
        self.unet = unet
        self.active: List[Tuple[str, float]] = []
        self._registry: Dict[str, LoRA] = {}
        # pristine copies of touched weights: full unmerge restores
        # bit-exactly ((w + d) - d is not w in floating point)
        self._pristine: Dict[str, torch.Tensor] = {}

    def register(self, lora: LoRA) -> None:
        self._registry[lora.name] = lora

    def get(self, name: str) -> LoRA:
        if name not in self._registry:
            path = lora_files().get(name)
            if path is not None:
                self._registry[name] = load_lora_file(path)
            else:
                # deterministic random fallback (no files in this environment)
                self._registry[name] = make_random_lora(name, self.unet)
        return self._registry[name]

    @torch.no_grad()
    def _apply(self, name: str, scale: float, sign: float) -> None:
        lora = self.get(name)
        mods = dict(self.unet.named_modules())
        for path, (a, b) in lora.tensors.items():
            mod = mods.get(path)
            if mod is None or not hasattr(mod, "weight"):
                continue
            d = lora.delta(path, mod.weight)
            if d is not None:
                if path not in self._pristine:
                    self._pristine[path] = mod.weight.detach().clone()
                mod.weight.add_(
                    (sign * scale) * d.to(mod.weight.device, mod.weight.dtype)
                )
                # invalidate any prepped-weight caches (SDConv2d)
                if hasattr(mod, "_wprep_cache"):
                    mod._wprep_cache = None

    @torch.no_grad()
    def set_active(self, wanted: List[Tuple[str, float]]) -> None:
        """Transition the merged set to ``wanted`` (unmerge removed, merge
        added, adjust rescaled)."""
        if wanted == self.active:
            return
        if not wanted and self._pristine:
            # bit-exact restore from the pristine snapshots
            mods = dict(self.unet.named_modules())
            for path, w in self._pristine.items():
                mod = mods.get(path)
                if mod is not None:
                    mod.weight.copy_(w)
                    if hasattr(mod, "_wprep_cache"):
                        mod._wprep_cache = None
            self._pristine.clear()
        else:
            for name, scale in self.active:
                self._apply(name, scale, -1.0)
            for name, scale in wanted:
                self._apply(name, scale, +1.0)
        self.active = list(wanted)
        # the fused q/k/v projection concats must be refreshed IN PLACE so
        # captured hipGraphs (which baked the concat buffers' addresses)
        # replay the new weights (unet.refresh_fused_projections docstring)
        from .unet import refresh_fused_projections

        refresh_fused_projections(self.unet)
        if wanted:
            log.info("active loras: %s", wanted)


def parse_prompt_loras(prompt: str) -> Tuple[str, List[Tuple[str, float]]]:
    """Extract ``<lora:name:scale>`` tags (sdwui syntax); returns the
    cleaned prompt and the adapter list."""
    import re

    loras: List[Tuple[str, float]] = []

    def grab(mt):
        name = mt.group(1)
        scale = float(mt.group(2)) if mt.group(2) else 1.0
        loras.append((name, scale))
        return ""

    cleaned = re.sub(r"<lora:([^:>]+)(?::([0-9.]+))?>", grab, prompt)
    return cleaned.strip(), loras
