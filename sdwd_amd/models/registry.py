"""Model registry: named checkpoint families with deterministic random init.

The reference synced *named* checkpoints between workers by name via
POST /options; here a name maps to an architecture config,
weights are random-init (no network for real checkpoints — BASELINE.md) but
deterministic per name, so every rank loading "sd15" holds identical
weights even without a broadcast, and the broadcast path (parallel/engine)
is still exercised to guarantee it.

Real weights can be loaded from a safetensors file when one exists locally.
"""
from __future__ import annotations

import zlib
from dataclasses import dataclass
from typing import Callable, Dict, Optional

import torch
import torch.nn as nn

from ..utils import get_logger
from .clip import CLIPTextEncoder
from .unet import UNetConfig, UNetModel
from .vae import AutoencoderKL, VAEConfig

log = get_logger("models")


@dataclass
class ModelBundle:
    name: str
    text_encoder: CLIPTextEncoder
    text_encoder_2: Optional[CLIPTextEncoder]  # SDXL second encoder
    unet: UNetModel
    vae: AutoencoderKL
    context_dim: int
    is_sdxl: bool = False
    # SDXL-refiner lineage: CLIP-G only (text_encoder is None), context
    # 1280, ADM = pooled + [orig_h, orig_w, crop_t, crop_l, aesthetic]
    is_refiner: bool = False
    # "eps" (noise prediction) or "v" (velocity, SD2.x-768 lineage);
    # the pipeline converts v -> eps algebraically before the sampler
    prediction_type: str = "eps"

    @property
    def latent_channels(self) -> int:
        # from the VAE, not unet.in_channels: inpainting UNets take
        # 2*latent+1 input channels but the latent state is still 4-wide
        return self.vae.cfg.latent_channels

    def to(self, device, dtype=None) -> "ModelBundle":
        for m in (self.text_encoder, self.text_encoder_2, self.unet, self.vae):
            if m is not None:
                m.to(device=device, dtype=dtype)
        return self

    def eval(self) -> "ModelBundle":
        for m in (self.text_encoder, self.text_encoder_2, self.unet, self.vae):
            if m is not None:
                m.eval()
        return self

    def parameters(self):
        for m in (self.text_encoder, self.text_encoder_2, self.unet, self.vae):
            if m is not None:
                yield from m.parameters()


def _seeded_init(module: nn.Module, seed: int) -> None:
    """Deterministic, device-independent random init.

    Every parameter must come from the seeded generator: PyTorch's default
    Linear/Conv bias init draws from the GLOBAL rng, which made two
    processes build different "identical" models (and made seeded
    generations irreproducible across runs). Norm weights stay at their
    deterministic default (ones)."""
    gen = torch.Generator().manual_seed(seed)
    with torch.no_grad():
        for name, p in module.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, mean=0.0, std=0.02, generator=gen)
            elif name.endswith(".bias") or name == "bias":
                nn.init.normal_(p, mean=0.0, std=0.01, generator=gen)
            # 1-d norm weights keep their default (ones)


def _build_sd15(name: str) -> ModelBundle:
    te = CLIPTextEncoder()
    unet = UNetModel(UNetConfig.sd15())
    vae = AutoencoderKL(VAEConfig.sd())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=768)


def _build_sd21(name: str) -> ModelBundle:
    """SD2.1-base shape: open_clip ViT-H text tower (width 1024, 24 layers,
    16 heads, fused qkv — the native layout), UNet with head_dim 64 and
    context 1024, same VAE. eps-prediction (the 512-base lineage)."""
    te = CLIPTextEncoder(d_model=1024, layers=24, heads=16)
    unet = UNetModel(
        UNetConfig(context_dim=1024, num_heads=0)  # num_heads 0 -> head_dim 64
    )
    vae = AutoencoderKL(VAEConfig.sd())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=1024)


def _build_sd15_inpaint(name: str) -> ModelBundle:
    """sd15-inpainting lineage: 9-channel UNet input (4 latent + 1 mask +
    4 masked-image latent, the runwayml inpainting conditioning)."""
    te = CLIPTextEncoder()
    unet = UNetModel(UNetConfig(in_channels=9))
    vae = AutoencoderKL(VAEConfig.sd())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=768)


def _build_sdxl_refiner(name: str) -> ModelBundle:
    """SDXL-refiner lineage (sgm refiner config): CLIP-G text tower ONLY
    (the refiner has no CLIP-L), context 1280, model_channels 384 with
    transformer depth 4 at the inner levels, ADM vector = pooled(1280) +
    5 x 256 fourier conds (orig size, crop, aesthetic score)."""
    te2 = CLIPTextEncoder(d_model=1280, layers=32, heads=20)
    unet = UNetModel(UNetConfig(
        model_channels=384,
        channel_mult=[1, 2, 4, 4],
        transformer_depth=[0, 4, 4, 0],
        context_dim=1280,
        num_heads=0,  # head_dim 64
        adm_in_channels=1280 + 5 * 256,
    ))
    vae = AutoencoderKL(VAEConfig.sd())
    vae.cfg.scale_factor = 0.13025
    for seed_off, m in enumerate((te2, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(
        name, None, te2, unet, vae, context_dim=1280, is_sdxl=True,
        is_refiner=True,
    )


def _build_sd21_inpaint(name: str) -> ModelBundle:
    """SD2-inpainting lineage: the sd21 stack with a 9-channel UNet."""
    te = CLIPTextEncoder(d_model=1024, layers=24, heads=16)
    unet = UNetModel(
        UNetConfig(in_channels=9, context_dim=1024, num_heads=0)
    )
    vae = AutoencoderKL(VAEConfig.sd())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=1024)


def _build_sdxl_inpaint(name: str) -> ModelBundle:
    """SDXL-inpainting lineage (diffusers sd-xl-inpainting layout): the
    XL UNet with a 9-channel input (4 latent + 1 mask + 4 masked-image
    latent); the same generic inpaint-conditioning path as sd15-inpaint
    (pipeline in_channels == 2*lat_c+1 detection)."""
    te = CLIPTextEncoder(d_model=768, layers=12, heads=12)
    te2 = CLIPTextEncoder(d_model=1280, layers=32, heads=20)
    cfg = UNetConfig.sdxl()
    cfg.in_channels = 9
    unet = UNetModel(cfg)
    vae = AutoencoderKL(VAEConfig.sd())
    vae.cfg.scale_factor = 0.13025
    for seed_off, m in enumerate((te, te2, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(
        name, te, te2, unet, vae, context_dim=2048, is_sdxl=True
    )


def _build_tiny_inpaint(name: str) -> ModelBundle:
    te = CLIPTextEncoder(d_model=64, layers=2, heads=2, max_len=77)
    cfg = UNetConfig.tiny()
    cfg.in_channels = 9
    unet = UNetModel(cfg)
    vae = AutoencoderKL(VAEConfig.tiny())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=64)


def _build_sd21v(name: str) -> ModelBundle:
    """SD2.1-768 lineage: same architecture as sd21, v-prediction."""
    b = _build_sd21(name)
    b.prediction_type = "v"
    return b


def _build_tiny_v(name: str) -> ModelBundle:
    """CPU-test v-prediction model (tiny arch, velocity output)."""
    b = _build_tiny(name)
    b.prediction_type = "v"
    return b


def _build_tiny_inpaint_v(name: str) -> ModelBundle:
    """CPU-test v-prediction 9-channel inpainting model (exercises the
    _to_eps-on-pre-concat-latents path in pipeline.py model_fn)."""
    b = _build_tiny_inpaint(name)
    b.prediction_type = "v"
    return b


def _build_sdxl(name: str) -> ModelBundle:
    te = CLIPTextEncoder(d_model=768, layers=12, heads=12)
    te2 = CLIPTextEncoder(d_model=1280, layers=32, heads=20)
    unet = UNetModel(UNetConfig.sdxl())
    vae = AutoencoderKL(VAEConfig.sd())
    vae.cfg.scale_factor = 0.13025
    for seed_off, m in enumerate((te, te2, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(
        name, te, te2, unet, vae, context_dim=2048, is_sdxl=True
    )


def _build_tiny_xl(name: str) -> ModelBundle:
    """Tiny model exercising the SDXL code path (dual encoders, penultimate
    hidden states, pooled+size ADM conditioning) at CPU-test scale."""
    from .unet import UNetConfig

    te = CLIPTextEncoder(d_model=32, layers=2, heads=2)
    te2 = CLIPTextEncoder(d_model=32, layers=2, heads=2)
    cfg = UNetConfig(
        model_channels=32,
        channel_mult=[1, 2],
        num_res_blocks=1,
        transformer_depth=[0, 1],
        context_dim=64,
        num_heads=0,  # head_dim 64 convention
        groups=8,
        adm_in_channels=32 + 6 * 256,
    )
    unet = UNetModel(cfg)
    vae = AutoencoderKL(VAEConfig.tiny())
    for seed_off, m in enumerate((te, te2, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, te2, unet, vae, context_dim=64, is_sdxl=True)


def _build_tiny_xl_inpaint(name: str) -> ModelBundle:
    """CPU-test XL + 9-channel inpaint interaction (the sdxl-inpaint
    lineage at tiny scale)."""
    b = _build_tiny_xl(name)
    from .unet import UNetModel as _UM

    cfg = b.unet.cfg
    cfg.in_channels = 9
    unet = _UM(cfg)
    _seeded_init(unet, zlib.crc32(name.encode()) % (2**31) + 7)
    return ModelBundle(
        name, b.text_encoder, b.text_encoder_2, unet, b.vae,
        context_dim=64, is_sdxl=True,
    )


def _build_tiny_xl_refiner(name: str) -> ModelBundle:
    """CPU-test refiner lineage (CLIP-G-only + aesthetic ADM at tiny
    scale; shares tiny's VAE so it can refine tiny/tiny-xl latents)."""
    te2 = CLIPTextEncoder(d_model=32, layers=2, heads=2)
    cfg = UNetConfig(
        model_channels=32,
        channel_mult=[1, 2],
        num_res_blocks=1,
        transformer_depth=[0, 1],
        context_dim=32,
        num_heads=0,
        groups=8,
        adm_in_channels=32 + 5 * 256,
    )
    unet = UNetModel(cfg)
    vae = AutoencoderKL(VAEConfig.tiny())
    for seed_off, m in enumerate((te2, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(
        name, None, te2, unet, vae, context_dim=32, is_sdxl=True,
        is_refiner=True,
    )


def _build_tiny(name: str) -> ModelBundle:
    te = CLIPTextEncoder(d_model=64, layers=2, heads=2, max_len=77)
    unet = UNetModel(UNetConfig.tiny())
    vae = AutoencoderKL(VAEConfig.tiny())
    for seed_off, m in enumerate((te, unet, vae)):
        _seeded_init(m, zlib.crc32(name.encode()) % (2**31) + seed_off)
    return ModelBundle(name, te, None, unet, vae, context_dim=64)


_BUILDERS: Dict[str, Callable[[str], ModelBundle]] = {
    "sd15": _build_sd15,
    "sd21": _build_sd21,
    "sd21v": _build_sd21v,
    "sd15-inpaint": _build_sd15_inpaint,
    "sdxl-inpaint": _build_sdxl_inpaint,
    "sdxl-refiner": _build_sdxl_refiner,
    "sd21-inpaint": _build_sd21_inpaint,
    "sdxl": _build_sdxl,
    "tiny-v": _build_tiny_v,
    "tiny-inpaint": _build_tiny_inpaint,
    "tiny-inpaint-v": _build_tiny_inpaint_v,
    "tiny": _build_tiny,
    "tiny-xl": _build_tiny_xl,
    "tiny-xl-inpaint": _build_tiny_xl_inpaint,
    "tiny-xl-refiner": _build_tiny_xl_refiner,
}

_cache: Dict[tuple, ModelBundle] = {}

# file-backed checkpoints (sdwui's checkpoint folder + refresh button):
# name (file stem) -> path, populated by refresh_checkpoint_files()
_FILE_MODELS: Dict[str, str] = {}


def checkpoint_dir() -> str:
    import os

    return os.environ.get("SDWD_CHECKPOINT_DIR", "checkpoints")


def refresh_checkpoint_files(dirpath: Optional[str] = None) -> list:
    """Scan the checkpoint directory for *.safetensors; each file becomes a
    loadable model name (its stem). Both formats load: native (saved by
    save_checkpoint) and sdwui/ldm (auto-detected by key prefix)."""
    import os

    global _FILE_MODELS
    d = dirpath or checkpoint_dir()
    found: Dict[str, str] = {}
    if os.path.isdir(d):
        for fn in sorted(os.listdir(d)):
            if fn.endswith(".safetensors"):
                found[os.path.splitext(fn)[0]] = os.path.join(d, fn)
    _FILE_MODELS = found
    return sorted(found)


def available_models() -> list:
    """ref worker.py:623-644 (GET /sd-models)."""
    return sorted(set(_BUILDERS) | set(_FILE_MODELS))


_VAE_FILES: Dict[str, str] = {}


def vae_dir() -> str:
    import os

    return os.environ.get("SDWD_VAE_DIR", "vae")


def refresh_vae_files(dirpath: Optional[str] = None) -> list:
    """Scan the VAE directory for standalone *.safetensors VAE files
    (the sdwui "SD VAE" dropdown the reference synced by name through
    load_options, ref worker.py:646-688). Each file's stem becomes a
    selectable VAE name; "auto" is the checkpoint's own VAE."""
    import os

    global _VAE_FILES
    d = dirpath or vae_dir()
    found: Dict[str, str] = {}
    if os.path.isdir(d):
        for fn in sorted(os.listdir(d)):
            if fn.endswith(".safetensors"):
                found[os.path.splitext(fn)[0]] = os.path.join(d, fn)
    _VAE_FILES = found
    return sorted(found)


def available_vaes() -> list:
    return ["auto"] + sorted(_VAE_FILES)


def load_vae_into(bundle, name: str) -> Dict[str, list]:
    """Swap ``bundle``'s VAE weights to the named standalone VAE file
    ("auto" restores the checkpoint's own VAE by rebuilding from its
    source). Returns the converter report."""
    from safetensors.torch import load_file

    from .convert import load_vae_state_dict

    if name not in _VAE_FILES:
        refresh_vae_files()
    if name not in _VAE_FILES:
        raise KeyError(f"unknown VAE '{name}'")
    state = load_file(_VAE_FILES[name])
    dev = next(bundle.vae.parameters()).device
    state = {k: v.to(dev) for k, v in state.items()}
    report = load_vae_state_dict(bundle.vae, state)
    if report["missing"]:
        raise ValueError(
            f"VAE '{name}' is missing {len(report['missing'])} tensors "
            f"(first: {report['missing'][:3]}) — wrong architecture?"
        )
    return report


def load_model(
    name: str, device="cpu", dtype: Optional[torch.dtype] = None,
    cache: bool = True,
) -> ModelBundle:
    # cache per (name, device, dtype): a shared single instance would be
    # silently .to()-moved under the feet of every other holder (e.g. a
    # LocalEngine with one pipeline per GPU in one process)
    key = (name, str(device), str(dtype))
    if cache and key in _cache:
        return _cache[key]
    if name not in _BUILDERS and name not in _FILE_MODELS:
        refresh_checkpoint_files()
    if name in _FILE_MODELS and name not in _BUILDERS:
        log.info("loading checkpoint file '%s'", _FILE_MODELS[name])
        bundle = load_checkpoint(_FILE_MODELS[name], device=device, dtype=dtype)
        bundle.name = name  # address by file stem, not the inferred arch
        if cache:
            _cache[key] = bundle
        return bundle
    if name not in _BUILDERS:
        raise KeyError(f"unknown model '{name}'; have {available_models()}")
    log.info("building model '%s' (random-init, deterministic)", name)
    bundle = _BUILDERS[name](name).eval()
    bundle.to(device, dtype)
    if cache:
        _cache[key] = bundle
    return bundle


_cn_cache: Dict[str, "object"] = {}


def load_controlnet(name: str = "controlnet-sd15", device="cpu", dtype=None):
    """ControlNet for a base arch (deterministic random init, like models).
    name: controlnet-<arch>, e.g. controlnet-sd15 / controlnet-tiny."""
    from .controlnet import ControlNetModel
    from .unet import UNetConfig

    if name in _cn_cache:
        mod = _cn_cache[name]
    else:
        arch = name.split("-", 1)[1] if "-" in name else "sd15"
        cfg = {
            "sd15": UNetConfig.sd15,
            "sdxl": UNetConfig.sdxl,
            "tiny": UNetConfig.tiny,
        }[arch]()
        mod = ControlNetModel(cfg, hint_factor=2 if arch == "tiny" else 8)
        _seeded_init(mod, zlib.crc32(name.encode()) % (2**31))
        mod.eval()
        _cn_cache[name] = mod
    mod.to(device=device, dtype=dtype)
    return mod


def clear_cache() -> None:
    _cache.clear()
    _cn_cache.clear()


# -- checkpoint files (ref C13 synced checkpoints by NAME over /options;
# here weights also round-trip to local safetensors files) -------------------
def save_checkpoint(bundle: ModelBundle, path: str) -> str:
    from safetensors.torch import save_file

    tensors = {}
    for prefix, mod in (
        ("text_encoder", bundle.text_encoder),
        ("text_encoder_2", bundle.text_encoder_2),
        ("unet", bundle.unet),
        ("vae", bundle.vae),
    ):
        if mod is None:
            continue
        for k, v in mod.state_dict().items():
            tensors[f"{prefix}.{k}"] = v.contiguous().cpu()
    save_file(tensors, path, metadata={"arch": bundle.name})
    return path


def load_checkpoint(path: str, device="cpu", dtype=None) -> ModelBundle:
    """Load a safetensors file saved by save_checkpoint; the architecture
    name is read from metadata (falls back to sd15)."""
    from safetensors import safe_open

    with safe_open(path, framework="pt") as f:
        meta = f.metadata() or {}
        arch = meta.get("arch", "sd15")
        keys = list(f.keys())
        if any(k.startswith("model.diffusion_model.") for k in keys):
            # a real sdwui/ldm checkpoint, not our native format: pick the
            # arch from the conv_in / cross-attention shapes
            from .convert import load_ldm_state_dict

            w_in = f.get_tensor(
                "model.diffusion_model.input_blocks.0.0.weight"
            )
            ch, in_ch = int(w_in.shape[0]), int(w_in.shape[1])
            is_xl = any(k.startswith("conditioner.") for k in keys)
            is_sd2 = any(k.startswith("cond_stage_model.model.") for k in keys)
            if ch == 384 and is_xl:
                arch = "sdxl-refiner"
            elif ch == 320:
                arch = "sdxl" if is_xl else ("sd21" if is_sd2 else "sd15")
                if in_ch == 9:  # the 9ch inpainting lineages
                    arch += "-inpaint"
            else:
                arch = "tiny-xl" if is_xl else (
                    "tiny-inpaint" if in_ch == 9 else "tiny"
                )
            bundle = _BUILDERS[arch](arch)
            load_ldm_state_dict(bundle, {k: f.get_tensor(k) for k in keys})
            bundle.eval().to(device, dtype)
            return bundle
        bundle = _BUILDERS[arch.split("/")[0] if arch in _BUILDERS else "sd15"](arch)
        by_prefix: Dict[str, Dict[str, torch.Tensor]] = {}
        for k in keys:
            prefix, rest = k.split(".", 1)
            by_prefix.setdefault(prefix, {})[rest] = f.get_tensor(k)
    for prefix, mod in (
        ("text_encoder", bundle.text_encoder),
        ("text_encoder_2", bundle.text_encoder_2),
        ("unet", bundle.unet),
        ("vae", bundle.vae),
    ):
        if mod is not None and prefix in by_prefix:
            mod.load_state_dict(by_prefix[prefix], strict=False)
    bundle.eval().to(device, dtype)
    return bundle
