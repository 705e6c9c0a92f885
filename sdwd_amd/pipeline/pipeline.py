"""txt2img / img2img pipelines — the compute the reference delegated per
POST /txt2img|/img2img runs here, in-process.

Determinism contract (ref C22): image k of a batch depends only on
(model, seed_k, subseed_k, request params) — never on batch position,
shard size or device count — so an 8-GPU gallery is image-for-image equal
to the 1-GPU gallery. Initial noise is drawn per-image from a CPU
generator seeded with that image's seed, then moved to the device.
"""
from __future__ import annotations

import math
import os
import time
from dataclasses import dataclass, field, replace
from typing import Callable, List, Optional

import torch

from ..models.registry import ModelBundle, load_model
from ..models import tokenizer
from ..utils import get_logger
from .graphs import GraphedDenoiser, GraphedModelFn, graphs_enabled
from .samplers import build_sampler
from .schedule import schedule_for, sigma_for_t

log = get_logger("pipeline")


@dataclass
class PipelineRequest:
    prompt: str = ""
    # optional per-image prompts (len == batch); overrides `prompt` and
    # disables editing/AND composition for the request (plain encode)
    prompts: Optional[List[str]] = None
    negative_prompt: str = ""
    steps: int = 20
    width: int = 512
    height: int = 512
    cfg_scale: float = 7.0
    sampler_name: str = "Euler a"
    scheduler: str = "Automatic"  # sdwui scheduler dropdown (sigma spacing)
    seeds: List[int] = field(default_factory=lambda: [0])
    subseeds: List[int] = field(default_factory=list)
    subseed_strength: float = 0.0
    # sdwui "Resize seed from": draw the initial noise at another
    # resolution's latent grid and resize it, so a composition found at
    # WxH roughly survives a resolution change
    seed_resize_from_w: int = 0
    seed_resize_from_h: int = 0
    # sdwui eta_noise_seed_delta: offsets the ancestral-noise seeds
    eta_noise_seed_delta: int = 0
    # img2img
    init_latents: Optional[torch.Tensor] = None  # pre-encoded [B,4,h,w]
    denoising_strength: float = 0.75
    # inpainting: uint8 mask, 255 = repaint region (sdwui convention)
    mask_image: Optional[torch.Tensor] = None    # [H,W] or [B,H,W]
    # sdwui "Masked content": 1 original (default), 2 latent noise,
    # 3 latent nothing ("fill"=0 is approximated upstream in the engine)
    inpainting_fill: int = 1
    # Soft inpainting (the webui-host built-in the reference forwards as an
    # alwayson-script payload): per-step SOFT latent blending instead of
    # hard-mask pinning, plus a pixel-space composite against the decoded
    # init at the end. Parameter names follow the host UI labels.
    soft_inpainting: bool = False
    si_schedule_bias: float = 1.0          # >1: repaint later / tighter band
    si_preservation_strength: float = 0.5  # latent-magnitude restoration 0..1
    si_transition_contrast_boost: float = 4.0  # sharpens the blend band
    si_mask_influence: float = 0.0         # pixel composite: mask vs diff
    si_difference_threshold: float = 0.5   # pixel diff at which gate = 0.5
    si_difference_contrast: float = 2.0    # sharpness of the pixel gate
    # hires fix (sdwui two-pass: base gen -> latent upscale -> img2img pass)
    enable_hr: bool = False
    hr_scale: float = 2.0
    hr_steps: int = 0  # 0 = same as steps
    # latent upscaler (sdwui hr_upscaler "Latent..." family):
    # nearest | bilinear | bicubic | bilinear-antialiased | bicubic-antialiased
    hr_upscaler: str = "nearest"
    # optional different conditioning for the hires second pass
    # (sdwui hr_prompt / hr_negative_prompt; "" = reuse the base prompt)
    hr_prompt: str = ""
    hr_negative_prompt: str = ""
    # explicit hires target size (sdwui hr_resize_x/y; 0 = use hr_scale)
    hr_resize_x: int = 0
    hr_resize_y: int = 0
    hr_sampler_name: str = ""  # "" = same sampler as the first pass
    # two-model refiner (sdwui refiner_checkpoint/refiner_switch_at): the
    # base model denoises the first switch_at fraction of steps, the
    # refiner model finishes (both share the latent space / VAE)
    refiner_model: str = ""
    refiner_switch_at: float = 0.8
    tiling: bool = False  # seamless tiles: wrap-around conv padding
    # k-diffusion stochasticity (sdwui Sampler parameters section)
    s_churn: float = 0.0
    s_tmin: float = 0.0
    s_tmax: float = 0.0   # 0 = unlimited (sdwui convention)
    s_noise: float = 1.0
    # skip the uncond eval below this sigma (sdwui s_min_uncond perf knob)
    s_min_uncond: float = 0.0
    # sdwui Eta: ancestral/SDE noise multiplier; None = sampler default
    eta: float = -1.0
    # controlnet (ref C17 executed natively); either the single-unit
    # legacy fields or a list of unit dicts with
    # {image, model, scale, guidance_start, guidance_end}
    control_image: Optional[torch.Tensor] = None  # [B,H,W,3] uint8
    control_model: str = ""  # e.g. "controlnet-sd15"; "" = off
    control_scale: float = 1.0
    control_units: Optional[List[dict]] = None
    clip_skip: int = 1  # 1 = final layer; 2 = penultimate (sdwui setting)
    # Regional Prompter matrix mode, executed natively (the reference
    # forwarded the extension's payload, C18): the prompt is split on
    # "BREAK" into [base?, region1, region2, ...]; regions tile the
    # canvas as columns or rows with the given ratios and blend through
    # masked cross-attention (attention couple). base_ratio > 0 keeps
    # that weight of the shared base prompt everywhere (and expects the
    # first BREAK chunk to BE the base prompt).
    regional_mode: str = ""        # "" off | "columns" | "rows"
    regional_ratios: str = "1,1"
    regional_base_ratio: float = 0.2

    @property
    def batch_size(self) -> int:
        return len(self.seeds)


@dataclass
class PipelineResult:
    images: torch.Tensor  # [B, H, W, 3] uint8 on CPU
    seeds: List[int]
    subseeds: List[int]
    infotexts: List[str]
    elapsed: float = 0.0
    interrupted: bool = False


def _slerp(a: torch.Tensor, b: torch.Tensor, t: float) -> torch.Tensor:
    """Spherical interpolation between noise tensors (subseed variation)."""
    af, bf = a.flatten().double(), b.flatten().double()
    dot = torch.dot(af / af.norm(), bf / bf.norm()).clamp(-1, 1)
    omega = torch.acos(dot)
    if omega.abs() < 1e-6:
        out = (1 - t) * af + t * bf
    else:
        so = torch.sin(omega)
        out = (math.sin((1 - t) * omega) / so) * af + (
            math.sin(t * omega) / so
        ) * bf
    return out.reshape(a.shape).to(a.dtype)


def _parse_regional(req: "PipelineRequest"):
    """Split the prompt on BREAK into (base, regions) per the ratios.

    With base_ratio > 0 the first chunk is the shared base prompt and the
    remaining chunks are the regions; with base_ratio == 0 every chunk is
    a region (base attention falls back to an empty prompt for uncovered
    pixels only). Returns None (regional off, warning logged) when the
    chunk count does not match the ratio count."""
    import re as _re

    chunks = [c.strip() for c in _re.split(r"\bBREAK\b", req.prompt)]
    try:
        ratios = [
            float(v) for v in str(req.regional_ratios).split(",") if v.strip()
        ]
    except ValueError:
        ratios = []
    if not ratios or any(r <= 0 for r in ratios):
        log.warning("regional: bad ratios %r — disabled", req.regional_ratios)
        return None
    base_w = min(max(float(req.regional_base_ratio), 0.0), 1.0)
    if base_w > 0 and len(chunks) == len(ratios) + 1:
        return chunks[0], chunks[1:], ratios, base_w
    if len(chunks) == len(ratios):
        return ("", chunks, ratios, base_w) if base_w == 0 else (
            chunks[0], chunks, ratios, base_w
        )
    log.warning(
        "regional: %d BREAK chunks vs %d ratios — disabled",
        len(chunks), len(ratios),
    )
    return None


def _region_masks(
    mode: str, ratios: List[float], lat_h: int, lat_w: int
) -> torch.Tensor:
    """[R, lat_h, lat_w] 0/1 masks tiling the canvas as columns or rows
    proportionally to the ratios (matrix-mode Regional Prompter)."""
    total = sum(ratios)
    extent = lat_w if mode == "columns" else lat_h
    nr = len(ratios)
    bounds = [0]
    acc = 0.0
    for r in ratios:
        acc += r
        bounds.append(int(round(extent * acc / total)))
    bounds[-1] = extent
    # every region keeps at least one line of the grid despite rounding
    for i in range(1, nr + 1):
        bounds[i] = max(bounds[i], bounds[i - 1] + 1)
    bounds[-1] = extent
    for i in range(nr, 0, -1):
        bounds[i - 1] = min(bounds[i - 1], bounds[i] - 1)
    masks = torch.zeros(nr, lat_h, lat_w)
    for i in range(nr):
        lo, hi = max(bounds[i], 0), bounds[i + 1]
        if mode == "columns":
            masks[i, :, lo:hi] = 1.0
        else:
            masks[i, lo:hi, :] = 1.0
    return masks


def _contrast(w: torch.Tensor, g: float) -> torch.Tensor:
    """Monotone contrast curve w^g / (w^g + (1-w)^g): fixed points at
    0, 0.5 and 1; g=1 is the identity, g>1 pushes values toward 0/1
    (sharpens a soft band without moving its midline)."""
    if g == 1.0:
        return w
    wg = w.clamp(0, 1).pow(g)
    return wg / (wg + (1.0 - w.clamp(0, 1)).pow(g)).clamp_min(1e-8)


_HR_MODES = {
    "nearest": ("nearest", False),
    "latent": ("nearest", False),  # sdwui "Latent" = nearest
    "bilinear": ("bilinear", False),
    "latent (bilinear)": ("bilinear", False),
    "bicubic": ("bicubic", False),
    "latent (bicubic)": ("bicubic", False),
    "bilinear-antialiased": ("bilinear", True),
    "latent (bilinear antialiased)": ("bilinear", True),
    "bicubic-antialiased": ("bicubic", True),
    "latent (bicubic antialiased)": ("bicubic", True),
}


def _upscale_latent(x: torch.Tensor, scale: float, upscaler: str,
                    size=None) -> torch.Tensor:
    """Latent-space upscale for the hires-fix first->second pass handoff
    (sdwui's "Latent ..." hr_upscaler family; ref CHANGELOG hires support).
    ``size`` (lat_h, lat_w) overrides ``scale`` (sdwui hr_resize_x/y)."""
    mode, aa = _HR_MODES.get((upscaler or "nearest").lower(), ("nearest", False))
    kwargs = {"antialias": True} if aa else {}
    if size is not None:
        return torch.nn.functional.interpolate(
            x, size=size, mode=mode, **kwargs
        )
    return torch.nn.functional.interpolate(
        x, scale_factor=scale, mode=mode, **kwargs
    )


def hires_active(enable_hr: bool, hr_scale: float, rx: int, ry: int) -> bool:
    """True when the hires second pass runs: sdwui's "resize to" mode
    (hr_resize_x/y set) enables the pass even with hr_scale<=1 (the UI
    sends hr_scale=0 in that mode)."""
    return bool(enable_hr) and (hr_scale > 1.0 or rx > 0 or ry > 0)


def hr_target_resolution(
    width: int, height: int, hr_scale: float, rx: int, ry: int, f: int = 8
):
    """sdwui hires target-resolution semantics
    (processing.py calculate_target_resolution):

    - rx == ry == 0: upscale by hr_scale;
    - one of rx/ry set: the other follows the source aspect ratio;
    - both set: upscale preserving aspect until the target is covered,
      then center-crop ("truncate") the latent down to the target.

    Returns (up_lat_h, up_lat_w, crop_lat_y, crop_lat_x): the latent
    dims to upscale to, and how many latent rows/cols to crop off
    (split top/bottom, left/right) before the second denoise pass.
    """
    if rx <= 0 and ry <= 0:
        ux, uy = int(width * hr_scale), int(height * hr_scale)
        tx = ty = 0
    elif ry <= 0:
        ux, uy = rx, rx * height // width
        tx = ty = 0
    elif rx <= 0:
        ux, uy = ry * width // height, ry
        tx = ty = 0
    else:
        src_ratio = width / height
        dst_ratio = rx / ry
        if src_ratio < dst_ratio:
            ux, uy = rx, rx * height // width
        else:
            ux, uy = ry * width // height, ry
        tx = (ux - rx) // f
        ty = (uy - ry) // f
    return uy // f, ux // f, max(0, ty), max(0, tx)


# sdwui's NON-"Latent" hr_upscalers work in pixel space (decode ->
# upscale -> re-encode); the model-free kernels map to torch modes and
# model-based upscalers (ESRGAN/SwinIR/LDSR — weights can't ship
# offline) fall back to bicubic-antialiased with a warning.
_PIXEL_HR_MODES = {
    "none": ("nearest", False),
    "lanczos": ("bicubic", True),  # closest torch kernel
    "pixel nearest": ("nearest", False),
    "pixel bilinear": ("bilinear", True),
    "pixel bicubic": ("bicubic", True),
}


def _apply_sampler_params(sampler, req: "PipelineRequest") -> None:
    sampler.s_churn = float(req.s_churn)
    sampler.s_tmin = float(req.s_tmin)
    sampler.s_tmax = float(req.s_tmax) if req.s_tmax > 0 else float("inf")
    sampler.s_noise = float(req.s_noise)
    if req.eta >= 0:  # sdwui Eta; -1 keeps the sampler's own default
        sampler.eta = float(req.eta)


def _image_noise(
    seed: int,
    subseed: int,
    subseed_strength: float,
    shape,
) -> torch.Tensor:
    """Per-image deterministic CPU noise (device-independent)."""
    g = torch.Generator("cpu").manual_seed(int(seed) & 0xFFFFFFFF)
    noise = torch.randn(shape, generator=g, dtype=torch.float32)
    if subseed_strength and subseed_strength > 0 and subseed >= 0:
        g2 = torch.Generator("cpu").manual_seed(int(subseed) & 0xFFFFFFFF)
        noise2 = torch.randn(shape, generator=g2, dtype=torch.float32)
        noise = _slerp(noise, noise2, float(subseed_strength))
    return noise


class StableDiffusionPipeline:
    """One model bundle on one device; reusable across requests."""

    def __init__(
        self,
        model: ModelBundle | str = "sd15",
        device: str | torch.device = "cpu",
        dtype: Optional[torch.dtype] = None,
    ) -> None:
        self.device = torch.device(device)
        if dtype is None:
            dtype = (
                torch.bfloat16 if self.device.type == "cuda" else torch.float32
            )
        self.dtype = dtype
        if isinstance(model, str):
            model = load_model(model, device=self.device, dtype=dtype)
        else:
            model.to(self.device, dtype)
        self.model = model
        if self.device.type == "cuda":
            # channels_last: MIOpen NHWC conv solvers are ~30% faster on
            # gfx950 and skip the NCHW<->NHWC transposes (conv_ab probe).
            # MIOpen benchmark-find is too slow on fresh machines (~10 min
            # of solver tuning per process); immediate FAST find + our own
            # implicit-GEMM convs for the hot 3x3/1x1 shapes instead.
            os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
            self.model.unet.to(memory_format=torch.channels_last)
            self.model.vae.to(memory_format=torch.channels_last)
        from ..models.lora import LoraManager

        self.lora = LoraManager(self.model.unet)
        self._tiling = False
        self._last_preview = None  # latest denoise-loop latents (preview)
        self._denoiser = GraphedDenoiser(
            lambda x, ts, ctx, y: self.model.unet(x, ts, ctx, y=y),
            self.device,
        )
        # whole-step graphs persist ACROSS generations (a fresh wrapper per
        # request would re-capture every generation: +3 UNet forwards)
        self._wholestep_cache: dict = {}
        self._canon_extra_cache: dict = {}

    # -- conditioning --------------------------------------------------------
    @torch.no_grad()
    def encode_prompts(
        self, prompts: List[str], negatives: List[str], clip_skip: int = 1,
        bundle=None,
    ) -> tuple:
        tokens, weights = tokenizer.encode_batch_weighted(
            prompts + negatives, device=self.device
        )
        # [N, K, 77]: long prompts spill into K chunks, each run through
        # CLIP separately and concatenated (sdwui unlimited prompt length)
        n, k, L = tokens.shape
        flat = tokens.reshape(n * k, L)
        m = bundle if bundle is not None else self.model

        def _cat_chunks(h):
            return h.reshape(n, k * L, h.shape[-1])

        if m.is_sdxl:
            h2 = _cat_chunks(m.text_encoder_2(flat, penultimate=True))
            if m.text_encoder is None:  # refiner lineage: CLIP-G only
                ctx = h2
            else:
                h1 = _cat_chunks(m.text_encoder(flat, penultimate=True))
                ctx = torch.cat([h1, h2], dim=-1)
            # pooled conditioning comes from the FIRST chunk's EOT
            first = tokens[:, 0]
            pooled = m.text_encoder_2.pooled(
                first, m.text_encoder_2(first)
            )
        else:
            ctx = _cat_chunks(m.text_encoder(flat, clip_skip=clip_skip))
            pooled = None
        # prompt-attention weights (sdwui semantics): scale the hidden
        # states of weighted tokens, then restore the original mean
        if (weights != 1.0).any():
            orig_mean = ctx.float().mean(dim=(1, 2), keepdim=True)
            ctx = ctx * weights[:, :, None].to(ctx.dtype)
            new_mean = ctx.float().mean(dim=(1, 2), keepdim=True)
            ctx = ctx * (orig_mean / new_mean.clamp_min(1e-9)).to(ctx.dtype)
        n = len(prompts)
        cond, uncond = ctx[:n], ctx[n:]
        pooled_cu = (pooled[:n], pooled[n:]) if pooled is not None else None
        return cond.to(self.dtype), uncond.to(self.dtype), pooled_cu

    def _sdxl_vector(self, req: PipelineRequest, pooled: torch.Tensor,
                     bundle=None, aesthetic: float = 6.0):
        """SDXL add-conditioning: pooled embed + fourier cond vectors.

        Base models take [orig_h, orig_w, crop_t, crop_l, tgt_h, tgt_w];
        the refiner lineage takes [orig_h, orig_w, crop_t, crop_l,
        aesthetic_score] (sgm convention: 6.0 for the cond row, 2.5 for
        the negative)."""
        m = bundle if bundle is not None else self.model
        if getattr(m, "is_refiner", False):
            conds = [req.height, req.width, 0, 0, aesthetic]
        else:
            conds = [
                req.height, req.width,  # original size
                0, 0,                   # crop top-left
                req.height, req.width,  # target size
            ]
        sizes = torch.tensor(
            conds, dtype=torch.float32, device=self.device,
        )
        from .. import ops

        emb = ops.timestep_embedding(sizes, 256).flatten()
        b = pooled.shape[0]
        return torch.cat(
            [pooled.float(), emb[None].expand(b, -1)], dim=-1
        ).to(self.dtype)

    def _build_ctx(self, p_text: str, n_text: str, req: "PipelineRequest",
                   b: int):
        """Conditioning for one prompt segment: handles AND composition.

        Returns (ctx, y, weights): ctx rows are [cond_1..cond_k, uncond],
        each broadcast to the batch; weights are the AND sub-prompt weights
        (k == 1, weight 1.0 for a plain prompt)."""
        from .prompt_schedule import split_and

        parts = split_and(p_text)
        prompts = [t for t, _ in parts]
        ws = [w for _, w in parts]
        c, u, pl = self.encode_prompts(prompts, [n_text], req.clip_skip)
        rows = [c[i : i + 1].expand(b, -1, -1) for i in range(len(prompts))]
        rows.append(u.expand(b, -1, -1))
        ctx = torch.cat(rows, dim=0)
        y = None
        if self.model.is_sdxl and pl is not None:
            pc, pu = pl
            yrows = [
                self._sdxl_vector(req, pc[i : i + 1].expand(b, -1))  # cond
                for i in range(len(prompts))
            ]
            yrows.append(
                self._sdxl_vector(req, pu.expand(b, -1), aesthetic=2.5)
            )
            y = torch.cat(yrows)
        return ctx, y, ws

    def _latent_mask(self, mask_image, b, lat_h, lat_w):
        """[H,W]/[B,H,W] uint8 mask -> [B,1,lat_h,lat_w] float in 0..1."""
        mk = mask_image
        if mk.dim() == 2:
            mk = mk[None]
        lat_mask = torch.nn.functional.interpolate(
            (mk.float() / 255.0)[:, None], size=(lat_h, lat_w),
            mode="area",
        ).clamp(0, 1).to(self.device)
        if lat_mask.shape[0] == 1 and b > 1:
            lat_mask = lat_mask.expand(b, -1, -1, -1)
        return lat_mask

    def _soft_composite(
        self, images: torch.Tensor, req: "PipelineRequest"
    ) -> torch.Tensor:
        """Soft-inpainting pixel composite: revert pixels the denoise
        barely changed to the decoded original, gated by
          gate = contrast(|new - orig| / (2 * difference_threshold))
        (gate = 0.5 exactly at the threshold) and pulled toward the raw
        mask by `mask influence`. Kills VAE round-trip drift outside the
        repaint region while keeping genuinely repainted pixels."""
        lat = req.init_latents.to(self.device, self.dtype)
        # per-image decode: like encode_image, keeps the reference pixels
        # identical under any shard split (batched conv numerics vary
        # with batch size at the ulp level)
        orig = torch.cat([
            (
                (self.model.vae.decode(lat[i : i + 1]).float() + 1.0)
                * 127.5
            )
            .clamp(0, 255)
            .to(torch.uint8)
            .permute(0, 2, 3, 1)
            .cpu()
            for i in range(lat.shape[0])
        ])
        if orig.shape != images.shape:  # defensive: odd sizes
            return images
        new_f = images.float()
        diff = (new_f - orig.float()).abs().mean(dim=-1, keepdim=True) / 255.0
        thr = max(float(req.si_difference_threshold), 1e-3)
        gate = _contrast(
            (diff / (2.0 * thr)).clamp(0, 1),
            max(float(req.si_difference_contrast), 1.0),
        )
        mk = req.mask_image
        if mk.dim() == 2:
            mk = mk[None]
        mk = torch.nn.functional.interpolate(
            (mk.float() / 255.0)[:, None],
            size=images.shape[1:3],
            mode="bilinear",
            align_corners=False,
        ).clamp(0, 1)
        if mk.shape[0] == 1 and images.shape[0] > 1:
            mk = mk.expand(images.shape[0], -1, -1, -1)
        mk = mk.permute(0, 2, 3, 1)
        # the mask floors the weight (repainted pixels are never reverted);
        # the difference gate only reclaims drift OUTSIDE the mask
        w = mk + (1.0 - mk) * gate
        mi = min(max(float(req.si_mask_influence), 0.0), 1.0)
        if mi > 0.0:
            w = (1.0 - mi) * w + mi * mk
        out = w * new_f + (1.0 - w) * orig.float()
        return out.round().clamp(0, 255).to(torch.uint8)

    def _pixel_hires(
        self, x: torch.Tensor, req: "PipelineRequest", hr_size, f: int
    ) -> torch.Tensor:
        """Pixel-space hires handoff (sdwui non-"Latent" hr_upscalers):
        decode the base latents, upscale the uint8 pixels (the PIL round
        trip sdwui performs), re-encode per image (shard-invariant,
        seeded). Model-based upscaler names fall back to
        bicubic-antialiased with a warning."""
        key = (req.hr_upscaler or "").lower()
        m, aa = _PIXEL_HR_MODES.get(key, ("bicubic", True))
        if key not in _PIXEL_HR_MODES:
            log.warning(
                "hr upscaler %r needs model weights that cannot ship "
                "offline; using bicubic-antialiased pixels", req.hr_upscaler,
            )
        if hr_size is not None:
            th, tw = hr_size
        else:
            th = int(x.shape[2] * req.hr_scale)
            tw = int(x.shape[3] * req.hr_scale)
        kwargs = {"antialias": True} if aa else {}
        outs = []
        for i in range(x.shape[0]):
            img = self.model.vae.decode(x[i : i + 1].to(self.dtype))
            img8 = (
                ((img.float() + 1.0) * 127.5).clamp(0, 255).to(torch.uint8)
            )
            up = torch.nn.functional.interpolate(
                img8.float(), size=(th * f, tw * f), mode=m, **kwargs
            )
            outs.append(
                up.clamp(0, 255).to(torch.uint8)[0].permute(1, 2, 0).cpu()
            )
        lat = self.encode_image(
            torch.stack(outs),
            seeds=[(int(s) ^ 0x9C51) & 0xFFFFFFFF for s in req.seeds],
        )
        return lat.to(self.dtype)

    # -- the denoise loop ----------------------------------------------------
    @torch.no_grad()
    def generate(
        self,
        req: PipelineRequest,
        interrupt: Optional[Callable[[], bool]] = None,
        step_callback: Optional[Callable[[int, int], None]] = None,
        decode: bool = True,
    ) -> PipelineResult:
        t0 = time.perf_counter()
        # sdwui <lora:name:scale> prompt tags: merge the adapter set into
        # the weights for this request (reversible; no per-step cost)
        from ..models.lora import parse_prompt_loras

        prompt, loras = parse_prompt_loras(req.prompt)
        if loras or self.lora.active:
            req = replace(req, prompt=prompt)
            self.lora.set_active(loras)
        if req.tiling != self._tiling:
            from ..models.layers import SDConv2d

            for m in list(self.model.unet.modules()) + list(
                self.model.vae.modules()
            ):
                if isinstance(m, SDConv2d):
                    m.circular = req.tiling
            # captured graphs baked the old padding path
            self._denoiser.cache.clear()
            self._wholestep_cache.clear()
            self._tiling = req.tiling
        b = req.batch_size
        f = self.model.vae.cfg.downsample_factor
        lat_h, lat_w = req.height // f, req.width // f
        lat_c = self.model.latent_channels
        subseeds = req.subseeds or [-1] * b

        # prompt editing ([from:to:when] / [a|b]) -> per-step text segments;
        # the common single-segment case stays on the encode-once path
        from .prompt_schedule import prompt_schedule

        per_image = bool(req.prompts) and len(req.prompts) == b
        if per_image:
            p_segs = [(0, req.prompts[0])]
            n_segs = [(0, req.negative_prompt)]
        else:
            p_segs = prompt_schedule(req.prompt, req.steps)
            n_segs = prompt_schedule(req.negative_prompt, req.steps)
        # img2img runs only the schedule tail; editing thresholds are
        # fractions of req.steps, so segments before the tail's first step
        # collapse into the base conditioning
        start_off = 0
        if req.init_latents is not None:
            start_off = max(
                0, req.steps - max(1, int(req.steps * req.denoising_strength))
            )

        def _seg_text_at(segs, step):
            return [txt for st, txt in segs if st <= step][-1]

        # one conditioning set per request (encode once, broadcast across
        # the batch); AND composition yields k cond rows + 1 uncond row.
        # Per-image prompts encode a cond row per image instead.
        if per_image:
            c_rows, u_rows, pl = self.encode_prompts(
                list(req.prompts), [req.negative_prompt] * b, req.clip_skip
            )
            ctx = torch.cat([c_rows, u_rows], dim=0)
            and_ws = [1.0]
            y = None
            if self.model.is_sdxl and pl is not None:
                y = torch.cat(
                    [
                        self._sdxl_vector(req, pl[0]),
                        self._sdxl_vector(req, pl[1], aesthetic=2.5),
                    ]
                )
        else:
            ctx, y, and_ws = self._build_ctx(
                _seg_text_at(p_segs, start_off),
                _seg_text_at(n_segs, start_off),
                req, b,
            )

        # Regional Prompter matrix mode: rebuild the cond rows from the
        # base chunk and wrap the conditioning so cross-attention blends
        # per-region contexts under columns/rows masks. Composes with CFG
        # and samplers; AND composition / prompt editing / per-image
        # prompts keep their own conditioning machinery and win.
        regional = (
            req.regional_mode in ("columns", "rows")
            and not per_image
            and and_ws == [1.0]
            and len(p_segs) == 1
            and len(n_segs) == 1
        )
        if regional:
            parsed = _parse_regional(req)
            if parsed is None:
                regional = False
            else:
                base_text, region_texts, ratios, base_w = parsed
                ctx, y, and_ws = self._build_ctx(
                    base_text, req.negative_prompt, req, b
                )
                region_rows, _, _ = self.encode_prompts(
                    region_texts, [req.negative_prompt], req.clip_skip
                )
                masks = _region_masks(
                    req.regional_mode, ratios, lat_h, lat_w
                )
                from ..models.unet import RegionalContext

                ctx = RegionalContext(
                    ctx, region_rows.to(ctx.dtype), masks.to(self.device),
                    rows=b, base_ratio=base_w, lat_hw=(lat_h, lat_w),
                )

        sched = schedule_for(req.sampler_name, req.steps, req.scheduler)
        sampler = build_sampler(req.sampler_name, sched)
        _apply_sampler_params(sampler, req)

        noise_shape = (lat_c, lat_h, lat_w)
        if req.seed_resize_from_w > 0 and req.seed_resize_from_h > 0:
            noise_shape = (
                lat_c, req.seed_resize_from_h // f, req.seed_resize_from_w // f
            )
        noise = torch.stack(
            [
                _image_noise(
                    req.seeds[i], subseeds[i], req.subseed_strength,
                    noise_shape,
                )
                for i in range(b)
            ]
        )
        if noise_shape != (lat_c, lat_h, lat_w):
            noise = torch.nn.functional.interpolate(
                noise, size=(lat_h, lat_w), mode="bilinear",
                antialias=False,
            )
        noise = noise.to(self.device, self.dtype)

        sig = sched.sigmas
        if req.init_latents is not None:
            # img2img: noise the init latents to the strength point, run the
            # tail of the schedule from there.
            start = start_off
            sched = type(sched)(
                sigmas=sched.sigmas[start:], timesteps=sched.timesteps[start:]
            )
            sampler = build_sampler(req.sampler_name, sched)
            _apply_sampler_params(sampler, req)
            s0 = float(sched.sigmas[0])
            init_lat0 = req.init_latents.to(self.device).float()
            if req.mask_image is not None and req.inpainting_fill in (2, 3):
                # masked content: replace the region's starting latents
                lm = self._latent_mask(
                    req.mask_image, b, lat_h, lat_w
                )
                if req.inpainting_fill == 2:  # latent noise
                    content = torch.stack([
                        torch.randn(
                            (lat_c, lat_h, lat_w),
                            generator=torch.Generator("cpu").manual_seed(
                                (int(sd) ^ 0xF111) & 0xFFFFFFFF
                            ),
                            dtype=torch.float32,
                        )
                        for sd in req.seeds
                    ]).to(self.device)
                else:  # latent nothing
                    content = torch.zeros_like(init_lat0)
                init_lat0 = (1.0 - lm) * init_lat0 + lm * content
            x = (init_lat0 + noise.float() * s0).to(self.dtype)
        else:
            x = (noise.float() * float(sig[0])).to(self.dtype)

        # ancestral noise: per-image generators stepped identically regardless
        # of shard composition
        ensd = int(req.eta_noise_seed_delta)
        gens = [
            torch.Generator("cpu").manual_seed(
                ((int(s) + ensd) ^ 0x5EED) & 0xFFFFFFFF
            )
            for s in req.seeds
        ]

        def noise_fn() -> torch.Tensor:
            n = torch.stack(
                [
                    torch.randn(
                        (lat_c, lat_h, lat_w), generator=g, dtype=torch.float32
                    )
                    for g in gens
                ]
            )
            return n.to(self.device, self.dtype)

        cfg = float(req.cfg_scale)
        unet = self.model.unet

        # prompt-editing segments: (t_threshold, ctx, y, weights) per
        # conditioning change, selected by the current timestep in model_fn
        seg_tensors: List[tuple] = []
        if (len(p_segs) > 1 or len(n_segs) > 1) and not per_image:
            ts_all = sched.timesteps.tolist()
            boundaries = sorted(
                {s for s, _ in p_segs} | {s for s, _ in n_segs}
            )
            enc_cache: dict = {}
            for s in boundaries:
                eff = s - start_off
                if eff <= 0 or eff >= len(ts_all):
                    continue  # the base conditioning already covers it
                p = [txt for st, txt in p_segs if st <= s][-1]
                n = [txt for st, txt in n_segs if st <= s][-1]
                if (p, n) not in enc_cache:
                    enc_cache[(p, n)] = self._build_ctx(p, n, req, b)
                ctx_b, y_b, ws_b = enc_cache[(p, n)]
                seg_tensors.append((ts_all[eff], ctx_b, y_b, ws_b))

        def _ctx_y_for(t: float):
            sel = None
            for s in seg_tensors:
                if t <= s[0] + 1e-6:
                    sel = s
            if sel is None:  # before the first switch: base conditioning
                return ctx, y, and_ws
            return sel[1], sel[2], sel[3]

        denoiser = self._denoiser

        # controlnet units: [(module, hint, scale, t_begin, t_end)] with a
        # per-unit guidance window in t-space (sdwui guidance_start/end)
        units = list(req.control_units or [])
        if req.control_model and req.control_image is not None:
            units.insert(0, {
                "image": req.control_image, "model": req.control_model,
                "scale": req.control_scale,
            })
        cn_units = []
        if units:
            from ..models.registry import load_controlnet

            ts_full = sched.timesteps.tolist()
            for u in units:
                img = u.get("image")
                name = u.get("model") or "controlnet-sd15"
                if img is None:
                    continue
                mod = load_controlnet(
                    name, device=self.device, dtype=self.dtype
                )
                if self.device.type == "cuda":
                    mod.to(memory_format=torch.channels_last)
                hint_u = (
                    img.permute(0, 3, 1, 2).float() / 255.0
                ).to(self.device, self.dtype)
                if hint_u.shape[0] == 1 and b > 1:
                    hint_u = hint_u.expand(b, -1, -1, -1)
                g0 = u.get("guidance_start", 0.0)
                g0 = 0.0 if g0 is None else float(g0)
                g1 = u.get("guidance_end", 1.0)
                g1 = 1.0 if g1 is None else float(g1)
                i0 = max(0, min(len(ts_full) - 1, int(g0 * len(ts_full))))
                i1 = int(g1 * len(ts_full))
                t_begin = ts_full[i0] if g0 > 0 else float("inf")
                t_end = (
                    ts_full[i1] if i1 < len(ts_full) else float("-inf")
                )
                cn_units.append((
                    mod, hint_u, float(u.get("scale", 1.0)), t_begin, t_end,
                ))
        controlnet = bool(cn_units)

        def _control_residuals(xk, ts, c_ctx, t):
            """Sum the active units' residuals at timestep t."""
            if not torch.is_tensor(c_ctx):  # RegionalContext: CN is a
                c_ctx = c_ctx.plain         # base-UNet copy, feed base rows
            total = None
            for mod, hint_u, scale, t_begin, t_end in cn_units:
                if not (t <= t_begin + 1e-6 and t > t_end):
                    continue
                k1 = xk.shape[0] // hint_u.shape[0]
                hk = torch.cat([hint_u] * k1, dim=0)
                xc = xk
                if xk.shape[1] != mod.cfg.in_channels:
                    # inpaint models carry extra conditioning channels the
                    # controlnet (a 4ch base-UNet copy) doesn't take
                    xc = xk[:, : mod.cfg.in_channels]
                ctrl = mod(xc, hk, ts, c_ctx, scale)
                if total is None:
                    total = ctrl
                else:
                    total = {
                        "down": [
                            a + bb for a, bb in zip(
                                total["down"], ctrl["down"]
                            )
                        ],
                        "mid": total["mid"] + ctrl["mid"],
                    }
            return total

        def _to_eps(out, x_scaled, t, pred_type):
            """v-prediction -> epsilon on the sampler's scaled input:
            denoised = c_skip*x + c_out*v gives eps = c_in*(sigma*x_scaled + v)
            (c_skip = 1/(s^2+1), c_out = -s/sqrt(s^2+1), c_in = 1/sqrt(s^2+1))."""
            if pred_type != "v":
                return out
            from .. import ops as _ops

            s = sigma_for_t(t)
            c_in = 1.0 / math.sqrt(s * s + 1.0)
            return _ops.lincomb(x_scaled, out, s * c_in, c_in)

        pred_type = self.model.prediction_type

        s_min_uncond = float(req.s_min_uncond)

        # 9-channel inpainting models (sd15-inpaint): the UNet input is
        # [z_t, mask(1ch), masked-init latents]; with no mask the
        # convention is mask=1 everywhere + zero masked-image (runwayml)
        is_inpaint_model = unet.cfg.in_channels == 2 * lat_c + 1
        _inpaint_cache: dict = {}

        def _inpaint_extra(h: int, w: int) -> torch.Tensor:
            key = (h, w)
            if key not in _inpaint_cache:
                if (req.init_latents is not None
                        and req.mask_image is not None):
                    lm = self._latent_mask(req.mask_image, b, lat_h, lat_w)
                    init_l = req.init_latents.to(self.device).float()
                    masked_z = init_l * (1.0 - (lm > 0.5).float())
                    ex = torch.cat([lm, masked_z], dim=1)
                    if (h, w) != (lat_h, lat_w):
                        ex = torch.nn.functional.interpolate(
                            ex, size=(h, w), mode="nearest"
                        )
                else:
                    ex = torch.cat(
                        [
                            torch.ones(b, 1, h, w, device=self.device),
                            torch.zeros(
                                b, lat_c, h, w, device=self.device
                            ),
                        ],
                        dim=1,
                    )
                _inpaint_cache[key] = ex.to(self.dtype).contiguous()
            return _inpaint_cache[key]

        # sdwui's s_min_uncond skips the uncond eval only on every OTHER
        # model eval (CFGDenoiser.forward: `self.step % 2`); the counter
        # ticks once per model_fn call like sdwui's self.step
        smu_step = [0]

        def model_fn(x_in: torch.Tensor, t: float) -> torch.Tensor:
            c_ctx, c_y, ws = _ctx_y_for(t)
            nb = x_in.shape[0]
            smu_step[0] += 1
            if (
                s_min_uncond > 0
                and len(ws) == 1
                and (smu_step[0] - 1) % 2
                and sigma_for_t(t) < s_min_uncond
            ):
                # sdwui s_min_uncond: at low noise the uncond eval barely
                # changes the output — skip it (halves the model cost)
                ts1 = torch.full(
                    (nb,), t, device=self.device, dtype=torch.float32,
                )
                yc = c_y[:nb] if c_y is not None else None
                x1 = x_in
                if is_inpaint_model:
                    x1 = torch.cat(
                        [x_in, _inpaint_extra(x_in.shape[2], x_in.shape[3])],
                        dim=1,
                    )
                ctrl1 = (
                    _control_residuals(x1, ts1, c_ctx[:nb], t)
                    if controlnet else None
                )
                if ctrl1 is not None:
                    out1 = unet(x1, ts1, c_ctx[:nb], y=yc, control=ctrl1)
                else:
                    out1 = denoiser(x1, ts1, c_ctx[:nb], yc)
                return _to_eps(out1, x_in, t, pred_type)
            k1 = len(ws) + 1  # k AND-conds + 1 uncond
            ts = torch.full(
                (nb * k1,), t, device=self.device, dtype=torch.float32,
            )
            xk = torch.cat([x_in] * k1, dim=0)
            x_rep = xk  # pre-concat latents: _to_eps needs lat_c channels
            if is_inpaint_model:
                ex = _inpaint_extra(x_in.shape[2], x_in.shape[3])
                xk = torch.cat([xk, torch.cat([ex] * k1, dim=0)], dim=1)
            ctrl = (
                _control_residuals(xk, ts, c_ctx, t) if controlnet else None
            )
            if ctrl is not None:
                eps = unet(xk, ts, c_ctx, y=c_y, control=ctrl)
            else:
                eps = denoiser(xk, ts, c_ctx, c_y)
            eps = _to_eps(eps, x_rep, t, pred_type)
            parts = eps.chunk(k1, dim=0)
            from .. import ops as _ops

            if k1 == 2:
                w = cfg * ws[0]
                return _ops.lincomb(parts[0], parts[1], w, 1.0 - w)
            # composable diffusion: u + cfg * sum_i w_i (c_i - u)
            out = parts[-1]
            for w, ec in zip(ws, parts[:-1]):
                d = _ops.lincomb(ec, parts[-1], 1.0, -1.0)
                out = _ops.lincomb(out, d, 1.0, cfg * w)
            return out

        if cfg == 1.0 and not seg_tensors and and_ws == [1.0] and not cn_units:

            def model_fn(x_in: torch.Tensor, t: float) -> torch.Tensor:  # noqa: F811
                nb = x_in.shape[0]
                ts = torch.full(
                    (nb,), t, device=self.device, dtype=torch.float32,
                )
                yc = y[:nb] if y is not None else None
                x1 = x_in
                if is_inpaint_model:
                    x1 = torch.cat(
                        [x_in, _inpaint_extra(x_in.shape[2], x_in.shape[3])],
                        dim=1,
                    )
                return _to_eps(
                    denoiser(x1, ts, ctx[:nb], yc), x_in, t, pred_type
                )

        # whole-step hipGraph (round-1 verdict #5): the CFG duplication,
        # the UNet forward and the guidance combine replay as ONE graph per
        # step; only the sampler's handful of elementwise kernels (and the
        # deterministic per-image CPU noise) stay eager - a design choice
        # that preserves the seed plan's bit-exact shard determinism.
        if (
            self.device.type == "cuda"
            and graphs_enabled()
            and os.environ.get("SDWD_WHOLESTEP", "1") not in ("", "0")
            and cfg != 1.0
            and not seg_tensors
            and and_ws == [1.0]
            and not cn_units
            and s_min_uncond == 0
            and pred_type != "v"
            and torch.is_tensor(ctx)  # RegionalContext runs eager
            and (
                not is_inpaint_model
                or (req.mask_image is None and req.init_latents is None)
            )
        ):
            # persistent per-(cfg, inpaint) graph: the core closure only
            # captures request-INDEPENDENT state (the raw UNet call, self's
            # canonical no-mask inpaint conditioning), so one capture
            # serves every later generation at the same shape
            wkey = (float(cfg), bool(is_inpaint_model))
            _graphed = self._wholestep_cache.get(wkey)
            if _graphed is None:
                _raw_fn = self._denoiser.fn  # eager UNet (no graph nesting)
                from .. import ops as _gops

                _lat_c = lat_c
                _wcfg = float(cfg)
                _inp = bool(is_inpaint_model)

                def _canon_extra(h, w):
                    key = (h, w)
                    t = self._canon_extra_cache.get(key)
                    if t is None:
                        t = torch.cat(
                            [
                                torch.ones(1, 1, h, w, device=self.device),
                                torch.zeros(
                                    1, _lat_c, h, w, device=self.device
                                ),
                            ],
                            dim=1,
                        ).to(self.dtype).contiguous()
                        self._canon_extra_cache[key] = t
                    return t

                def _cfg_core(x_in, t0d, ctx2, y2):
                    nb = x_in.shape[0]
                    ts = t0d.expand(nb * 2)
                    xk = torch.cat([x_in, x_in], dim=0)
                    if _inp:
                        ex = _canon_extra(x_in.shape[2], x_in.shape[3])
                        xk = torch.cat(
                            [xk, ex.expand(nb * 2, -1, -1, -1)], dim=1
                        )
                    eps = _raw_fn(xk, ts, ctx2, y2)
                    e_c, e_u = eps.chunk(2, dim=0)
                    return _gops.lincomb(e_c, e_u, _wcfg, 1.0 - _wcfg)

                _graphed = GraphedModelFn(_cfg_core, self.device)
                self._wholestep_cache[wkey] = _graphed

            _wrap = _graphed

            def model_fn(x_in: torch.Tensor, t: float) -> torch.Tensor:  # noqa: F811
                ctx2, y2, _ws = _ctx_y_for(t)
                _wrap.bind(ctx2, y2)
                return _wrap(x_in, t)

        if req.refiner_model:
            # two-model handoff: t descends through the schedule, so the
            # refiner takes over once t falls to the switch timestep
            base_fn = model_fn
            refiner = load_model(
                req.refiner_model, device=self.device, dtype=self.dtype
            )
            if self.device.type == "cuda":
                refiner.unet.to(memory_format=torch.channels_last)
            r_c1, r_u1, r_p1 = self.encode_prompts(
                [req.prompt], [req.negative_prompt], req.clip_skip,
                bundle=refiner,
            )
            r_ctx = torch.cat(
                [r_c1.expand(b, -1, -1), r_u1.expand(b, -1, -1)], dim=0
            )
            r_y = None
            if refiner.is_sdxl and r_p1 is not None:
                r_y = torch.cat(
                    [
                        self._sdxl_vector(
                            req, r_p1[0].expand(b, -1), bundle=refiner
                        ),
                        self._sdxl_vector(
                            req, r_p1[1].expand(b, -1), bundle=refiner,
                            aesthetic=2.5,
                        ),
                    ]
                )
            ts_list = sched.timesteps.tolist()
            si = max(0, int(round(len(ts_list) * req.refiner_switch_at)))
            # switch_at is the fraction of steps the BASE model runs;
            # 1.0 means the refiner never fires
            t_switch = ts_list[si] if si < len(ts_list) else float("-inf")
            r_unet = refiner.unet

            def model_fn(x_in: torch.Tensor, t: float) -> torch.Tensor:  # noqa: F811
                if t > t_switch:
                    return base_fn(x_in, t)
                ts = torch.full(
                    (x_in.shape[0] * 2,), t, device=self.device,
                    dtype=torch.float32,
                )
                x2 = torch.cat([x_in, x_in], dim=0)
                eps = r_unet(x2, ts, r_ctx, y=r_y)
                eps = _to_eps(eps, x2, t, refiner.prediction_type)
                eps_c, eps_u = eps.chunk(2, dim=0)
                from .. import ops as _ops

                return _ops.lincomb(eps_c, eps_u, cfg, 1.0 - cfg)

        was_interrupted = False

        def _interrupt() -> bool:
            nonlocal was_interrupted
            if interrupt is not None and interrupt():
                was_interrupted = True
                return True
            return False

        post_step = None
        if req.mask_image is not None and req.init_latents is not None:
            # latent-space inpainting: outside the mask the trajectory is
            # pinned to the init re-noised at the current sigma (sdwui
            # masked-img2img semantics; deterministic via the init noise)
            lat_mask = self._latent_mask(req.mask_image, b, lat_h, lat_w)
            init_lat = req.init_latents.to(self.device).float()
            noise_f32 = noise.float()

            if req.soft_inpainting:
                # Soft inpainting: the repaint weight is the soft mask
                # value raised to a sigma-dependent exponent
                #   w(m, s) = contrast(m ** (bias * (1 + 2*s/s_max)))
                # so early steps (high sigma) strongly favour the original
                # (context forms first) and the final blend keeps a soft
                # m**bias edge instead of a hard 0/1 seam. The magnitude of
                # the blended latent is then pulled toward the weighted mix
                # of the input magnitudes (preservation strength), which
                # counters the detail washout a plain lerp of latents
                # causes in the transition band.
                sig0 = max(float(sched.sigmas[0]), 1e-6)
                bias = max(float(req.si_schedule_bias), 1e-3)
                boost = max(float(req.si_transition_contrast_boost), 1.0)
                keep_s = min(max(float(req.si_preservation_strength), 0.0), 1.0)

                def post_step(xc, sigma_next):
                    keep = init_lat + noise_f32 * sigma_next
                    xf = xc.float()
                    e = bias * (1.0 + 2.0 * float(sigma_next) / sig0)
                    w = lat_mask.pow(e)
                    w = _contrast(w, boost)
                    mixed = w * xf + (1.0 - w) * keep
                    if keep_s > 0.0:
                        cur = mixed.norm(p=2, dim=1, keepdim=True)
                        want = w * xf.norm(p=2, dim=1, keepdim=True) + (
                            1.0 - w
                        ) * keep.norm(p=2, dim=1, keepdim=True)
                        ratio = (want / cur.clamp_min(1e-5)).clamp(0.25, 4.0)
                        mixed = mixed * (
                            1.0 + keep_s * (ratio - 1.0)
                        )
                    return mixed.to(xc.dtype)

            else:
                def post_step(xc, sigma_next):
                    keep = init_lat + noise_f32 * sigma_next
                    return (
                        lat_mask * xc.float() + (1.0 - lat_mask) * keep
                    ).to(xc.dtype)

        orig_post = post_step

        def post_step(xc, sigma_next):  # noqa: F811 - compose preview store
            if orig_post is not None:
                xc = orig_post(xc, sigma_next)
            self._last_preview = xc
            return xc

        x = sampler.sample(
            model_fn, x, noise_fn=noise_fn, callback=step_callback,
            interrupt=_interrupt, post_step=post_step,
        )

        # hires fix: latent-upscale the base result and run a second,
        # strength-limited denoise pass at the scaled resolution
        # (ref eta_hr, worker.py:205-228 predicts exactly this shape).
        if (
            hires_active(
                req.enable_hr, req.hr_scale, req.hr_resize_x, req.hr_resize_y
            )
            and not was_interrupted
        ):
            hr_steps = req.hr_steps or req.steps
            if req.hr_prompt or req.hr_negative_prompt:
                # the hires pass denoises under its own conditioning;
                # model_fn reads ctx/y/and_ws through the closure
                from ..models.lora import parse_prompt_loras

                hp = req.hr_prompt or req.prompt
                hp, _ = parse_prompt_loras(hp)  # tags stripped, set unchanged
                hn = req.hr_negative_prompt or req.negative_prompt
                ctx, y, and_ws = self._build_ctx(hp, hn, req, b)
                seg_tensors.clear()
            uh, uw, crop_y, crop_x = hr_target_resolution(
                req.width, req.height, req.hr_scale,
                req.hr_resize_x, req.hr_resize_y, f,
            )
            hr_size = (uh, uw)
            up_key = (req.hr_upscaler or "nearest").lower()
            if up_key in _HR_MODES:
                x = _upscale_latent(
                    x.float(), req.hr_scale, req.hr_upscaler, size=hr_size
                ).to(self.dtype)
            else:
                # sdwui non-"Latent" upscaler: pixel space round trip
                x = self._pixel_hires(x, req, hr_size, f)
            if crop_y or crop_x:
                # sdwui "truncate": both hr_resize dims set -> the
                # aspect-covering upscale is center-cropped to the target
                # BEFORE the second pass denoises at that size
                x = x[
                    :, :,
                    crop_y // 2 : x.shape[2] - (crop_y - crop_y // 2),
                    crop_x // 2 : x.shape[3] - (crop_x - crop_x // 2),
                ]
            hr_sampler = req.hr_sampler_name or req.sampler_name
            hsched = schedule_for(hr_sampler, hr_steps, req.scheduler)
            start = max(
                0, hr_steps - max(1, int(hr_steps * req.denoising_strength))
            )
            hsched = type(hsched)(
                sigmas=hsched.sigmas[start:], timesteps=hsched.timesteps[start:]
            )
            hsampler = build_sampler(hr_sampler, hsched)
            _apply_sampler_params(hsampler, req)
            hh, hw = x.shape[2], x.shape[3]
            hr_noise = torch.stack(
                [
                    _image_noise(
                        (int(req.seeds[i]) ^ 0x12E50),
                        subseeds[i],
                        0.0,
                        (lat_c, hh, hw),
                    )
                    for i in range(b)
                ]
            ).to(self.device, self.dtype)
            s0 = float(hsched.sigmas[0])
            x = (x.float() + hr_noise.float() * s0).to(self.dtype)

            def hr_noise_fn() -> torch.Tensor:
                n = torch.stack(
                    [
                        torch.randn(
                            (lat_c, hh, hw), generator=g, dtype=torch.float32
                        )
                        for g in gens
                    ]
                )
                return n.to(self.device, self.dtype)

            def hr_post(xc, sigma_next):
                # live preview only (the inpainting re-imposition, if any,
                # is keyed to the base-resolution mask)
                self._last_preview = xc
                return xc

            x = hsampler.sample(
                model_fn, x, noise_fn=hr_noise_fn, callback=step_callback,
                interrupt=_interrupt, post_step=hr_post,
            )

        if decode:
            # chunked decode: bounds decoder activation memory at large
            # batches (and sidesteps a torch NHWC-upsample grid-size limit
            # seen at batch 64 x 512^2); scale the chunk down with pixels
            px = x.shape[2] * x.shape[3] * f * f  # actual final latent dims
            default = max(1, int(16 * (512 * 512) / max(px, 1)))
            chunk = int(os.environ.get("SDWD_DECODE_CHUNK", str(default)))
            outs = []
            for i in range(0, x.shape[0], chunk):
                pixels = self.model.vae.decode(x[i : i + chunk])
                outs.append(
                    ((pixels.float() + 1.0) * 127.5)
                    .clamp(0, 255)
                    .to(torch.uint8)
                    .permute(0, 2, 3, 1)
                    .cpu()
                )
            images = torch.cat(outs) if len(outs) > 1 else outs[0]
            if (
                req.soft_inpainting
                and req.mask_image is not None
                and req.init_latents is not None
                and not hires_active(
                    req.enable_hr, req.hr_scale,
                    req.hr_resize_x, req.hr_resize_y,
                )
            ):
                images = self._soft_composite(images, req)
        else:
            images = x.cpu()

        elapsed = time.perf_counter() - t0
        # sdwui infotext format: optional fields appear only when active,
        # so downstream "send to txt2img"-style parsers reproduce the run
        extra = ""
        if req.clip_skip > 1:
            extra += f", Clip skip: {req.clip_skip}"
        if req.init_latents is not None:
            extra += f", Denoising strength: {req.denoising_strength}"
        if req.soft_inpainting and req.mask_image is not None:
            extra += (
                f", Soft inpainting: True"
                f", Schedule bias: {req.si_schedule_bias}"
                f", Preservation strength: {req.si_preservation_strength}"
            )
        if regional:
            extra += (
                f", RP Active: True, RP Matrix submode: {req.regional_mode}"
                f", RP Ratios: \"{req.regional_ratios}\""
                f", RP Base Ratios: {req.regional_base_ratio}"
            )
        if req.subseed_strength > 0 and req.subseeds:
            extra += f", Variation seed strength: {req.subseed_strength}"
        if req.eta >= 0:
            extra += f", Eta: {req.eta}"
        if hires_active(
            req.enable_hr, req.hr_scale, req.hr_resize_x, req.hr_resize_y
        ):
            if req.hr_resize_x > 0 or req.hr_resize_y > 0:
                # sdwui "resize to" mode reports the target, not a scale
                extra += (
                    f", Hires resize: {req.hr_resize_x}x{req.hr_resize_y}"
                )
            else:
                extra += f", Hires upscale: {req.hr_scale}"
            extra += (
                f", Hires steps: {req.hr_steps or req.steps}"
                f", Hires upscaler: {req.hr_upscaler}"
            )
        if req.refiner_model:
            extra += (
                f", Refiner: {req.refiner_model}"
                f", Refiner switch at: {req.refiner_switch_at}"
            )
        if req.tiling:
            extra += ", Tiling: True"
        prompt_of = (
            (lambda i: req.prompts[i]) if per_image else (lambda i: req.prompt)
        )
        infotexts = [
            f"{prompt_of(i)}\nNegative prompt: {req.negative_prompt}\n"
            f"Steps: {req.steps}, Sampler: {req.sampler_name}, "
            f"CFG scale: {req.cfg_scale}, Seed: {req.seeds[i]}, "
            f"Size: {req.width}x{req.height}, Model: {self.model.name}"
            + (
                f", Variation seed: {subseeds[i]}"
                if req.subseed_strength > 0 and subseeds[i] != -1 else ""
            )
            + extra
            for i in range(b)
        ]
        return PipelineResult(
            images=images,
            seeds=list(req.seeds),
            subseeds=list(subseeds),
            infotexts=infotexts,
            elapsed=elapsed,
            interrupted=was_interrupted,
        )

    # approximate latent->RGB projection (the well-known SD 4ch linear
    # map; sdwui's "cheap" live-preview mode)
    _L2RGB = torch.tensor([
        [0.298, 0.207, 0.208],
        [0.187, 0.286, 0.173],
        [-0.158, 0.189, 0.264],
        [-0.184, -0.271, -0.473],
    ])

    @torch.no_grad()
    def preview_image(self) -> Optional[torch.Tensor]:
        """Cheap preview of the generation in flight: [h,w,3] uint8 at
        latent resolution (no VAE decode), or None."""
        lat = self._last_preview
        if lat is None:
            return None
        l4 = lat[0, :4].float().cpu()  # first image, first 4 channels
        rgb = torch.einsum("chw,cr->rhw", l4, self._L2RGB)
        rgb = ((rgb / 3.0 + 0.5) * 255.0).clamp(0, 255)
        return rgb.permute(1, 2, 0).to(torch.uint8)

    @torch.no_grad()
    def encode_image(
        self, images: torch.Tensor, seeds: Optional[List[int]] = None
    ) -> torch.Tensor:
        """[B,H,W,3] uint8 -> latents [B,4,h,w] (img2img entry).

        The encoder's sampling noise is seeded per image for determinism.
        """
        x = (
            images.permute(0, 3, 1, 2).float() / 127.5 - 1.0
        ).to(self.device, self.dtype)
        if seeds:
            # PER-IMAGE encoder forwards: batched conv numerics depend on
            # the batch size (CPU blocking; library GEMM selection), so a
            # shard of 2 would encode differently from the same images
            # inside a batch of 4 and break the C22 exactness contract
            # (N-GPU gallery == 1-GPU batch). Encoding
            # image-by-image makes the latents shard-placement-invariant
            # by construction; the encoder runs once per job and each
            # image still fills the GPU (M = H*W rows at the top level).
            # Sampling noise is seeded per image (device-independent).
            lats = []
            for i, sd in enumerate(seeds):
                moments = self.model.vae.encoder(x[i : i + 1])
                mean, logvar = moments.chunk(2, dim=1)
                std = torch.exp(0.5 * logvar.float().clamp(-30, 20))
                noise = torch.randn(
                    mean.shape[1:],
                    generator=torch.Generator("cpu").manual_seed(
                        (int(sd) ^ 0xE4C0DE) & 0xFFFFFFFF
                    ),
                    dtype=torch.float32,
                )[None].to(self.device)
                lats.append(
                    (mean.float() + std * noise)
                    * self.model.vae.cfg.scale_factor
                )
            return torch.cat(lats).to(self.dtype)
        return self.model.vae.encode(x)
