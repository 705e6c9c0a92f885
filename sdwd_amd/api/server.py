"""sdapi/v1-compatible HTTP surface.

The reference *consumed* these routes on remote webui instances
(txt2img, img2img, options, memory, sd-models, interrupt,
refresh-checkpoints, server-restart, progress); this engine *serves* them, so
existing sdwui API clients can point at this node and drive the whole-node
sharded engine. Plus a /sdwd/status route exposing the reference's Status-tab
data (worker states, speeds, log ring buffer — ref ui.py:230-249).
"""
from __future__ import annotations

import base64
import json
import os
import threading
import time
from typing import Any, Dict, List, Optional

import torch
from fastapi import FastAPI, HTTPException
from pydantic import BaseModel, Field

from ..config.models import SettingsModel
from ..core import State
from ..models.registry import available_models
from ..parallel import GenerationRequest, LocalEngine
from ..pipeline.samplers import sampler_names
from ..utils import get_logger, ring_buffer
from ..utils.images import decode_png, encode_png

log = get_logger("api")


class Txt2ImgRequest(BaseModel):
    prompt: str = ""
    negative_prompt: str = ""
    seed: int = -1
    subseed: int = -1
    subseed_strength: float = 0.0
    seed_resize_from_w: int = Field(default=0, ge=0)
    seed_resize_from_h: int = Field(default=0, ge=0)
    steps: int = 20
    cfg_scale: float = 7.0
    width: int = 512
    height: int = 512
    batch_size: int = Field(default=1, ge=1, le=4096)
    n_iter: int = Field(default=1, ge=1, le=64)
    sampler_name: str = "Euler a"
    sampler_index: Optional[str] = None  # legacy alias
    scheduler: str = "Automatic"
    styles: List[str] = Field(default_factory=list)
    tiling: bool = False
    clip_skip: int = 1
    s_churn: float = 0.0
    s_tmin: float = 0.0
    s_tmax: float = 0.0
    s_noise: float = 1.0
    s_min_uncond: float = 0.0
    eta: float = -1.0  # sdwui Eta for ancestral/DDIM; -1/unset = default
    # hires fix (sdwui fields)
    enable_hr: bool = False
    hr_scale: float = 2.0
    hr_upscaler: str = "Latent"
    hr_second_pass_steps: int = 0
    hr_prompt: str = ""
    hr_negative_prompt: str = ""
    hr_resize_x: int = Field(default=0, ge=0)
    hr_resize_y: int = Field(default=0, ge=0)
    hr_sampler_name: str = ""
    denoising_strength: float = 0.75
    # alwayson scripts (ref C17/C18: the reference forwarded these; we
    # execute the controlnet unit natively, other scripts are ignored
    # with a warning as the reference's compat filter did)
    alwayson_scripts: Dict[str, Any] = Field(default_factory=dict)
    # sdwui refiner fields
    refiner_checkpoint: str = ""
    refiner_switch_at: float = 0.8
    # sdwui per-request overrides (sd_model_checkpoint,
    # CLIP_stop_at_last_layers are honored; the rest are ignored)
    override_settings: Dict[str, Any] = Field(default_factory=dict)
    # sdwui selectable script: X/Y/Z plot, Prompt matrix and
    # Prompts-from-file run natively (parallel/xyz.py,
    # parallel/builtin_scripts.py); any other name is rejected loudly
    script_name: Optional[str] = ""  # sdwui clients send null for "none"
    script_args: List[Any] = Field(default_factory=list)
    send_images: bool = True   # omit base64 images from the response
    save_images: bool = False  # persist PNGs server-side (SDWD_OUTPUT_DIR)


class Img2ImgRequest(Txt2ImgRequest):
    init_images: List[str] = Field(default_factory=list)  # base64 PNG
    denoising_strength: float = 0.75
    mask: Optional[str] = None  # base64 PNG, white = repaint
    inpaint_full_res: bool = False  # sdwui "Inpaint area: Only masked"
    inpaint_full_res_padding: int = 32
    mask_blur: int = 4
    inpainting_fill: int = 1  # 0 fill, 1 original, 2 latent noise, 3 nothing
    inpainting_mask_invert: int = 0  # 1 = inpaint NOT-masked region
    color_correction: bool = False
    # sdwui resize_mode: 0 just resize, 1 crop and resize, 2 resize and fill
    resize_mode: int = 0


class OptionsRequest(BaseModel):
    sd_model_checkpoint: Optional[str] = None
    sd_vae: Optional[str] = None

    model_config = {"extra": "allow"}


class ServerState:
    def __init__(self, engine: LocalEngine):
        self.engine = engine
        self.current_model = engine.model_name
        self.busy = False
        self.progress = 0.0
        self.started_at = 0.0
        self.lock = threading.Lock()
        # global defaults settable via POST /sdapi/v1/options
        self.default_clip_skip = 1
        self.ensd = 0


def _b64_png(img: torch.Tensor) -> str:
    return base64.b64encode(encode_png(img)).decode()


def _decode_b64_png(data: str) -> torch.Tensor:
    if "," in data[:64]:  # data URI prefix
        data = data.split(",", 1)[1]
    return decode_png(base64.b64decode(data))


def _resize_init(img: torch.Tensor, w: int, h: int, mode: int) -> torch.Tensor:
    """sdwui img2img resize modes for an init image of any size:
    0 = just resize (stretch), 1 = crop and resize (cover, center-crop),
    2 = resize and fill (contain, edges replicated)."""
    import torch.nn.functional as F

    ih, iw = img.shape[0], img.shape[1]
    if (ih, iw) == (h, w):
        return img
    x = img.float().permute(2, 0, 1)[None]

    def interp(t, size):
        return F.interpolate(t, size=size, mode="bilinear", antialias=True)

    if mode == 1:  # cover then center-crop
        scale = max(w / iw, h / ih)
        rh, rw = int(round(ih * scale)), int(round(iw * scale))
        x = interp(x, (rh, rw))
        top, left = (rh - h) // 2, (rw - w) // 2
        x = x[:, :, top : top + h, left : left + w]
    elif mode == 2:  # contain then replicate-pad
        scale = min(w / iw, h / ih)
        rh, rw = int(round(ih * scale)), int(round(iw * scale))
        x = interp(x, (rh, rw))
        pt = (h - rh) // 2
        pl = (w - rw) // 2
        x = F.pad(
            x, (pl, w - rw - pl, pt, h - rh - pt), mode="replicate"
        )
    else:  # just resize
        x = interp(x, (h, w))
    return x[0].permute(1, 2, 0).clamp(0, 255).to(torch.uint8)


# alwayson scripts executed natively (C18): ControlNet (full guidance
# windows) and Dynamic Prompts (wildcard/variant expansion). Everything
# else is logged and skipped, like the reference's compat filter when a
# remote lacked the script - docs/usage.md documents the boundary.
_NATIVE_ALWAYSON = (
    "controlnet", "dynamic prompts", "soft inpainting", "regional prompter",
)


def _parse_regional_prompter(alwayson: Dict[str, Any]) -> Dict[str, Any]:
    """Regional Prompter matrix-mode payload, dict form:
    {"regional prompter": {"args": [{"active": true, "mode": "Columns",
    "ratios": "1,1", "base_ratio": 0.2, "use_base": true}]}}. Only the
    matrix (columns/rows) mode executes natively; mask/prompt modes are
    logged and skipped like any unsupported script."""
    for name, body in (alwayson or {}).items():
        if name.lower().replace("-", " ") != "regional prompter":
            continue
        args = (body or {}).get("args", [])
        d = args[0] if args and isinstance(args[0], dict) else {}
        if not d or not d.get("active", True):
            return {}
        mode = str(d.get("mode", d.get("Matrix mode", "columns"))).lower()
        if mode not in ("columns", "rows"):
            log.warning("regional prompter: mode %r unsupported", mode)
            return {}
        out: Dict[str, Any] = {
            "regional_mode": mode,
            "regional_ratios": str(d.get("ratios", "1,1")),
        }
        base = d.get("base_ratio", 0.2)
        if not d.get("use_base", True):
            base = 0.0
        out["regional_base_ratio"] = float(base)
        return out
    return {}


def _dynamic_prompts_enabled(alwayson: Dict[str, Any]) -> bool:
    """sd-dynamic-prompts payload: {"dynamic prompts": {"args": [True, ...]}}
    (first arg = enabled, like the extension's process() signature)."""
    for name, body in (alwayson or {}).items():
        if name.lower().replace("-", " ") == "dynamic prompts":
            args = (body or {}).get("args", [True])
            return bool(args[0]) if args else True
    return False


def _expand_dynamic(gen) -> None:
    """Per-image seeded expansion (determinism contract C22: image k
    depends on seed_k only, never on shard placement)."""
    from ..core.seeds import fix_seed
    from ..pipeline.wildcards import expand, has_dynamic_syntax

    if not has_dynamic_syntax(gen.prompt):
        return
    gen.seed = fix_seed(gen.seed)
    gen.prompts = [
        expand(gen.prompt, gen.seed + i) for i in range(gen.batch_size)
    ]


_SI_KEYS = (  # host UI labels -> GenerationRequest fields, in API arg order
    ("Schedule bias", "si_schedule_bias"),
    ("Preservation strength", "si_preservation_strength"),
    ("Transition contrast boost", "si_transition_contrast_boost"),
    ("Mask influence", "si_mask_influence"),
    ("Difference threshold", "si_difference_threshold"),
    ("Difference contrast", "si_difference_contrast"),
)


def _parse_soft_inpainting(alwayson: Dict[str, Any]) -> Dict[str, float]:
    """Host built-in soft-inpainting payload:
    {"soft inpainting": {"args": [{"Soft inpainting": true,
    "Schedule bias": 1, ...}]}} — a single dict arg keyed by UI label
    (the host's API convention), or the same values positionally with a
    leading enabled flag. Returns GenerationRequest field overrides."""
    for name, body in (alwayson or {}).items():
        if name.lower().replace("-", " ") != "soft inpainting":
            continue
        args = (body or {}).get("args", [])
        out: Dict[str, float] = {"soft_inpainting": True}
        if args and isinstance(args[0], dict):
            d = args[0]
            if not d.get("Soft inpainting", True):
                return {}
            for label, field in _SI_KEYS:
                if label in d:
                    out[field] = float(d[label])
        elif args:
            if not args[0]:
                return {}
            for (_, field), v in zip(_SI_KEYS, args[1:]):
                out[field] = float(v)
        return out
    return {}


def _parse_controlnet(alwayson: Dict[str, Any]):
    """sdwui controlnet payload: {"controlnet": {"args": [unit, ...]}}
    (ref control_net.py:20-79 packed this; every unit is executed natively
    with its weight and guidance window). Unknown scripts are logged and
    skipped (ref C18 compat filter)."""
    units = []
    for name, body in (alwayson or {}).items():
        if name.lower() not in _NATIVE_ALWAYSON:
            log.warning("ignoring unsupported alwayson script '%s'", name)
            continue
        if name.lower() != "controlnet":
            continue
        for unit in (body or {}).get("args", []):
            img_b64 = unit.get("input_image") or unit.get("image")
            if not img_b64:
                continue
            units.append({
                "image": _decode_b64_png(img_b64)[None],
                "model": unit.get("model", "controlnet-sd15"),
                "scale": float(unit.get("weight", 1.0)),
                "guidance_start": float(unit.get("guidance_start", 0.0)),
                "guidance_end": float(unit.get("guidance_end", 1.0)),
            })
    return units


def create_app(engine: Optional[LocalEngine] = None,
               model: str = "sd15") -> FastAPI:
    if engine is None:
        engine = LocalEngine(model=model)
    state = ServerState(engine)
    app = FastAPI(title="sdwd_amd", version="0.1.0")
    app.state.engine = engine

    # HTTP basic auth, sdwui --api-auth parity (the reference's worker
    # records carried user:pass credentials, pmodels.py:12-34):
    # SDWD_API_AUTH="user:pass" protects every route.
    auth_cfg = os.environ.get("SDWD_API_AUTH", "")
    if auth_cfg and ":" in auth_cfg:
        import hmac
        import base64 as _b64

        expected = auth_cfg.encode()

        @app.middleware("http")
        async def _basic_auth(request, call_next):
            from fastapi.responses import JSONResponse

            hdr = request.headers.get("authorization", "")
            ok = False
            if hdr.lower().startswith("basic "):
                try:
                    got = _b64.b64decode(hdr[6:])
                    ok = hmac.compare_digest(got, expected)
                except Exception:
                    ok = False
            if not ok:
                return JSONResponse(
                    {"detail": "unauthorized"},
                    status_code=401,
                    headers={"WWW-Authenticate": "Basic"},
                )
            return await call_next(request)

    def run_generation(gen: GenerationRequest, send_images: bool = True,
                       save_images: bool = False,
                       vae_override: str = "") -> Dict[str, Any]:
        # one generation at a time (the reference serialized on the host's
        # queue_lock, world.py:244,273); concurrent requests queue here
        with state.lock:
            state.busy = True
            state.started_at = time.time()
            prev_vae = getattr(engine, "vae_name", "auto")
            swap = bool(vae_override) and vae_override != prev_vae
            try:
                if swap:  # sdwui override_settings.sd_vae: per-request
                    engine.set_vae(vae_override)
                result = engine.generate(gen)
                state.current_model = engine.model_name
            finally:
                if swap:
                    engine.set_vae(prev_vae)
                state.busy = False
        if save_images:
            from ..utils.images import save_png

            outdir = os.environ.get("SDWD_OUTPUT_DIR", "outputs")
            os.makedirs(outdir, exist_ok=True)
            stamp = time.strftime("%Y%m%d-%H%M%S")
            for i in range(result.images.shape[0]):
                save_png(
                    result.images[i],
                    os.path.join(
                        outdir, f"{stamp}-{result.seeds[i]}-{i:03d}.png"
                    ),
                    parameters=result.infotexts[i],
                )
        images = []
        if send_images:
            images = [
                base64.b64encode(
                    encode_png(result.images[i], result.infotexts[i])
                ).decode()
                for i in range(result.images.shape[0])
            ]
            if result.grid is not None:
                images.insert(0, _b64_png(result.grid))
        info = {
            "all_seeds": result.seeds,
            "all_subseeds": [-1] * len(result.seeds),
            "all_prompts": (
                list(gen.prompts)
                if gen.prompts and len(gen.prompts) == len(result.seeds)
                else [gen.prompt] * len(result.seeds)
            ),
            "all_negative_prompts": [gen.negative_prompt] * len(result.seeds),
            "infotexts": result.infotexts,
            "job_summary": result.job_summary,
            "elapsed": result.elapsed,
        }
        params = {
            k: (None if isinstance(v, torch.Tensor) else v)
            for k, v in gen.__dict__.items()
        }
        if params.get("control_units"):
            params["control_units"] = [
                {kk: (None if isinstance(vv, torch.Tensor) else vv)
                 for kk, vv in u.items()}
                for u in params["control_units"]
            ]
        return {
            "images": images,
            "parameters": params,
            "info": json.dumps(info),
        }

    _XYZ_NAMES = {"x/y/z plot", "xyz plot", "xyz grid", "x/y/z", "xyz"}
    _MATRIX_NAMES = {"prompt matrix"}
    _FROMFILE_NAMES = {"prompts from file or textbox", "prompts from file"}

    def _script_response(out: Dict[str, Any], gen: GenerationRequest,
                         req: Txt2ImgRequest,
                         extra_info: Optional[Dict[str, Any]] = None
                         ) -> Dict[str, Any]:
        """Shared response assembly for selectable-script runs."""
        images = []
        if req.send_images:
            if out.get("grid") is not None:
                images.append(_b64_png(out["grid"]))
            for sg in out.get("sub_grids", []):
                images.append(_b64_png(sg))
            images.extend(
                base64.b64encode(encode_png(img, info)).decode()
                for img, info in zip(out["images"], out["infotexts"])
            )
        info = {
            "all_seeds": out["seeds"],
            "all_subseeds": [-1] * len(out["seeds"]),
            "all_prompts": out.get(
                "prompts", [gen.prompt] * len(out["seeds"])
            ),
            "all_negative_prompts": [gen.negative_prompt] * len(out["seeds"]),
            "infotexts": out["infotexts"],
            "interrupted": out["interrupted"],
        }
        if extra_info:
            info.update(extra_info)
        return {
            "images": images,
            "parameters": {"script_name": req.script_name},
            "info": json.dumps(info),
        }

    def _run_prompt_matrix_script(gen: GenerationRequest,
                                  req: Txt2ImgRequest) -> Dict[str, Any]:
        """sdwui Prompt matrix script_args:
        [put_at_start, different_seeds, prompt_type, variations_delimiter,
        margin_size] (trailing args optional)."""
        from ..parallel.builtin_scripts import run_prompt_matrix

        a = req.script_args
        g = lambda i, d=None: a[i] if i < len(a) else d  # noqa: E731
        try:
            with state.lock:
                state.busy = True
                state.started_at = time.time()
                try:
                    out = run_prompt_matrix(
                        engine, gen,
                        put_at_start=bool(g(0, False)),
                        different_seeds=bool(g(1, False)),
                        prompt_type=str(g(2, "positive") or "positive").lower(),
                        variations_delimiter=str(g(3, "comma") or "comma"),
                    )
                finally:
                    state.busy = False
        except ValueError as exc:
            raise HTTPException(422, str(exc))
        return _script_response(out, gen, req)

    def _run_prompts_file_script(gen: GenerationRequest,
                                 req: Txt2ImgRequest) -> Dict[str, Any]:
        """sdwui Prompts-from-file script_args:
        [checkbox_iterate, checkbox_iterate_batches, prompt_txt]; the
        text may also arrive as the sole string argument."""
        from ..parallel.builtin_scripts import run_prompts_from_file

        a = req.script_args
        text = next((v for v in a if isinstance(v, str) and v.strip()), "")
        bools = [v for v in a if isinstance(v, bool)]
        it = bools[0] if bools else False
        itb = bools[1] if len(bools) > 1 else False
        try:
            with state.lock:
                state.busy = True
                state.started_at = time.time()
                try:
                    out = run_prompts_from_file(
                        engine, gen, text,
                        checkbox_iterate=it, checkbox_iterate_batches=itb,
                    )
                finally:
                    state.busy = False
        except ValueError as exc:
            raise HTTPException(422, str(exc))
        return _script_response(out, gen, req)

    def _parse_xyz_args(args: List[Any]) -> Dict[str, Any]:
        """Positional sdwui xyz_grid script_args, both layouts:

        new (dropdown) — [x_type, x_values, x_values_dropdown, y_type,
            y_values, y_values_dropdown, z_type, z_values,
            z_values_dropdown, draw_legend, no_fixed_seeds,
            include_lone_images, include_sub_grids, margin, csv_mode];
        old — [x_type, x_values, y_type, y_values, z_type, z_values,
            draw_legend, include_lone_images, include_sub_grids,
            no_fixed_seeds].

        Axis types are int indices into our published AXIS_OPTIONS or
        axis names (version-proof). A dropdown list, when non-empty,
        takes precedence over the paired values string.
        """
        if not args:
            raise HTTPException(
                422, "X/Y/Z plot needs script_args (x_type, x_values, ...)"
            )

        def _vals(raw, drop):
            return drop if isinstance(drop, (list, tuple)) and drop else raw

        dropdown = len(args) >= 9 and (
            isinstance(args[2], (list, tuple))
            or isinstance(args[5], (list, tuple))
            or isinstance(args[8], (list, tuple))
        )
        # heuristic: 15-arg payloads are always the dropdown layout
        dropdown = dropdown or len(args) >= 13
        g = lambda i, d=None: args[i] if i < len(args) else d  # noqa: E731
        if dropdown:
            out = dict(
                x_axis=g(0, 0), x_values=_vals(g(1, ""), g(2)),
                y_axis=g(3, 0), y_values=_vals(g(4, ""), g(5)),
                z_axis=g(6, 0), z_values=_vals(g(7, ""), g(8)),
                no_fixed_seeds=bool(g(10, False)),
                include_lone_images=bool(g(11, False)),
                include_sub_grids=bool(g(12, False)),
            )
        else:
            out = dict(
                x_axis=g(0, 0), x_values=g(1, ""),
                y_axis=g(2, 0), y_values=g(3, ""),
                z_axis=g(4, 0), z_values=g(5, ""),
                include_lone_images=bool(g(7, False)),
                include_sub_grids=bool(g(8, False)),
                no_fixed_seeds=bool(g(9, False)),
            )
        return out

    def run_script(gen: GenerationRequest, req: Txt2ImgRequest
                   ) -> Dict[str, Any]:
        """Dispatch a sdwui selectable script (script_name). Only the
        natively-implemented X/Y/Z plot runs; anything else is rejected
        loudly (same degradation the reference applied when a remote
        lacked a script, distributed.py:199-234)."""
        from ..parallel.xyz import run_xyz

        name = req.script_name.strip().lower()
        if name in _MATRIX_NAMES:
            return _run_prompt_matrix_script(gen, req)
        if name in _FROMFILE_NAMES:
            return _run_prompts_file_script(gen, req)
        if name not in _XYZ_NAMES:
            raise HTTPException(
                422,
                f"selectable script {req.script_name!r} is not available "
                "natively; supported: X/Y/Z plot, Prompt matrix, "
                "Prompts from file or textbox",
            )
        kw = _parse_xyz_args(req.script_args)
        # pre-validate axes whose bad values would otherwise produce a
        # misleading grid (sampler falls back to Euler a by design) or
        # fail mid-sweep (unknown checkpoint)
        from ..parallel.xyz import parse_axis_values, resolve_axis
        from ..pipeline import sampler_names

        for axis_key, vals_key in (
            ("x_axis", "x_values"), ("y_axis", "y_values"),
            ("z_axis", "z_values"),
        ):
            try:
                ax = resolve_axis(kw[axis_key])
                vals = parse_axis_values(ax.kind, kw[vals_key])
            except ValueError as exc:
                raise HTTPException(422, str(exc))
            if ax.name == "Checkpoint name":
                for v in vals:
                    if v not in available_models():
                        raise HTTPException(422, f"unknown model {v!r}")
            if ax.name == "Sampler":
                known = {s.lower() for s in sampler_names()}
                for v in vals:
                    if str(v).lower() not in known:
                        raise HTTPException(422, f"unknown sampler {v!r}")
        try:
            with state.lock:
                state.busy = True
                state.started_at = time.time()
                try:
                    out = run_xyz(engine, gen, **kw)
                finally:
                    state.busy = False
        except ValueError as exc:
            raise HTTPException(422, str(exc))
        return _script_response(
            out, gen, req, extra_info={"xyz_plot": out["labels"]}
        )

    @app.get("/")
    def index():
        """Built-in control surface (ref C19)."""
        from fastapi.responses import HTMLResponse

        from .ui import PAGE

        return HTMLResponse(PAGE)

    def _overrides(req: Txt2ImgRequest):
        """Per-request override_settings (sdwui convention)."""
        ov = req.override_settings or {}
        model = str(ov.get("sd_model_checkpoint") or "")
        if model and model not in available_models():
            raise HTTPException(404, f"unknown model {model}")
        # per-request override > request field > global option default
        clip_skip = int(
            ov.get("CLIP_stop_at_last_layers")
            or (req.clip_skip if req.clip_skip > 1 else 0)
            or state.default_clip_skip
        )
        ensd = int(ov.get("eta_noise_seed_delta") or state.ensd)
        if req.styles:
            from ..pipeline.styles import all_styles, apply_styles, refresh_styles

            if not all_styles():
                refresh_styles()
            req.prompt, req.negative_prompt = apply_styles(
                req.prompt, req.negative_prompt, req.styles
            )
        if req.refiner_checkpoint and (
            req.refiner_checkpoint not in available_models()
        ):
            raise HTTPException(404, f"unknown refiner {req.refiner_checkpoint}")
        vae = str(ov.get("sd_vae") or "")
        if vae in ("Automatic", "None"):
            vae = "auto"
        if vae and vae != "auto":
            from ..models.registry import available_vaes, refresh_vae_files

            if vae not in available_vaes():
                refresh_vae_files()
            if vae not in available_vaes():
                raise HTTPException(404, f"unknown VAE {vae}")
        return model, clip_skip, ensd, vae

    @app.post("/sdapi/v1/txt2img")
    def txt2img(req: Txt2ImgRequest):
        control_units = _parse_controlnet(req.alwayson_scripts)
        model, clip_skip, ensd, vae_ov = _overrides(req)
        gen = GenerationRequest(
            prompt=req.prompt,
            negative_prompt=req.negative_prompt,
            batch_size=req.batch_size * max(1, req.n_iter),
            width=req.width,
            height=req.height,
            steps=req.steps,
            cfg_scale=req.cfg_scale,
            sampler_name=req.sampler_name or req.sampler_index or "Euler a",
            scheduler=req.scheduler,
            seed=req.seed,
            subseed=req.subseed,
            subseed_strength=req.subseed_strength,
            seed_resize_from_w=req.seed_resize_from_w,
            seed_resize_from_h=req.seed_resize_from_h,
            eta_noise_seed_delta=ensd,
            tiling=req.tiling,
            s_churn=req.s_churn,
            s_tmin=req.s_tmin,
            s_tmax=req.s_tmax,
            s_noise=req.s_noise,
            s_min_uncond=req.s_min_uncond,
            eta=req.eta,
            enable_hr=req.enable_hr,
            hr_scale=req.hr_scale,
            hr_steps=req.hr_second_pass_steps,
            hr_upscaler=req.hr_upscaler,
            hr_prompt=req.hr_prompt,
            hr_negative_prompt=req.hr_negative_prompt,
            hr_resize_x=req.hr_resize_x,
            hr_resize_y=req.hr_resize_y,
            hr_sampler_name=req.hr_sampler_name,
            denoising_strength=req.denoising_strength,
            clip_skip=clip_skip,
            control_units=control_units,
            model=model,
            refiner_model=req.refiner_checkpoint,
            refiner_switch_at=req.refiner_switch_at,
            **_parse_regional_prompter(req.alwayson_scripts),
        )
        if _dynamic_prompts_enabled(req.alwayson_scripts):
            _expand_dynamic(gen)
        if req.script_name:
            return run_script(gen, req)
        return run_generation(gen, req.send_images, req.save_images,
                              vae_override=vae_ov)

    @app.post("/sdapi/v1/img2img")
    def img2img(req: Img2ImgRequest):
        if not req.init_images:
            raise HTTPException(422, "init_images required")
        model, clip_skip, ensd, vae_ov = _overrides(req)
        try:
            inits = torch.stack(
                [
                    _resize_init(
                        _decode_b64_png(d), req.width, req.height,
                        req.resize_mode,
                    )
                    for d in req.init_images
                ]
            )
        except Exception as exc:
            raise HTTPException(422, f"bad init image: {exc}")
        mask_image = None
        if req.mask:
            try:
                m = _decode_b64_png(req.mask)
                mask_image = _resize_init(
                    m, req.width, req.height, 0
                ).float().mean(-1).to(torch.uint8)
            except Exception as exc:
                raise HTTPException(422, f"bad mask image: {exc}")
        gen = GenerationRequest(
            prompt=req.prompt,
            negative_prompt=req.negative_prompt,
            batch_size=req.batch_size * max(1, req.n_iter),
            width=req.width,
            height=req.height,
            steps=req.steps,
            cfg_scale=req.cfg_scale,
            sampler_name=req.sampler_name or req.sampler_index or "Euler a",
            scheduler=req.scheduler,
            seed=req.seed,
            subseed=req.subseed,
            subseed_strength=req.subseed_strength,
            seed_resize_from_w=req.seed_resize_from_w,
            seed_resize_from_h=req.seed_resize_from_h,
            eta_noise_seed_delta=ensd,
            tiling=req.tiling,
            s_churn=req.s_churn,
            s_tmin=req.s_tmin,
            s_tmax=req.s_tmax,
            s_noise=req.s_noise,
            s_min_uncond=req.s_min_uncond,
            eta=req.eta,
            init_images=inits,
            denoising_strength=req.denoising_strength,
            mask_image=mask_image,
            clip_skip=clip_skip,
            model=model,
            inpaint_full_res=req.inpaint_full_res,
            inpaint_full_res_padding=req.inpaint_full_res_padding,
            mask_blur=req.mask_blur,
            inpainting_fill=req.inpainting_fill,
            inpainting_mask_invert=req.inpainting_mask_invert,
            color_correction=req.color_correction,
            **_parse_soft_inpainting(req.alwayson_scripts),
            **_parse_regional_prompter(req.alwayson_scripts),
        )
        if req.script_name:
            return run_script(gen, req)
        return run_generation(gen, req.send_images, req.save_images,
                              vae_override=vae_ov)

    @app.post("/sdapi/v1/options")
    def set_options(req: OptionsRequest):
        extras = req.model_extra or {}
        if "CLIP_stop_at_last_layers" in extras:
            state.default_clip_skip = max(
                1, int(extras["CLIP_stop_at_last_layers"] or 1)
            )
        if "eta_noise_seed_delta" in extras:
            state.ensd = int(extras["eta_noise_seed_delta"] or 0)
        if req.sd_model_checkpoint and req.sd_model_checkpoint != state.current_model:
            name = req.sd_model_checkpoint
            if name not in available_models():
                raise HTTPException(404, f"unknown model {name}")
            engine.set_model(name)
            state.current_model = name
        if req.sd_vae:
            from ..models.registry import available_vaes, refresh_vae_files

            vae = req.sd_vae
            if vae in ("Automatic", "None"):  # sdwui dropdown spellings
                vae = "auto"
            if vae not in available_vaes():
                refresh_vae_files()
            if vae not in available_vaes():
                raise HTTPException(404, f"unknown VAE {vae}")
            try:
                engine.set_vae(vae)
            except (KeyError, ValueError) as exc:
                raise HTTPException(422, str(exc))
        return {}

    @app.get("/sdapi/v1/options")
    def get_options():
        return {
            "sd_model_checkpoint": state.current_model,
            "sd_vae": getattr(engine, "vae_name", "auto"),
            "CLIP_stop_at_last_layers": state.default_clip_skip,
            "eta_noise_seed_delta": state.ensd,
        }

    @app.get("/sdapi/v1/sd-models")
    def sd_models():
        return [
            {"title": m, "model_name": m, "filename": f"{m}.safetensors"}
            for m in available_models()
        ]

    @app.get("/sdapi/v1/samplers")
    def samplers():
        return [{"name": s, "aliases": [s]} for s in sampler_names()]

    @app.post("/sdapi/v1/extra-single-image")
    def extra_single_image(body: Dict[str, Any]):
        """Pixel-space upscale (sdwui extras tab surface; the model-free
        upscalers: Nearest / Bilinear / Bicubic / Lanczos-approx)."""
        try:
            img = _decode_b64_png(body.get("image", ""))
        except Exception as exc:
            raise HTTPException(422, f"bad image: {exc}")
        scale = float(body.get("upscaling_resize", 2.0))
        w2 = int(body.get("upscaling_resize_w", 0))
        h2 = int(body.get("upscaling_resize_h", 0))
        mode = str(body.get("upscaler_1", "Bilinear")).lower()
        mode_map = {
            "nearest": ("nearest", False),
            "none": ("nearest", False),
            "bilinear": ("bilinear", True),
            "bicubic": ("bicubic", True),
            "lanczos": ("bicubic", True),  # closest torch kernel
        }
        m, aa = mode_map.get(mode, ("bilinear", True))
        x = img.permute(2, 0, 1)[None].float()
        kwargs: Dict[str, Any] = {"mode": m}
        if aa:
            kwargs["antialias"] = True
        if w2 and h2:
            out = torch.nn.functional.interpolate(x, size=(h2, w2), **kwargs)
        else:
            out = torch.nn.functional.interpolate(
                x, scale_factor=scale, **kwargs
            )
        out8 = out.clamp(0, 255).to(torch.uint8)[0].permute(1, 2, 0)
        return {"image": _b64_png(out8), "html_info": ""}

    @app.post("/sdapi/v1/extra-batch-images")
    def extra_batch_images(body: Dict[str, Any]):
        """Batch version of extra-single-image (same pixel upscalers)."""
        images = body.get("imageList") or body.get("image_list") or []
        out = []
        for entry in images:
            data = entry.get("data") if isinstance(entry, dict) else entry
            sub = dict(body)
            sub["image"] = data
            out.append(extra_single_image(sub)["image"])
        return {"images": out, "html_info": ""}

    @app.get("/sdapi/v1/prompt-styles")
    def prompt_styles():
        from ..pipeline.styles import all_styles, refresh_styles

        if not all_styles():
            refresh_styles()
        return [
            {"name": n, "prompt": p, "negative_prompt": np}
            for n, (p, np) in sorted(all_styles().items())
        ]

    # -- static enumerations sdwui GUIs query at startup ---------------------
    @app.get("/sdapi/v1/upscalers")
    def upscalers():
        return [
            {"name": n, "model_name": None, "model_path": None, "scale": 4}
            for n in ("None", "Nearest", "Bilinear", "Bicubic", "Lanczos")
        ]

    @app.get("/sdapi/v1/latent-upscale-modes")
    def latent_upscale_modes():
        return [
            {"name": n}
            for n in ("Latent", "Latent (bilinear)", "Latent (bicubic)",
                      "Latent (bilinear antialiased)",
                      "Latent (bicubic antialiased)")
        ]

    @app.get("/sdapi/v1/face-restorers")
    def face_restorers():
        return [{"name": "None", "cmd_dir": None}]

    @app.get("/sdapi/v1/embeddings")
    def embeddings():
        from ..models.embeddings import loaded

        return {
            "loaded": {
                n: {"step": None, "sd_checkpoint": None,
                    "sd_checkpoint_name": None, "shape": None, "vectors": k}
                for n, k in loaded().items()
            },
            "skipped": {},
        }

    @app.post("/sdapi/v1/refresh-embeddings")
    def refresh_embeddings():
        from ..models.embeddings import refresh_embedding_files

        return {"found": refresh_embedding_files()}

    @app.get("/sdapi/v1/hypernetworks")
    def hypernetworks():
        return []

    @app.get("/sdapi/v1/scripts")
    def scripts():
        # the natively-executed alwayson set (C18) — the reference PROBED
        # each remote's script list through exactly this surface
        # (worker.py:375-404), so report what this engine runs in-process
        selectable = [
            "x/y/z plot", "prompt matrix", "prompts from file or textbox",
        ]
        return {
            "txt2img": list(_NATIVE_ALWAYSON) + selectable,
            "img2img": list(_NATIVE_ALWAYSON) + selectable,
        }

    @app.get("/sdapi/v1/script-info")
    def script_info():
        from ..parallel.xyz import AXIS_OPTIONS

        xyz_args = [
            {"label": f"{i}={o.name}", "kind": o.kind}
            for i, o in enumerate(AXIS_OPTIONS)
        ]
        return [
            {"name": n, "is_alwayson": True, "is_img2img": True,
             "args": []}
            for n in _NATIVE_ALWAYSON
        ] + [
            {"name": n, "is_alwayson": True, "is_img2img": False,
             "args": []}
            for n in _NATIVE_ALWAYSON
        ] + [
            {"name": "x/y/z plot", "is_alwayson": False,
             "is_img2img": im, "args": xyz_args}
            for im in (True, False)
        ] + [
            {"name": n, "is_alwayson": False, "is_img2img": im, "args": []}
            for n in ("prompt matrix", "prompts from file or textbox")
            for im in (True, False)
        ]

    @app.get("/sdapi/v1/cmd-flags")
    def cmd_flags():
        return {
            "api": True,
            "listen": None,
            "port": None,
            "device_id": None,
            "medvram": False,
            "lowvram": False,
            "xformers": False,
        }

    @app.get("/sdapi/v1/sd-vae")
    def sd_vae():
        from ..models.registry import available_vaes, refresh_vae_files

        refresh_vae_files()
        return [
            {"model_name": n,
             "filename": "" if n == "auto" else f"{n}.safetensors"}
            for n in available_vaes()
        ]

    @app.post("/sdapi/v1/refresh-vae")
    def refresh_vae():
        from ..models.registry import refresh_vae_files

        return {"found": refresh_vae_files()}

    @app.get("/sdapi/v1/schedulers")
    def schedulers():
        from ..pipeline.schedule import scheduler_names

        return [
            {"name": n.lower().replace(" ", "_"), "label": n}
            for n in scheduler_names()
        ]

    @app.get("/sdapi/v1/memory")
    def memory():
        cuda: Dict[str, Any] = {}
        if torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info(0)
            cuda = {"system": {"free": free, "used": total - free,
                               "total": total}}
        return {"ram": {}, "cuda": cuda}

    @app.post("/sdapi/v1/interrupt")
    def interrupt():
        engine.interrupt()
        return {}

    @app.post("/sdapi/v1/skip")
    def skip():
        """sdwui 'skip' ends the current iteration and continues the job;
        with n_iter folded into one sharded batch the nearest semantic is
        interrupting the in-flight generation (partial gallery returns)."""
        engine.interrupt()
        return {}

    @app.post("/sdapi/v1/unload-checkpoint")
    def unload_checkpoint():
        """Move every rank's weights to host memory and release VRAM
        (sdwui parity; reload-checkpoint restores)."""
        for pipe in engine.pipes.values():
            pipe.model.to("cpu")
            pipe._denoiser.cache.clear()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
        return {}

    @app.post("/sdapi/v1/reload-checkpoint")
    def reload_checkpoint():
        for pipe in engine.pipes.values():
            pipe.model.to(pipe.device, pipe.dtype)
        return {}

    @app.post("/sdapi/v1/interrogate")
    def interrogate():
        # no caption model ships in this offline environment
        raise HTTPException(
            501, "interrogate requires a caption model (not available)"
        )

    @app.post("/sdapi/v1/refresh-checkpoints")
    def refresh_checkpoints():
        from ..models.registry import refresh_checkpoint_files

        return {"found": refresh_checkpoint_files()}

    @app.post("/sdapi/v1/refresh-loras")
    def refresh_loras():
        from ..models.lora import refresh_lora_files

        return {"found": sorted(refresh_lora_files())}

    @app.get("/sdapi/v1/loras")
    def loras():
        """Available LoRA adapters: files from SDWD_LORA_DIR plus any
        registered in the engine's manager (sdwui /sdapi/v1/loras shape)."""
        from ..models.lora import lora_files, refresh_lora_files

        if not lora_files():
            refresh_lora_files()
        files = lora_files()
        names = set(files)
        for pipe in engine.pipes.values():
            names.update(pipe.lora._registry)
        return [
            {"name": n, "alias": n, "path": files.get(n, "")}
            for n in sorted(names)
        ]

    @app.post("/sdapi/v1/png-info")
    def png_info(body: Dict[str, Any]):
        """Read back the 'parameters' infotext from a generated PNG
        (sdwui /sdapi/v1/png-info)."""
        from ..utils.images import png_parameters

        data = body.get("image", "")
        try:
            raw = base64.b64decode(data.split(",", 1)[-1])
            info = png_parameters(raw)
        except Exception as exc:
            raise HTTPException(422, f"bad png: {exc}")
        from ..utils.images import parse_infotext

        parsed = parse_infotext(info) if info else {}
        return {"info": info or "", "items": {}, "parameters": parsed}

    @app.get("/sdapi/v1/progress")
    def progress():
        frac = engine.progress() if state.busy else 1.0
        current = None
        if state.busy:
            # cheap live preview (sdwui current_image): approximate
            # latent->RGB of rank 0's in-flight latents, no VAE decode
            try:
                pv = engine.pipes["gpu0"].preview_image()
                if pv is not None:
                    current = _b64_png(pv)
            except Exception:  # preview must never break progress polling
                current = None
        return {
            "progress": frac,
            "eta_relative": 0.0,
            "state": {
                "job": "generate" if state.busy else "",
                "interrupted": engine.world.interrupted.is_set(),
            },
            "current_image": current,
        }

    @app.post("/sdapi/v1/server-restart")
    def server_restart():
        """Soft restart (ref worker.py:690-717): rebuild every rank's
        pipeline, clear caches, reset worker states."""
        from ..models.registry import clear_cache
        from ..pipeline import StableDiffusionPipeline

        log.info("soft restart: rebuilding pipelines")
        engine.world.clear_interrupt()
        clear_cache()
        for label, pipe in list(engine.pipes.items()):
            engine.pipes[label] = StableDiffusionPipeline(
                state.current_model, device=pipe.device, dtype=pipe.dtype
            )
        for w in engine.world.workers:
            if w.state is not State.DISABLED:
                w.sm.force(State.IDLE)
        return {}

    @app.get("/sdwd/status")
    def status():
        """The reference's Status tab as JSON (ref ui.py:230-249)."""
        return {
            "workers": [
                {
                    "label": w.label,
                    "device": w.device,
                    "state": w.state.value,
                    "avg_ipm": w.eta.avg_ipm,
                    "mpe": w.eta.mpe(),
                    "pixel_cap": w.pixel_cap,
                    "model_override": getattr(w, "model_override", "")
                    or "",
                }
                for w in engine.world.workers
            ],
            "speed_summary": engine.world.speed_summary(),
            "log": ring_buffer(),
            "model": state.current_model,
            "vae": getattr(engine, "vae_name", "auto"),
            "busy": state.busy,
        }

    @app.post("/sdwd/sync-script")
    def sync_script():
        """Run the user's sync* script (ref C21, ui.py:26-55)."""
        from ..utils.sync_scripts import run_sync_script

        rc, output = run_sync_script()
        return {"returncode": rc, "output": output[-4000:]}

    @app.post("/sdwd/benchmark")
    def benchmark(rebenchmark: bool = True):
        speeds = engine.benchmark(rebenchmark=rebenchmark)
        engine.world.save()  # ref world.py:278 persisted post-benchmark
        return {"speeds": speeds}

    @app.get("/sdwd/settings")
    def get_settings():
        """Current engine settings (the reference's Settings tab state,
        ref ui.py:363-391)."""
        return engine.world.settings.model_dump()

    @app.post("/sdwd/settings")
    def set_settings(body: Dict[str, Any]):
        """Update engine settings live and persist them (ref ui.py Settings
        tab + update_world handlers). Unknown keys are rejected."""
        current = engine.world.settings.model_dump()
        unknown = set(body) - set(current)
        if unknown:
            raise HTTPException(422, f"unknown settings: {sorted(unknown)}")
        current.update(body)
        try:
            engine.world.settings = SettingsModel.model_validate(current)
        except Exception as exc:
            raise HTTPException(422, str(exc))
        engine.world.save()
        return engine.world.settings.model_dump()

    @app.post("/sdwd/reset-mpe")
    def reset_mpe():
        """Clear every rank's ETA error-correction history (ref 2.0.0
        debug utility)."""
        for w in engine.world.workers:
            w.eta.reset_errors()
        engine.world.save()
        return {}

    @app.post("/sdwd/release-lock")
    def release_lock():
        """Debug escape hatch (ref ui.py:69-70 force-released the host's
        queue lock): clears a stuck busy flag and replaces the lock."""
        state.busy = False
        state.lock = threading.Lock()
        log.warning("generation lock force-released")
        return {}

    @app.post("/sdwd/restart-workers")
    def restart_workers():
        """The Utils tab's 'restart all remotes' (ref 2.1.0,
        worker.py:690-717 POSTed /server-restart to every remote): for
        in-node ranks a restart is a state reset — interrupt flags clear,
        UNAVAILABLE/INTERRUPTED ranks return to IDLE and rejoin
        scheduling (their next liveness probe re-verifies them)."""
        from ..core.state import State as WState

        eng = app.state.engine
        eng.world.interrupted.clear()
        restarted = []
        for w in eng.world.workers:
            if w.state in (WState.UNAVAILABLE, WState.INTERRUPTED,
                           WState.WORKING):
                w.set_state(WState.IDLE)
                restarted.append(w.label)
        log.info("restart-workers: %s", restarted or "none needed")
        return {"restarted": restarted}

    @app.get("/sdwd/benchmark-payload")
    def get_benchmark_payload():
        """The canonical benchmark payload (ref shared.py:63-77 constants,
        editable via config like the reference's Benchmark_Payload)."""
        return engine.world.benchmark_payload.model_dump()

    @app.post("/sdwd/benchmark-payload")
    def set_benchmark_payload(body: Dict[str, Any]):
        from ..config.models import BenchmarkPayload

        current = engine.world.benchmark_payload.model_dump()
        unknown = set(body) - set(current)
        if unknown:
            raise HTTPException(422, f"unknown fields: {sorted(unknown)}")
        current.update(body)
        try:
            engine.world.benchmark_payload = BenchmarkPayload.model_validate(
                current
            )
        except Exception as exc:
            raise HTTPException(422, str(exc))
        engine.world.save()
        return engine.world.benchmark_payload.model_dump()

    @app.post("/sdwd/worker/{label}/config")
    def worker_config(label: str, body: Dict[str, Any]):
        """Per-worker pixel cap + checkpoint override (ref ui.py:161-171,
        313-319: both editable in the reference's Worker Config tab)."""
        w = engine.world.get_worker(label)
        if w is None:
            raise HTTPException(404, label)
        unknown = set(body) - {"pixel_cap", "model_override"}
        if unknown:
            raise HTTPException(422, f"unknown fields: {sorted(unknown)}")
        if "pixel_cap" in body:
            cap = body["pixel_cap"]
            if cap in (None, "", 0, "0"):
                w.pixel_cap = 0
            else:
                try:
                    cap = int(cap)
                except (TypeError, ValueError):
                    raise HTTPException(422, "pixel_cap must be an int")
                if cap < 0:
                    raise HTTPException(422, "pixel_cap must be >= 0")
                w.pixel_cap = cap
        if "model_override" in body:
            name = (body["model_override"] or "").strip()
            if name and name not in available_models():
                raise HTTPException(404, f"unknown model: {name}")
            w.model_override = name or None
        engine.world.save()
        return {
            "label": w.label,
            "pixel_cap": w.pixel_cap,
            "model_override": getattr(w, "model_override", "") or "",
        }

    @app.post("/sdwd/worker/{label}/enable")
    def enable_worker(label: str):
        w = engine.world.get_worker(label)
        if w is None:
            raise HTTPException(404, label)
        w.set_state(State.IDLE)
        return {}

    @app.post("/sdwd/worker/{label}/disable")
    def disable_worker(label: str):
        w = engine.world.get_worker(label)
        if w is None:
            raise HTTPException(404, label)
        w.set_state(State.DISABLED, strict=False)
        return {}

    return app


def install_signal_handlers(engine: LocalEngine) -> None:
    """Save config then chain the previous handler on SIGINT/SIGTERM
    (ref distributed.py:359-375)."""
    import signal

    previous = {}

    def handler(signum, frame):
        try:
            if engine.world.config_path:
                engine.world.save()
                log.info("config saved on signal %d", signum)
        finally:
            prev = previous.get(signum)
            if callable(prev):
                prev(signum, frame)
            else:
                raise SystemExit(0)

    for sig in (signal.SIGINT, signal.SIGTERM):
        previous[sig] = signal.getsignal(sig)
        signal.signal(sig, handler)


def main() -> None:  # pragma: no cover - manual entry
    import argparse

    import uvicorn

    from ..config import add_flags, default_config_path, export_env

    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=7860)
    ap.add_argument("--model", default="sd15")
    add_flags(ap)
    args = ap.parse_args()
    export_env(args)
    engine = LocalEngine(model=args.model)
    engine.world.config_path = args.sdwd_config or default_config_path()
    engine.world.load(engine.world.config_path)  # persisted speeds/settings
    if not engine.world.workers:  # fresh config: build from devices
        from ..core import World

        engine.world = World.from_devices(
            len(engine.devices), config_path=engine.world.config_path
        )
    install_signal_handlers(engine)
    uvicorn.run(create_app(engine=engine), host=args.host, port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
