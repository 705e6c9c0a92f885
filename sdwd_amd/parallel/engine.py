"""Generation engines: the reference's dispatch/gather path (C2/C9/C10/C20)
re-designed for one node.

Two engines share the scheduler (core.World) and pipeline:

* LocalEngine — one process drives N devices with one thread per rank
  (the reference's threading model, distributed.py:316-318), full recovery
  semantics: a failing rank's shard is requeued onto survivors
  (improving on ref worker.py:498-500 which dropped it).

* DistributedEngine — one process per GPU under torchrun, RCCL over xGMI:
  plan broadcast from rank 0, per-rank shard execution, padded all_gather
  of uint8 images, weight broadcast at load (C13), TCPStore out-of-band
  control plane for interrupt (C20) — collectives never carry control.
"""
from __future__ import annotations

import os
import threading
import time
from dataclasses import dataclass, field, replace as replace_dc
from typing import Dict, List, Optional

import torch

from ..core import GenRequest, Job, State, World, Worker
from ..pipeline import PipelineRequest, StableDiffusionPipeline
from ..pipeline.pipeline import hires_active, hr_target_resolution
from ..utils import get_logger
from ..utils.images import make_grid
from . import group as pg

log = get_logger("engine")


def _final_output_hw(gen) -> "tuple[int, int]":
    """Pixel (H, W) the pipeline actually decodes for this request — the
    latent floor math, including sdwui hires resize-to/truncate semantics
    (pipeline.hr_target_resolution). Placeholder/empty-shard tensors sized
    from this always match generating ranks' output shapes."""
    f = 8
    if hires_active(
        gen.enable_hr, gen.hr_scale, gen.hr_resize_x, gen.hr_resize_y
    ):
        uh, uw, ty, tx = hr_target_resolution(
            gen.width, gen.height, gen.hr_scale,
            gen.hr_resize_x, gen.hr_resize_y, f,
        )
        return (uh - ty) * f, (uw - tx) * f
    return (gen.height // f) * f, (gen.width // f) * f


@dataclass
class GenerationRequest:
    """User-facing request = scheduling fields + pipeline fields."""

    prompt: str = ""
    # optional distinct prompt per gallery image (len == batch_size);
    # sharded with the seeds so the N-GPU gallery stays 1-GPU-identical
    prompts: Optional[List[str]] = None
    negative_prompt: str = ""
    batch_size: int = 1
    width: int = 512
    height: int = 512
    steps: int = 20
    cfg_scale: float = 7.0
    sampler_name: str = "Euler a"
    scheduler: str = "Automatic"
    seed: int = -1
    subseed: int = -1
    subseed_strength: float = 0.0
    seed_resize_from_w: int = 0
    seed_resize_from_h: int = 0
    eta_noise_seed_delta: int = 0
    init_images: Optional[torch.Tensor] = None  # [B,H,W,3] uint8 (img2img)
    denoising_strength: float = 0.75
    mask_image: Optional[torch.Tensor] = None   # [H,W] uint8, 255=repaint
    enable_hr: bool = False
    hr_scale: float = 2.0
    hr_steps: int = 0
    hr_upscaler: str = "nearest"
    hr_prompt: str = ""
    hr_negative_prompt: str = ""
    hr_resize_x: int = 0
    hr_resize_y: int = 0
    hr_sampler_name: str = ""
    control_image: Optional[torch.Tensor] = None
    control_model: str = ""
    control_scale: float = 1.0
    control_units: Optional[List[dict]] = None  # multi-unit controlnet
    clip_skip: int = 1
    model: str = ""  # hot-swap to this checkpoint first ("" = keep current)
    refiner_model: str = ""  # two-model refiner handoff (sdwui refiner)
    refiner_switch_at: float = 0.8
    # sdwui "Inpaint area: Only masked": crop the padded mask bbox, run the
    # generation on the crop at full W x H, paste the result back
    inpaint_full_res: bool = False
    inpaint_full_res_padding: int = 32
    mask_blur: int = 0  # gaussian blur radius on the mask (sdwui mask_blur)
    tiling: bool = False
    # sdwui "Masked content": 0 fill (mean color), 1 original,
    # 2 latent noise, 3 latent nothing
    inpainting_fill: int = 1
    # sdwui "Inpaint masked / not masked" (1 inverts the mask)
    inpainting_mask_invert: int = 0
    # sampler stochasticity + uncond-skip perf knob (sdwui Sampler params)
    s_churn: float = 0.0
    s_tmin: float = 0.0
    s_tmax: float = 0.0
    s_noise: float = 1.0
    s_min_uncond: float = 0.0
    eta: float = -1.0  # sdwui Eta (ancestral/SDE); -1 = sampler default
    # sdwui img2img color correction: match output statistics to the init
    color_correction: bool = False
    # Regional Prompter matrix mode (see PipelineRequest for semantics)
    regional_mode: str = ""
    regional_ratios: str = "1,1"
    regional_base_ratio: float = 0.2
    # soft inpainting (host built-in; see PipelineRequest for semantics)
    soft_inpainting: bool = False
    si_schedule_bias: float = 1.0
    si_preservation_strength: float = 0.5
    si_transition_contrast_boost: float = 4.0
    si_mask_influence: float = 0.0
    si_difference_threshold: float = 0.5
    si_difference_contrast: float = 2.0

    def sched(self) -> GenRequest:
        hr_on = hires_active(
            self.enable_hr, self.hr_scale, self.hr_resize_x, self.hr_resize_y
        )
        eff_scale = 0.0
        if hr_on:
            # ETA model wants one linear scale factor; "resize to" mode
            # (hr_resize_x/y) implies it from the actual output pixels
            oh, ow = _final_output_hw(self)
            eff_scale = (
                (oh * ow)
                / max(1, ((self.height // 8) * 8) * ((self.width // 8) * 8))
            ) ** 0.5
        return GenRequest(
            task="img2img" if self.init_images is not None else "txt2img",
            batch_size=self.batch_size,
            width=self.width,
            height=self.height,
            steps=self.steps,
            sampler_name=self.sampler_name,
            seed=self.seed,
            subseed=self.subseed,
            subseed_strength=self.subseed_strength,
            hr_scale=eff_scale,
            hr_steps=(self.hr_steps or self.steps) if hr_on else 0,
        )


@dataclass
class GalleryResult:
    images: torch.Tensor  # [N,H,W,3] uint8 cpu, requested batch first
    seeds: List[int]
    infotexts: List[str]
    grid: Optional[torch.Tensor] = None
    elapsed: float = 0.0
    job_summary: List[str] = field(default_factory=list)
    interrupted: bool = False


def _job_pipeline_request(
    gen: GenerationRequest, job: Job, init_latents=None
) -> PipelineRequest:
    job_prompts = None
    if gen.prompts:
        n = len(gen.prompts)
        job_prompts = [
            gen.prompts[(job.gallery_offset + i) % n]
            for i in range(job.batch_size)
        ]
    return PipelineRequest(
        prompt=gen.prompt,
        prompts=job_prompts,
        negative_prompt=gen.negative_prompt,
        steps=job.step_override or gen.steps,
        width=gen.width,
        height=gen.height,
        cfg_scale=gen.cfg_scale,
        sampler_name=gen.sampler_name,
        scheduler=gen.scheduler,
        seeds=list(job.seeds),
        subseeds=list(job.subseeds),
        subseed_strength=gen.subseed_strength,
        seed_resize_from_w=gen.seed_resize_from_w,
        seed_resize_from_h=gen.seed_resize_from_h,
        eta_noise_seed_delta=gen.eta_noise_seed_delta,
        init_latents=init_latents,
        denoising_strength=gen.denoising_strength,
        mask_image=gen.mask_image,
        inpainting_fill=gen.inpainting_fill,
        enable_hr=gen.enable_hr,
        hr_scale=gen.hr_scale,
        hr_steps=gen.hr_steps,
        hr_upscaler=gen.hr_upscaler,
        hr_prompt=gen.hr_prompt,
        hr_negative_prompt=gen.hr_negative_prompt,
        hr_resize_x=gen.hr_resize_x,
        hr_resize_y=gen.hr_resize_y,
        hr_sampler_name=gen.hr_sampler_name,
        control_image=gen.control_image,
        control_model=gen.control_model,
        control_scale=gen.control_scale,
        control_units=gen.control_units,
        clip_skip=gen.clip_skip,
        refiner_model=gen.refiner_model,
        refiner_switch_at=gen.refiner_switch_at,
        tiling=gen.tiling,
        s_churn=gen.s_churn,
        s_tmin=gen.s_tmin,
        s_tmax=gen.s_tmax,
        s_noise=gen.s_noise,
        s_min_uncond=gen.s_min_uncond,
        eta=gen.eta,
        regional_mode=gen.regional_mode,
        regional_ratios=gen.regional_ratios,
        regional_base_ratio=gen.regional_base_ratio,
        soft_inpainting=gen.soft_inpainting,
        si_schedule_bias=gen.si_schedule_bias,
        si_preservation_strength=gen.si_preservation_strength,
        si_transition_contrast_boost=gen.si_transition_contrast_boost,
        si_mask_influence=gen.si_mask_influence,
        si_difference_threshold=gen.si_difference_threshold,
        si_difference_contrast=gen.si_difference_contrast,
    )


def _learn_pixel_cap(worker, gen: GenerationRequest, batch: int) -> bool:
    """After an out-of-memory failure, cap the rank's future jobs below the
    attempted size instead of sidelining a healthy GPU (an improvement on
    the reference, whose pixel caps were manual-only, ui.py:313-319).
    Returns True if a cap was applied."""
    attempted = batch * gen.width * gen.height
    if attempted <= 0:
        return False
    cap = int(attempted * 0.8)
    if worker.pixel_cap and worker.pixel_cap <= cap:
        cap = int(worker.pixel_cap * 0.8)  # cap again, smaller
    if cap < gen.width * gen.height:
        return False  # can't even fit one image; leave UNAVAILABLE
    worker.pixel_cap = cap
    log.warning(
        "%s: OOM at %d px; learned pixel cap %d", worker.label, attempted, cap
    )
    return True


def _blur_mask(mask: torch.Tensor, radius: int) -> torch.Tensor:
    """Separable gaussian blur on a [H,W] uint8 mask (sdwui mask_blur:
    softens the inpaint boundary in both the latent mask and the paste)."""
    import torch.nn.functional as F

    if radius <= 0:
        return mask
    sigma = max(0.5, radius / 2.0)
    k = int(2 * round(3 * sigma) + 1)
    xs = torch.arange(k, dtype=torch.float32) - (k - 1) / 2
    g = torch.exp(-0.5 * (xs / sigma) ** 2)
    g = (g / g.sum()).reshape(1, 1, 1, k)
    m = mask.float()[None, None]
    m = F.conv2d(F.pad(m, (k // 2, k // 2, 0, 0), mode="replicate"), g)
    m = F.conv2d(
        F.pad(m, (0, 0, k // 2, k // 2), mode="replicate"),
        g.reshape(1, 1, k, 1),
    )
    return m[0, 0].clamp(0, 255).to(torch.uint8)


def _preprocess_mask(gen: GenerationRequest) -> GenerationRequest:
    if gen.mask_image is None:
        return gen
    m = gen.mask_image
    if m.dim() == 3:
        m = m[0]
    if gen.inpainting_mask_invert:
        m = 255 - m
        gen = replace_dc(gen, mask_image=m)
    if gen.mask_blur > 0:
        gen = replace_dc(gen, mask_image=_blur_mask(m, gen.mask_blur))
        m = gen.mask_image
    if gen.inpainting_fill == 0 and gen.init_images is not None:
        # "fill": seed the masked region with the image's unmasked mean
        # color before encoding (approximates sdwui's blurred fill)
        sel = (m > 127)
        if sel.any() and (~sel).any():
            imgs = gen.init_images.float().clone()
            for i in range(imgs.shape[0]):
                mean = imgs[i][~sel].reshape(-1, 3).mean(dim=0)
                imgs[i][sel] = mean
            gen = replace_dc(
                gen, init_images=imgs.clamp(0, 255).to(torch.uint8)
            )
    return gen


def _crop_for_inpaint_full_res(gen: GenerationRequest):
    """-> (cropped_gen, paste_ctx) or (gen, None).

    Crops the padded mask bounding box (expanded toward the W:H aspect),
    resizes init+mask crops to the full generation size, and returns what
    _paste_inpaint_full_res needs to put results back (sdwui
    "Only masked" semantics)."""
    import torch.nn.functional as F

    if not (
        gen.inpaint_full_res
        and gen.init_images is not None
        and gen.mask_image is not None
    ):
        return gen, None
    mask = gen.mask_image
    if mask.dim() == 3:
        mask = mask[0]
    h, w = mask.shape
    ys, xs = (mask > 127).nonzero(as_tuple=True)
    if len(ys) == 0:
        return gen, None
    pad = max(0, int(gen.inpaint_full_res_padding))
    y0, y1 = max(0, int(ys.min()) - pad), min(h, int(ys.max()) + 1 + pad)
    x0, x1 = max(0, int(xs.min()) - pad), min(w, int(xs.max()) + 1 + pad)
    # expand the short side toward the generation aspect ratio
    target_ar = gen.width / gen.height
    ch, cw = y1 - y0, x1 - x0
    if cw / ch < target_ar:  # too narrow -> widen
        want = min(w, int(round(ch * target_ar)))
        grow = want - cw
        x0 = max(0, x0 - grow // 2)
        x1 = min(w, x0 + want)
        x0 = max(0, x1 - want)
    elif cw / ch > target_ar:  # too short -> heighten
        want = min(h, int(round(cw / target_ar)))
        grow = want - ch
        y0 = max(0, y0 - grow // 2)
        y1 = min(h, y0 + want)
        y0 = max(0, y1 - want)

    def _resize(img_f32, size):
        return F.interpolate(
            img_f32, size=size, mode="bilinear", antialias=True
        )

    inits = gen.init_images.float().permute(0, 3, 1, 2)  # [B,3,H,W]
    crop = _resize(
        inits[:, :, y0:y1, x0:x1], (gen.height, gen.width)
    ).clamp(0, 255).to(torch.uint8).permute(0, 2, 3, 1)
    mcrop = _resize(
        mask[None, None, y0:y1, x0:x1].float(), (gen.height, gen.width)
    )[0, 0].clamp(0, 255).to(torch.uint8)
    new_gen = replace_dc(gen, init_images=crop, mask_image=mcrop)
    paste_ctx = {
        "orig_inits": gen.init_images,
        "box": (y0, y1, x0, x1),
        "mask": (mask[y0:y1, x0:x1].float() / 255.0),
    }
    return new_gen, paste_ctx


def _apply_color_correction(result: "GalleryResult",
                            gen: GenerationRequest) -> None:
    """sdwui img2img color correction: match each output's channel
    statistics to its init image (in place, after assembly/paste)."""
    from ..utils.images import color_correct

    if not (gen.color_correction and gen.init_images is not None):
        return
    n_init = gen.init_images.shape[0]
    imgs = []
    for i in range(result.images.shape[0]):
        ref = gen.init_images[i % n_init]
        imgs.append(color_correct(result.images[i], ref))
    result.images = torch.stack(imgs)
    result.grid = make_grid(result.images) if len(imgs) > 1 else None


def _paste_inpaint_full_res(result: "GalleryResult", paste_ctx) -> None:
    """Paste generated crops back into the original images (in place)."""
    import torch.nn.functional as F

    if paste_ctx is None:
        return
    y0, y1, x0, x1 = paste_ctx["box"]
    m = paste_ctx["mask"][..., None]  # [ch,cw,1] in 0..1
    orig = paste_ctx["orig_inits"]
    n_orig = orig.shape[0]
    out = []
    for i in range(result.images.shape[0]):
        base = orig[i % n_orig].float().clone()
        region = F.interpolate(
            result.images[i].float().permute(2, 0, 1)[None],
            size=(y1 - y0, x1 - x0), mode="bilinear", antialias=True,
        )[0].permute(1, 2, 0)
        patch = base[y0:y1, x0:x1]
        base[y0:y1, x0:x1] = m * region + (1.0 - m) * patch
        out.append(base.clamp(0, 255).to(torch.uint8))
    result.images = torch.stack(out)
    result.grid = make_grid(result.images) if len(out) > 1 else None


class _EngineBase:
    # Master-side postprocess hooks (ref 2.3.0 "compatibility for
    # extensions which mostly only do postprocessing", e.g. ADetailer:
    # the reference ran those on the MASTER after the gather, never on
    # the remotes). Each hook is called on the fully assembled
    # GalleryResult (rank 0 in torchrun mode) and may edit it in place;
    # a hook that raises is logged and skipped so a broken postprocessor
    # cannot lose the batch.
    @property
    def postprocess_hooks(self) -> List:
        hooks = self.__dict__.get("_pp_hooks")
        if hooks is None:
            hooks = self.__dict__["_pp_hooks"] = []
        return hooks

    def _run_postprocess_hooks(self, result: "GalleryResult") -> None:
        hooks = self.__dict__.get("_pp_hooks") or []
        if not hooks:
            return
        for hook in list(hooks):
            try:
                hook(result)
            except Exception:
                log.exception("postprocess hook %r failed; skipped", hook)
        if result.images is not None and result.images.shape[0] > 1:
            result.grid = make_grid(result.images)

    def _assemble(
        self,
        gen: GenerationRequest,
        jobs: List[Job],
        shards: Dict[str, torch.Tensor],
        infos: Dict[str, List[str]],
        elapsed: float,
        interrupted: bool = False,
    ) -> GalleryResult:
        """Order shards by gallery offset, fix up infotexts with the worker
        label (ref distributed.py:343-349), build the grid."""
        total = sum(j.batch_size for j in jobs)
        h, w = _final_output_hw(gen)
        for sh in shards.values():  # trust actual shard shape (fractional hr)
            if sh is not None and sh.numel():
                h, w = sh.shape[1], sh.shape[2]
                break
        images = torch.zeros(total, h, w, 3, dtype=torch.uint8)
        seeds = [0] * total
        infotexts = [""] * total
        summary = []
        for job in sorted(jobs, key=lambda j: j.gallery_offset):
            shard = shards.get(id(job), shards.get(job.worker_label))
            o = job.gallery_offset
            n = job.batch_size
            if shard is not None and shard.shape[0] >= n:
                images[o : o + n] = shard[:n]
            job_infos = infos.get(id(job), infos.get(job.worker_label)) or []
            for i in range(n):
                seeds[o + i] = job.seeds[i] if i < len(job.seeds) else 0
                base = job_infos[i] if i < len(job_infos) else ""
                # same line as the param list, comma-joined — exactly the
                # reference's fixup (distributed.py:347-349) and the only
                # placement sdwui's infotext parser round-trips
                infotexts[o + i] = (
                    f"{base}, Worker Label: {job.worker_label}"
                    if base else f"Worker Label: {job.worker_label}"
                )
            tag = " (complementary)" if job.complementary else ""
            summary.append(
                f"{job.worker_label}: {n} image(s){tag}, "
                f"predicted {job.predicted_eta:.2f}s, actual {job.elapsed:.2f}s"
            )
        grid = make_grid(images) if total > 1 else None
        return GalleryResult(
            images=images,
            seeds=seeds,
            infotexts=infotexts,
            grid=grid,
            elapsed=elapsed,
            job_summary=summary,
            interrupted=interrupted,
        )

    def _record_eta(self, world: World, job: Job) -> None:
        w = world.get_worker(job.worker_label)
        if w is not None and job.predicted_eta > 0 and job.elapsed > 0:
            w.eta.record_outcome(job.predicted_eta, job.elapsed)


class LocalEngine(_EngineBase):
    """One process, one thread per rank (device)."""

    def __init__(
        self,
        model: str = "sd15",
        devices: Optional[List[str]] = None,
        world: Optional[World] = None,
        dtype: Optional[torch.dtype] = None,
    ) -> None:
        if devices is None:
            # --sdwd-devices / SDWD_DEVICES: ordinal subset (the flag that
            # replaces the reference's --distributed-remotes host list)
            sel = os.environ.get("SDWD_DEVICES", "").strip()
            if torch.cuda.is_available():
                if sel:
                    devices = [
                        f"cuda:{int(o)}" for o in sel.split(",") if o.strip()
                    ]
                else:
                    devices = [
                        f"cuda:{i}" for i in range(torch.cuda.device_count())
                    ]
            else:
                devices = ["cpu"] * max(1, len(sel.split(","))) if sel \
                    else ["cpu"]
        self.devices = devices
        self.model_name = model
        self.vae_name = "auto"
        self.world = world or World.from_devices(len(devices))
        self.pipes: Dict[str, StableDiffusionPipeline] = {}
        self._live_jobs = []
        self.dtype = dtype
        for i, dev in enumerate(devices):
            self.pipes[f"gpu{i}"] = StableDiffusionPipeline(
                model, device=dev, dtype=dtype
            )
        self._fail_injection: Dict[str, object] = {}
        self._oom_capped: set = set()
        if torch.cuda.is_available():
            # startup liveness sweep (ref distributed.py:48-52 pinged all
            # remotes at init); CPU ranks have no memory probe to ping
            self.world.ping(indiscriminate=True)

    def set_model(self, name: str) -> None:
        """Hot-swap the checkpoint on every rank (ref C13: the reference
        pushed the model name to all remotes over POST /options,
        world.py:784-811)."""
        if not name or name == self.model_name:
            return
        log.info("switching model to %s on all ranks", name)
        for label, pipe in self.pipes.items():
            self.pipes[label] = StableDiffusionPipeline(
                name, device=pipe.device, dtype=pipe.dtype
            )
        self.model_name = name
        self.vae_name = "auto"  # fresh bundles carry their own VAE

    def set_vae(self, name: str) -> None:
        """sdwui "SD VAE" selection (ref C13: the reference synced the VAE
        by name to every remote through load_options, worker.py:646-688):
        swap every rank's VAE weights to a standalone file from
        SDWD_VAE_DIR; "auto" restores each checkpoint's own VAE from a
        snapshot taken before the first swap."""
        from ..models.registry import load_vae_into

        name = name or "auto"
        if name == getattr(self, "vae_name", "auto"):
            return
        log.info("switching VAE to %s on all ranks", name)
        for pipe in self.pipes.values():
            vae = pipe.model.vae
            if not hasattr(pipe.model, "_own_vae_sd"):
                pipe.model._own_vae_sd = {
                    k: v.detach().cpu().clone()
                    for k, v in vae.state_dict().items()
                }
            if name == "auto":
                vae.load_state_dict(pipe.model._own_vae_sd)
            else:
                load_vae_into(pipe.model, name)
        self.vae_name = name

    # benchmark runner wired into core.Worker.benchmark (ref C8)
    def _bench_runner(self, worker: Worker, payload) -> float:
        pipe = self.pipes[worker.label]
        req = PipelineRequest(
            prompt=payload.prompt,
            negative_prompt=payload.negative_prompt,
            steps=payload.steps,
            width=payload.width,
            height=payload.height,
            sampler_name=payload.sampler_name,
            seeds=list(range(payload.batch_size)),
        )
        t0 = time.perf_counter()
        pipe.generate(req, decode=True)
        if pipe.device.type == "cuda":
            torch.cuda.synchronize(pipe.device)
        return time.perf_counter() - t0

    def benchmark(self, rebenchmark: bool = False) -> Dict[str, float]:
        return self.world.benchmark(self._bench_runner, rebenchmark=rebenchmark)

    def inject_failure(self, label: str, oom: bool = False) -> None:
        """Test hook: make the next shard on this rank raise (optionally
        as an out-of-memory error to exercise pixel-cap learning)."""
        self._fail_injection[label] = "oom" if oom else True

    def _run_job(
        self,
        gen: GenerationRequest,
        job: Job,
        shards: Dict[str, torch.Tensor],
        infos: Dict[str, List[str]],
        errors: Dict[str, Exception],
    ) -> None:
        worker = self.world.get_worker(job.worker_label)
        pipe = self.pipes[job.worker_label]
        # per-worker checkpoint override (ref ui.py:161-171: a remote could
        # pin its own model); rebuilt lazily when it changes
        want = worker.model_override or self.model_name
        if pipe.model.name != want:
            log.info("%s: model override -> %s", job.worker_label, want)
            pipe = StableDiffusionPipeline(
                want, device=pipe.device, dtype=pipe.dtype
            )
            self.pipes[job.worker_label] = pipe
        try:
            kind = self._fail_injection.pop(job.worker_label, False)
            if kind == "oom":
                raise torch.cuda.OutOfMemoryError("injected OOM")
            if kind:
                raise RuntimeError("injected failure")
            worker.set_state(State.WORKING)
            init_latents = None
            if gen.init_images is not None:
                src = gen.init_images
                idx = [
                    (job.gallery_offset + i) % src.shape[0]
                    for i in range(job.batch_size)
                ]
                init_latents = pipe.encode_image(src[idx], seeds=job.seeds)
            t0 = time.perf_counter()

            def on_step(i, n, _job=job):
                _job.steps_done = i
                _job.steps_total = n

            res = pipe.generate(
                _job_pipeline_request(gen, job, init_latents),
                interrupt=worker.interrupt_event.is_set,
                step_callback=on_step,
            )
            job.elapsed = time.perf_counter() - t0
            shards[id(job)] = res.images
            infos[id(job)] = res.infotexts
            self._record_eta(self.world, job)
            worker.set_state(
                State.INTERRUPTED if res.interrupted else State.IDLE
            )
        except Exception as exc:  # noqa: BLE001 - recovery path
            log.warning("rank %s failed: %s", job.worker_label, exc)
            errors[job.worker_label] = exc
            worker.set_state(State.UNAVAILABLE)
            if isinstance(exc, torch.cuda.OutOfMemoryError) and (
                _learn_pixel_cap(worker, gen, job.batch_size)
            ):
                # healthy GPU, job just too big: re-admit it AFTER this
                # generation's requeue (which excludes UNAVAILABLE ranks)
                self._oom_capped.add(job.worker_label)

    def progress(self) -> float:
        """0..1 across the jobs of the generation in flight (ref /progress)."""
        jobs = self._live_jobs
        if not jobs:
            return 1.0
        fracs = [
            (j.steps_done / j.steps_total) if j.steps_total else 0.0
            for j in jobs
        ]
        return sum(fracs) / len(fracs)

    def generate(self, gen: GenerationRequest) -> GalleryResult:
        t0 = time.perf_counter()
        self.set_model(gen.model)
        gen = _preprocess_mask(gen)
        gen, paste_ctx = _crop_for_inpaint_full_res(gen)
        self.world.clear_interrupt()
        jobs = self.world.make_jobs(gen.sched())
        self._live_jobs = jobs
        shards: Dict[str, torch.Tensor] = {}
        infos: Dict[str, List[str]] = {}
        errors: Dict[str, Exception] = {}

        threads = [
            threading.Thread(
                target=self._run_job, args=(gen, j, shards, infos, errors)
            )
            for j in jobs
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join()  # the gather barrier (ref distributed.py:137-142)

        # recovery: requeue failed shards onto survivors (seeds preserved)
        for job in [j for j in jobs if j.worker_label in errors]:
            if job.complementary:
                jobs.remove(job)
                continue
            try:
                retries = self.world.requeue_failed(job, gen.sched())
            except RuntimeError:
                log.error("no survivors; shard lost")
                continue
            jobs.remove(job)
            jobs.extend(retries)
            rts = [
                threading.Thread(
                    target=self._run_job, args=(gen, j, shards, infos, errors)
                )
                for j in retries
            ]
            for t in rts:
                t.start()
            for t in rts:
                t.join()

        # OOM-capped ranks come back for the next plan (now capped)
        for label in self._oom_capped:
            w = self.world.get_worker(label)
            if w is not None and w.state is State.UNAVAILABLE:
                w.set_state(State.IDLE, strict=False)
        self._oom_capped.clear()

        interrupted = self.world.interrupted.is_set()
        result = self._assemble(
            gen, jobs, shards, infos, time.perf_counter() - t0, interrupted
        )
        _paste_inpaint_full_res(result, paste_ctx)
        _apply_color_correction(result, gen)
        self._run_postprocess_hooks(result)
        self._autosave()
        return result

    def _autosave(self) -> None:
        """Persist worker speeds/MPE after each run (ref distributed.py:357,
        gated like --distributed-remotes-autosave)."""
        import os as _os

        if self.world.config_path and _os.environ.get(
            "SDWD_AUTOSAVE", "1"
        ) not in ("", "0", "false"):
            try:
                self.world.save()
            except OSError as exc:
                log.warning("config autosave failed: %s", exc)

    def interrupt(self) -> None:
        self.world.interrupt_all()


class DistributedEngine(_EngineBase):
    """One process per GPU (torchrun), RCCL over xGMI. Every rank calls
    generate() with the same request; rank 0's plan is broadcast; images are
    all-gathered; every rank returns the full gallery (rank 0 saves it)."""

    def __init__(
        self,
        model: str = "sd15",
        dtype: Optional[torch.dtype] = None,
        backend: Optional[str] = None,
    ) -> None:
        self.is_dist = pg.init_group(backend=backend)
        import torch.distributed as dist

        self.rank = dist.get_rank() if self.is_dist else 0
        self.world_size = dist.get_world_size() if self.is_dist else 1
        if torch.cuda.is_available():
            self.device = torch.device("cuda", pg.env_local_rank())
            torch.cuda.set_device(self.device)
        else:
            self.device = torch.device("cpu")
        self.label = f"gpu{self.rank}"
        self.model_name = model
        self._dtype = dtype
        self.pipe = StableDiffusionPipeline(model, device=self.device, dtype=dtype)
        self._sync_pipe_weights()
        self.world = World.from_devices(self.world_size)
        self._store = None
        if self.is_dist:
            try:
                from torch.distributed.distributed_c10d import _get_default_store

                self._store = _get_default_store()
            except Exception:  # pragma: no cover
                self._store = None
        self._interrupt_epoch = 0

    def _sync_pipe_weights(self) -> None:
        # weight sync from rank 0 (C13): registry init is deterministic, the
        # broadcast makes identity unconditional.
        for m in (
            self.pipe.model.text_encoder,
            self.pipe.model.text_encoder_2,
            self.pipe.model.unet,
            self.pipe.model.vae,
        ):
            if m is not None:
                pg.sync_weights(m)

    def set_model(self, name: str) -> None:
        """Collective model hot-swap (C13): every rank rebuilds its pipeline
        and re-syncs weights from rank 0. Must be called on ALL ranks
        (generate() routes it through the plan broadcast so this holds)."""
        if not name or name == self.model_name:
            return
        log.info("rank %d switching model to %s", self.rank, name)
        self.pipe = StableDiffusionPipeline(
            name, device=self.device, dtype=self._dtype
        )
        self._sync_pipe_weights()
        self.model_name = name

    # -- heartbeat (C11): ranks stamp the store; anyone can read liveness ----
    def heartbeat(self) -> None:
        if self._store is not None:
            try:
                self._store.set(f"sdwd_hb_{self.rank}", str(time.time()))
            except Exception:
                pass

    def rank_progress(self) -> Dict[int, str]:
        """Last published 'step/total' per rank (torchrun status surface)."""
        out: Dict[int, str] = {}
        if self._store is None:
            return out
        for r in range(self.world_size):
            try:
                if self._store.check([f"sdwd_prog_{r}"]):
                    out[r] = self._store.get(f"sdwd_prog_{r}").decode()
            except Exception:
                continue
        return out

    def heartbeats(self) -> Dict[int, float]:
        """age (seconds) of each rank's last heartbeat; missing = never."""
        out: Dict[int, float] = {}
        if self._store is None:
            return out
        now = time.time()
        for r in range(self.world_size):
            try:
                if self._store.check([f"sdwd_hb_{r}"]):
                    out[r] = now - float(self._store.get(f"sdwd_hb_{r}"))
            except Exception:
                continue
        return out

    # -- out-of-band interrupt (C20): TCPStore control plane -----------------
    def interrupt(self) -> None:
        self._interrupt_epoch += 1
        if self._store is not None:
            self._store.set("sdwd_interrupt", str(self._interrupt_epoch))

    def _interrupted(self) -> bool:
        """Polled between denoise steps; the store check is a TCP round
        trip, so throttle to the reference's 0.5 s cadence
        (ref worker.py:444-448)."""
        if self._store is None:
            return False
        now = time.monotonic()
        if now - getattr(self, "_int_polled", 0.0) < 0.5:
            return getattr(self, "_int_cached", False)
        self._int_polled = now
        val = False
        try:
            if self._store.check(["sdwd_interrupt"]):
                val = int(self._store.get("sdwd_interrupt")) > 0
        except Exception:
            val = False
        self._int_cached = val
        return val

    def _clear_interrupt(self) -> None:
        self._int_cached = False
        self._int_polled = 0.0
        if self._store is not None and self.rank == 0:
            try:
                self._store.set("sdwd_interrupt", "0")
            except Exception:
                pass

    # -- benchmark (C8): each rank times itself, speeds are all-gathered -----
    def benchmark(self, rebenchmark: bool = False) -> Dict[str, float]:
        me = self.world.get_worker(self.label)
        if rebenchmark or me.eta.avg_ipm <= 0:
            payload = self.world.benchmark_payload
            req = PipelineRequest(
                prompt=payload.prompt,
                steps=payload.steps,
                width=payload.width,
                height=payload.height,
                sampler_name=payload.sampler_name,
                seeds=list(range(payload.batch_size)),
            )
            from ..config import BENCHMARK_TIMED_SAMPLES, BENCHMARK_WARMUP_SAMPLES

            for _ in range(BENCHMARK_WARMUP_SAMPLES):
                self.pipe.generate(req)
            samples = []
            for _ in range(BENCHMARK_TIMED_SAMPLES):
                t0 = time.perf_counter()
                self.pipe.generate(req)
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                samples.append(
                    payload.batch_size * 60.0 / (time.perf_counter() - t0)
                )
            my_ipm = sum(samples) / len(samples)
        else:
            my_ipm = me.eta.avg_ipm
        speeds = pg.allgather_floats([my_ipm], self.device)
        for r, (ipm,) in enumerate(speeds):
            self.world.get_worker(f"gpu{r}").eta.avg_ipm = ipm
        return {w.label: w.eta.avg_ipm for w in self.world.workers}

    def _weight_digest(self) -> float:
        """Cheap per-pipe weight digest (sum of |param|) for the cross-rank
        identity check; cached until the pipe object is rebuilt."""
        cached = getattr(self.pipe, "_sdwd_digest", None)
        if cached is None:
            total = 0.0
            for m in (
                self.pipe.model.text_encoder,
                self.pipe.model.text_encoder_2,
                self.pipe.model.unet,
                self.pipe.model.vae,
            ):
                if m is None:
                    continue
                acc = None
                for p in m.parameters():
                    s = p.detach().float().abs().sum()
                    acc = s if acc is None else acc + s
                if acc is not None:
                    total += float(acc.item())
            self.pipe._sdwd_digest = cached = total
        return cached

    def _exec_shard(self, gen: GenerationRequest, job) -> dict:
        """Run one shard on this rank. Returns images/infos/status; never
        raises (failures are reported through ok=0 so the collectives that
        follow stay symmetric across ranks)."""
        h, w = _final_output_hw(gen)
        out = {
            "images": torch.zeros(0, h, w, 3, dtype=torch.uint8),
            "infos": [],
            "ok": 1.0,
            "elapsed": 0.0,
            "cap": 0.0,
            "interrupted": False,
        }
        if job is None or job.batch_size <= 0:
            return out
        try:
            init_latents = None
            if gen.init_images is not None:
                src = gen.init_images
                idx = [
                    (job.gallery_offset + i) % src.shape[0]
                    for i in range(job.batch_size)
                ]
                init_latents = self.pipe.encode_image(
                    src[idx], seeds=job.seeds
                )
            ts = time.perf_counter()

            last_pub = [0.0]

            def on_step(i, n):
                # publish rank progress out-of-band, throttled to the
                # interrupt-poll cadence (C19 status for torchrun mode)
                now = time.monotonic()
                if self._store is not None and (
                    now - last_pub[0] > 0.5 or i == n
                ):
                    last_pub[0] = now
                    try:
                        self._store.set(
                            f"sdwd_prog_{self.rank}", f"{i}/{n}"
                        )
                    except Exception:
                        pass

            res = self.pipe.generate(
                _job_pipeline_request(gen, job, init_latents),
                interrupt=self._interrupted,
                step_callback=on_step,
            )
            out["elapsed"] = time.perf_counter() - ts
            out["images"] = res.images
            out["infos"] = res.infotexts
            out["interrupted"] = res.interrupted
        except Exception as exc:  # noqa: BLE001
            log.warning("rank %d shard failed: %s", self.rank, exc)
            out["ok"] = 0.0
            if isinstance(exc, torch.cuda.OutOfMemoryError):
                me = self.world.get_worker(self.label)
                if _learn_pixel_cap(me, gen, job.batch_size):
                    out["cap"] = float(me.pixel_cap)
        return out

    def generate(self, gen: GenerationRequest) -> GalleryResult:
        t0 = time.perf_counter()
        self.heartbeat()
        self._clear_interrupt()
        if self.rank == 0:
            plan = (gen.model, self.world.make_jobs(gen.sched()))
        else:
            plan = None
        model_name, jobs = pg.broadcast_object(plan)
        self.set_model(model_name)
        # per-worker checkpoint override (ref ui.py:161-171): purely local —
        # no collective weight sync, the deterministic registry (or the
        # checkpoint file) guarantees any rank builds identical weights
        override = getattr(
            self.world.get_worker(self.label), "model_override", None
        )
        want = override or self.model_name
        if self.pipe.model.name != want:
            log.info("rank %d: model override -> %s", self.rank, want)
            self.pipe = StableDiffusionPipeline(
                want, device=self.device, dtype=self._dtype
            )
        # cross-rank weight-identity check: ranks that claim the same model
        # name must hold identical weights (registry init is deterministic;
        # file-backed checkpoints could silently diverge, e.g. a stale file
        # on one node path). Cheap: the digest is cached per pipe build.
        if self.is_dist:
            pairs = pg.allgather_object((want, self._weight_digest()))
            by_name: Dict[str, float] = {}
            for r, (name, d) in enumerate(pairs):
                if name in by_name:
                    ref = by_name[name]
                    if abs(ref - d) > 1e-3 * max(1.0, abs(ref)):
                        raise RuntimeError(
                            f"weight divergence for model '{name}' across "
                            f"ranks ({ref:.6g} vs {d:.6g} on rank {r}); "
                            "resync the checkpoint files"
                        )
                else:
                    by_name[name] = d
        gen = _preprocess_mask(gen)
        gen, paste_ctx = _crop_for_inpaint_full_res(gen)

        mine = next((j for j in jobs if j.worker_label == self.label), None)
        r1 = self._exec_shard(gen, mine)
        interrupted = r1["interrupted"]

        # status + images + infotexts to everyone (rank 0 consumes; the
        # infotext gather keeps ref parity: distributed.py:343-349 preserved
        # per-image infotexts across all workers)
        sizes = [
            next(
                (
                    j.batch_size
                    for j in jobs
                    if j.worker_label == f"gpu{r}" and j.batch_size > 0
                ),
                0,
            )
            for r in range(self.world_size)
        ]
        status = pg.allgather_floats(
            [r1["ok"], r1["elapsed"], r1["cap"]], self.device
        )
        # every rank applies learned pixel caps so future plans agree
        for r in range(self.world_size):
            if len(status[r]) > 2 and status[r][2] > 0:
                self.world.get_worker(f"gpu{r}").pixel_cap = int(status[r][2])
        gathered = pg.gather_images(
            r1["images"].to(self.device), sizes, self.device
        )
        all_images = gathered.cpu() if gathered is not None else r1["images"]
        all_infos = pg.allgather_object(r1["infos"])

        # slice the gathered stack back into per-rank shards
        shards: Dict[str, torch.Tensor] = {}
        infos: Dict[str, List[str]] = {}
        pos = 0
        for r in range(self.world_size):
            n = sizes[r]
            shards[f"gpu{r}"] = all_images[pos : pos + n]
            infos[f"gpu{r}"] = list(all_infos[r]) if r < len(all_infos) else []
            pos += n

        # eta bookkeeping + failure detection on EVERY rank (identical
        # allgathered inputs keep the replicated World state in agreement)
        failed: List[Job] = []
        for r in range(self.world_size):
            ok, elapsed = status[r][0], status[r][1]
            job = next(
                (j for j in jobs if j.worker_label == f"gpu{r}"), None
            )
            if job is None:
                continue
            job.elapsed = elapsed
            if job.predicted_eta > 0 and elapsed > 0:
                self._record_eta(self.world, job)
            if ok < 0.5:
                self.world.get_worker(f"gpu{r}").set_state(State.UNAVAILABLE)
                if not job.complementary:
                    failed.append(job)
                elif self.rank == 0:
                    jobs.remove(job)  # bonus images are simply dropped

        # recovery: reshard failed shards over the survivors — the same
        # semantics LocalEngine has (seeds preserved, benchmark-weighted),
        # instead of a serial re-run on rank 0. Rank 0 plans; the plan is
        # broadcast so every rank runs its part of the retry collectives.
        retries: List[Job] = []
        if self.rank == 0:
            for job in failed:
                try:
                    rjobs = self.world.requeue_failed(job, gen.sched())
                except RuntimeError:
                    log.error(
                        "no survivors; shard of %s lost", job.worker_label
                    )
                    continue
                log.warning(
                    "resharding failed shard of %s over %s",
                    job.worker_label,
                    [j.worker_label for j in rjobs],
                )
                jobs.remove(job)
                jobs.extend(rjobs)
                retries.extend(rjobs)
        if self.is_dist and failed:
            retries = pg.broadcast_object(retries if self.rank == 0 else None)

        if retries:
            my_retries = [
                j for j in retries if j.worker_label == self.label
            ]
            imgs2, infos2 = [], []
            ok2 = 1.0
            for job in my_retries:
                rr = self._exec_shard(gen, job)
                imgs2.append(rr["images"])
                infos2.extend(rr["infos"])
                ok2 = min(ok2, rr["ok"])
                interrupted = interrupted or rr["interrupted"]
            my_img2 = (
                torch.cat(imgs2, dim=0)
                if imgs2
                else torch.zeros(0, 1, 1, 3, dtype=torch.uint8)
            )
            sizes2 = [
                sum(
                    j.batch_size
                    for j in retries
                    if j.worker_label == f"gpu{r}"
                )
                for r in range(self.world_size)
            ]
            status2 = pg.allgather_floats([ok2], self.device)
            g2 = pg.gather_images(
                my_img2.to(self.device), sizes2, self.device
            )
            ai2 = g2.cpu() if g2 is not None else my_img2
            inf2 = pg.allgather_object(infos2)
            for r in range(self.world_size):
                if status2[r][0] < 0.5:
                    self.world.get_worker(f"gpu{r}").set_state(
                        State.UNAVAILABLE, strict=False
                    )
            if self.rank == 0:
                pos = 0
                for r in range(self.world_size):
                    ipos = 0
                    for job in retries:
                        if job.worker_label != f"gpu{r}":
                            continue
                        shards[id(job)] = ai2[pos : pos + job.batch_size]
                        infos[id(job)] = inf2[r][
                            ipos : ipos + job.batch_size
                        ]
                        pos += job.batch_size
                        ipos += job.batch_size

        result = self._assemble(
            gen, jobs, shards, infos, time.perf_counter() - t0, interrupted
        )
        _paste_inpaint_full_res(result, paste_ctx)
        _apply_color_correction(result, gen)
        if self.rank == 0:
            self._run_postprocess_hooks(result)
        return result
