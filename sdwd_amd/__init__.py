"""sdwd_amd — MI355X-native distributed Stable Diffusion batch-inference engine.

A brand-new, single-node, in-process framework with the capabilities of the
``stable-diffusion-webui-distributed`` webui extension: a
benchmark-weighted batch sharder scatters txt2img/img2img work across the GPUs
of one node, gathers the decoded images into a single gallery, tracks
per-worker state, predicts ETAs with rolling error correction, supports
interrupt, complementary-image production, step scaling, pixel caps and
persistent JSON config — but where the reference delegated all compute to
remote webui instances over HTTPS/JSON (reference scripts/spartan/worker.py),
this engine runs its own SD pipeline on PyTorch-ROCm with hand-written
CDNA4 (gfx950) HIP kernels and RCCL collectives over xGMI.

Layout:
    config/    pydantic config schema + CLI flags        (ref C12, C16)
    core/      scheduler: World/Worker/Job, ETA, states  (ref C3-C8, C14, C22)
    models/    CLIP text encoder, UNet, VAE (SD1.5/SDXL)
    ops/       HIP kernel extension + op entry points
    pipeline/  samplers, noise schedule, txt2img/img2img
    parallel/  RCCL process group, dispatch/gather, health
    api/       sdapi/v1-compatible HTTP surface
    utils/     logging (console+file+ring buffer), image grid/infotext
"""

__version__ = "0.1.0"
