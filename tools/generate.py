#!/usr/bin/env python3
"""Command-line txt2img / img2img over all visible GPUs.

Examples:
  python tools/generate.py --prompt "a herd of cows" --batch 8 --out out/
  python tools/generate.py --prompt "re-style" --init photo.png \
      --strength 0.6 --out out/
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from sdwd_amd.config import add_flags  # noqa: E402
from sdwd_amd.parallel import GenerationRequest, LocalEngine  # noqa: E402
from sdwd_amd.utils.images import decode_png, save_png  # noqa: E402


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--prompt", default="")
    ap.add_argument("--prompts-file", default="",
                    help="one prompt per line -> one image per prompt")
    ap.add_argument("--negative", default="")
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--width", type=int, default=512)
    ap.add_argument("--height", type=int, default=512)
    ap.add_argument("--cfg", type=float, default=7.0)
    ap.add_argument("--sampler", default="Euler a")
    ap.add_argument("--scheduler", default="Automatic",
                    help="Uniform | Karras | Exponential | SGM Uniform")
    ap.add_argument("--clip-skip", type=int, default=1)
    ap.add_argument("--hr-upscaler", default="Latent")
    ap.add_argument("--refiner", default="", help="refiner model name")
    ap.add_argument("--refiner-switch-at", type=float, default=0.8)
    ap.add_argument("--seed", type=int, default=-1)
    ap.add_argument("--model", default="sd15")
    ap.add_argument("--init", default="", help="PNG for img2img")
    ap.add_argument("--strength", type=float, default=0.75)
    ap.add_argument("--hires", action="store_true")
    ap.add_argument("--hr-scale", type=float, default=2.0)
    ap.add_argument("--hr-resize", default="",
                    help="WxH hires target (sdwui resize-to mode)")
    ap.add_argument("--eta", type=float, default=-1.0,
                    help="ancestral/SDE noise multiplier (sdwui Eta)")
    ap.add_argument("--vae", default="",
                    help="standalone VAE name from SDWD_VAE_DIR")
    ap.add_argument("--out", default="outputs")
    ap.add_argument("--benchmark", action="store_true",
                    help="re-benchmark ranks before generating")
    add_flags(ap)
    args = ap.parse_args()

    if not args.prompt and not args.prompts_file:
        ap.error("--prompt or --prompts-file required")
    prompts = None
    if args.prompts_file:
        with open(args.prompts_file) as fh:
            prompts = [ln.strip() for ln in fh if ln.strip()]
        args.batch = len(prompts)
        if not args.prompt:
            args.prompt = prompts[0]
    engine = LocalEngine(model=args.model)
    if args.vae:
        engine.set_vae(args.vae)
    print(f"{len(engine.devices)} rank(s): {engine.devices}")
    if args.benchmark:
        print("benchmarking...", engine.benchmark(rebenchmark=True))

    init_images = None
    if args.init:
        with open(args.init, "rb") as fh:
            img = decode_png(fh.read())
        init_images = img[None]

    res = engine.generate(
        GenerationRequest(
            prompt=args.prompt,
            prompts=prompts,
            negative_prompt=args.negative,
            batch_size=args.batch,
            width=args.width,
            height=args.height,
            steps=args.steps,
            cfg_scale=args.cfg,
            sampler_name=args.sampler,
            scheduler=args.scheduler,
            clip_skip=args.clip_skip,
            seed=args.seed,
            init_images=init_images,
            denoising_strength=args.strength,
            enable_hr=args.hires or bool(args.hr_resize),
            hr_scale=0.0 if args.hr_resize else args.hr_scale,
            hr_resize_x=int(args.hr_resize.split('x')[0]) if args.hr_resize else 0,
            hr_resize_y=int(args.hr_resize.split('x')[1]) if args.hr_resize else 0,
            eta=args.eta,
            hr_upscaler=args.hr_upscaler,
            refiner_model=args.refiner,
            refiner_switch_at=args.refiner_switch_at,
        )
    )
    os.makedirs(args.out, exist_ok=True)
    for i in range(res.images.shape[0]):
        path = os.path.join(args.out, f"{res.seeds[i]}_{i:03d}.png")
        save_png(res.images[i], path, parameters=res.infotexts[i])
        print(f"saved {path}")
    if res.grid is not None:
        save_png(res.grid, os.path.join(args.out, "grid.png"))
        print(f"saved {os.path.join(args.out, 'grid.png')}")
    print("\n".join(res.job_summary))
    print(f"total {res.elapsed:.2f}s")
    return 0


if __name__ == "__main__":
    sys.exit(main())
