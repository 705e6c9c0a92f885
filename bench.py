#!/usr/bin/env python3
"""Benchmark: the BASELINE.json headline metric.

metric: images/sec (whole node), SD1.5 txt2img 512x512 20-step batch=64,
at 1/2/4/8 GPUs (strong scaling: the 64-image batch is sharded across N
ranks by the benchmark-weighted scheduler). Synthetic prompts, random-init
weights (no network for datasets/checkpoints), bf16 on GPU.

Run:  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...
(one rank per GPU over RCCL; RANK/WORLD_SIZE read from the env).
One bench "step" = one full generation of the global batch (plan ->
CLIP encode -> 20-step denoise -> VAE decode -> image gather).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3, help="timed bench steps")
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--model", type=str, default="sd15")
    ap.add_argument("--global-batch", type=int, default=64)
    ap.add_argument("--width", type=int, default=512)
    ap.add_argument("--height", type=int, default=512)
    ap.add_argument("--denoise-steps", type=int, default=20)
    ap.add_argument("--sampler", type=str, default="Euler a")
    ap.add_argument("--cfg", type=float, default=7.0)
    ap.add_argument("--task", choices=["txt2img", "img2img"], default="txt2img",
                    help="img2img adds VAE encode + strength-limited denoise")
    ap.add_argument("--strength", type=float, default=0.75)
    args = ap.parse_args()

    from sdwd_amd.config import add_flags  # noqa: F401  (flag surface)
    from sdwd_amd.parallel import DistributedEngine, GenerationRequest, barrier

    have_gpu = torch.cuda.is_available()
    engine = DistributedEngine(model=args.model)
    rank = engine.rank
    world = engine.world_size
    dev = engine.device

    # homogeneous node: seed equal speeds so the weighted planner runs its
    # real path (it reduces to equal split + remainder round-robin)
    for w in engine.world.workers:
        w.eta.avg_ipm = 60.0

    init_images = None
    if args.task == "img2img":
        g = torch.Generator().manual_seed(7)
        init_images = torch.randint(
            0, 255, (args.global_batch, args.height, args.width, 3),
            generator=g, dtype=torch.uint8,
        )
    req = GenerationRequest(
        prompt="A herd of cows grazing at the bottom of a sunny valley",
        negative_prompt="blurry, low quality",
        batch_size=args.global_batch,
        width=args.width,
        height=args.height,
        steps=args.denoise_steps,
        cfg_scale=args.cfg,
        sampler_name=args.sampler,
        seed=1234,
        init_images=init_images,
        denoising_strength=args.strength,
    )

    def sync():
        if have_gpu:
            torch.cuda.synchronize(dev)

    for _ in range(args.warmup):
        engine.generate(req)
    sync()
    barrier()
    sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        res = engine.generate(req)
    sync()
    barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (the slowest rank defines the job)
    from sdwd_amd.parallel import allgather_floats

    all_elapsed = allgather_floats([elapsed], dev if have_gpu else "cpu")
    elapsed = max(v[0] for v in all_elapsed)

    if rank == 0:
        total_images = args.global_batch * args.steps
        value = total_images / elapsed
        out = {
            "metric": "images/sec (whole node) SD1.5 txt2img 512x512 "
                      "20-step batch=64 at 1/2/4/8 GPU",
            "value": round(value, 4),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000.0 / args.steps, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic prompts, random-init weights",
            "config": {
                "task": args.task,
                "model": args.model,
                "global_batch": args.global_batch,
                "seq_len": args.width,
                "resolution": f"{args.width}x{args.height}",
                "denoise_steps": args.denoise_steps,
                "sampler": args.sampler,
                "cfg_scale": args.cfg,
                "parallelism": f"dp{world} (benchmark-weighted batch shard)",
            },
        }
        print(json.dumps(out))
        sys.stdout.flush()
    return 0


if __name__ == "__main__":
    sys.exit(main())
