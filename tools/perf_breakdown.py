#!/usr/bin/env python3
"""Stage-level timing breakdown on one GPU: CLIP encode, UNet step (eager vs
hipGraph), VAE decode, flash attention vs composed fallback. Guides kernel
optimization (run under gpurun; writes JSON to stdout)."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, warmup=2, iters=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64, help="images (pre-CFG)")
    ap.add_argument("--res", type=int, default=512)
    ap.add_argument("--only", choices=["all", "unet", "vae"], default="all")
    ap.add_argument("--nchw", action="store_true",
                    help="keep NCHW (disable channels_last)")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    dt = torch.bfloat16
    from sdwd_amd.models import load_model
    from sdwd_amd import ops

    m = load_model("sd15", device=dev, dtype=dt)
    if not args.nchw:
        torch.backends.cudnn.benchmark = True
        m.unet.to(memory_format=torch.channels_last)
        m.vae.to(memory_format=torch.channels_last)
    b2 = args.batch * 2  # CFG doubling
    lat = args.res // 8
    x = torch.randn(b2, 4, lat, lat, device=dev, dtype=dt)
    ts = torch.full((b2,), 500.0, device=dev)
    ctx = torch.randn(b2, 77, 768, device=dev, dtype=dt)
    out = {}

    with torch.no_grad():
        out["unet_eager_ms"] = timeit(lambda: m.unet(x, ts, ctx)) * 1000
        if args.only == "unet":
            print(json.dumps(out, indent=1))
            return
        if args.only == "vae":
            z = torch.randn(args.batch, 4, args.res // 8, args.res // 8,
                            device=dev, dtype=dt)
            out["vae_decode_ms"] = timeit(
                lambda: m.vae.decode(z), iters=3) * 1000
            print(json.dumps(out, indent=1))
            return

        from sdwd_amd.pipeline.graphs import GraphedDenoiser

        g = GraphedDenoiser(lambda a, b, c, d: m.unet(a, b, c, y=d), dev)
        out["unet_graph_ms"] = timeit(lambda: g(x, ts, ctx, None)) * 1000

        z = torch.randn(args.batch, 4, lat, lat, device=dev, dtype=dt)
        out["vae_decode_ms"] = timeit(lambda: m.vae.decode(z), iters=3) * 1000

        tok = torch.randint(0, 49000, (b2, 77), device=dev)
        out["clip_ms"] = timeit(lambda: m.text_encoder(tok)) * 1000

        # attention microbench: SD1.5 level-1 self-attn shape
        q = torch.randn(b2 * 8, 1, 4096, 40, device=dev, dtype=dt)
        k, v = torch.randn_like(q), torch.randn_like(q)
        qq = q.view(b2, 8, 4096, 40)
        kk, vv = k.view_as(qq), v.view_as(qq)
        t_flash = timeit(lambda: ops.attention(qq, kk, vv), iters=3)
        out["attn_L1_flash_ms"] = t_flash * 1000
        flops = 4 * 4096 * 4096 * 48 * b2 * 8
        out["attn_L1_flash_tflops"] = flops / t_flash / 1e12

        q2 = torch.randn(b2, 8, 1024, 80, device=dev, dtype=dt)
        k2, v2 = torch.randn_like(q2), torch.randn_like(q2)
        t2 = timeit(lambda: ops.attention(q2, k2, v2), iters=3)
        out["attn_L2_flash_ms"] = t2 * 1000
        out["attn_L2_flash_tflops"] = 4 * 1024 * 1024 * 80 * b2 * 8 / t2 / 1e12

        # GroupNorm bandwidth
        gx = torch.randn(b2, 320, 64, 64, device=dev, dtype=dt)
        w = torch.ones(320, device=dev)
        bb = torch.zeros(320, device=dev)
        tg = timeit(lambda: ops.group_norm_silu(gx, w, bb, 32))
        out["gn_silu_ms"] = tg * 1000
        out["gn_silu_gbps"] = gx.numel() * 2 * 3 / tg / 1e9  # 2 reads + 1 write

    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
