#!/usr/bin/env python3
"""Conv solver A/B on MI355X: NCHW vs channels_last, benchmark-find on/off,
on the SD1.5/VAE hot conv shapes. Prints JSON {config: ms}."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

SHAPES = [
    # (N, Cin, H, W, Cout, k, stride) — UNet + VAE decoder hot shapes, batch 32 (16+CFG)
    (32, 320, 64, 64, 320, 3, 1),
    (32, 640, 32, 32, 640, 3, 1),
    (32, 1280, 16, 16, 1280, 3, 1),
    (32, 1280, 8, 8, 1280, 3, 1),
    (16, 512, 128, 128, 512, 3, 1),   # VAE decoder
    (16, 256, 256, 256, 256, 3, 1),
    (16, 128, 512, 512, 128, 3, 1),
]


def timeit(fn, warm=3, iters=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(tag, channels_last, benchmark):
    torch.backends.cudnn.benchmark = benchmark
    out = {}
    total = 0.0
    for (n, ci, h, w, co, k, s) in SHAPES:
        x = torch.randn(n, ci, h, w, device="cuda", dtype=torch.bfloat16)
        wt = torch.randn(co, ci, k, k, device="cuda", dtype=torch.bfloat16)
        if channels_last:
            x = x.to(memory_format=torch.channels_last)
            wt = wt.to(memory_format=torch.channels_last)
        t = timeit(lambda: F.conv2d(x, wt, stride=s, padding=k // 2))
        flops = 2 * n * co * ci * k * k * (h // s) * (w // s)
        out[f"{n}x{ci}x{h}x{w}->{co}"] = {
            "ms": round(t * 1e3, 3),
            "tflops": round(flops / t / 1e12, 1),
        }
        total += t
    out["_total_ms"] = round(total * 1e3, 2)
    print(json.dumps({tag: out}))


if __name__ == "__main__":
    tag = sys.argv[1] if len(sys.argv) > 1 else "default"
    cl = "--cl" in sys.argv
    bench = "--benchmark" in sys.argv
    run(tag, cl, bench)
