#!/usr/bin/env python3
"""Our conv3x3 vs MIOpen F.conv2d on the SD hot shapes (channels_last)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from sdwd_amd import ops

SHAPES = [
    (128, 320, 64, 64, 320, 1),
    (128, 640, 32, 32, 640, 1),
    (128, 1280, 16, 16, 1280, 1),
    (128, 1280, 8, 8, 1280, 1),
    (128, 2560, 8, 8, 1280, 1),
    (128, 1920, 16, 16, 1280, 1),
    (128, 960, 32, 32, 640, 1),
    (128, 640, 64, 64, 320, 1),
    (128, 320, 64, 64, 320, 2),
    (16, 512, 64, 64, 512, 1),    # VAE mid
    (16, 512, 128, 128, 512, 1),
    (16, 256, 256, 256, 256, 1),
    (16, 128, 512, 512, 128, 1),
]


def t(fn, warm=2, it=6):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it


out = {}
tot_ours = tot_miopen = 0.0
for (N, Ci, H, W, Co, s) in SHAPES:
    x = torch.randn(N, Ci, H, W, device="cuda", dtype=torch.bfloat16)
    xc = x.contiguous(memory_format=torch.channels_last)
    w = torch.randn(Co, Ci, 3, 3, device="cuda", dtype=torch.bfloat16) * 0.02
    b = torch.randn(Co, device="cuda", dtype=torch.bfloat16)
    wp = w.permute(0, 2, 3, 1).contiguous()
    wcl = w.contiguous(memory_format=torch.channels_last)
    t_ours = t(lambda: ops.conv3x3(xc, wp, b, None, s))
    t_mi = t(lambda: F.conv2d(xc, wcl, b, stride=s, padding=1))
    flops = 2 * N * Co * Ci * 9 * (H // s) * (W // s)
    key = f"{N}x{Ci}x{H}x{W}->{Co}/s{s}"
    out[key] = {
        "ours_ms": round(t_ours * 1e3, 3),
        "ours_tf": round(flops / t_ours / 1e12, 1),
        "miopen_ms": round(t_mi * 1e3, 3),
        "miopen_tf": round(flops / t_mi / 1e12, 1),
    }
    tot_ours += t_ours
    tot_miopen += t_mi
out["_total"] = {"ours_ms": round(tot_ours * 1e3, 2),
                 "miopen_ms": round(tot_miopen * 1e3, 2)}
print(json.dumps(out, indent=1))
