#!/usr/bin/env python3
"""Our linear GEMM vs hipBLASLt (F.linear) on the SD projection shapes."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from sdwd_amd import ops

# (M, K, N): qkv/out projections + GEGLU/FF at b2=128 shapes
SHAPES = [
    (128 * 4096, 320, 960),    # fused self-attn QKV (round 2)
    (128 * 4096, 320, 320),    # out-proj
    (128 * 4096, 320, 2560),   # GEGLU proj
    (128 * 4096, 1280, 320),   # FF out
    (128 * 1024, 640, 1920),   # fused QKV L2
    (128 * 1024, 640, 640),
    (128 * 1024, 640, 5120),
    (128 * 1024, 2560, 640),
    (128 * 256, 1280, 3840),   # fused QKV L3
    (128 * 256, 1280, 1280),
    (128 * 256, 1280, 10240),
    (128 * 256, 5120, 1280),
]


def t(fn, warm=2, it=8):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it


out = {}
for (M, K, N) in SHAPES:
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.1
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    t_v2 = t(lambda: ops.ext().gemm_v2(x, w, b))
    t_blas = t(lambda: F.linear(x, w, b))
    # numerics check (gemm_v2, the candidate)
    o = ops.ext().gemm_v2(x, w, b).float()
    r = F.linear(x.float(), w.float(), b.float())
    err = ((o - r).abs().max() / r.abs().max()).item()
    fl = 2.0 * M * K * N
    out[f"{M}x{K}x{N}"] = {
        "v2_ms": round(t_v2 * 1e3, 3), "v2_tf": round(fl / t_v2 / 1e12, 1),
        "blas_ms": round(t_blas * 1e3, 3), "blas_tf": round(fl / t_blas / 1e12, 1),
        "err": round(err, 4),
    }
    assert err < 0.02, f"gemm_v2 numerics off at {M}x{K}x{N}: {err}" 
print(json.dumps(out, indent=1))
