#!/usr/bin/env python3
"""Attention kernel microbench at the SD UNet shapes.

Reports per-shape ms and TF/s (useful FLOPs, true head dim D, not the
padded kernel extent) for the in-tree flash kernel. --check compares
against a fp32 eager reference first. Shapes follow the batch-64
CFG-doubled 512x512 SD1.5 step (B=128) plus SDXL/cross-attn variants.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from sdwd_amd import ops  # noqa: E402

# (tag, B, H, Sq, Sk, D)
SHAPES = [
    ("sd15-self-320", 128, 8, 4096, 4096, 40),
    ("sd15-self-640", 128, 8, 1024, 1024, 80),
    ("sd15-self-1280", 128, 8, 256, 256, 160),
    ("sd15-cross-320", 128, 8, 4096, 77, 40),
    ("sd15-cross-640", 128, 8, 1024, 77, 80),
    ("sdxl-self-640", 64, 10, 4096, 4096, 64),
    ("sdxl-self-1280", 64, 20, 1024, 1024, 64),
]


def flops(B, H, Sq, Sk, D):
    return 2 * B * H * Sq * Sk * D * 2  # QK^T + PV


def bench(fn, warm=3, it=10):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--only", default=None, help="substring filter on tag")
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()
    assert torch.cuda.is_available(), "GPU required"
    results = {}
    for tag, B, H, Sq, Sk, D in SHAPES:
        if args.only and args.only not in tag:
            continue
        g = torch.Generator(device="cuda").manual_seed(hash(tag) % 2**31)
        q = torch.randn(B, H, Sq, D, device="cuda", dtype=torch.bfloat16,
                        generator=g) * 0.5
        k = torch.randn(B, H, Sk, D, device="cuda", dtype=torch.bfloat16,
                        generator=g) * 0.5
        v = torch.randn(B, H, Sk, D, device="cuda", dtype=torch.bfloat16,
                        generator=g)
        scale = D ** -0.5
        if args.check:
            # reference on a 2-batch slice: the eager fp32 score matrix at
            # full B would need tens of GB
            with torch.no_grad():
                qs, ks, vs = q[:2], k[:2], v[:2]
                ref = torch.softmax(
                    (qs.float() @ ks.float().transpose(-1, -2)) * scale,
                    dim=-1,
                ) @ vs.float()
                got = ops.attention(qs, ks, vs, scale).float()
                err = (got - ref).abs().max().item()
                rel = err / ref.abs().max().item()
            print(f"{tag}: max abs err {err:.4e} (rel {rel:.3e})")
            assert rel < 2e-2, f"{tag} numerics off"
            del ref, got, qs, ks, vs
        ms = bench(lambda: ops.attention(q, k, v, scale),
                   it=args.iters) * 1e3
        tf = flops(B, H, Sq, Sk, D) / (ms * 1e-3) / 1e12
        results[tag] = {"ms": round(ms, 3), "tf": round(tf, 1)}
        print(f"{tag}: {ms:.3f} ms  {tf:.1f} TF/s (useful, D={D})")
    print(json.dumps(results))


if __name__ == "__main__":
    main()
