#!/usr/bin/env python3
"""Decode ds_read_b64_tr_b16 semantics on real hardware.

LDS bf16 element e holds value e; each lane issues one tr read at
addr = base + lane*stride. The dump shows, per (lane, j), WHICH LDS
element landed there — settling the transpose mapping for the attention
V path (the guide's worked example is not on disk).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from sdwd_amd.ops import ext  # noqa: E402


def dump(stride, base=0):
    out = ext().probe_tr16(stride, base).cpu().to(torch.int32)
    print(f"--- stride={stride}B base={base}B: lane -> [elem j=0..3]")
    for l in range(64):
        print(f"lane {l:2d}: {out[l].tolist()}")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    dump(8)    # each lane its own consecutive 8B row
    dump(0)    # uniform address
    dump(16)   # 16B-strided rows
