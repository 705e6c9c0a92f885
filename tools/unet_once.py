#!/usr/bin/env python3
"""One warmed SD1.5 UNet forward (for compact PMC captures)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from sdwd_amd.models import load_model

m = load_model("sd15", device="cuda", dtype=torch.bfloat16)
m.unet.to(memory_format=torch.channels_last)
x = torch.randn(64, 4, 64, 64, device="cuda", dtype=torch.bfloat16)
t = torch.full((64,), 500.0, device="cuda")
ctx = torch.randn(64, 77, 768, device="cuda", dtype=torch.bfloat16)
with torch.no_grad():
    m.unet(x, t, ctx)          # warmup (excluded? no - but one extra pass ok)
    torch.cuda.synchronize()
    m.unet(x, t, ctx)
    torch.cuda.synchronize()
print("done")
