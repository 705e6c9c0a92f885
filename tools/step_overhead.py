#!/usr/bin/env python3
"""Where does a small-shard (batch-8) generation spend its non-kernel
time? Times (a) bare whole-step graph replays, (b) the full pipeline
generate, (c) its stages — the gap guides host-overhead work (the N=8
scaling shard runs this size)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from sdwd_amd.pipeline import PipelineRequest, StableDiffusionPipeline

assert torch.cuda.is_available()
B = int(os.environ.get("B", "8"))
pipe = StableDiffusionPipeline("sd15", device="cuda:0")
req = PipelineRequest(
    prompt="overhead probe", steps=20, width=512, height=512,
    seeds=list(range(B)),
)
# warmup (captures graphs)
pipe.generate(req)
torch.cuda.synchronize()

t0 = time.perf_counter()
res = pipe.generate(req)
torch.cuda.synchronize()
t_gen = time.perf_counter() - t0
print(f"full generate: {t_gen*1e3:.1f} ms")

# bare replay loop of the captured whole-step graph at the same shape
g = next(iter(pipe._wholestep_cache.values()))
e = next(iter(g.cache.values()))
x = torch.randn_like(e.x)
torch.cuda.synchronize()
t0 = time.perf_counter()
for i in range(20):
    e.x.copy_(x)
    e.t.fill_(500.0)
    e.graph.replay()
torch.cuda.synchronize()
t_replay = time.perf_counter() - t0
print(f"20 bare graph replays: {t_replay*1e3:.1f} ms")

# stage timing
t0 = time.perf_counter()
c, u, _ = pipe.encode_prompts([req.prompt], [req.negative_prompt], 1)
torch.cuda.synchronize()
print(f"clip encode: {(time.perf_counter()-t0)*1e3:.1f} ms")
z = torch.randn(B, 4, 64, 64, device="cuda", dtype=torch.bfloat16)
t0 = time.perf_counter()
img = pipe.model.vae.decode(z)
torch.cuda.synchronize()
print(f"vae decode: {(time.perf_counter()-t0)*1e3:.1f} ms")
print(f"=> non-denoise overhead ~= {t_gen*1e3 - t_replay*1e3:.1f} ms "
      f"(incl. clip/vae/noise/uint8/host)")
