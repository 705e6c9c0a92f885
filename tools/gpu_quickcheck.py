"""Targeted GPU validation of recent changes (VAEDownsample pad path,
UniPC / DPM++ 2S a samplers, refiner handoff) at small scale."""
import torch

from sdwd_amd.pipeline import PipelineRequest, StableDiffusionPipeline

pipe = StableDiffusionPipeline("sd15", device="cuda:0")
img = torch.randint(0, 255, (2, 512, 512, 3), dtype=torch.uint8)
lat = pipe.encode_image(img, seeds=[1, 2])  # VAEDownsample asym pad on GPU
assert lat.shape == (2, 4, 64, 64) and torch.isfinite(lat.float()).all()
print("encode ok", lat.float().abs().mean().item())
for name in ("UniPC", "DPM++ 2S a", "Euler a"):
    res = pipe.generate(PipelineRequest(
        prompt="q", steps=3, width=512, height=512, seeds=[1, 2],
        sampler_name=name, init_latents=lat if name == "Euler a" else None,
        denoising_strength=0.7,
    ))
    assert res.images.shape == (2, 512, 512, 3), name
    assert torch.isfinite(res.images.float()).all(), name
    assert res.images.float().std() > 1.0, name
    print(name, "ok")
res = pipe.generate(PipelineRequest(
    prompt="q", steps=4, width=512, height=512, seeds=[5],
    refiner_model="sd15", refiner_switch_at=0.5,
))
assert torch.isfinite(res.images.float()).all()
print("refiner ok")
print("ALL GPU QUICKCHECKS PASSED")
