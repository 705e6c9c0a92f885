"""VAE encoder/decoder (AutoencoderKL-shaped) — the latent<->pixel stage of
the delegated pipeline (decoder for txt2img, encoder+decoder
for img2img).

SD shape: base 128 channels, mult [1,2,4,4], 2 res blocks per level,
attention mid-block, 4-channel latents, scale factor 0.18215.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

import torch
import torch.nn as nn

from .. import ops
from .layers import FusedGroupNorm, SDConv2d

SD_VAE_SCALE = 0.18215
SDXL_VAE_SCALE = 0.13025


@dataclass
class VAEConfig:
    base_channels: int = 128
    channel_mult: List[int] = field(default_factory=lambda: [1, 2, 4, 4])
    num_res_blocks: int = 2
    latent_channels: int = 4
    groups: int = 32
    scale_factor: float = SD_VAE_SCALE

    @property
    def downsample_factor(self) -> int:
        return 2 ** (len(self.channel_mult) - 1)

    @classmethod
    def sd(cls) -> "VAEConfig":
        return cls()

    @classmethod
    def tiny(cls) -> "VAEConfig":
        return cls(
            base_channels=16,
            channel_mult=[1, 2],
            num_res_blocks=1,
            groups=8,
        )


class VAEResBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, groups: int):
        super().__init__()
        self.norm1 = FusedGroupNorm(in_ch, groups, silu=True)
        self.conv1 = SDConv2d(in_ch, out_ch, 3, padding=1)
        self.norm2 = FusedGroupNorm(out_ch, groups, silu=True)
        self.conv2 = SDConv2d(out_ch, out_ch, 3, padding=1)
        self.skip = (
            SDConv2d(in_ch, out_ch, 1) if in_ch != out_ch else nn.Identity()
        )

    def forward(self, x):
        h = self.conv1(self.norm1(x), collect_gn=True)
        return self.conv2(
            self.norm2(h), residual=self.skip(x), collect_gn=True
        )


class VAEAttention(nn.Module):
    """Single-head spatial attention mid-block (seq = H*W, one head)."""

    def __init__(self, ch: int, groups: int):
        super().__init__()
        self.norm = FusedGroupNorm(ch, groups, silu=False)
        self.q = nn.Linear(ch, ch)
        self.k = nn.Linear(ch, ch)
        self.v = nn.Linear(ch, ch)
        self.out = nn.Linear(ch, ch)

    def forward(self, x):
        b, c, h, w = x.shape
        n = self.norm(x).permute(0, 2, 3, 1).reshape(b, h * w, c)
        q = self.q(n)[:, None]  # [b, 1(head), s, c]
        k = self.k(n)[:, None]
        v = self.v(n)[:, None]
        o = ops.attention(q, k, v)[:, 0]
        o = self.out(o).reshape(b, h, w, c).permute(0, 3, 1, 2)
        return x + o


class VAEDownsample(nn.Module):
    """ldm VAE downsample: zero pad (0,1,0,1) then a stride-2 valid conv.
    The asymmetric pad shifts the sampling grid half a latent pixel vs a
    symmetric pad-1 conv, so real checkpoint weights only reproduce the
    reference's latents with this exact arithmetic.

    Implemented as a stride-1 pad-1 conv subsampled at odd indices —
    output i then reads rows 2i..2i+2 with a zero row/col past the edge,
    which is the identical computation, and it keeps the conv on the
    pad-1 paths (the HIP kernel is pad-1-only, and MIOpen's pad-0
    stride-2 NHWC bf16 fallback raises on small channel counts)."""

    def __init__(self, ch: int):
        super().__init__()
        self.conv = SDConv2d(ch, ch, 3, stride=1, padding=1)

    def forward(self, x):
        out = self.conv(x)[:, :, 1::2, 1::2]
        # the strided view is non-contiguous; the fused GN kernel (and the
        # NHWC conv path) need a packed layout
        if out.is_cuda:
            return out.contiguous(memory_format=torch.channels_last)
        return out.contiguous()


class VAEEncoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = cfg.base_channels
        self.conv_in = SDConv2d(3, ch, 3, padding=1)
        blocks: List[nn.Module] = []
        cur = ch
        for lvl, mult in enumerate(cfg.channel_mult):
            out_ch = ch * mult
            for _ in range(cfg.num_res_blocks):
                blocks.append(VAEResBlock(cur, out_ch, cfg.groups))
                cur = out_ch
            if lvl != len(cfg.channel_mult) - 1:
                blocks.append(VAEDownsample(cur))
        self.blocks = nn.ModuleList(blocks)
        self.mid = nn.ModuleList(
            [
                VAEResBlock(cur, cur, cfg.groups),
                VAEAttention(cur, cfg.groups),
                VAEResBlock(cur, cur, cfg.groups),
            ]
        )
        self.norm_out = FusedGroupNorm(cur, cfg.groups, silu=True)
        self.conv_out = SDConv2d(cur, 2 * cfg.latent_channels, 3, padding=1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """[B,3,H,W] -> moments [B, 2*latent, H/8, W/8]."""
        h = self.conv_in(x)
        for blk in self.blocks:
            h = blk(h)
        for blk in self.mid:
            h = blk(h)
        return self.conv_out(self.norm_out(h))


class VAEDecoder(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        ch = cfg.base_channels
        cur = ch * cfg.channel_mult[-1]
        self.conv_in = SDConv2d(cfg.latent_channels, cur, 3, padding=1)
        self.mid = nn.ModuleList(
            [
                VAEResBlock(cur, cur, cfg.groups),
                VAEAttention(cur, cfg.groups),
                VAEResBlock(cur, cur, cfg.groups),
            ]
        )
        blocks: List[nn.Module] = []
        for lvl in reversed(range(len(cfg.channel_mult))):
            out_ch = ch * cfg.channel_mult[lvl]
            for _ in range(cfg.num_res_blocks + 1):
                blocks.append(VAEResBlock(cur, out_ch, cfg.groups))
                cur = out_ch
            if lvl != 0:
                blocks.append(_Up(cur))
        self.blocks = nn.ModuleList(blocks)
        self.norm_out = FusedGroupNorm(cur, cfg.groups, silu=True)
        self.conv_out = SDConv2d(cur, 3, 3, padding=1)

    def forward(self, z: torch.Tensor) -> torch.Tensor:
        """latents [B,4,h,w] -> pixels [B,3,8h,8w] in [-1,1]."""
        h = self.conv_in(z)
        for blk in self.mid:
            h = blk(h)
        for blk in self.blocks:
            h = blk(h)
        return self.conv_out(self.norm_out(h))


class _Up(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv = SDConv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        return self.conv.forward_upsampled2x(x)


class AutoencoderKL(nn.Module):
    def __init__(self, cfg: VAEConfig):
        super().__init__()
        self.cfg = cfg
        self.encoder = VAEEncoder(cfg)
        self.decoder = VAEDecoder(cfg)

    def encode(self, x: torch.Tensor, sample: bool = True,
               generator: torch.Generator | None = None) -> torch.Tensor:
        moments = self.encoder(x)
        mean, logvar = moments.chunk(2, dim=1)
        if not sample:
            return mean * self.cfg.scale_factor
        logvar = logvar.clamp(-30.0, 20.0)
        std = torch.exp(0.5 * logvar.float()).to(mean.dtype)
        noise = torch.randn(
            mean.shape, generator=generator, device=mean.device,
            dtype=torch.float32,
        ).to(mean.dtype)
        return (mean + std * noise) * self.cfg.scale_factor

    def decode(self, z: torch.Tensor) -> torch.Tensor:
        return self.decoder(z / self.cfg.scale_factor)
