"""ControlNet for the SD UNet (ref C17: control_net.py packed CN payloads
for remote execution; this engine executes ControlNet natively).

Standard architecture: a trainable copy of the UNet encoder + mid block, a
small hint encoder that maps the (pixel-space) control image to the latent
resolution, and zero-initialised 1x1 convs on every skip output. The UNet
consumes the residuals via its ``control`` argument.
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn

from .. import ops
from .layers import SDConv2d
from .unet import ResBlock, SpatialTransformer, Downsample, UNetConfig, _Seq


def zero_conv(ch: int) -> nn.Conv2d:
    conv = SDConv2d(ch, ch, 1)
    nn.init.zeros_(conv.weight)
    nn.init.zeros_(conv.bias)
    return conv


class HintEncoder(nn.Module):
    """[B,3,H,W] control image -> [B, model_channels, H/8, W/8]."""

    def __init__(self, model_channels: int, factor: int = 8):
        super().__init__()
        chans = [16, 32, 96, 256]
        layers: List[nn.Module] = [SDConv2d(3, chans[0], 3, padding=1)]
        cur = chans[0]
        n_stride = {8: 3, 4: 2, 2: 1, 1: 0}.get(factor, 3)
        for i in range(3):
            nxt = chans[min(i + 1, len(chans) - 1)]
            stride = 2 if i < n_stride else 1
            layers += [nn.SiLU(), SDConv2d(cur, nxt, 3, padding=1, stride=stride)]
            cur = nxt
        layers += [nn.SiLU(), zero_conv_out(cur, model_channels)]
        self.body = nn.Sequential(*layers)

    def forward(self, hint: torch.Tensor) -> torch.Tensor:
        return self.body(hint)


def zero_conv_out(in_ch: int, out_ch: int) -> nn.Conv2d:
    conv = SDConv2d(in_ch, out_ch, 3, padding=1)
    nn.init.zeros_(conv.weight)
    nn.init.zeros_(conv.bias)
    return conv


class ControlNetModel(nn.Module):
    def __init__(self, cfg: UNetConfig, hint_factor: int = 8):
        super().__init__()
        self.cfg = cfg
        self.hint_factor = hint_factor
        ch = cfg.model_channels
        time_dim = ch * 4
        self.time_mlp = nn.Sequential(
            nn.Linear(ch, time_dim), nn.SiLU(), nn.Linear(time_dim, time_dim)
        )
        self.conv_in = SDConv2d(cfg.in_channels, ch, 3, padding=1)
        self.hint_encoder = HintEncoder(ch, hint_factor)

        self.down = nn.ModuleList()
        self.zero_convs = nn.ModuleList([zero_conv(ch)])
        cur = ch
        levels = len(cfg.channel_mult)
        for lvl, mult in enumerate(cfg.channel_mult):
            out_ch = ch * mult
            for _ in range(cfg.num_res_blocks):
                mods: List[nn.Module] = [
                    ResBlock(cur, out_ch, time_dim, cfg.groups)
                ]
                cur = out_ch
                depth = cfg.transformer_depth[lvl]
                if depth > 0:
                    mods.append(
                        SpatialTransformer(
                            cur, cfg.context_dim, cfg.heads_for(cur), depth,
                            cfg.groups,
                        )
                    )
                self.down.append(_Seq(*mods))
                self.zero_convs.append(zero_conv(cur))
            if lvl != levels - 1:
                self.down.append(_Seq(Downsample(cur)))
                self.zero_convs.append(zero_conv(cur))

        mid_depth = cfg.transformer_depth[-1] or 1
        self.mid = _Seq(
            ResBlock(cur, cur, time_dim, cfg.groups),
            SpatialTransformer(
                cur, cfg.context_dim, cfg.heads_for(cur), mid_depth, cfg.groups
            ),
            ResBlock(cur, cur, time_dim, cfg.groups),
        )
        self.mid_zero = zero_conv(cur)

    def forward(
        self,
        x: torch.Tensor,
        hint: torch.Tensor,
        timesteps: torch.Tensor,
        context: torch.Tensor,
        conditioning_scale: float = 1.0,
    ) -> Dict[str, object]:
        """Returns {"down": [skip residuals...], "mid": residual}."""
        temb = ops.timestep_embedding(timesteps, self.cfg.model_channels)
        emb = self.time_mlp(temb.to(x.dtype))
        h = self.conv_in(x) + self.hint_encoder(hint)
        outs = [self.zero_convs[0](h)]
        for blk, zc in zip(self.down, list(self.zero_convs)[1:]):
            h = blk(h, emb, context)
            outs.append(zc(h))
        h = self.mid(h, emb, context)
        mid = self.mid_zero(h)
        if conditioning_scale != 1.0:
            outs = [o * conditioning_scale for o in outs]
            mid = mid * conditioning_scale
        return {"down": outs, "mid": mid}
