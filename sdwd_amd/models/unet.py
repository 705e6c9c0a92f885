"""UNet denoiser — the hot path the reference delegated to remote webui
instances per POST /txt2img (SURVEY.md §2.5 derives this compute surface).

Architecture-compatible with SD1.5 (model_channels 320, channel_mult
[1,2,4,4], 2 res blocks, spatial transformers at downsample 1/2/4, context
768, 8 heads) and SDXL-base (channel_mult [1,2,4], transformer depths
[0,2,10], context 2048, additional pooled/size conditioning), parameterized
so tiny CPU-test configs use the same code.

GPU hot ops route through sdwd_amd.ops: fused GroupNorm+SiLU, flash
attention (MFMA), GEGLU, fused LayerNorm. Projections/convs use
hipBLASLt/MIOpen via torch (library GEMMs are allowed; convs get
hand-written implicit-GEMM kernels in a later pass).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.nn as nn

from .. import ops
from .layers import GEGLU, FusedGroupNorm, FusedLayerNorm, SDConv2d


@dataclass
class UNetConfig:
    in_channels: int = 4
    out_channels: int = 4
    model_channels: int = 320
    channel_mult: List[int] = field(default_factory=lambda: [1, 2, 4, 4])
    num_res_blocks: int = 2
    # transformer depth per level; 0 = no attention at that level
    transformer_depth: List[int] = field(default_factory=lambda: [1, 1, 1, 0])
    context_dim: int = 768
    num_heads: int = 8
    groups: int = 32
    # SDXL extras: dimension of the pooled+size conditioning vector (0 = off)
    adm_in_channels: int = 0

    @classmethod
    def sd15(cls) -> "UNetConfig":
        return cls()

    @classmethod
    def sdxl(cls) -> "UNetConfig":
        return cls(
            model_channels=320,
            channel_mult=[1, 2, 4],
            transformer_depth=[0, 2, 10],
            context_dim=2048,
            num_heads=0,  # 0 -> fixed head_dim 64 (SDXL convention)
            adm_in_channels=2816,
        )

    @classmethod
    def tiny(cls) -> "UNetConfig":
        """CPU-test config (BASELINE config #1 plumbing scale)."""
        return cls(
            model_channels=32,
            channel_mult=[1, 2],
            num_res_blocks=1,
            transformer_depth=[1, 1],
            context_dim=64,
            num_heads=2,
            groups=8,
        )

    def heads_for(self, channels: int) -> int:
        if self.num_heads > 0:
            return self.num_heads
        return max(1, channels // 64)  # SDXL: head_dim 64


class ResBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, emb_ch: int, groups: int):
        super().__init__()
        self.norm1 = FusedGroupNorm(in_ch, groups, silu=True)
        self.conv1 = SDConv2d(in_ch, out_ch, 3, padding=1)
        self.emb_proj = nn.Linear(emb_ch, out_ch)
        self.norm2 = FusedGroupNorm(out_ch, groups, silu=True)
        self.conv2 = SDConv2d(out_ch, out_ch, 3, padding=1)
        self.skip = (
            SDConv2d(in_ch, out_ch, 1) if in_ch != out_ch else nn.Identity()
        )

    def forward(self, x: torch.Tensor, emb: torch.Tensor) -> torch.Tensor:
        # the time-embedding channel bias rides conv1's epilogue; the
        # skip-residual add rides conv2's (both fused on GPU)
        emb_b = self.emb_proj(ops.silu(emb))
        # collect_gn: the conv epilogues emit the NEXT GroupNorm's partial
        # sums (norm2 consumes conv1's; the next block's norm1 consumes
        # conv2's when the tensor reaches it unmodified)
        h = self.conv1(self.norm1(x), chan_bias=emb_b, collect_gn=True)
        skip = self.skip(x)
        return self.conv2(self.norm2(h), residual=skip, collect_gn=True)


class RegionalContext:
    """Attention-couple regional prompting (the Regional Prompter
    extension's matrix mode, executed natively — the reference only
    FORWARDED its payload, distributed.py:199-234).

    Wraps the normal conditioning rows (`plain`, [N,77,C]) plus R
    per-region contexts ([R,77,C]) and R spatial masks on the latent
    grid. Cross-attention evaluates `plain` for every row as usual, then
    re-blends the first `rows` rows (the cond rows) from per-region
    attention outputs weighted by the masks pooled to that resolution:

        out = (base + (1-base)*(1-cover)) * attn(base_ctx)
              + (1-base) * sum_r mask_r * attn(ctx_r)

    so uncovered pixels and `base_ratio` keep the shared base prompt.
    Only cross-attention interprets the wrapper; everything else treats
    conditioning opaquely, and hipGraph wrappers fall back to eager."""

    def __init__(self, plain, region_ctx, masks, rows, base_ratio, lat_hw):
        self.plain = plain            # [N, 77, C]
        self.region_ctx = region_ctx  # [R, 77, C]
        self.masks = masks            # [R, lat_h, lat_w] float, sum<=1/pixel
        self.rows = int(rows)         # leading rows blended regionally
        self.base_ratio = float(base_ratio)
        self.lat_hw = lat_hw          # latent grid the masks are drawn on
        self._cache = {}              # seq_len -> [R, s] pooled masks

    def __getitem__(self, sl):
        plain = self.plain[sl]
        out = RegionalContext(
            plain, self.region_ctx, self.masks,
            min(self.rows, plain.shape[0]), self.base_ratio, self.lat_hw,
        )
        out._cache = self._cache
        return out

    @property
    def shape(self):
        return self.plain.shape

    def to(self, *a, **kw):
        out = RegionalContext(
            self.plain.to(*a, **kw), self.region_ctx.to(*a, **kw),
            self.masks, self.rows, self.base_ratio, self.lat_hw,
        )
        out._cache = self._cache
        return out

    def masks_for(self, s: int, device, dtype) -> torch.Tensor:
        """[R, s] masks at the attention resolution with h*w == s (the
        UNet halves the even latent grid per level)."""
        got = self._cache.get(s)
        if got is not None and got.device == device:
            return got
        lh, lw = self.lat_hw
        k = 0
        while k < 6 and (lh >> k) * (lw >> k) != s:
            k += 1
        h, w = max(1, lh >> k), max(1, lw >> k)
        if h * w != s:  # non-halving grid: nearest square-ish fallback
            import math

            h = max(1, int(round(math.sqrt(s * lh / max(lw, 1)))))
            while s % h:
                h -= 1
            w = s // h
        m = torch.nn.functional.adaptive_avg_pool2d(
            self.masks[None].float(), (h, w)
        )[0]
        m = m.reshape(m.shape[0], s).to(device=device, dtype=torch.float32)
        self._cache[s] = m
        return m


class CrossAttention(nn.Module):
    def __init__(self, query_dim: int, context_dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.d_head = query_dim // heads
        self.to_q = nn.Linear(query_dim, query_dim, bias=False)
        self.to_k = nn.Linear(context_dim, query_dim, bias=False)
        self.to_v = nn.Linear(context_dim, query_dim, bias=False)
        self.to_out = nn.Linear(query_dim, query_dim)

    def _wcat(self, names):
        """Concatenated projection weight, cached by source data_ptrs AND
        in-place versions (LoRA applies with add_/copy_, which keeps the
        pointer — keying on the pointer alone served a stale concat and
        silently dropped LoRA deltas on q/k/v): one GEMM instead of 2-3
        reads the (huge, M=B*S) activation once."""
        ws = [getattr(self, n).weight for n in names]
        key = tuple((w.data_ptr(), w._version) for w in ws)
        cache = getattr(self, "_wcat_cache", None)
        if cache is None or cache[0] != key:
            if cache is not None and cache[1].shape[0] == sum(
                w.shape[0] for w in ws
            ):
                # refresh IN PLACE: captured hipGraphs baked this buffer's
                # address, so a reallocation would leave graph replays on
                # the pre-update weights while eager code moved on
                cat = cache[1]
                torch.cat([w.detach() for w in ws], dim=0, out=cat)
            else:
                cat = torch.cat([w.detach() for w in ws], dim=0)
            self._wcat_cache = cache = (key, cat, tuple(names))
        return cache[1]


    def forward(
        self, x: torch.Tensor, context: Optional[torch.Tensor] = None
    ) -> torch.Tensor:
        import torch.nn.functional as F

        b, s, d = x.shape
        hd = (self.heads, self.d_head)
        if context is None:
            # self-attention: one fused QKV GEMM (the skinny K=channels
            # projections are A-traffic-bound at S=4096); the q/k/v slices
            # stay strided views - the flash kernel reads strides directly
            qkv = F.linear(x, self._wcat(("to_q", "to_k", "to_v")))
            q = qkv[..., :d].unflatten(-1, hd)
            k = qkv[..., d:2 * d].unflatten(-1, hd)
            v = qkv[..., 2 * d:].unflatten(-1, hd)
        elif torch.is_tensor(context):
            q = self.to_q(x).unflatten(-1, hd)
            kv = F.linear(context, self._wcat(("to_k", "to_v")))
            k = kv[..., :d].unflatten(-1, hd)
            v = kv[..., d:].unflatten(-1, hd)
        else:  # RegionalContext: base attention + masked per-region blend
            rc = context
            wkv = self._wcat(("to_k", "to_v"))
            q = self.to_q(x).unflatten(-1, hd)
            kv = F.linear(rc.plain, wkv)
            k = kv[..., :d].unflatten(-1, hd)
            v = kv[..., d:].unflatten(-1, hd)
            out = ops.attention_bshd(q, k, v).reshape(b, s, d)
            nr = rc.rows
            if nr > 0 and rc.region_ctx is not None:
                m = rc.masks_for(s, x.device, x.dtype)  # [R, s]
                qr = q[:nr]
                acc = None
                for r in range(rc.region_ctx.shape[0]):
                    kvr = F.linear(
                        rc.region_ctx[r : r + 1].expand(nr, -1, -1), wkv
                    )
                    kr = kvr[..., :d].unflatten(-1, hd)
                    vr = kvr[..., d:].unflatten(-1, hd)
                    orr = ops.attention_bshd(qr, kr, vr).reshape(nr, s, d)
                    term = orr * m[r][None, :, None]
                    acc = term if acc is None else acc + term
                bw = rc.base_ratio
                keep = (bw + (1.0 - bw) * (1.0 - m.sum(0).clamp(0, 1)))[
                    None, :, None
                ]
                blended = (
                    out[:nr].float() * keep + (1.0 - bw) * acc.float()
                ).to(out.dtype)
                out = torch.cat([blended, out[nr:]], dim=0)
            return self.to_out(out)
        out = ops.attention_bshd(q, k, v)
        return self.to_out(out.reshape(b, s, d))



def refresh_fused_projections(module: torch.nn.Module) -> None:
    """Re-run every CrossAttention's fused-projection concat after an
    in-place weight update (LoRA set_active). The refresh writes into the
    SAME storage, so hipGraphs that baked the buffer's address replay the
    updated weights; without this, graphed generations would keep the
    previous LoRA state on q/k/v."""
    for mod in module.modules():
        cache = getattr(mod, "_wcat_cache", None)
        if cache is not None:
            mod._wcat(cache[2])


class BasicTransformerBlock(nn.Module):
    def __init__(self, dim: int, context_dim: int, heads: int):
        super().__init__()
        self.norm1 = FusedLayerNorm(dim)
        self.attn1 = CrossAttention(dim, dim, heads)  # self
        self.norm2 = FusedLayerNorm(dim)
        self.attn2 = CrossAttention(dim, context_dim, heads)  # cross
        self.norm3 = FusedLayerNorm(dim)
        self.ff = nn.Sequential(GEGLU(dim, dim * 4), nn.Linear(dim * 4, dim))

    def forward(self, x, context):
        from .layers import _fp32_cached

        h = self.attn1(self.norm1(x))
        if x.is_cuda:
            w2, b2 = _fp32_cached(self.norm2)
            w3, b3 = _fp32_cached(self.norm3)
        else:
            w2, b2 = self.norm2.weight, self.norm2.bias
            w3, b3 = self.norm3.weight, self.norm3.bias
        # fused residual-add + pre-norm (one HBM round trip saved per hop)
        x, n2 = ops.add_layer_norm(x, h, w2, b2, self.norm2.eps)
        h = self.attn2(n2, context)
        x, n3 = ops.add_layer_norm(x, h, w3, b3, self.norm3.eps)
        return x + self.ff(n3)


class SpatialTransformer(nn.Module):
    def __init__(
        self, channels: int, context_dim: int, heads: int, depth: int, groups: int
    ):
        super().__init__()
        self.norm = FusedGroupNorm(channels, groups, silu=False)
        self.proj_in = nn.Linear(channels, channels)
        self.blocks = nn.ModuleList(
            BasicTransformerBlock(channels, context_dim, heads)
            for _ in range(depth)
        )
        self.proj_out = nn.Linear(channels, channels)

    def forward(self, x: torch.Tensor, context: torch.Tensor) -> torch.Tensor:
        b, c, h, w = x.shape
        residual = x
        x = self.norm(x)
        x = x.permute(0, 2, 3, 1).reshape(b, h * w, c)
        x = self.proj_in(x)
        for blk in self.blocks:
            x = blk(x, context)
        x = self.proj_out(x)
        x = x.reshape(b, h, w, c).permute(0, 3, 1, 2)
        return x + residual


class Downsample(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv = SDConv2d(ch, ch, 3, stride=2, padding=1)

    def forward(self, x):
        return self.conv(x)


class Upsample(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv = SDConv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        return self.conv.forward_upsampled2x(x)


class _Seq(nn.Module):
    """Runs children, feeding emb to ResBlocks and context to transformers."""

    def __init__(self, *mods: nn.Module):
        super().__init__()
        self.mods = nn.ModuleList(mods)

    def forward(self, x, emb, context):
        for m in self.mods:
            if isinstance(m, ResBlock):
                x = m(x, emb)
            elif isinstance(m, SpatialTransformer):
                x = m(x, context)
            else:
                x = m(x)
        return x


class UNetModel(nn.Module):
    def __init__(self, cfg: UNetConfig):
        super().__init__()
        self.cfg = cfg
        ch = cfg.model_channels
        time_dim = ch * 4
        self.time_mlp = nn.Sequential(
            nn.Linear(ch, time_dim), nn.SiLU(), nn.Linear(time_dim, time_dim)
        )
        if cfg.adm_in_channels:
            self.label_mlp = nn.Sequential(
                nn.Linear(cfg.adm_in_channels, time_dim),
                nn.SiLU(),
                nn.Linear(time_dim, time_dim),
            )
        else:
            self.label_mlp = None

        self.conv_in = SDConv2d(cfg.in_channels, ch, 3, padding=1)

        self.down = nn.ModuleList()
        skip_chs = [ch]
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
                skip_chs.append(cur)
            if lvl != levels - 1:
                self.down.append(_Seq(Downsample(cur)))
                skip_chs.append(cur)

        mid_depth = cfg.transformer_depth[-1] or 1
        self.mid = _Seq(
            ResBlock(cur, cur, time_dim, cfg.groups),
            SpatialTransformer(
                cur, cfg.context_dim, cfg.heads_for(cur), mid_depth, cfg.groups
            ),
            ResBlock(cur, cur, time_dim, cfg.groups),
        )

        self.up = nn.ModuleList()
        for lvl in reversed(range(levels)):
            out_ch = ch * cfg.channel_mult[lvl]
            for i in range(cfg.num_res_blocks + 1):
                skip = skip_chs.pop()
                mods = [ResBlock(cur + skip, out_ch, time_dim, cfg.groups)]
                cur = out_ch
                depth = cfg.transformer_depth[lvl]
                if depth > 0:
                    mods.append(
                        SpatialTransformer(
                            cur, cfg.context_dim, cfg.heads_for(cur), depth,
                            cfg.groups,
                        )
                    )
                if lvl != 0 and i == cfg.num_res_blocks:
                    mods.append(Upsample(cur))
                self.up.append(_Seq(*mods))

        self.norm_out = FusedGroupNorm(cur, cfg.groups, silu=True)
        self.conv_out = SDConv2d(cur, cfg.out_channels, 3, padding=1)

    def forward(
        self,
        x: torch.Tensor,
        timesteps: torch.Tensor,
        context: torch.Tensor,
        y: Optional[torch.Tensor] = None,
        control: Optional[dict] = None,
    ) -> torch.Tensor:
        """control: {"down": [residual per skip], "mid": residual} from a
        ControlNetModel (models/controlnet.py)."""
        temb = ops.timestep_embedding(timesteps, self.cfg.model_channels)
        emb = self.time_mlp(temb.to(x.dtype))
        if self.label_mlp is not None and y is not None:
            emb = emb + self.label_mlp(y.to(x.dtype))

        h = self.conv_in(x)
        skips = [h]
        for blk in self.down:
            h = blk(h, emb, context)
            skips.append(h)
        # cldm semantics: the mid block sees the UNMODIFIED last down
        # activation; down residuals apply only at the up-path skip-concat
        # and the mid residual after the mid block.
        h = self.mid(h, emb, context)
        if control is not None:
            h = h + control["mid"]
            skips = [s + c for s, c in zip(skips, control["down"])]
        for blk in self.up:
            h = ops.cat_channels_gn(h, skips.pop())
            h = blk(h, emb, context)
        return self.conv_out(self.norm_out(h))
