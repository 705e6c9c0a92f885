"""Parameter-holding wrappers around the fused ops (ops/__init__.py)."""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


def _fp32_cached(mod: nn.Module):
    """fp32 copies of (weight, bias), re-cast only when the params change
    (by data_ptr): saves thousands of tiny cast kernels per generation."""
    key = (mod.weight.data_ptr(), mod.bias.data_ptr(),
           mod.weight._version, mod.bias._version)
    cache = getattr(mod, "_fp32_cache", None)
    if cache is None or cache[0] != key:
        cache = (key, mod.weight.detach().float(),
                 mod.bias.detach().float())
        mod._fp32_cache = cache
    return cache[1], cache[2]


class FusedGroupNorm(nn.Module):
    """GroupNorm with an optionally fused SiLU (one HIP kernel on GPU)."""

    def __init__(self, channels: int, groups: int = 32, silu: bool = False,
                 eps: float = 1e-5):
        super().__init__()
        self.groups = groups
        self.silu = silu
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w, b = (_fp32_cached(self) if x.is_cuda
                else (self.weight, self.bias))
        return ops.group_norm_silu(
            x, w, b, self.groups, self.eps, self.silu
        )


class FusedLayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w, b = (_fp32_cached(self) if x.is_cuda
                else (self.weight, self.bias))
        return ops.layer_norm(x, w, b, self.eps)


class GEGLU(nn.Module):
    def __init__(self, dim_in: int, dim_out: int):
        super().__init__()
        self.proj = nn.Linear(dim_in, dim_out * 2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.geglu(self.proj(x))


class SDConv2d(nn.Conv2d):
    """Conv2d dispatching to the MI355X-native paths:

    * 3x3 pad-1 stride-1/2 with Cin % 64 == 0, bf16 channels_last on GPU ->
      the implicit-GEMM MFMA kernel (ops/hip/conv.hip), with bias and an
      optional residual add fused into the epilogue;
    * 1x1 stride-1 on GPU -> a hipBLASLt GEMM over the NHWC rows;
    * everything else (stem convs, CPU) -> F.conv2d (MIOpen / native).
    """

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._wprep_cache = None

    def _wprep(self) -> torch.Tensor:
        w = self.weight
        cache = self._wprep_cache
        if (
            cache is None
            or cache[0] != (w.data_ptr(), w._version)
            or cache[1] != w.dtype
        ):
            prep = w.permute(0, 2, 3, 1).contiguous()  # [Cout,3,3,Cin]
            self._wprep_cache = ((w.data_ptr(), w._version), w.dtype, prep)
            return prep
        return cache[2]

    def forward_upsampled2x(self, x: torch.Tensor) -> torch.Tensor:
        """conv(nearest2x(x)) with the upsample folded into the conv's
        im2col addressing on the GPU path (no 4x intermediate)."""
        if (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and self.kernel_size == (3, 3)
            and self.padding == (1, 1)
            and self.stride == (1, 1)
            and self.groups == 1
            and self.in_channels % 64 == 0
            and not getattr(self, "circular", False)
        ):
            from .. import ops

            xc = x.contiguous(memory_format=torch.channels_last)
            return ops.ups2x_conv3x3(
                xc, self._wprep(), self.bias, collect_gn=True
            )
        x = torch.nn.functional.interpolate(
            x, scale_factor=2, mode="nearest"
        )
        return self(x)

    def forward(  # type: ignore[override]
        self,
        x: torch.Tensor,
        residual: torch.Tensor | None = None,
        chan_bias: torch.Tensor | None = None,
        collect_gn: bool = False,
    ) -> torch.Tensor:
        import torch.nn.functional as F

        from .. import ops

        k = self.kernel_size
        if getattr(self, "circular", False) and self.padding != (0, 0):
            # seamless-tiling mode (sdwui "Tiling"): wrap-around padding;
            # runs the library conv (the HIP kernel is zero-pad only)
            ph, pw = self.padding
            xp = F.pad(x, (pw, pw, ph, ph), mode="circular")
            out = F.conv2d(
                xp, self.weight, self.bias, self.stride, 0,
                self.dilation, self.groups,
            )
            if chan_bias is not None:
                out = out + chan_bias.to(out.dtype)[:, :, None, None]
            if residual is not None:
                out = out + residual
            return out
        if x.is_cuda and x.dtype == torch.bfloat16:
            if (
                k == (3, 3)
                and self.padding == (1, 1)
                and self.stride[0] in (1, 2)
                and self.stride[0] == self.stride[1]
                and self.in_channels % 64 == 0
                and self.groups == 1
            ):
                xc = x.contiguous(memory_format=torch.channels_last)
                res = None
                if residual is not None:
                    res = residual.contiguous(
                        memory_format=torch.channels_last
                    )
                return ops.conv3x3(
                    xc, self._wprep(), self.bias, res, self.stride[0],
                    chan_bias, collect_gn=collect_gn,
                )
            if (
                k == (3, 3)
                and self.padding == (1, 1)
                and self.stride[0] in (1, 2)
                and self.stride[0] == self.stride[1]
                and self.groups == 1
                and residual is None
                and chan_bias is None
                and ops.conv3x3_small_supported(
                    self.in_channels, self.out_channels
                )
            ):
                # stem/IO convs (Cin 3/4/9): native direct kernel instead
                # of MIOpen's fallback solvers
                xc = x.contiguous(memory_format=torch.channels_last)
                return ops.conv3x3_small(
                    xc, self._wprep(), self.bias, self.stride[0]
                )
            if k == (1, 1) and self.stride == (1, 1) and self.groups == 1:
                xc = x.contiguous(memory_format=torch.channels_last)
                n, c, h, w = xc.shape
                rows = xc.permute(0, 2, 3, 1).reshape(n * h * w, c)
                out = F.linear(
                    rows, self.weight.reshape(self.out_channels, c), self.bias
                )
                out = (
                    out.reshape(n, h, w, self.out_channels)
                    .permute(0, 3, 1, 2)
                )
                if residual is not None:
                    out = out + residual
                return out
        out = F.conv2d(
            x, self.weight, self.bias, self.stride, self.padding,
            self.dilation, self.groups,
        )
        if chan_bias is not None:
            out = out + chan_bias.to(out.dtype)[:, :, None, None]
        if residual is not None:
            out = out + residual
        return out
