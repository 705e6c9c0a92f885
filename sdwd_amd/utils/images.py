"""Image utilities: grid assembly (ref world.py:588-591) and PNG encoding
(the reference shipped base64 PNGs over HTTP, distributed.py:103-106; here
PNG is only an output format). Pure python + zlib — no Pillow in the image.
"""
from __future__ import annotations

import math
import struct
import zlib
from typing import Optional

import torch


def make_grid(images: torch.Tensor, rows: Optional[int] = None) -> torch.Tensor:
    """[N,H,W,3] uint8 -> one [GH,GW,3] grid tensor (sdwui grid layout:
    near-square, row-major)."""
    n, h, w, c = images.shape
    if rows is None:
        rows = int(math.sqrt(n))
        rows = max(1, min(rows, n))
    cols = math.ceil(n / rows)
    grid = torch.zeros(rows * h, cols * w, c, dtype=torch.uint8)
    for i in range(n):
        r, col = divmod(i, cols)
        grid[r * h : (r + 1) * h, col * w : (col + 1) * w] = images[i]
    return grid


def encode_png(image: torch.Tensor, parameters: Optional[str] = None) -> bytes:
    """[H,W,3] uint8 -> PNG bytes (filter 0, zlib level 6).

    ``parameters`` is embedded as a tEXt chunk under the "parameters" key —
    the sdwui infotext convention, so downstream tools can read the
    prompt/seed/sampler back out of the file."""
    h, w, c = image.shape
    assert c == 3 and image.dtype == torch.uint8
    raw = image.contiguous().numpy().tobytes()
    stride = w * 3
    scanlines = b"".join(
        b"\x00" + raw[y * stride : (y + 1) * stride] for y in range(h)
    )

    def chunk(tag: bytes, payload: bytes) -> bytes:
        return (
            struct.pack(">I", len(payload))
            + tag
            + payload
            + struct.pack(">I", zlib.crc32(tag + payload) & 0xFFFFFFFF)
        )

    ihdr = struct.pack(">IIBBBBB", w, h, 8, 2, 0, 0, 0)
    text = b""
    if parameters:
        payload = b"parameters\x00" + parameters.encode("latin-1", "replace")
        text = chunk(b"tEXt", payload)
    return (
        b"\x89PNG\r\n\x1a\n"
        + chunk(b"IHDR", ihdr)
        + text
        + chunk(b"IDAT", zlib.compress(scanlines, 6))
        + chunk(b"IEND", b"")
    )


def png_parameters(data: bytes) -> Optional[str]:
    """Read back the tEXt "parameters" infotext, if present."""
    pos = 8
    while pos + 8 <= len(data):
        (length,) = struct.unpack(">I", data[pos : pos + 4])
        tag = data[pos + 4 : pos + 8]
        if tag == b"tEXt":
            payload = data[pos + 8 : pos + 8 + length]
            key, _, value = payload.partition(b"\x00")
            if key == b"parameters":
                return value.decode("latin-1")
        if tag == b"IEND":
            break
        pos += 12 + length
    return None


def color_correct(out: torch.Tensor, ref: torch.Tensor) -> torch.Tensor:
    """Match per-channel mean/std of ``out`` to ``ref`` ([H,W,3] uint8 each;
    sdwui's img2img color correction, approximated in RGB)."""
    o = out.float()
    r = ref.float()
    om = o.mean(dim=(0, 1))
    osd = o.std(dim=(0, 1)).clamp_min(1e-5)
    rm = r.mean(dim=(0, 1))
    rsd = r.std(dim=(0, 1))
    corrected = (o - om) / osd * rsd + rm
    return corrected.clamp(0, 255).to(torch.uint8)


def parse_infotext(text: str) -> dict:
    """Parse an sdwui 'parameters' infotext back into TYPED request fields
    (snake_case keys, ints/floats coerced) — built on the raw grammar
    parser in utils/infotext.py, which preserves sdwui key spelling and
    handles double-quoted values."""
    from .infotext import parse_infotext as parse_raw

    raw = parse_raw(text)
    out: dict = {
        "prompt": raw.pop("prompt", ""),
        "negative_prompt": raw.pop("negative_prompt", ""),
    }
    for key, val in raw.items():
        out[key.strip().lower().replace(" ", "_")] = val
    if "size" in out and "x" in str(out["size"]):
        w, h = out.pop("size").split("x")
        out["width"], out["height"] = int(w), int(h)
    for k in ("steps", "seed", "clip_skip"):
        if k in out:
            out[k] = int(out[k])
    for k in ("cfg_scale", "denoising_strength", "eta"):
        if k in out:
            out[k] = float(out[k])
    return out


def save_png(
    image: torch.Tensor, path: str, parameters: Optional[str] = None
) -> str:
    with open(path, "wb") as fh:
        fh.write(encode_png(image, parameters))
    return path


def decode_png(data: bytes) -> torch.Tensor:
    """PNG (8-bit RGB, filter 0/1/2/3/4) -> [H,W,3] uint8. Minimal reader
    for round-tripping our own encoder's output and API uploads."""
    assert data[:8] == b"\x89PNG\r\n\x1a\n", "not a PNG"
    pos = 8
    idat = b""
    w = h = 0
    while pos < len(data):
        (length,) = struct.unpack(">I", data[pos : pos + 4])
        tag = data[pos + 4 : pos + 8]
        payload = data[pos + 8 : pos + 8 + length]
        if tag == b"IHDR":
            w, h, depth, color = struct.unpack(">IIBB", payload[:10])
            if depth != 8 or color != 2:
                raise ValueError("only 8-bit RGB PNGs supported")
        elif tag == b"IDAT":
            idat += payload
        elif tag == b"IEND":
            break
        pos += 12 + length
    raw = zlib.decompress(idat)
    stride = w * 3
    out = bytearray(h * stride)
    prev = bytearray(stride)
    for y in range(h):
        f = raw[y * (stride + 1)]
        line = bytearray(raw[y * (stride + 1) + 1 : (y + 1) * (stride + 1)])
        if f == 1:  # sub
            for i in range(3, stride):
                line[i] = (line[i] + line[i - 3]) & 0xFF
        elif f == 2:  # up
            for i in range(stride):
                line[i] = (line[i] + prev[i]) & 0xFF
        elif f == 3:  # average
            for i in range(stride):
                a = line[i - 3] if i >= 3 else 0
                line[i] = (line[i] + ((a + prev[i]) >> 1)) & 0xFF
        elif f == 4:  # paeth
            for i in range(stride):
                a = line[i - 3] if i >= 3 else 0
                b = prev[i]
                cdiag = prev[i - 3] if i >= 3 else 0
                p = a + b - cdiag
                pa, pb, pc = abs(p - a), abs(p - b), abs(p - cdiag)
                pred = a if (pa <= pb and pa <= pc) else (b if pb <= pc else cdiag)
                line[i] = (line[i] + pred) & 0xFF
        out[y * stride : (y + 1) * stride] = line
        prev = line
    import numpy as np

    arr = np.frombuffer(bytes(out), dtype=np.uint8).reshape(h, w, 3)
    return torch.from_numpy(arr.copy())
