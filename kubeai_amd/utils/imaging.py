"""Minimal image decoding for multimodal requests — no PIL in the image.

Supports the content-part form the reference forwards verbatim to vLLM
(`/root/reference/api/openai/v1/chat_completions.go:350-515` keeps
image_url parts intact): `data:` URLs carrying PNG (8-bit gray/RGB/RGBA,
non-interlaced) or binary PPM (P6). The decoder is self-contained
(zlib + struct); anything else is rejected with a clear error so the
client sees a 400, not a silent drop.
"""
from __future__ import annotations

import base64
import binascii
import struct
import zlib

import torch

# CLIP normalization (the vision towers we ship are CLIP-ViT shaped)
_CLIP_MEAN = (0.48145466, 0.4578275, 0.40821073)
_CLIP_STD = (0.26862954, 0.26130258, 0.27577711)


class ImageError(ValueError):
    """Client-facing image decoding problem (maps to HTTP 400)."""


def _paeth(a: int, b: int, c: int) -> int:
    p = a + b - c
    pa, pb, pc = abs(p - a), abs(p - b), abs(p - c)
    if pa <= pb and pa <= pc:
        return a
    return b if pb <= pc else c


def decode_png(data: bytes) -> torch.Tensor:
    """8-bit non-interlaced PNG -> uint8 tensor [H, W, C]."""
    if data[:8] != b"\x89PNG\r\n\x1a\n":
        raise ImageError("not a PNG file")
    pos = 8
    width = height = None
    bit_depth = color_type = None
    idat = b""
    while pos + 8 <= len(data):
        (length,) = struct.unpack(">I", data[pos : pos + 4])
        ctype = data[pos + 4 : pos + 8]
        body = data[pos + 8 : pos + 8 + length]
        pos += 12 + length  # length + type + crc
        if ctype == b"IHDR":
            width, height, bit_depth, color_type, _comp, _filt, interlace = (
                struct.unpack(">IIBBBBB", body)
            )
            if bit_depth != 8:
                raise ImageError(f"unsupported PNG bit depth {bit_depth}")
            if interlace:
                raise ImageError("interlaced PNG not supported")
            if color_type not in (0, 2, 6):  # gray, RGB, RGBA
                raise ImageError(f"unsupported PNG color type {color_type}")
        elif ctype == b"IDAT":
            idat += body
        elif ctype == b"IEND":
            break
    if width is None:
        raise ImageError("PNG missing IHDR")
    channels = {0: 1, 2: 3, 6: 4}[color_type]
    try:
        raw = zlib.decompress(idat)
    except zlib.error as e:
        raise ImageError(f"PNG deflate stream corrupt: {e}") from None
    stride = width * channels
    if len(raw) != (stride + 1) * height:
        raise ImageError("PNG data size mismatch")
    out = bytearray(stride * height)
    prev = bytearray(stride)
    for y in range(height):
        row_start = y * (stride + 1)
        filt = raw[row_start]
        row = bytearray(raw[row_start + 1 : row_start + 1 + stride])
        if filt == 0:
            pass
        elif filt == 1:  # sub
            for i in range(channels, stride):
                row[i] = (row[i] + row[i - channels]) & 0xFF
        elif filt == 2:  # up
            for i in range(stride):
                row[i] = (row[i] + prev[i]) & 0xFF
        elif filt == 3:  # average
            for i in range(stride):
                left = row[i - channels] if i >= channels else 0
                row[i] = (row[i] + ((left + prev[i]) >> 1)) & 0xFF
        elif filt == 4:  # paeth
            for i in range(stride):
                left = row[i - channels] if i >= channels else 0
                pl = prev[i - channels] if i >= channels else 0
                row[i] = (row[i] + _paeth(left, prev[i], pl)) & 0xFF
        else:
            raise ImageError(f"unknown PNG filter {filt}")
        out[y * stride : (y + 1) * stride] = row
        prev = row
    t = torch.frombuffer(bytes(out), dtype=torch.uint8)
    return t.reshape(height, width, channels)


def decode_ppm(data: bytes) -> torch.Tensor:
    """Binary PPM (P6, maxval 255) -> uint8 tensor [H, W, 3]."""
    if not data.startswith(b"P6"):
        raise ImageError("not a binary PPM (P6)")
    # header: P6 <w> <h> <maxval> with comments allowed
    fields: list[int] = []
    pos = 2
    while len(fields) < 3 and pos < len(data):
        while pos < len(data) and data[pos : pos + 1].isspace():
            pos += 1
        if data[pos : pos + 1] == b"#":
            while pos < len(data) and data[pos] != 0x0A:
                pos += 1
            continue
        start = pos
        while pos < len(data) and not data[pos : pos + 1].isspace():
            pos += 1
        fields.append(int(data[start:pos]))
    pos += 1  # single whitespace after maxval
    w, h, maxval = fields
    if maxval != 255:
        raise ImageError("PPM maxval must be 255")
    body = data[pos : pos + w * h * 3]
    if len(body) != w * h * 3:
        raise ImageError("PPM data truncated")
    return torch.frombuffer(body, dtype=torch.uint8).reshape(h, w, 3)


def decode_image_bytes(data: bytes) -> torch.Tensor:
    """Sniff the container and decode to uint8 [H, W, C]."""
    if data[:8] == b"\x89PNG\r\n\x1a\n":
        return decode_png(data)
    if data[:2] == b"P6":
        return decode_ppm(data)
    if data[:2] == b"\xff\xd8":
        raise ImageError(
            "JPEG is not supported by the in-house engine; send PNG "
            "(or use a vLLM-engine model for JPEG inputs)"
        )
    raise ImageError("unrecognized image format (PNG and binary PPM supported)")


def image_from_url(url: str) -> tuple[torch.Tensor, bytes]:
    """OpenAI image_url content part -> (uint8 [H,W,C], raw bytes).

    Only data: URLs are accepted — the engine has no egress, same as the
    reference's network-isolated deployments.
    """
    if not url.startswith("data:"):
        raise ImageError("only data: image URLs are supported (no egress)")
    header, _, payload = url.partition(",")
    if not payload:
        raise ImageError("malformed data URL")
    try:
        raw = (
            base64.b64decode(payload, validate=True)
            if ";base64" in header
            else payload.encode("latin-1")
        )
    except (binascii.Error, UnicodeEncodeError) as e:
        raise ImageError(f"bad data URL payload: {e}") from None
    return decode_image_bytes(raw), raw


def preprocess(img: torch.Tensor, image_size: int) -> torch.Tensor:
    """uint8 [H,W,C] -> CLIP-normalized float32 [3, S, S]."""
    if img.ndim != 3:
        raise ImageError("expected HWC image")
    c = img.shape[2]
    x = img.float().permute(2, 0, 1) / 255.0  # [C,H,W]
    if c == 1:
        x = x.expand(3, -1, -1).contiguous()
    elif c == 4:
        x = x[:3]  # drop alpha
    elif c != 3:
        raise ImageError(f"unsupported channel count {c}")
    x = torch.nn.functional.interpolate(
        x.unsqueeze(0), size=(image_size, image_size), mode="bilinear",
        align_corners=False, antialias=True,
    ).squeeze(0)
    mean = torch.tensor(_CLIP_MEAN).view(3, 1, 1)
    std = torch.tensor(_CLIP_STD).view(3, 1, 1)
    return (x - mean) / std


def encode_ppm(img: torch.Tensor) -> bytes:
    """uint8 [H,W,3] -> binary PPM bytes (test/client helper)."""
    h, w, c = img.shape
    assert c == 3
    return b"P6\n%d %d\n255\n" % (w, h) + bytes(img.reshape(-1).tolist())


def encode_png(img: torch.Tensor) -> bytes:
    """uint8 [H,W,C] -> PNG bytes, filter 0 only (test/client helper)."""
    h, w, c = img.shape
    color_type = {1: 0, 3: 2, 4: 6}[c]

    def chunk(ctype: bytes, body: bytes) -> bytes:
        return (
            struct.pack(">I", len(body))
            + ctype
            + body
            + struct.pack(">I", zlib.crc32(ctype + body) & 0xFFFFFFFF)
        )

    ihdr = struct.pack(">IIBBBBB", w, h, 8, color_type, 0, 0, 0)
    rows = b"".join(
        b"\x00" + bytes(img[y].reshape(-1).tolist()) for y in range(h)
    )
    return (
        b"\x89PNG\r\n\x1a\n"
        + chunk(b"IHDR", ihdr)
        + chunk(b"IDAT", zlib.compress(rows))
        + chunk(b"IEND", b"")
    )
