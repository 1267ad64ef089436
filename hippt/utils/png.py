"""Minimal PNG read/write using only the standard library (zlib/struct).

Capability parity: the reference uses stb_image/stb_image_write for texture
loading and render output (src/impl/textures.cu:35-49,141-169).  This module
covers RGB/RGBA 8-bit PNGs, which is all the framework emits and consumes
offline (HDR envmaps use .npy).
"""
from __future__ import annotations

import struct
import zlib

import numpy as np


def write_png(path: str, img: np.ndarray) -> None:
    """img: (h, w, 3|4) uint8 or float in [0,1]."""
    if img.dtype != np.uint8:
        img = (np.clip(img, 0.0, 1.0) * 255.0 + 0.5).astype(np.uint8)
    h, w = img.shape[:2]
    if img.ndim == 2:
        img = np.repeat(img[:, :, None], 3, axis=2)
    ch = img.shape[2]
    color_type = {3: 2, 4: 6}[ch]
    raw = b"".join(b"\x00" + img[y].tobytes() for y in range(h))

    def chunk(tag: bytes, data: bytes) -> bytes:
        return (struct.pack(">I", len(data)) + tag + data +
                struct.pack(">I", zlib.crc32(tag + data) & 0xFFFFFFFF))

    with open(path, "wb") as f:
        f.write(b"\x89PNG\r\n\x1a\n")
        f.write(chunk(b"IHDR", struct.pack(">IIBBBBB", w, h, 8, color_type, 0, 0, 0)))
        f.write(chunk(b"IDAT", zlib.compress(raw, 6)))
        f.write(chunk(b"IEND", b""))


def read_png(path: str) -> np.ndarray:
    """Returns (h, w, 4) float32 in [0,1] (alpha=1 if absent)."""
    with open(path, "rb") as f:
        data = f.read()
    assert data[:8] == b"\x89PNG\r\n\x1a\n", "not a PNG"
    pos = 8
    w = h = bit_depth = color_type = None
    idat = b""
    palette = None
    while pos < len(data):
        (length,) = struct.unpack(">I", data[pos:pos + 4])
        tag = data[pos + 4:pos + 8]
        payload = data[pos + 8:pos + 8 + length]
        if tag == b"IHDR":
            w, h, bit_depth, color_type, _, _, interlace = struct.unpack(">IIBBBBB", payload)
            assert bit_depth == 8 and interlace == 0, "only 8-bit non-interlaced PNG"
        elif tag == b"PLTE":
            palette = np.frombuffer(payload, np.uint8).reshape(-1, 3)
        elif tag == b"IDAT":
            idat += payload
        elif tag == b"IEND":
            break
        pos += 12 + length
    raw = zlib.decompress(idat)
    nch = {0: 1, 2: 3, 3: 1, 4: 2, 6: 4}[color_type]
    stride = w * nch
    img = np.zeros((h, stride), np.uint8)
    prev = np.zeros(stride, np.int32)
    off = 0
    for y in range(h):
        ft = raw[off]
        line = np.frombuffer(raw[off + 1:off + 1 + stride], np.uint8).astype(np.int32)
        off += 1 + stride
        if ft == 0:
            cur = line
        elif ft == 1:
            cur = line.copy()
            for i in range(nch, stride):
                cur[i] = (cur[i] + cur[i - nch]) & 0xFF
        elif ft == 2:
            cur = (line + prev) & 0xFF
        elif ft == 3:
            cur = line.copy()
            for i in range(stride):
                a = cur[i - nch] if i >= nch else 0
                cur[i] = (cur[i] + ((a + prev[i]) >> 1)) & 0xFF
        elif ft == 4:
            cur = line.copy()
            for i in range(stride):
                a = cur[i - nch] if i >= nch else 0
                b = prev[i]
                c = prev[i - nch] if i >= nch else 0
                p = a + b - c
                pa, pb, pc = abs(p - a), abs(p - b), abs(p - c)
                pred = a if (pa <= pb and pa <= pc) else (b if pb <= pc else c)
                cur[i] = (cur[i] + pred) & 0xFF
        else:
            raise ValueError(f"bad filter {ft}")
        img[y] = cur.astype(np.uint8)
        prev = cur
    img = img.reshape(h, w, nch)
    if color_type == 3:
        img = palette[img[:, :, 0]]
        nch = 3
    out = np.ones((h, w, 4), np.float32)
    if nch == 1:
        out[:, :, :3] = img.astype(np.float32) / 255.0
    elif nch == 2:
        out[:, :, :3] = img[:, :, :1].astype(np.float32) / 255.0
        out[:, :, 3] = img[:, :, 1].astype(np.float32) / 255.0
    else:
        out[:, :, :img.shape[2]] = img.astype(np.float32) / 255.0
    return out


def tonemap(accum: np.ndarray, gamma: float = 2.1,
            exposure: float = 1.0) -> np.ndarray:
    """accum: (h,w,4) radiance sums + counts -> (h,w,3) uint8 (reference
    DeviceImage::export_cpu + to_int gamma 1/2.1).  `exposure` scales
    radiance before gamma (bright-light scenes like the reference's
    bunny.xml, scaler 50, clip at exposure 1)."""
    cnt = np.maximum(accum[:, :, 3:4], 1e-9)
    rgb = accum[:, :, :3] / cnt * exposure
    rgb = np.clip(rgb, 0.0, None) ** (1.0 / gamma)
    return (np.clip(rgb, 0.0, 1.0) * 255.0 + 0.5).astype(np.uint8)
