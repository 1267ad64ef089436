"""Assemble rendered PNG frames into an animated GIF (stdlib only).

Capability parity: reference scripts/video.py (png->jpg/video via external
tools); headless nodes here have no ffmpeg, so we emit GIF89a directly.

Usage: python scripts/video.py 'serial_out/*.png' -o out.gif --fps 8
"""
import argparse
import glob
import os
import struct
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from hippt.utils.png import read_png  # noqa: E402


def quantize(img):
    """(h,w,3) float -> 6x7x6 color cube palette indices."""
    r = np.clip(img[:, :, 0] * 5.999, 0, 5).astype(np.uint8)
    g = np.clip(img[:, :, 1] * 6.999, 0, 6).astype(np.uint8)
    b = np.clip(img[:, :, 2] * 5.999, 0, 5).astype(np.uint8)
    return (r * 42 + g * 6 + b).astype(np.uint8)


def palette():
    pal = []
    for r in range(6):
        for g in range(7):
            for b in range(6):
                pal += [int(r * 255 / 5), int(g * 255 / 6), int(b * 255 / 5)]
    pal += [0, 0, 0] * (256 - 252)
    return bytes(pal)


def lzw_encode(data, min_code_size=8):
    clear, end = 1 << min_code_size, (1 << min_code_size) + 1
    dict_size = end + 1
    table = {bytes([i]): i for i in range(1 << min_code_size)}
    out_bits = []
    code_size = min_code_size + 1

    def emit(code):
        out_bits.append((code, code_size))

    emit(clear)
    w = b""
    for ch in data:
        wc = w + bytes([ch])
        if wc in table:
            w = wc
        else:
            emit(table[w])
            if dict_size < 4096:
                table[wc] = dict_size
                dict_size += 1
                if dict_size > (1 << code_size) and code_size < 12:
                    code_size += 1
            else:
                emit(clear)
                table = {bytes([i]): i for i in range(1 << min_code_size)}
                dict_size = end + 1
                code_size = min_code_size + 1
            w = bytes([ch])
    if w:
        emit(table[w])
    emit(end)
    # pack bits LSB-first
    buf = bytearray()
    acc = n = 0
    for code, size in out_bits:
        acc |= code << n
        n += size
        while n >= 8:
            buf.append(acc & 0xFF)
            acc >>= 8
            n -= 8
    if n:
        buf.append(acc & 0xFF)
    return bytes(buf)


def write_gif(path, frames, fps=8):
    h, w = frames[0].shape[:2]
    delay = max(2, int(100 / fps))
    with open(path, "wb") as f:
        f.write(b"GIF89a")
        f.write(struct.pack("<HHBBB", w, h, 0xF7, 0, 0))
        f.write(palette())
        f.write(b"\x21\xFF\x0BNETSCAPE2.0\x03\x01\x00\x00\x00")  # loop forever
        for fr in frames:
            idx = quantize(fr)
            f.write(b"\x21\xF9\x04\x04" + struct.pack("<H", delay) + b"\x00\x00")
            f.write(b"\x2C" + struct.pack("<HHHHB", 0, 0, w, h, 0))
            f.write(b"\x08")  # LZW min code size
            data = lzw_encode(idx.reshape(-1))
            for i in range(0, len(data), 255):
                chunk = data[i:i + 255]
                f.write(bytes([len(chunk)]) + chunk)
            f.write(b"\x00")
        f.write(b"\x3B")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("pattern")
    ap.add_argument("-o", "--output", default="out.gif")
    ap.add_argument("--fps", type=float, default=8)
    args = ap.parse_args()
    files = sorted(glob.glob(args.pattern))
    if not files:
        raise SystemExit(f"no frames match {args.pattern}")
    frames = [read_png(f)[:, :, :3] for f in files]
    write_gif(args.output, frames, args.fps)
    print(f"wrote {args.output} ({len(frames)} frames)")


if __name__ == "__main__":
    main()
