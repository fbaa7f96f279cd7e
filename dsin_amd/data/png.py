"""Minimal pure-Python PNG codec (no PIL in this environment).

Supports what the KITTI pipeline needs: 8-bit grayscale/RGB/RGBA decode
(all five scanline filters) and RGB encode. Replaces tf.image.decode_png /
the PIL save path of the reference (src/DataProvider.py:23-30,
src/utils.py:102-111).
"""

from __future__ import annotations

import struct
import zlib

import numpy as np

_SIG = b"\x89PNG\r\n\x1a\n"


def read_png(path: str) -> np.ndarray:
    """Returns (H, W, C) uint8 with C in {1, 3, 4}."""
    with open(path, "rb") as f:
        data = f.read()
    if data[:8] != _SIG:
        raise ValueError(f"{path}: not a PNG")
    pos = 8
    width = height = bitdepth = colortype = None
    idat = []
    palette = None
    while pos < len(data):
        (length,) = struct.unpack(">I", data[pos:pos + 4])
        ctype = data[pos + 4:pos + 8]
        chunk = data[pos + 8:pos + 8 + length]
        pos += 12 + length
        if ctype == b"IHDR":
            width, height, bitdepth, colortype, comp, filt, interlace = \
                struct.unpack(">IIBBBBB", chunk)
            if bitdepth != 8 or interlace != 0:
                raise NotImplementedError(f"{path}: bitdepth={bitdepth} interlace={interlace}")
        elif ctype == b"PLTE":
            palette = np.frombuffer(chunk, np.uint8).reshape(-1, 3)
        elif ctype == b"IDAT":
            idat.append(chunk)
        elif ctype == b"IEND":
            break
    raw = zlib.decompress(b"".join(idat))
    nch = {0: 1, 2: 3, 3: 1, 4: 2, 6: 4}[colortype]
    stride = width * nch
    out = np.empty((height, stride), np.uint8)
    prev = np.zeros(stride, np.int32)
    pos = 0
    for y in range(height):
        ftype = raw[pos]
        line = np.frombuffer(raw, np.uint8, stride, pos + 1).astype(np.int32)
        pos += 1 + stride
        if ftype == 0:
            cur = line
        elif ftype == 1:  # Sub
            cur = line.copy()
            for i in range(nch, stride):
                cur[i] = (cur[i] + cur[i - nch]) & 0xFF
        elif ftype == 2:  # Up
            cur = (line + prev) & 0xFF
        elif ftype == 3:  # Average
            cur = line.copy()
            for i in range(stride):
                left = cur[i - nch] if i >= nch else 0
                cur[i] = (cur[i] + ((left + prev[i]) >> 1)) & 0xFF
        elif ftype == 4:  # Paeth
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
            raise ValueError(f"{path}: bad filter {ftype}")
        out[y] = cur.astype(np.uint8)
        prev = cur
    img = out.reshape(height, width, nch)
    if colortype == 3:
        if palette is None:
            raise ValueError(f"{path}: paletted PNG without PLTE")
        img = palette[img[..., 0]]
    return img


def write_png(path: str, img: np.ndarray, compress_level: int = 6) -> None:
    """img: (H, W) or (H, W, C) uint8 with C in {1, 3, 4}."""
    img = np.asarray(img)
    if img.dtype != np.uint8:
        img = np.clip(img, 0, 255).astype(np.uint8)
    if img.ndim == 2:
        img = img[:, :, None]
    h, w, c = img.shape
    colortype = {1: 0, 3: 2, 4: 6}[c]
    raw = bytearray()
    for y in range(h):
        raw.append(0)  # filter: None
        raw.extend(img[y].tobytes())
    compressed = zlib.compress(bytes(raw), compress_level)

    def chunk(ctype: bytes, payload: bytes) -> bytes:
        return (struct.pack(">I", len(payload)) + ctype + payload
                + struct.pack(">I", zlib.crc32(ctype + payload) & 0xFFFFFFFF))

    ihdr = struct.pack(">IIBBBBB", w, h, 8, colortype, 0, 0, 0)
    with open(path, "wb") as f:
        f.write(_SIG + chunk(b"IHDR", ihdr) + chunk(b"IDAT", compressed)
                + chunk(b"IEND", b""))
