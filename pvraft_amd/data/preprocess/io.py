"""Minimal readers for the raw FT3D / KITTI scene-flow formats.

Dependency-free (numpy + zlib): PFM (FlyingThings3D disparity), .flo
(Middlebury optical flow), and non-interlaced 8/16-bit grayscale/RGB PNG
(KITTI disparity/flow maps; PIL cannot decode 16-bit RGB, so PNG decoding
is done directly).  Capability parity with reference data_preprocess/IO.py
and python_pfm.py.
"""

from __future__ import annotations

import re
import struct
import zlib

import numpy as np

FLO_MAGIC = 202021.25


def read_pfm(path: str) -> np.ndarray:
    """PFM -> float32 array (H, W) or (H, W, 3); rows are stored bottom-up."""
    with open(path, "rb") as f:
        header = f.readline().rstrip()
        if header == b"PF":
            channels = 3
        elif header == b"Pf":
            channels = 1
        else:
            raise ValueError(f"{path}: not a PFM file")
        dims = f.readline()
        while dims.startswith(b"#"):
            dims = f.readline()
        m = re.match(rb"^(\d+)\s+(\d+)\s*$", dims)
        if not m:
            raise ValueError(f"{path}: malformed PFM dimensions")
        w, h = int(m.group(1)), int(m.group(2))
        scale = float(f.readline().rstrip())
        endian = "<" if scale < 0 else ">"
        data = np.frombuffer(f.read(), dtype=endian + "f4", count=w * h * channels)
    data = data.reshape(h, w, channels) if channels == 3 else data.reshape(h, w)
    return np.flipud(data).astype(np.float32)


def read_flo(path: str) -> np.ndarray:
    """Middlebury .flo -> float32 (H, W, 2)."""
    with open(path, "rb") as f:
        magic = struct.unpack("<f", f.read(4))[0]
        if abs(magic - FLO_MAGIC) > 1e-3:
            raise ValueError(f"{path}: bad .flo magic {magic}")
        w, h = struct.unpack("<ii", f.read(8))
        data = np.frombuffer(f.read(), dtype="<f4", count=w * h * 2)
    return data.reshape(h, w, 2).astype(np.float32)


_PNG_SIG = b"\x89PNG\r\n\x1a\n"


def read_png(path: str) -> np.ndarray:
    """Non-interlaced 8/16-bit grayscale or RGB PNG -> uint8/uint16 array."""
    with open(path, "rb") as f:
        raw = f.read()
    if raw[:8] != _PNG_SIG:
        raise ValueError(f"{path}: not a PNG")
    pos = 8
    idat = b""
    w = h = bitdepth = colortype = None
    while pos < len(raw):
        (length,) = struct.unpack(">I", raw[pos : pos + 4])
        ctype = raw[pos + 4 : pos + 8]
        data = raw[pos + 8 : pos + 8 + length]
        pos += 12 + length
        if ctype == b"IHDR":
            w, h, bitdepth, colortype, comp, filt, interlace = struct.unpack(">IIBBBBB", data)
            if interlace != 0:
                raise ValueError(f"{path}: interlaced PNG unsupported")
            if colortype not in (0, 2):
                raise ValueError(f"{path}: only grayscale/RGB PNG supported (got {colortype})")
        elif ctype == b"IDAT":
            idat += data
        elif ctype == b"IEND":
            break
    channels = 1 if colortype == 0 else 3
    sample_bytes = bitdepth // 8
    stride = w * channels * sample_bytes
    decomp = zlib.decompress(idat)

    out = np.empty((h, stride), dtype=np.uint8)
    bpp = channels * sample_bytes  # filter step
    prev = np.zeros(stride, dtype=np.uint8)
    for row in range(h):
        ftype = decomp[row * (stride + 1)]
        line = np.frombuffer(
            decomp, dtype=np.uint8, count=stride, offset=row * (stride + 1) + 1
        ).copy()
        if ftype == 0:
            pass
        elif ftype == 1:  # Sub
            for i in range(bpp, stride):
                line[i] = (line[i] + line[i - bpp]) & 0xFF
        elif ftype == 2:  # Up
            line = (line.astype(np.uint16) + prev).astype(np.uint8)
        elif ftype == 3:  # Average
            for i in range(stride):
                left = int(line[i - bpp]) if i >= bpp else 0
                line[i] = (line[i] + ((left + int(prev[i])) >> 1)) & 0xFF
        elif ftype == 4:  # Paeth
            for i in range(stride):
                a = int(line[i - bpp]) if i >= bpp else 0
                b = int(prev[i])
                c = int(prev[i - bpp]) if i >= bpp else 0
                p = a + b - c
                pa, pb, pc = abs(p - a), abs(p - b), abs(p - c)
                pred = a if (pa <= pb and pa <= pc) else (b if pb <= pc else c)
                line[i] = (line[i] + pred) & 0xFF
        else:
            raise ValueError(f"{path}: unknown PNG filter {ftype}")
        out[row] = line
        prev = line

    if bitdepth == 16:
        arr = out.reshape(h, w, channels, 2)
        arr = (arr[..., 0].astype(np.uint16) << 8) | arr[..., 1].astype(np.uint16)
    else:
        arr = out.reshape(h, w, channels).astype(np.uint8)
    return arr[..., 0] if channels == 1 else arr
