"""Flow / image file I/O — byte-identical formats to the reference
`core/utils/frame_utils.py` (Middlebury .flo, PFM, KITTI 16-bit PNG,
FlyingThings webp/npz), implemented on numpy + PIL (no OpenCV dependency).
"""

from os.path import splitext

import numpy as np
from PIL import Image

# Middlebury .flo layout: 4-byte float sanity tag, then little-endian
# int32 width, int32 height, then h*w (u, v) float32 pairs row-major.
FLO_TAG = 202021.25


def readFlow(fn):
    """Read a Middlebury .flo file into an HxWx2 float32 array."""
    with open(fn, "rb") as f:
        head = f.read(12)
    if len(head) < 12 or np.frombuffer(head, "<f4", 1)[0] != FLO_TAG:
        print("Magic number incorrect. Invalid .flo file")
        return None
    w, h = np.frombuffer(head, "<i4", 2, offset=4)
    body = np.fromfile(fn, "<f4", offset=12)
    return body[:2 * w * h].reshape(int(h), int(w), 2)


def writeFlow(filename, uv, v=None):
    """Write a Middlebury .flo file. Accepts either one HxWx2 array or
    separate u, v planes (reference API — frame_utils.py:70-99)."""
    if v is not None:
        uv = np.stack([uv, v], axis=-1)
    assert uv.ndim == 3 and uv.shape[2] == 2
    h, w = uv.shape[:2]
    with open(filename, "wb") as f:
        f.write(np.float32(FLO_TAG).tobytes())
        f.write(np.asarray([w, h], "<i4").tobytes())
        # HWC float32 is already the interleaved (u, v)-per-pixel layout
        f.write(np.ascontiguousarray(uv, "<f4").tobytes())


def readPFM(file):
    """Read a PFM image (FlyingThings3D flow GT). Header = type line
    ('PF' color / 'Pf' gray), dimensions line, scale line (sign encodes
    endianness); pixel rows are stored bottom-to-top."""
    with open(file, "rb") as f:
        kind = f.readline().strip()
        if kind not in (b"PF", b"Pf"):
            raise ValueError(f"{file}: not a PFM file")
        dims = f.readline().split()
        if len(dims) != 2:
            raise ValueError(f"{file}: malformed PFM dimensions")
        w, h = (int(d) for d in dims)
        scale = float(f.readline())
        dtype = np.dtype("<f4" if scale < 0 else ">f4")
        channels = 3 if kind == b"PF" else 1
        pixels = np.fromfile(f, dtype, w * h * channels)
    img = pixels.reshape((h, w, 3) if channels == 3 else (h, w))
    return img[::-1]


def _read_png16_bgr(filename):
    """Read a 16-bit RGB png returning float32 HxWx3 (u, v, valid channels in
    the KITTI encoding). PIL converts 16-bit RGB to 8-bit silently, so the
    PNG is decoded manually (IHDR/IDAT parse + scanline unfiltering)."""
    import struct
    import zlib

    with open(filename, "rb") as f:
        data = f.read()
    assert data[:8] == b"\x89PNG\r\n\x1a\n", f"{filename}: not a png"

    pos, w = 8, None
    idat = []
    while pos < len(data):
        (length,) = struct.unpack(">I", data[pos:pos + 4])
        tag = data[pos + 4:pos + 8]
        chunk = data[pos + 8:pos + 8 + length]
        if tag == b"IHDR":
            w, h, depth, ctype = struct.unpack(">IIBB", chunk[:10])
            assert depth == 16 and ctype == 2, \
                f"{filename}: expected 16-bit RGB, got depth={depth} type={ctype}"
        elif tag == b"IDAT":
            idat.append(chunk)
        elif tag == b"IEND":
            break
        pos += 12 + length

    raw = zlib.decompress(b"".join(idat))
    bpp = 6  # 3 channels x 2 bytes
    stride = w * bpp
    out = np.empty((h, stride), dtype=np.uint8)
    prev = np.zeros(stride, dtype=np.uint8)
    off = 0
    for y in range(h):
        ftype = raw[off]
        line = np.frombuffer(raw[off + 1:off + 1 + stride], dtype=np.uint8).copy()
        off += 1 + stride
        if ftype == 0:
            pass
        elif ftype == 2:  # Up
            line += prev
        elif ftype in (1, 3, 4):  # Sub / Average / Paeth need sequential pass
            line = line.astype(np.int32)
            pr = prev.astype(np.int32)
            rec = np.zeros(stride, dtype=np.int32)
            for i in range(stride):
                a = rec[i - bpp] if i >= bpp else 0
                b = pr[i]
                c = pr[i - bpp] if i >= bpp else 0
                if ftype == 1:
                    pred = a
                elif ftype == 3:
                    pred = (a + b) // 2
                else:
                    p = a + b - c
                    pa, pb, pc = abs(p - a), abs(p - b), abs(p - c)
                    pred = a if (pa <= pb and pa <= pc) else (b if pb <= pc else c)
                rec[i] = (line[i] + pred) & 0xFF
            line = rec.astype(np.uint8)
        else:
            raise ValueError(f"{filename}: unsupported png filter {ftype}")
        out[y] = line
        prev = out[y]

    arr = out.reshape(h, w, 3, 2)
    vals = arr[..., 0].astype(np.uint16) << 8 | arr[..., 1]
    return vals.astype(np.float32)


def readFlowKITTI(filename):
    """KITTI flow png: 16-bit RGB where flow = (value - 2^15)/64, third
    channel is the valid mask (parity: frame_utils.py:102-107; the reference
    reads BGR via cv2 then reverses to RGB — PIL reads RGB directly)."""
    flow = _read_png16_bgr(filename)
    flow, valid = flow[:, :, :2], flow[:, :, 2]
    flow = (flow - 2 ** 15) / 64.0
    return flow, valid


def readDispKITTI(filename):
    disp = np.array(Image.open(filename)).astype(np.float32) / 256.0
    valid = disp > 0.0
    flow = np.stack([-disp, np.zeros_like(disp)], -1)
    return flow, valid


def writeFlowKITTI(filename, uv):
    """Write KITTI 16-bit flow png (parity: frame_utils.py:116-120)."""
    uv = 64.0 * uv + 2 ** 15
    valid = np.ones([uv.shape[0], uv.shape[1], 1])
    uv = np.concatenate([uv, valid], axis=-1).astype(np.uint16)
    # PIL has no native 16-bit RGB writer; encode the PNG manually.
    _write_png16_rgb(filename, uv)


def _write_png16_rgb(filename, arr):
    """Minimal 16-bit RGB PNG encoder (zlib, no filtering)."""
    import struct
    import zlib

    h, w, c = arr.shape
    assert c == 3 and arr.dtype == np.uint16
    raw = b"".join(b"\x00" + arr[i].astype(">u2").tobytes() for i in range(h))

    def chunk(tag, data):
        block = tag + data
        return (struct.pack(">I", len(data)) + block
                + struct.pack(">I", zlib.crc32(block) & 0xFFFFFFFF))

    ihdr = struct.pack(">IIBBBBB", w, h, 16, 2, 0, 0, 0)
    png = (b"\x89PNG\r\n\x1a\n" + chunk(b"IHDR", ihdr)
           + chunk(b"IDAT", zlib.compress(raw, 6)) + chunk(b"IEND", b""))
    with open(filename, "wb") as f:
        f.write(png)


def read_gen(file_name, pil=False):
    """Dispatch on extension (parity: frame_utils.py:123-139)."""
    ext = splitext(file_name)[-1]
    if ext in (".png", ".jpeg", ".ppm", ".jpg", ".webp"):
        return Image.open(file_name)
    if ext in (".bin", ".raw"):
        return np.load(file_name)
    if ext == ".flo":
        return readFlow(file_name).astype(np.float32)
    if ext == ".pfm":
        flow = readPFM(file_name).astype(np.float32)
        return flow if len(flow.shape) == 2 else flow[:, :, :-1]
    if ext == ".npz":
        return np.load(file_name)["optical_flow"].astype(np.float32).transpose(1, 2, 0)
    return []
