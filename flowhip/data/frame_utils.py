"""Flow / image file I/O — byte-identical formats to the reference
`core/utils/frame_utils.py` (Middlebury .flo, PFM, KITTI 16-bit PNG,
FlyingThings webp/npz), implemented on numpy + PIL (no OpenCV dependency).
"""

import re
from os.path import splitext

import numpy as np
from PIL import Image

TAG_CHAR = np.array([202021.25], np.float32)


def readFlow(fn):
    """Read a Middlebury .flo file (magic 202021.25, little-endian)."""
    with open(fn, "rb") as f:
        magic = np.fromfile(f, np.float32, count=1)
        if magic.size == 0 or magic[0] != 202021.25:
            print("Magic number incorrect. Invalid .flo file")
            return None
        w = int(np.fromfile(f, np.int32, count=1)[0])
        h = int(np.fromfile(f, np.int32, count=1)[0])
        data = np.fromfile(f, np.float32, count=2 * w * h)
        return np.resize(data, (h, w, 2))


def writeFlow(filename, uv, v=None):
    """Write a Middlebury .flo file (parity: frame_utils.py:70-99)."""
    if v is None:
        assert uv.ndim == 3 and uv.shape[2] == 2
        u = uv[:, :, 0]
        v = uv[:, :, 1]
    else:
        u = uv
    assert u.shape == v.shape
    height, width = u.shape
    with open(filename, "wb") as f:
        f.write(TAG_CHAR.tobytes())
        np.array(width, np.int32).tofile(f)
        np.array(height, np.int32).tofile(f)
        tmp = np.zeros((height, width * 2), np.float32)
        tmp[:, 0::2] = u
        tmp[:, 1::2] = v
        tmp.tofile(f)


def readPFM(file):
    """Read a PFM file (FlyingThings3D flow)."""
    with open(file, "rb") as f:
        header = f.readline().rstrip()
        if header == b"PF":
            color = True
        elif header == b"Pf":
            color = False
        else:
            raise Exception("Not a PFM file.")

        dim_match = re.match(rb"^(\d+)\s(\d+)\s$", f.readline())
        if not dim_match:
            raise Exception("Malformed PFM header.")
        width, height = map(int, dim_match.groups())

        scale = float(f.readline().rstrip())
        endian = "<" if scale < 0 else ">"

        data = np.fromfile(f, endian + "f")
    shape = (height, width, 3) if color else (height, width)
    return np.flipud(np.reshape(data, shape))


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
