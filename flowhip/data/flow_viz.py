"""Optical-flow color-wheel visualization (Middlebury convention).

Behavioral parity with the reference `core/utils/flow_viz.py`:
`flow_to_color` (Baker et al. color wheel, :22-137) and the VCN-style
`flow_to_image` with UNKNOWN_FLOW_THRESH (:140-275). numpy only.
"""

import numpy as np

UNKNOWN_FLOW_THRESH = 1e7


def make_colorwheel():
    """55-color RY/YG/GC/CB/BM/MR wheel (Baker et al., ICCV 2007)."""
    RY, YG, GC, CB, BM, MR = 15, 6, 4, 11, 13, 6
    ncols = RY + YG + GC + CB + BM + MR
    colorwheel = np.zeros((ncols, 3))
    col = 0

    colorwheel[0:RY, 0] = 255
    colorwheel[0:RY, 1] = np.floor(255 * np.arange(0, RY) / RY)
    col += RY
    colorwheel[col:col + YG, 0] = 255 - np.floor(255 * np.arange(0, YG) / YG)
    colorwheel[col:col + YG, 1] = 255
    col += YG
    colorwheel[col:col + GC, 1] = 255
    colorwheel[col:col + GC, 2] = np.floor(255 * np.arange(0, GC) / GC)
    col += GC
    colorwheel[col:col + CB, 1] = 255 - np.floor(255 * np.arange(CB) / CB)
    colorwheel[col:col + CB, 2] = 255
    col += CB
    colorwheel[col:col + BM, 2] = 255
    colorwheel[col:col + BM, 0] = np.floor(255 * np.arange(0, BM) / BM)
    col += BM
    colorwheel[col:col + MR, 2] = 255 - np.floor(255 * np.arange(MR) / MR)
    colorwheel[col:col + MR, 0] = 255
    return colorwheel


def flow_compute_color(u, v, convert_to_bgr=False, legacy_offset=False):
    """Map normalized (u, v) to wheel colors.

    The reference ships two disagreeing Matlab ports: flow_to_color's
    (core/utils/flow_viz.py:88-92) keeps the 1-based `+1` offset and wraps
    k1 at ncols -> 1, rotating the wheel one slot; flow_to_image's
    compute_color (:216-227) is the faithful port (fk+1, wrap at ncols+1,
    indexed with k-1 — equivalent to the plain 0-based form). Both are
    reproduced byte-for-byte: `legacy_offset=True` selects the rotated
    flow_to_color variant (in range because callers normalize by
    rad_max + eps, so |a| < 1 strictly)."""
    flow_image = np.zeros((u.shape[0], u.shape[1], 3), np.uint8)
    colorwheel = make_colorwheel()
    ncols = colorwheel.shape[0]

    rad = np.sqrt(np.square(u) + np.square(v))
    a = np.arctan2(-v, -u) / np.pi

    fk = (a + 1) / 2 * (ncols - 1)
    if legacy_offset:
        fk = fk + 1
    k0 = np.floor(fk).astype(np.int32)
    k1 = k0 + 1
    k1[k1 == ncols] = 1 if legacy_offset else 0
    f = fk - k0
    if legacy_offset:
        k0 = k0 % ncols
        k1 = k1 % ncols

    for i in range(colorwheel.shape[1]):
        tmp = colorwheel[:, i]
        col0 = tmp[k0] / 255.0
        col1 = tmp[k1] / 255.0
        col = (1 - f) * col0 + f * col1

        idx = rad <= 1
        col[idx] = 1 - rad[idx] * (1 - col[idx])
        col[~idx] = col[~idx] * 0.75  # out of range

        ch_idx = 2 - i if convert_to_bgr else i
        flow_image[:, :, ch_idx] = np.floor(255 * col)
    return flow_image


def flow_to_color(flow_uv, clip_flow=None, convert_to_bgr=False):
    """HxWx2 flow -> HxWx3 uint8 color image, normalized by max radius."""
    assert flow_uv.ndim == 3 and flow_uv.shape[2] == 2
    if clip_flow is not None:
        flow_uv = np.clip(flow_uv, 0, clip_flow)

    u = flow_uv[:, :, 0]
    v = flow_uv[:, :, 1]
    rad = np.sqrt(np.square(u) + np.square(v))
    rad_max = np.max(rad)
    epsilon = 1e-5
    u = u / (rad_max + epsilon)
    v = v / (rad_max + epsilon)
    return flow_compute_color(u, v, convert_to_bgr, legacy_offset=True)


def flow_to_image(flow):
    """VCN-style variant with unknown-flow masking (reference :140-275)."""
    u = flow[:, :, 0]
    v = flow[:, :, 1]

    # non-finite values count as unknown flow (NaN would otherwise poison
    # the color-wheel index cast)
    idxUnknown = ((np.abs(u) > UNKNOWN_FLOW_THRESH)
                  | (np.abs(v) > UNKNOWN_FLOW_THRESH)
                  | ~np.isfinite(u) | ~np.isfinite(v))
    u = np.where(idxUnknown, 0.0, u)
    v = np.where(idxUnknown, 0.0, v)

    rad = np.sqrt(u ** 2 + v ** 2)
    maxrad = max(-1, np.max(rad))

    u = u / (maxrad + np.finfo(float).eps)
    v = v / (maxrad + np.finfo(float).eps)

    img = flow_compute_color(u, v)
    img[idxUnknown] = 0
    return np.uint8(img)


def make_color_wheel():
    """Name-compat with the reference's VCN-port naming (flow_viz.py:158):
    both reference ports build the identical 55-entry wheel."""
    return make_colorwheel()


def compute_color(u, v):
    """Name-compat with the reference's VCN port (flow_viz.py:199-237): the
    faithful 0-based wheel indexing — our default variant. The reference
    returns a float array of integral values; matched here."""
    return flow_compute_color(u, v).astype(np.float64)
