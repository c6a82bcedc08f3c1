"""Cross-checks of the torch reference ops (flowhip/ops/torch_ref.py)
against independent loop-based NumPy implementations on fuzzed small
shapes. torch_ref is the numerical oracle for every HIP kernel, so it gets
its own independent verification here (SURVEY.md §4.2 item 2 contracts,
written from the math, not from either implementation).
"""

import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

from flowhip.ops import torch_ref


def naive_corr_volume(f1, f2):
    """C[b, i, j] = <f1[b,:,i], f2[b,:,j]> / sqrt(D), loops only."""
    B, D, H, W = f1.shape
    P = H * W
    out = np.zeros((B, P, P), dtype=np.float64)
    f1f = f1.reshape(B, D, P)
    f2f = f2.reshape(B, D, P)
    for b in range(B):
        for i in range(P):
            for j in range(P):
                out[b, i, j] = np.dot(f1f[b, :, i], f2f[b, :, j])
    return out / np.sqrt(D)


def naive_bilinear(map2d, x, y):
    """align_corners=True grid_sample semantics on pixel coords with zero
    padding, scalar implementation."""
    H, W = map2d.shape
    x0, y0 = int(np.floor(x)), int(np.floor(y))
    wx1, wy1 = x - x0, y - y0
    v = 0.0
    for (xi, wx) in ((x0, 1 - wx1), (x0 + 1, wx1)):
        for (yi, wy) in ((y0, 1 - wy1), (y0 + 1, wy1)):
            if 0 <= xi < W and 0 <= yi < H:
                v += wx * wy * map2d[yi, xi]
    return v


@settings(max_examples=10, deadline=None)
@given(st.integers(2, 4), st.integers(2, 4), st.integers(2, 5),
       st.integers(0, 2 ** 31 - 1))
def test_corr_volume_matches_naive(h, w, d, seed):
    rng = np.random.default_rng(seed)
    f1 = rng.standard_normal((1, d, h, w)).astype(np.float32)
    f2 = rng.standard_normal((1, d, h, w)).astype(np.float32)
    got = torch_ref.corr_volume(torch.from_numpy(f1), torch.from_numpy(f2))
    # torch_ref returns (B*P, 1, H, W): volume C[i] viewed as image-2 maps
    got = got.reshape(1, h * w, h * w).numpy()
    ref = naive_corr_volume(f1, f2)
    np.testing.assert_allclose(got, ref, atol=1e-5)


@settings(max_examples=10, deadline=None)
@given(st.integers(6, 10), st.integers(6, 12), st.integers(0, 2 ** 31 - 1))
def test_corr_lookup_matches_naive_taps(h, w, seed):
    """Every output channel of the fused lookup equals a scalar bilinear
    sample at coords/2^l + (dx, dy) — x-offset-major channel order."""
    rng = np.random.default_rng(seed)
    radius, levels = 2, 2
    P = h * w
    pyramid = [torch.from_numpy(
        rng.standard_normal((P, 1, h >> l, w >> l)).astype(np.float32))
        for l in range(levels)]
    coords = torch.from_numpy(
        (rng.random((1, 2, h, w)) * [[[[w]], [[h]]]]).astype(np.float32))

    out = torch_ref.corr_lookup(pyramid, coords, radius).numpy()
    K = 2 * radius + 1

    for _ in range(12):  # spot-check random taps
        i = rng.integers(P)
        l = rng.integers(levels)
        a = rng.integers(K)
        c = rng.integers(K)
        y, x = divmod(int(i), w)
        cx = coords[0, 0, y, x].item() / (1 << l) + (a - radius)
        cy = coords[0, 1, y, x].item() / (1 << l) + (c - radius)
        ref = naive_bilinear(pyramid[l][i, 0].numpy(), cx, cy)
        ch = l * K * K + a * K + c
        np.testing.assert_allclose(out[0, ch, y, x], ref, atol=1e-5)


@settings(max_examples=10, deadline=None)
@given(st.integers(3, 8), st.integers(3, 8), st.integers(1, 2),
       st.integers(1, 2), st.sampled_from([1, 3]),
       st.integers(0, 2 ** 31 - 1))
def test_nconv2d_matches_naive(h, w, ci, co, k, seed):
    rng = np.random.default_rng(seed)
    data = rng.standard_normal((1, ci, h, w)).astype(np.float32)
    conf = rng.random((1, ci, h, w)).astype(np.float32)
    wt = (rng.random((co, ci, k, k)) + 0.1).astype(np.float32)
    pad = k // 2

    out, cout = torch_ref.nconv2d(
        torch.from_numpy(data), torch.from_numpy(conf), torch.from_numpy(wt),
        padding=pad, prop_conf=True)

    for _ in range(8):
        o = rng.integers(co)
        y = rng.integers(h)
        x = rng.integers(w)
        nomin = denom = 0.0
        for c in range(ci):
            for ky in range(k):
                for kx in range(k):
                    yy, xx = y + ky - pad, x + kx - pad
                    if 0 <= yy < h and 0 <= xx < w:
                        cv = conf[0, c, yy, xx]
                        nomin += wt[o, c, ky, kx] * data[0, c, yy, xx] * cv
                        denom += wt[o, c, ky, kx] * cv
        np.testing.assert_allclose(out[0, o, y, x].item(),
                                   nomin / (denom + 1e-20), atol=1e-4)
        np.testing.assert_allclose(cout[0, o, y, x].item(),
                                   denom / wt[o].sum(), atol=1e-4)


@settings(max_examples=8, deadline=None)
@given(st.integers(3, 5), st.integers(3, 5), st.integers(0, 2 ** 31 - 1))
def test_convex_upsample_partition_of_unity(h, w, seed):
    """INTERIOR coarse pixels of a constant field reproduce factor*flow
    exactly: the softmax weights are a convex combination over the 3x3
    neighborhood and the op scales flow by `factor` internally
    (reference raft.py:79). Border pixels mix the unfold's zero padding —
    same as the reference — so only the interior is checked."""
    rng = np.random.default_rng(seed)
    flow = np.broadcast_to(
        rng.standard_normal((1, 2, 1, 1)).astype(np.float32),
        (1, 2, h, w)).copy()
    mask = rng.standard_normal((1, 576, h, w)).astype(np.float32)
    out = torch_ref.convex_upsample(torch.from_numpy(flow),
                                    torch.from_numpy(mask), 8)
    assert out.shape == (1, 2, 8 * h, 8 * w)
    interior = out[:, :, 8:-8, 8:-8].numpy()
    np.testing.assert_allclose(
        interior, np.broadcast_to(8.0 * flow[:, :, :1, :1], interior.shape),
        atol=1e-4)


@settings(max_examples=10, deadline=None)
@given(st.integers(4, 9), st.integers(4, 9), st.integers(0, 2 ** 31 - 1))
def test_corr_pyramid_matches_naive(h, w, seed):
    """Each level is a 2x2 mean of the previous (floor semantics: trailing
    odd row/col dropped, as F.avg_pool2d(2,2))."""
    rng = np.random.default_rng(seed)
    corr = rng.standard_normal((3, 1, h, w)).astype(np.float32)
    levels = torch_ref.corr_pyramid(torch.from_numpy(corr), 3)
    assert len(levels) == 3
    cur = corr
    for lvl in levels[1:]:
        hh, ww = cur.shape[-2] // 2, cur.shape[-1] // 2
        ref = np.zeros((3, 1, hh, ww), dtype=np.float64)
        for y in range(hh):
            for x in range(ww):
                ref[:, :, y, x] = cur[:, :, 2 * y:2 * y + 2,
                                      2 * x:2 * x + 2].mean(axis=(-1, -2))
        np.testing.assert_allclose(lvl.numpy(), ref, atol=1e-5)
        cur = lvl.numpy()


@settings(max_examples=10, deadline=None)
@given(st.integers(4, 10), st.integers(4, 10), st.integers(0, 2 ** 31 - 1))
def test_conf_pool_matches_naive(h, w, seed):
    """data_ds keeps the value at the argmax-confidence position of each
    2x2 block (row-major first-max ties); conf_ds = max/4."""
    rng = np.random.default_rng(seed)
    data = rng.standard_normal((1, 2, h, w)).astype(np.float32)
    conf = rng.random((1, 2, h, w)).astype(np.float32)
    dds, cds = torch_ref.conf_pool(torch.from_numpy(data),
                                   torch.from_numpy(conf))
    for c in range(2):
        for y in range(h // 2):
            for x in range(w // 2):
                block_c = conf[0, c, 2 * y:2 * y + 2, 2 * x:2 * x + 2]
                block_d = data[0, c, 2 * y:2 * y + 2, 2 * x:2 * x + 2]
                flat = block_c.reshape(-1)
                arg = int(np.argmax(flat))  # first max, row-major
                np.testing.assert_allclose(cds[0, c, y, x].item(),
                                           flat[arg] / 4, atol=1e-6)
                np.testing.assert_allclose(dds[0, c, y, x].item(),
                                           block_d.reshape(-1)[arg],
                                           atol=1e-6)


@settings(max_examples=10, deadline=None)
@given(st.integers(2, 6), st.integers(2, 6), st.sampled_from([2, 4]),
       st.integers(0, 2 ** 31 - 1))
def test_zero_inject_matches_naive(h, w, s, seed):
    rng = np.random.default_rng(seed)
    x = rng.standard_normal((1, 2, h, w)).astype(np.float32)
    out = torch_ref.zero_inject(torch.from_numpy(x), s, s).numpy()
    assert out.shape == (1, 2, h * s, w * s)
    ref = np.zeros_like(out)
    ref[:, :, s // 2::s, s // 2::s] = x
    np.testing.assert_allclose(out, ref)


@settings(max_examples=6, deadline=None)
@given(st.integers(16, 24), st.integers(16, 24), st.integers(0, 2 ** 31 - 1))
def test_corr_lookup_flagship_radius_levels(h, w, seed):
    """Flagship configuration (radius 4, 4 levels = 324 channels): taps
    still match the scalar bilinear oracle at every level."""
    rng = np.random.default_rng(seed)
    radius, levels = 4, 4
    P = h * w
    pyramid = [torch.from_numpy(
        rng.standard_normal((P, 1, max(h >> l, 1), max(w >> l, 1)))
        .astype(np.float32)) for l in range(levels)]
    coords = torch.from_numpy(
        (rng.random((1, 2, h, w)) * [[[[w]], [[h]]]]).astype(np.float32))

    out = torch_ref.corr_lookup(pyramid, coords, radius).numpy()
    K = 2 * radius + 1
    assert out.shape == (1, levels * K * K, h, w)

    for _ in range(10):
        i = rng.integers(P)
        l = rng.integers(levels)
        a = rng.integers(K)
        c = rng.integers(K)
        y, x = divmod(int(i), w)
        cx = coords[0, 0, y, x].item() / (1 << l) + (a - radius)
        cy = coords[0, 1, y, x].item() / (1 << l) + (c - radius)
        ref = naive_bilinear(pyramid[l][i, 0].numpy(), cx, cy)
        ch = l * K * K + a * K + c
        np.testing.assert_allclose(out[0, ch, y, x], ref, atol=1e-5)


@settings(max_examples=10, deadline=None)
@given(st.integers(1, 5), st.integers(4, 8), st.integers(4, 8),
       st.integers(0, 2 ** 31 - 1))
def test_sequence_loss_matches_naive(n_preds, h, w, seed):
    """gamma-weighted L1 + final-pred metrics vs an independent NumPy
    implementation of the reference formula (train.py:46-71)."""
    rng = np.random.default_rng(seed)
    gamma = 0.85
    gt = rng.standard_normal((2, 2, h, w)).astype(np.float32)
    gt[0, :, 0, 0] = 300.0  # ||gt|| ~ 424 > 400: excluded by max_flow
    valid = (rng.random((2, h, w)) > 0.3).astype(np.float32)
    preds = [rng.standard_normal((2, 2, h, w)).astype(np.float32)
             for _ in range(n_preds)]

    loss, metrics = torch_ref.sequence_loss(
        [torch.from_numpy(p) for p in preds], torch.from_numpy(gt),
        torch.from_numpy(valid), gamma)

    mag = np.sqrt((gt ** 2).sum(axis=1))
    v = (valid >= 0.5) & (mag < 400.0)
    expect = 0.0
    for i, p in enumerate(preds):
        w_i = gamma ** (n_preds - i - 1)
        expect += w_i * (v[:, None] * np.abs(p - gt)).mean()
    np.testing.assert_allclose(loss.item(), expect, rtol=1e-5)

    epe = np.sqrt(((preds[-1] - gt) ** 2).sum(axis=1)).reshape(-1)[v.reshape(-1)]
    np.testing.assert_allclose(metrics["epe"], epe.mean(), rtol=1e-5)
    np.testing.assert_allclose(metrics["3px"], (epe < 3).mean(), rtol=1e-6)


def test_nconv2d_bias_and_no_prop():
    """bias branch adds after normalization; prop_conf=False returns
    cout=None (nconv_modules.py:176-199)."""
    rng = np.random.default_rng(0)
    data = torch.from_numpy(rng.standard_normal((1, 2, 6, 6)).astype(np.float32))
    conf = torch.from_numpy(rng.random((1, 2, 6, 6)).astype(np.float32))
    wt = torch.from_numpy((rng.random((3, 2, 3, 3)) + 0.1).astype(np.float32))
    bias = torch.from_numpy(rng.standard_normal(3).astype(np.float32))

    out_nb, cout = torch_ref.nconv2d(data, conf, wt, None, padding=1)
    out_b, cout_none = torch_ref.nconv2d(data, conf, wt, bias, padding=1,
                                         prop_conf=False)
    assert cout_none is None
    np.testing.assert_allclose(
        (out_b - out_nb).numpy(),
        np.broadcast_to(bias.numpy()[None, :, None, None], out_b.shape),
        atol=1e-6)
    assert cout.shape == out_nb.shape


@settings(max_examples=8, deadline=None)
@given(st.integers(2, 4), st.integers(2, 4), st.sampled_from([2, 8]),
       st.integers(0, 2 ** 31 - 1))
def test_convex_upsample_matches_naive(h, w, factor, seed):
    """Full scalar oracle for kernel #11 (reference raft.py:73-84): output
    subpixel (dy,dx) of coarse pixel (y,x) = softmax-weighted sum of
    factor*flow over the zero-padded 3x3 neighborhood."""
    rng = np.random.default_rng(seed)
    flow = rng.standard_normal((1, 2, h, w)).astype(np.float32)
    mask = rng.standard_normal((1, 9 * factor * factor, h, w)).astype(np.float32)

    out = torch_ref.convex_upsample(torch.from_numpy(flow),
                                    torch.from_numpy(mask), factor).numpy()
    assert out.shape == (1, 2, factor * h, factor * w)

    m = mask.reshape(1, 1, 9, factor, factor, h, w)
    for _ in range(10):
        y = rng.integers(h)
        x = rng.integers(w)
        dy = rng.integers(factor)
        dx = rng.integers(factor)
        c = rng.integers(2)
        logits = m[0, 0, :, dy, dx, y, x]
        weights = np.exp(logits - logits.max())
        weights /= weights.sum()
        val = 0.0
        # unfold(3x3, pad 1) tap order: k = ky*3 + kx, tap = (y+ky-1, x+kx-1)
        for k in range(9):
            yy = y + k // 3 - 1
            xx = x + k % 3 - 1
            if 0 <= yy < h and 0 <= xx < w:
                val += weights[k] * factor * flow[0, c, yy, xx]
        np.testing.assert_allclose(out[0, c, y * factor + dy, x * factor + dx],
                                   val, atol=1e-5)


@settings(max_examples=8, deadline=None)
@given(st.integers(8, 16), st.integers(8, 16), st.integers(0, 2 ** 31 - 1))
def test_corr_lookup_adversarial_coords(h, w, seed):
    """Boundary-hostile coords — exact integers, negatives, far out of
    range — where bilinear implementations typically diverge (floor vs
    trunc, edge-tap weighting). All levels kept >=2 px per side (the
    documented input constraint, PARITY.md deviation 6)."""
    rng = np.random.default_rng(seed)
    radius, levels = 2, 2
    P = h * w
    pyramid = [torch.from_numpy(
        rng.standard_normal((P, 1, h >> l, w >> l)).astype(np.float32))
        for l in range(levels)]
    base = rng.random((1, 2, h, w)) * [[[[2 * w]], [[2 * h]]]] \
        - [[[[w / 2]], [[h / 2]]]]
    mask = rng.random((1, 2, h, w)) < 0.3
    base[mask] = np.round(base[mask])
    coords = torch.from_numpy(base.astype(np.float32))

    out = torch_ref.corr_lookup(pyramid, coords, radius).numpy()
    K = 2 * radius + 1
    for _ in range(12):
        i = rng.integers(P)
        l = rng.integers(levels)
        a = rng.integers(K)
        c = rng.integers(K)
        y, x = divmod(int(i), w)
        cx = coords[0, 0, y, x].item() / (1 << l) + (a - radius)
        cy = coords[0, 1, y, x].item() / (1 << l) + (c - radius)
        ref = naive_bilinear(pyramid[l][i, 0].numpy(), cx, cy)
        ch = l * K * K + a * K + c
        np.testing.assert_allclose(out[0, ch, y, x], ref, atol=2e-5)
