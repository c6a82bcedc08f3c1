"""Unit tests of the torch reference ops against independent naive math.

These pin the contracts that the HIP kernels are later tested against
(SURVEY.md §4.2 items 1-2).
"""

import math

import pytest
import torch
import torch.nn.functional as F

from flowhip.ops import torch_ref


def test_corr_volume_matches_einsum():
    torch.manual_seed(0)
    B, D, H, W = 2, 8, 3, 5
    f1 = torch.randn(B, D, H, W)
    f2 = torch.randn(B, D, H, W)
    out = torch_ref.corr_volume(f1, f2)
    assert out.shape == (B * H * W, 1, H, W)

    ref = torch.einsum("bdi,bdj->bij", f1.reshape(B, D, -1),
                       f2.reshape(B, D, -1)) / math.sqrt(D)
    assert torch.allclose(out.reshape(B, H * W, H * W), ref, atol=1e-5)


def test_corr_pyramid_levels():
    corr = torch.randn(6, 1, 8, 12)
    pyr = torch_ref.corr_pyramid(corr, 4)
    assert [p.shape[-2:] for p in pyr] == [(8, 12), (4, 6), (2, 3), (1, 1)]
    assert torch.allclose(pyr[1], F.avg_pool2d(corr, 2, 2))


def _naive_lookup(pyramid, coords, radius):
    """Independent loop implementation of the 4-level window lookup.

    Channel ordering quirk (reference corr.py:31-37): delta is built as
    meshgrid(dy, dx) but ADDED to (x, y)-ordered coords, so channel
    c = a*K + b samples at (x + (a-r), y + (b-r)) — the window offsets are
    x-offset-MAJOR. Coverage is identical (symmetric window) but the
    channel layout must match for weight compatibility.
    """
    B, _, H, W = coords.shape
    L = len(pyramid)
    K = 2 * radius + 1
    out = torch.zeros(B, L * K * K, H, W)
    for b in range(B):
        for y in range(H):
            for x in range(W):
                i = y * W + x
                cx, cy = coords[b, 0, y, x].item(), coords[b, 1, y, x].item()
                for l in range(L):
                    lvl = pyramid[l][b * H * W + i, 0]
                    Hl, Wl = lvl.shape
                    for ox in range(-radius, radius + 1):    # x offset (major)
                        for oy in range(-radius, radius + 1):  # y offset
                            sx = cx / 2 ** l + ox
                            sy = cy / 2 ** l + oy
                            x0, y0 = math.floor(sx), math.floor(sy)
                            v = 0.0
                            for (xi, yi, wgt) in [
                                    (x0, y0, (1 - (sx - x0)) * (1 - (sy - y0))),
                                    (x0 + 1, y0, (sx - x0) * (1 - (sy - y0))),
                                    (x0, y0 + 1, (1 - (sx - x0)) * (sy - y0)),
                                    (x0 + 1, y0 + 1, (sx - x0) * (sy - y0))]:
                                if 0 <= xi < Wl and 0 <= yi < Hl:
                                    v += wgt * lvl[yi, xi].item()
                            ch = l * K * K + (ox + radius) * K + (oy + radius)
                            out[b, ch, y, x] = v
    return out


def test_corr_lookup_matches_naive():
    torch.manual_seed(1)
    B, H, W, r = 1, 4, 4, 1  # both pyramid levels stay >=2px
    P = H * W
    l0 = torch.randn(B * P, 1, H, W)
    pyramid = torch_ref.corr_pyramid(l0, 2)
    coords = torch.rand(B, 2, H, W) * 3

    out = torch_ref.corr_lookup(pyramid, coords, r)
    naive = _naive_lookup(pyramid, coords, r)
    assert out.shape == naive.shape
    assert torch.allclose(out, naive, atol=1e-4)


def test_nconv2d_math():
    torch.manual_seed(2)
    data = torch.randn(2, 1, 6, 6)
    conf = torch.rand(2, 1, 6, 6)
    weight = torch.rand(2, 1, 3, 3) + 0.1
    out, cout = torch_ref.nconv2d(data, conf, weight, padding=1)

    denom = F.conv2d(conf, weight, padding=1)
    nomin = F.conv2d(data * conf, weight, padding=1)
    assert torch.allclose(out, nomin / (denom + 1e-20))
    s = weight.sum(dim=(1, 2, 3)).view(1, -1, 1, 1)
    assert torch.allclose(cout, denom / s)


def test_nconv2d_constant_data_is_preserved():
    # normalized conv of a constant field under any confidence = the constant
    data = torch.full((1, 1, 8, 8), 3.17)
    conf = torch.rand(1, 1, 8, 8)
    weight = torch.rand(1, 1, 5, 5) + 0.01
    out, _ = torch_ref.nconv2d(data, conf, weight, padding=2)
    assert torch.allclose(out, data, atol=1e-4)


def test_conf_pool_conf_based():
    data = torch.tensor([[[[1., 2.], [3., 4.]]]])
    conf = torch.tensor([[[[.1, .9], [.2, .3]]]])
    d, c = torch_ref.conf_pool(data, conf)
    assert torch.allclose(c, torch.tensor([[[[0.9 / 4]]]]))
    assert torch.allclose(d, torch.tensor([[[[2.]]]]))  # value at argmax conf


def test_zero_inject():
    x = torch.arange(4, dtype=torch.float32).reshape(1, 1, 2, 2)
    out = torch_ref.zero_inject(x, 4, 4)
    assert out.shape == (1, 1, 8, 8)
    assert out.sum() == x.sum()
    assert out[0, 0, 2, 2] == 0.0 and out[0, 0, 2, 6] == 1.0
    assert out[0, 0, 6, 2] == 2.0 and out[0, 0, 6, 6] == 3.0


def test_convex_upsample_uniform_mask_is_interp():
    torch.manual_seed(3)
    flow = torch.randn(1, 2, 4, 4)
    mask = torch.zeros(1, 9 * 64, 4, 4)  # uniform softmax = mean of 3x3
    out = torch_ref.convex_upsample(flow, mask, 8)
    assert out.shape == (1, 2, 32, 32)
    # uniform convex combination = avg-pool3x3 of 8*flow, replicated 8x8
    avg = F.avg_pool2d(F.pad(8 * flow, (1, 1, 1, 1)), 3, stride=1) * 9 / 9
    manual = F.conv2d(F.pad(8 * flow, (1, 1, 1, 1)),
                      torch.ones(1, 1, 3, 3).expand(2, 1, 3, 3) / 9, groups=2)
    assert torch.allclose(out[:, :, ::8, ::8], manual, atol=1e-5)
    assert torch.allclose(out[:, :, 3::8, 5::8], manual, atol=1e-5)


def test_sequence_loss_weighting():
    gt = torch.zeros(1, 2, 4, 4)
    valid = torch.ones(1, 4, 4)
    p1 = torch.ones(1, 2, 4, 4)
    p2 = 2 * torch.ones(1, 2, 4, 4)
    loss, metrics = torch_ref.sequence_loss([p1, p2], gt, valid, gamma=0.5)
    # loss = 0.5^1 * mean|p1| + 0.5^0 * mean|p2| = 0.5*1 + 1*2
    assert abs(loss.item() - 2.5) < 1e-6
    assert abs(metrics["epe"] - math.sqrt(8)) < 1e-5
    assert metrics["1px"] == 0.0 and metrics["3px"] == 1.0


def test_sequence_loss_excludes_large_flow():
    gt = torch.zeros(1, 2, 2, 2)
    gt[0, 0, 0, 0] = 500.0  # exceeds MAX_FLOW
    valid = torch.ones(1, 2, 2)
    pred = torch.zeros(1, 2, 2, 2)
    loss, metrics = torch_ref.sequence_loss([pred], gt, valid)
    # excluded pixel contributes 0 despite |pred-gt|=500
    assert loss.item() < 1e-6


def test_sequence_loss_all_invalid():
    """Zero valid pixels: the masked loss is 0. Metrics over an empty valid
    set are NaN in the reference (train.py:61-69 .mean() on empty) and this
    torch path mirrors that; the fused GPU kernel clamps the count to 1
    (metrics 0) — a benign deviation on a degenerate input."""
    import torch
    from flowhip.ops import torch_ref
    preds = [torch.randn(1, 2, 8, 8) for _ in range(3)]
    gt = torch.randn(1, 2, 8, 8)
    valid = torch.zeros(1, 8, 8)
    loss, metrics = torch_ref.sequence_loss(preds, gt, valid, 0.8)
    assert loss.item() == 0.0


def test_sequence_loss_max_flow_exclusion():
    import torch
    from flowhip.ops import torch_ref
    preds = [torch.zeros(1, 2, 4, 4)]
    gt = torch.zeros(1, 2, 4, 4)
    gt[0, 0, 0, 0] = 1000.0  # ||gt|| >= 400 -> excluded
    valid = torch.ones(1, 4, 4)
    loss, metrics = torch_ref.sequence_loss(preds, gt, valid, 0.8)
    # excluded pixel contributes nothing to the (masked) loss
    assert loss.item() == 0.0


def test_up2x_cat_fallback_matches_eager():
    import torch
    import torch.nn.functional as F
    from flowhip import ops
    low = torch.randn(2, 3, 5, 7)   # odd sizes: CPU fallback path
    skip = torch.randn(2, 2, 10, 14)
    got = ops.up2x_cat(low, skip)
    up = F.interpolate(low, size=(10, 14), mode="nearest")
    ref = torch.cat((up, skip), 1)
    torch.testing.assert_close(got, ref)


def test_zero_inject_explicit_out_size():
    import torch
    from flowhip.ops import torch_ref
    x = torch.arange(6.0).view(1, 1, 2, 3)
    out = torch_ref.zero_inject(x, 2, 2, out_h=5, out_w=7)
    assert out.shape == (1, 1, 5, 7)
    # samples at stride 2, offset 1
    assert out[0, 0, 1, 1] == x[0, 0, 0, 0]
    assert out[0, 0, 3, 5] == x[0, 0, 1, 2]
    assert out.sum() == x.sum()


def test_extension_symbol_surface():
    """flowhip._C must expose every kernel entry the op layer dispatches to
    (catches bindings drift at CPU-test time, before any GPU run)."""
    import flowhip._C as C
    expected = [
        "bgemm_nt", "corr_lookup_fwd", "corr_lookup_bwd",
        "corr_pyramid_fwd", "corr_pyramid_bwd",
        "convex_up_fwd", "convex_up_bwd",
        "nconv_fwd", "nconv_bwd", "nconv_bwd_prep",
        "conf_pool_fwd", "conf_pool_bwd",
        "zero_inject_fwd", "zero_inject_bwd",
        "gru_gate1_fwd", "gru_gate1_bwd", "gru_gate2_fwd", "gru_gate2_bwd",
        "seq_loss_fwd", "seq_loss_bwd",
        "conv_gemm_fwd", "conv_gemm_fwd2", "conv_gemm_wrw", "conv_gemm_pack",
        "instnorm_cl_fwd", "instnorm_cl_bwd",
        "transpose_cast_bf16", "col_sum_bf16",
        "area_up2x_fwd", "area_up2x_bwd", "up2x_cat_fwd",
        "packernel_fwd", "packernel_bwd", "pacconv_fwd", "pacconv_bwd",
    ]
    missing = [name for name in expected if not hasattr(C, name)]
    assert not missing, missing
