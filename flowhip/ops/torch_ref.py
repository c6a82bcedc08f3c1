"""Pure-PyTorch reference implementations of every flowhip op.

These serve three roles:
  1. the CPU execution path (no GPU required — tests, plumbing, debugging);
  2. the numerical oracle that every HIP kernel is unit-tested against
     (SURVEY.md §4.2 item 2);
  3. the documentation of the exact math each kernel computes, with citations
     into the reference repo.

Every function here is differentiable through torch autograd.
"""

import math

import torch
import torch.nn.functional as F

from ..utils.geometry import bilinear_sampler


# ---------------------------------------------------------------------------
# Correlation volume / pyramid / lookup  (reference core/corr.py)
# ---------------------------------------------------------------------------

def corr_volume(fmap1, fmap2):
    """All-pairs correlation: C[b, i, j] = <f1[b,:,i], f2[b,:,j]> / sqrt(D).

    In:  fmap1, fmap2 (B, D, H, W).
    Out: (B*H*W, 1, H, W) — the reference's post-reshape layout
         (core/corr.py:12-16, 47-55); scaling by 1/sqrt(D) per corr.py:55.
    """
    batch, dim, ht, wd = fmap1.shape
    f1 = fmap1.reshape(batch, dim, ht * wd)
    f2 = fmap2.reshape(batch, dim, ht * wd)
    corr = torch.matmul(f1.transpose(1, 2), f2) / math.sqrt(dim)
    return corr.reshape(batch * ht * wd, 1, ht, wd)


def corr_pyramid(corr, num_levels=4):
    """Build the average-pool pyramid over the *second* image's spatial dims.

    In:  corr (B*H*W, 1, H, W) from `corr_volume`.
    Out: list of `num_levels` tensors, level l at (H/2^l, W/2^l).
    Parity: core/corr.py:18-21.
    """
    pyramid = [corr]
    for _ in range(num_levels - 1):
        corr = F.avg_pool2d(corr, 2, stride=2)
        pyramid.append(corr)
    return pyramid


def corr_lookup(pyramid, coords, radius):
    """Window lookup: for each target pixel, bilinearly sample a
    (2r+1)^2 window around `coords / 2^l` at every pyramid level.

    In:  pyramid — list of L tensors (B*H1*W1, 1, Hl, Wl);
         coords (B, 2, H1, W1) pixel coordinates into level 0.
    Out: (B, L*(2r+1)^2, H1, W1); channel = l*(2r+1)^2 + dy_idx*(2r+1) + dx_idx.
    Parity: core/corr.py:23-44 (delta from meshgrid(dy, dx) 'ij'), sampling via
    bilinear_sampler (= grid_sample align_corners=True, zero padding).
    """
    r = radius
    coords = coords.permute(0, 2, 3, 1)  # (B, H1, W1, 2)
    batch, h1, w1, _ = coords.shape

    out_pyramid = []
    for i, corr in enumerate(pyramid):
        dx = torch.linspace(-r, r, 2 * r + 1, device=coords.device, dtype=coords.dtype)
        dy = torch.linspace(-r, r, 2 * r + 1, device=coords.device, dtype=coords.dtype)
        delta = torch.stack(torch.meshgrid(dy, dx, indexing="ij"), axis=-1)

        centroid_lvl = coords.reshape(batch * h1 * w1, 1, 1, 2) / 2 ** i
        delta_lvl = delta.view(1, 2 * r + 1, 2 * r + 1, 2)
        coords_lvl = centroid_lvl + delta_lvl

        sampled = bilinear_sampler(corr, coords_lvl)
        out_pyramid.append(sampled.view(batch, h1, w1, -1))

    out = torch.cat(out_pyramid, dim=-1)
    return out.permute(0, 3, 1, 2).contiguous().float()


# ---------------------------------------------------------------------------
# Normalized convolution  (reference core/nconv_modules.py:140-216)
# ---------------------------------------------------------------------------

def nconv2d(data, conf, weight, bias=None, stride=1, padding=0, dilation=1,
            groups=1, eps=1e-20, prop_conf=True):
    """Confidence-normalized convolution with confidence propagation.

    out  = conv(data*conf, w) / (conv(conf, w) + eps)  [+ bias]
    cout = conv(conf, w) / sum_per_outchannel(w)

    `weight` is the *effective* (non-negative) weight — the softplus
    reparameterization (EnforcePos, nconv_modules.py:218-265) is applied by
    the calling module, not here.
    Parity: nconv_modules.py:164-199.
    """
    denom = F.conv2d(conf, weight, None, stride, padding, dilation, groups)
    nomin = F.conv2d(data * conf, weight, None, stride, padding, dilation, groups)
    nconv = nomin / (denom + eps)

    if bias is not None:
        nconv = nconv + bias.view(1, -1, 1, 1)

    if prop_conf:
        s = weight.reshape(weight.shape[0], -1).sum(dim=-1).view(1, -1, 1, 1)
        cout = denom / s
    else:
        cout = None
    return nconv, cout


def conf_pool(data, conf, ds_factor=2, pooling_type="conf_based"):
    """Confidence-based 2x downsampling of a (data, conf) pair.

    conf is max-pooled (and divided by 4 — the Jacobian determinant of the
    scale change, nconv_modules.py:97); data keeps the values at the argmax-
    confidence positions (`conf_based`) or is max-pooled itself.
    Parity: nconv_modules.py:94-104 and retrieve_elements_from_indices :19-22.
    """
    conf_ds, idx = F.max_pool2d(conf, ds_factor, ds_factor, return_indices=True)
    conf_ds = conf_ds / 4
    if pooling_type == "conf_based":
        flat = data.flatten(start_dim=2)
        data_ds = flat.gather(dim=2, index=idx.flatten(start_dim=2)).view_as(idx)
    elif pooling_type == "max_pooling":
        data_ds = F.max_pool2d(data, ds_factor, ds_factor)
    else:
        raise NotImplementedError(
            "Choose pooling_type from [conf_based, max_pooling]!")
    return data_ds, conf_ds


# ---------------------------------------------------------------------------
# Sparse zero-injection upsample  (reference core/upsampler.py:179-210)
# ---------------------------------------------------------------------------

def zero_inject(inp, scale_h, scale_w, out_h=None, out_w=None):
    """Place low-res samples on a zero high-res grid at stride s, offset s//2.

    out[:, :, sH//2::sH, sW//2::sW] = inp   (upsampler.py:208)
    """
    b, c, ih, iw = inp.shape
    oh = out_h if out_h is not None else ih * scale_h
    ow = out_w if out_w is not None else iw * scale_w
    out = inp.new_zeros((b, c, oh, ow))
    out[:, :, scale_h // 2::scale_h, scale_w // 2::scale_w] = inp
    return out


# ---------------------------------------------------------------------------
# Convex-combination upsample  (reference core/raft.py:73-84)
# ---------------------------------------------------------------------------

def convex_upsample(flow, mask, factor=8):
    """x`factor` upsample of flow as a learned convex combination of each
    coarse pixel's 3x3 neighborhood; mask holds 9 logits per output subpixel.

    In:  flow (N, 2, H, W); mask (N, 9*factor*factor, H, W).
    Out: (N, 2, factor*H, factor*W); flow values scaled by `factor`.
    Parity: raft.py:73-84 (softmax over the 9 taps, unfold 3x3 pad 1,
    permute to pixel-shuffle layout).
    """
    N, _, H, W = flow.shape
    mask = mask.view(N, 1, 9, factor, factor, H, W)
    mask = torch.softmax(mask, dim=2)

    up_flow = F.unfold(factor * flow, [3, 3], padding=1)
    up_flow = up_flow.view(N, 2, 9, 1, 1, H, W)

    up_flow = torch.sum(mask * up_flow, dim=2)
    up_flow = up_flow.permute(0, 1, 4, 2, 5, 3)
    return up_flow.reshape(N, 2, factor * H, factor * W)


# ---------------------------------------------------------------------------
# Sequence loss  (reference train.py:43-71)
# ---------------------------------------------------------------------------

MAX_FLOW = 400


def sequence_loss(flow_preds, flow_gt, valid, gamma=0.8, max_flow=MAX_FLOW):
    """Exponentially weighted L1 over the iteration sequence + EPE metrics.

    Weight of prediction i (of n) is gamma^(n-1-i) — later iterations weigh
    more. Pixels with valid < 0.5 or ||gt|| >= max_flow are excluded.
    Metrics are computed on the final prediction only.
    Parity: train.py:46-71.
    """
    n_predictions = len(flow_preds)
    flow_loss = 0.0

    mag = torch.sum(flow_gt ** 2, dim=1).sqrt()
    valid = (valid >= 0.5) & (mag < max_flow)

    for i in range(n_predictions):
        i_weight = gamma ** (n_predictions - i - 1)
        i_loss = (flow_preds[i] - flow_gt).abs()
        flow_loss += i_weight * (valid[:, None] * i_loss).mean()

    epe = torch.sum((flow_preds[-1] - flow_gt) ** 2, dim=1).sqrt()
    epe = epe.view(-1)[valid.view(-1)]

    metrics = {
        "epe": epe.mean().item(),
        "1px": (epe < 1).float().mean().item(),
        "3px": (epe < 3).float().mean().item(),
        "5px": (epe < 5).float().mean().item(),
    }
    return flow_loss, metrics
