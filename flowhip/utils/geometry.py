"""Geometry / sampling utilities (layer L0 of SURVEY.md).

Behavioral parity with the reference `core/utils/utils.py` (see file:line cites
on each function); implementations are written fresh for PyTorch 2.x
(explicit `indexing='ij'`, no deprecated idioms).
"""

import numpy as np
import torch
import torch.nn.functional as F


class InputPadder:
    """Pads images so H and W are divisible by 8.

    Parity: reference `core/utils/utils.py:7-26`. `sintel` mode centers the
    vertical padding; any other mode (`kitti`) pads only at the bottom.
    Replicate padding, and `unpad` crops back to the original extent.
    """

    def __init__(self, dims, mode="sintel"):
        self.ht, self.wd = dims[-2:]
        pad_ht = (((self.ht // 8) + 1) * 8 - self.ht) % 8
        pad_wd = (((self.wd // 8) + 1) * 8 - self.wd) % 8
        if mode == "sintel":
            self._pad = [pad_wd // 2, pad_wd - pad_wd // 2, pad_ht // 2, pad_ht - pad_ht // 2]
        else:
            self._pad = [pad_wd // 2, pad_wd - pad_wd // 2, 0, pad_ht]

    def pad(self, *inputs):
        return [F.pad(x, self._pad, mode="replicate") for x in inputs]

    def unpad(self, x):
        ht, wd = x.shape[-2:]
        c = [self._pad[2], ht - self._pad[3], self._pad[0], wd - self._pad[1]]
        return x[..., c[0]:c[1], c[2]:c[3]]


def forward_interpolate(flow):
    """Warm-start splat: forward-warp a flow field onto the regular grid by
    nearest-neighbor scattered interpolation (CPU, scipy).

    Parity: reference `core/utils/utils.py:28-56` (used by the Sintel
    submission writer for warm-started inference).
    """
    from scipy import interpolate as scipy_interpolate

    flow = flow.detach().cpu().numpy()
    dx, dy = flow[0], flow[1]

    ht, wd = dx.shape
    x0, y0 = np.meshgrid(np.arange(wd), np.arange(ht))

    x1 = (x0 + dx).reshape(-1)
    y1 = (y0 + dy).reshape(-1)
    dx = dx.reshape(-1)
    dy = dy.reshape(-1)

    valid = (x1 > 0) & (x1 < wd) & (y1 > 0) & (y1 < ht)
    x1, y1, dx, dy = x1[valid], y1[valid], dx[valid], dy[valid]

    flow_x = scipy_interpolate.griddata((x1, y1), dx, (x0, y0), method="nearest", fill_value=0)
    flow_y = scipy_interpolate.griddata((x1, y1), dy, (x0, y0), method="nearest", fill_value=0)

    return torch.from_numpy(np.stack([flow_x, flow_y], axis=0)).float()


def bilinear_sampler(img, coords, mode="bilinear", mask=False):
    """Sample `img` at pixel coordinates `coords` (…,2 = (x, y)).

    grid_sample semantics with align_corners=True and zero padding outside.
    Parity: reference `core/utils/utils.py:59-73`.
    """
    H, W = img.shape[-2:]
    xgrid, ygrid = coords.split([1, 1], dim=-1)
    xgrid = 2 * xgrid / (W - 1) - 1
    ygrid = 2 * ygrid / (H - 1) - 1

    grid = torch.cat([xgrid, ygrid], dim=-1)
    img = F.grid_sample(img, grid, align_corners=True)

    if mask:
        mask = (xgrid > -1) & (ygrid > -1) & (xgrid < 1) & (ygrid < 1)
        return img, mask.float()

    return img


def coords_grid(batch, ht, wd, device=None, dtype=torch.float32):
    """(batch, 2, ht, wd) grid of pixel coordinates; channel 0 = x, 1 = y.

    Parity: reference `core/utils/utils.py:76-79` (meshgrid is 'ij' there by
    the pre-1.10 default; stacked reversed so x comes first).
    """
    y, x = torch.meshgrid(
        torch.arange(ht, device=device, dtype=dtype),
        torch.arange(wd, device=device, dtype=dtype),
        indexing="ij",
    )
    coords = torch.stack([x, y], dim=0)
    return coords[None].repeat(batch, 1, 1, 1)


def upflow8(flow, mode="bilinear", align_corners=True):
    """Bilinear x8 upsample of a flow field, scaling the vectors by 8.

    Parity: reference `core/utils/utils.py:82-84`.
    """
    new_size = (8 * flow.shape[2], 8 * flow.shape[3])
    return 8 * F.interpolate(flow, size=new_size, mode=mode, align_corners=align_corners)
