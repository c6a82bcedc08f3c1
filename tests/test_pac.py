"""PAC op tests against naive math (reference core/pac_modules.py contracts)."""

import math

import torch
import torch.nn.functional as F

from flowhip.nn.pac import (
    PacConv2d,
    PacConvTranspose2d,
    PacPool2d,
    nd2col,
    packernel2d,
    pacconv2d,
    pacpool2d,
)


def test_nd2col_shapes():
    x = torch.randn(2, 3, 10, 12)
    cols = nd2col(x, 3, stride=1, padding=1)
    assert cols.shape == (2, 3, 3, 3, 10, 12)
    # center tap of the window is the pixel itself
    assert torch.allclose(cols[:, :, 1, 1], x)


def test_nd2col_transposed_geometry():
    x = torch.randn(1, 2, 5, 7)
    cols = nd2col(x, 5, stride=2, padding=1, output_padding=1, transposed=True)
    # L_out = (L_in-1)*s - 2p + d(k-1) + 1 + op
    assert cols.shape[-2:] == ((5 - 1) * 2 - 2 + 4 + 1 + 1,
                               (7 - 1) * 2 - 2 + 4 + 1 + 1)


def test_packernel_gaussian_matches_naive():
    torch.manual_seed(0)
    g = torch.randn(1, 2, 6, 6)
    k, _ = packernel2d(g, kernel_size=3, stride=1, padding=1)
    assert k.shape == (1, 1, 3, 3, 6, 6)
    # naive check at an interior pixel
    y, x = 2, 3
    for dy in range(3):
        for dx in range(3):
            yy, xx = y + dy - 1, x + dx - 1
            d2 = ((g[0, :, yy, xx] - g[0, :, y, x]) ** 2).sum()
            expect = math.exp(-0.5 * d2.item())
            assert abs(k[0, 0, dy, dx, y, x].item() - expect) < 1e-5
    # center tap is always exp(0)=1
    assert torch.allclose(k[0, 0, 1, 1], torch.ones(6, 6))


def test_pacconv2d_uniform_guidance_is_conv():
    """With constant guidance the adapting kernel is all-ones ->
    pacconv == plain conv."""
    torch.manual_seed(1)
    x = torch.randn(2, 3, 8, 8)
    g = torch.ones(2, 4, 8, 8)
    w = torch.randn(5, 3, 3, 3)
    b = torch.randn(5)
    k, _ = packernel2d(g, kernel_size=3, stride=1, padding=1)
    out = pacconv2d(x, k, w, b, stride=1, padding=1)
    ref = F.conv2d(x, w, b, padding=1)
    assert torch.allclose(out, ref, atol=1e-5)


def test_pacconvtranspose_shape_matches_convtranspose():
    m = PacConvTranspose2d(4, 6, kernel_size=5, stride=2, padding=2,
                           output_padding=1)
    x = torch.randn(1, 4, 8, 8)
    g = torch.randn(1, 3, 16, 16)
    out = m(x, g)
    ref_shape = F.conv_transpose2d(x, torch.randn(4, 6, 5, 5), stride=2,
                                   padding=2, output_padding=1).shape
    assert out.shape == ref_shape


def test_pacpool_uniform_guidance_normalized_is_avgpool():
    x = torch.randn(1, 2, 8, 8)
    g = torch.ones(1, 3, 8, 8)
    m = PacPool2d(3, stride=1, padding=0, normalize_kernel=True)
    out = m(x, g)
    ref = F.avg_pool2d(x, 3, stride=1)
    assert torch.allclose(out, ref, atol=1e-5)


def test_pac_backward_flows():
    m = PacConv2d(2, 2, 3, padding=1)
    x = torch.randn(1, 2, 6, 6, requires_grad=True)
    g = torch.randn(1, 2, 6, 6, requires_grad=True)
    m(x, g).sum().backward()
    assert x.grad is not None and g.grad is not None
    assert m.weight.grad is not None


def test_pac_state_dict_keys():
    m = PacConvTranspose2d(3, 3, 5, stride=2, padding=2, output_padding=1)
    keys = set(m.state_dict().keys())
    assert keys == {"weight", "bias"}
    assert m.weight.shape == (3, 3, 5, 5)  # (in, out, k, k) for transposed


def test_np_gaussian_2d():
    """Reference pac_modules.py:38-49 utility: normalized, symmetric,
    peak-centered."""
    from flowhip.nn.pac import np_gaussian_2d

    g = np_gaussian_2d(5)
    assert g.shape == (5, 5)
    assert abs(g.sum() - 1.0) < 1e-6
    assert g[2, 2] == g.max()
    import numpy as np
    np.testing.assert_allclose(g, g.T, atol=0)
    g3 = np_gaussian_2d(3, sigma=1.0)
    assert g3.shape == (3, 3) and abs(g3.sum() - 1.0) < 1e-6
