import torch

from flowhip.utils.geometry import InputPadder, bilinear_sampler, coords_grid, upflow8


def test_coords_grid():
    g = coords_grid(2, 3, 4)
    assert g.shape == (2, 2, 3, 4)
    # channel 0 = x, channel 1 = y
    assert torch.equal(g[0, 0, 0], torch.tensor([0., 1., 2., 3.]))
    assert torch.equal(g[0, 1, :, 0], torch.tensor([0., 1., 2.]))
    assert torch.equal(g[0], g[1])


def test_bilinear_sampler_identity():
    img = torch.arange(24, dtype=torch.float32).reshape(1, 1, 4, 6)
    coords = coords_grid(1, 4, 6).permute(0, 2, 3, 1)  # (1,H,W,2) xy
    out = bilinear_sampler(img, coords)
    assert torch.allclose(out, img, atol=1e-5)


def test_bilinear_sampler_zero_outside():
    img = torch.ones(1, 1, 4, 4)
    coords = torch.tensor([[[[-10.0, -10.0]]]])  # far out of bounds
    out = bilinear_sampler(img, coords)
    assert out.abs().max().item() == 0.0


def test_input_padder_roundtrip():
    x = torch.rand(1, 3, 437, 1021)
    padder = InputPadder(x.shape)
    (xp,) = padder.pad(x)
    assert xp.shape[-1] % 8 == 0 and xp.shape[-2] % 8 == 0
    assert torch.equal(padder.unpad(xp), x)


def test_input_padder_kitti_mode():
    x = torch.rand(1, 3, 370, 1226)
    padder = InputPadder(x.shape, mode="kitti")
    (xp,) = padder.pad(x)
    # kitti pads only at the bottom vertically
    assert torch.equal(xp[..., :370, 3:-3], x)


def test_upflow8_shape_and_scale():
    flow = torch.ones(1, 2, 4, 5)
    up = upflow8(flow)
    assert up.shape == (1, 2, 32, 40)
    assert torch.allclose(up, 8 * torch.ones_like(up))


def test_forward_interpolate_zero_flow_identity():
    """Warm-start splat: zero flow maps every pixel onto itself, so the
    interpolated field is (numerically) zero (reference utils.py:28-56)."""
    import torch
    from flowhip.utils.geometry import forward_interpolate
    flow = torch.zeros(2, 12, 16)
    out = forward_interpolate(flow)
    assert out.shape == (2, 12, 16)
    assert torch.allclose(out, torch.zeros_like(out), atol=1e-5)


def test_forward_interpolate_constant_shift():
    """A constant integer shift moves the field; in-range target pixels keep
    the constant value (nearest-neighbor griddata fill elsewhere)."""
    import torch
    from flowhip.utils.geometry import forward_interpolate
    flow = torch.zeros(2, 10, 14)
    flow[0] = 3.0  # shift x by +3
    out = forward_interpolate(flow)
    assert out.shape == (2, 10, 14)
    # splatted positions carry the same constant flow
    assert torch.allclose(out[0], torch.full_like(out[0], 3.0), atol=1e-4)


def test_bilinear_sampler_mask_branch():
    """mask=True returns strict-interior validity in normalized coords
    (reference utils.py:69-71)."""
    img = torch.randn(1, 1, 5, 7)
    coords = torch.tensor([[[[0.0, 0.0], [6.0, 4.0], [-1.0, 2.0],
                             [3.0, 2.0]]]])
    out, mask = bilinear_sampler(img, coords, mask=True)
    # corners sit exactly on the +-1 normalized boundary -> excluded (strict)
    assert mask.flatten().tolist() == [0.0, 0.0, 0.0, 1.0]
    assert out.shape[:2] == (1, 1)


def test_input_padder_noop_when_divisible():
    """Dims already divisible by 8 pad by zero in both modes."""
    x = torch.randn(1, 3, 64, 96)
    for mode in ("sintel", "kitti"):
        p = InputPadder(x.shape, mode=mode)
        (y,) = p.pad(x.clone())
        assert y.shape == x.shape
        assert torch.equal(p.unpad(y), x)


from hypothesis import given, settings, strategies as st


@settings(max_examples=40, deadline=None)
@given(st.integers(3, 100), st.integers(3, 100),
       st.sampled_from(["sintel", "kitti"]))
def test_input_padder_property(h, w, mode):
    """For ANY input size: padded dims divide by 8, per-side pads < 8,
    and unpad(pad(x)) is exactly x (reference utils.py:7-26 contract)."""
    from flowhip.utils.geometry import InputPadder

    x = torch.arange(h * w, dtype=torch.float32).reshape(1, 1, h, w)
    padder = InputPadder(x.shape, mode=mode)
    (y,) = padder.pad(x)
    assert y.shape[-2] % 8 == 0 and y.shape[-1] % 8 == 0
    assert y.shape[-2] - h < 8 and y.shape[-1] - w < 8
    back = padder.unpad(y)
    assert back.shape == x.shape
    torch.testing.assert_close(back, x, rtol=0, atol=0)


@settings(max_examples=25, deadline=None)
@given(st.integers(4, 40), st.integers(4, 40), st.integers(0, 2 ** 31 - 1))
def test_bilinear_sampler_identity_property(h, w, seed):
    """Sampling at the identity grid returns the input exactly for any
    size (align_corners=True pixel-coordinate convention)."""
    from flowhip.utils.geometry import bilinear_sampler, coords_grid

    g = torch.Generator().manual_seed(seed)
    img = torch.randn(1, 3, h, w, generator=g)
    coords = coords_grid(1, h, w).permute(0, 2, 3, 1)
    out = bilinear_sampler(img, coords)
    # pixel -> [-1,1] -> pixel round-trip carries ~1e-5 fp32 rounding
    torch.testing.assert_close(out, img, rtol=0, atol=1e-4)
