"""Direct numerical parity against the reference implementation.

These tests import the PUBLIC reference snapshot mounted at /root/reference
(read-only) and compare it to this framework end-to-end on CPU: identical
state-dict surface (keys + shapes), identical forward outputs under shared
weights, and identical gradients — the strongest form of the SURVEY.md §4.2
"vs reference" contract. Skipped wherever the reference isn't mounted (e.g.
on GPU boxes, which only receive this repo).

No reference code is copied or adapted here; it is imported as an external
oracle, exactly like the plain-PyTorch oracles in test_oracle_crosscheck.py.
"""

import os
import sys

import pytest
import torch

REF_CORE = "/root/reference/core"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(REF_CORE), reason="reference snapshot not mounted")


@pytest.fixture(scope="module")
def ref_modules():
    sys.path.insert(0, REF_CORE)
    try:
        import raft as ref_raft
        import raft_nc_dbl as ref_nc_dbl
        yield ref_raft, ref_nc_dbl
    finally:
        sys.path.remove(REF_CORE)


def _ncup_args(**kw):
    from flowhip.config.args import default_ncup_args
    args = default_ncup_args(model="raft_nc_dbl", **kw)
    args.mixed_precision = False
    return args


def test_ncup_state_dict_surface_matches_reference(ref_modules):
    """Every parameter/buffer key and shape is identical to the reference
    RAFT_NC_DBL (checkpoints interchange 1:1)."""
    from flowhip.models import build_model
    _, ref_nc_dbl = ref_modules

    args = _ncup_args()
    ours = {k: tuple(v.shape) for k, v in build_model(args).state_dict().items()}
    ref = {k: tuple(v.shape) for k, v in ref_nc_dbl.RAFT(args).state_dict().items()}
    assert ours == ref
    assert len(ours) > 150  # full model, not a stub


@pytest.mark.parametrize("small", [True, False])
def test_raft_state_dict_surface_matches_reference(ref_modules, small):
    from flowhip.config.args import default_ncup_args
    from flowhip.models import build_model
    ref_raft, _ = ref_modules

    args = default_ncup_args(model="raft", small=small)
    args.mixed_precision = False
    ours = {k: tuple(v.shape) for k, v in build_model(args).state_dict().items()}
    ref = {k: tuple(v.shape) for k, v in ref_raft.RAFT(args).state_dict().items()}
    assert ours == ref


@pytest.mark.timeout(900)
def test_ncup_forward_and_grads_match_reference(ref_modules):
    """Shared weights => same flow predictions (<=1e-4) and same gradients,
    over 3 refinement iterations at the minimum supported size. Also the
    direct check that skipping the reference's dead deepest encoder stage
    (flowhip/nn/nconv.py docstring) preserves outputs AND gradients."""
    from flowhip.models import build_model
    _, ref_nc_dbl = ref_modules

    torch.manual_seed(0)
    args = _ncup_args()
    ours = build_model(args)
    ref = ref_nc_dbl.RAFT(args)
    ref.load_state_dict(ours.state_dict())
    ours.train()
    ref.train()

    img1 = torch.rand(1, 3, 128, 128) * 255
    img2 = torch.rand(1, 3, 128, 128) * 255

    preds_ref = ref(img1, img2, iters=3)
    preds_ours = ours(img1, img2, iters=3)
    assert len(preds_ref) == len(preds_ours) == 3
    for a, b in zip(preds_ref, preds_ours):
        assert (a - b).abs().max().item() < 1e-4

    # identical scalar objective => comparable grads
    sum(p.abs().mean() for p in preds_ref).backward()
    sum(p.abs().mean() for p in preds_ours).backward()

    ref_named = dict(ref.named_parameters())
    rels = []
    for name, p in ours.named_parameters():
        g_ours, g_ref = p.grad, ref_named[name].grad
        if g_ours is None and g_ref is None:
            continue
        assert g_ours is not None and g_ref is not None, name
        ref_norm = g_ref.norm().item()
        if ref_norm < 1e-6:
            # analytically-zero grads (e.g. conv bias followed by a
            # normalization): both sides emit ~1e-9 fp32 noise whose ratio
            # is meaningless — just require ours to be equally negligible
            assert g_ours.norm().item() < 1e-6, name
            continue
        rel = (g_ours - g_ref).norm().item() / ref_norm
        # fnet gradients flow exclusively through the corr lookup, whose
        # bilinear-tap derivative is discontinuous at integer coords: ~1e-5
        # forward coordinate differences flip a few tap cells, giving a
        # uniform ~5e-3 rel diff on fnet.* (measured; the reference's own
        # thread-order noise on those layers is ~1e-6, and cnet — not
        # behind the lookup — matches to <5e-3). Bound the tail at 2e-2
        # and the bulk (median) at 1e-4 to still catch systematic errors.
        assert rel < 2e-2, f"{name}: rel grad diff {rel:.2e}"
        rels.append(rel)
    assert len(rels) > 100
    rels.sort()
    assert rels[len(rels) // 2] < 1e-4, f"median rel grad diff {rels[len(rels)//2]:.2e}"


@pytest.mark.timeout(900)
def test_raft_basic_forward_matches_reference(ref_modules):
    """RAFT-basic (convex-upsample path): shared weights => same outputs in
    train and test mode."""
    from flowhip.config.args import default_ncup_args
    from flowhip.models import build_model
    ref_raft, _ = ref_modules

    torch.manual_seed(0)
    args = default_ncup_args(model="raft", small=False)
    args.mixed_precision = False
    args.align_corners = False  # reference train.py:294 flag (accessed by ref)
    ours = build_model(args)
    ref = ref_raft.RAFT(args)
    ref.load_state_dict(ours.state_dict())

    img1 = torch.rand(1, 3, 128, 128) * 255
    img2 = torch.rand(1, 3, 128, 128) * 255
    preds_ref = ref(img1, img2, iters=2)
    preds_ours = ours(img1, img2, iters=2)
    for a, b in zip(preds_ref, preds_ours):
        assert (a - b).abs().max().item() < 1e-4

    low_r, up_r = ref(img1, img2, iters=2, test_mode=True)
    low_o, up_o = ours(img1, img2, iters=2, test_mode=True)
    assert (low_r - low_o).abs().max().item() < 1e-4
    assert (up_r - up_o).abs().max().item() < 1e-4


def test_raft_small_reference_is_broken_ours_works(ref_modules):
    """Reference defect: raft.py:134 passes align_corners= to upflow8, whose
    only definition (utils/utils.py:82) doesn't accept it — the reference's
    RAFT-small (mask-less) upsample path always raises TypeError as shipped.
    We keep the working utils.py default (align_corners=True, original-RAFT
    behavior). Documented in PARITY.md."""
    from flowhip.config.args import default_ncup_args
    from flowhip.models import build_model
    ref_raft, _ = ref_modules

    torch.manual_seed(0)
    args = default_ncup_args(model="raft", small=True)
    args.mixed_precision = False
    args.align_corners = False
    img1 = torch.rand(1, 3, 128, 128) * 255
    img2 = torch.rand(1, 3, 128, 128) * 255

    ours = build_model(args)
    out = ours(img1, img2, iters=2)
    assert len(out) == 2 and out[-1].shape == (1, 2, 128, 128)

    ref = ref_raft.RAFT(args)
    with pytest.raises(TypeError):
        ref(img1, img2, iters=2)


@pytest.mark.parametrize("shared", [True, False])
@pytest.mark.parametrize("pool", ["conf_based", "max_pooling"])
def test_nconv_unet_bitexact_vs_reference(ref_modules, shared, pool):
    """NConvUNet forward is BIT-exact vs the reference in every
    shared-encoder / pooling configuration — direct proof that eliding the
    reference's dead deepest encoder stage (see flowhip/nn/nconv.py) changes
    nothing."""
    sys.path.insert(0, REF_CORE)
    try:
        import nconv_modules as ref_nconv
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn.nconv import NConvUNet

    torch.manual_seed(1)
    ours = NConvUNet(in_ch=1, channels_multiplier=2, num_downsampling=3,
                     shared_encoder=shared, data_pooling=pool)
    ref = ref_nconv.NConvUNet(in_ch=1, channels_multiplier=2,
                              num_downsampling=3, shared_encoder=shared,
                              data_pooling=pool)
    ref.load_state_dict(ours.state_dict())

    data = torch.randn(2, 1, 64, 64)
    conf = torch.rand(2, 1, 64, 64) * (torch.rand(2, 1, 64, 64) > 0.5)
    x_o, c_o = ours((data, conf))
    x_r, c_r = ref((data, conf))
    assert torch.equal(x_o, x_r)
    assert torch.equal(c_o, c_r)


def test_pac_ops_bitexact_vs_reference(ref_modules):
    """packernel2d / pacconv2d and both guided-upsampling heads are bit-exact
    vs the reference on CPU under shared weights."""
    sys.path.insert(0, REF_CORE)
    try:
        import pac_modules as ref_pac
        import pac_upsampler as ref_pup
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import pac as our_pac
    from flowhip.nn import pac_upsampler as our_pup

    torch.manual_seed(0)
    guide = torch.randn(2, 4, 16, 16)
    x = torch.randn(2, 6, 16, 16)
    w = torch.randn(8, 6, 3, 3)
    b = torch.randn(8)
    k_o, _ = our_pac.packernel2d(guide, kernel_size=3, stride=1, padding=1,
                                 dilation=1)
    k_r, _ = ref_pac.packernel2d(guide, kernel_size=3, stride=1, padding=1,
                                 dilation=1)
    assert torch.equal(k_o, k_r)
    y_o = our_pac.pacconv2d(x, k_o, w, b, stride=1, padding=1, dilation=1)
    y_r = ref_pac.pacconv2d(x, k_r, w, b, stride=1, padding=1, dilation=1)
    assert torch.equal(y_o, y_r)

    torch.manual_seed(2)
    up_o = our_pup.PacJointUpsample(factor=4, channels=1, guide_channels=3)
    up_r = ref_pup.PacJointUpsample(factor=4, channels=1, guide_channels=3)
    up_r.load_state_dict(up_o.state_dict())
    lr = torch.randn(2, 1, 8, 8)
    hr_guide = torch.randn(2, 3, 32, 32)
    assert torch.equal(up_o(lr, hr_guide), up_r(lr, hr_guide))

    torch.manual_seed(3)
    dj_o = our_pup.DJIF(factor=4, channels=1, guide_channels=3)
    dj_r = ref_pup.DJIF(factor=4, channels=1, guide_channels=3)
    dj_r.load_state_dict(dj_o.state_dict())
    assert torch.equal(dj_o(lr, hr_guide), dj_r(lr, hr_guide))


def test_reflective_flag_surface_matches_reference():
    """The reflective-CLI flag surface for the three module families is
    name-for-name identical to what the reference's introspection machinery
    generates (core/utils/args.py driven exactly as train.py:299-342 drives
    it, with the NCUP classes selected). Defaults deliberately differ: ours
    are the SHIPPED NCUP configuration (SURVEY §2.5 / PARITY.md), the
    reference's are the raw constructor defaults."""
    import argparse

    saved_argv = sys.argv
    sys.path.insert(0, REF_CORE)
    try:
        sys.argv = ["x", "--final_upsampling", "NConvUpsampler",
                    "--interp_net", "NConvUNet",
                    "--weights_est_net", "Simple"]
        from utils.args import (_add_arguments_for_module, str2bool,
                                str2intlist)
        import interp_weights_est as ref_iwe
        import nconv_modules as ref_nconv
        import upsampler as ref_up

        parser = argparse.ArgumentParser()
        _add_arguments_for_module(
            parser, ref_up, name="final_upsampling", default_class=None,
            exclude_classes=["_*"],
            exclude_params=["self", "args", "interpolation_net",
                            "weights_est_net", "size"],
            forced_default_types={"scale": int,
                                  "use_data_for_guidance": str2bool,
                                  "channels_to_batch": str2bool,
                                  "use_residuals": str2bool,
                                  "est_on_high_res": str2bool})
        _add_arguments_for_module(
            parser, ref_nconv, name="interp_net", default_class=None,
            exclude_classes=["_*"], exclude_params=["self", "args"],
            forced_default_types={"encoder_fiter_sz": int,
                                  "decoder_fiter_sz": int,
                                  "out_filter_size": int,
                                  "use_double_conv": str2bool,
                                  "use_bias": str2bool})
        _add_arguments_for_module(
            parser, ref_iwe, name="weights_est_net", default_class=None,
            exclude_classes=["_*"],
            exclude_params=["self", "args", "out_ch", "final_act"],
            unknown_default_types={"num_ch": str2intlist,
                                   "filter_sz": str2intlist},
            forced_default_types={"dilation": str2intlist})
    finally:
        sys.argv = saved_argv
        sys.path.remove(REF_CORE)

    ref_opts = sorted(o for a in parser._actions for o in a.option_strings
                      if o.startswith("--") and o != "--help")

    from flowhip.config import build_train_parser
    ours = build_train_parser(argv=[])
    our_opts = sorted(o for a in ours._actions for o in a.option_strings
                      if o.startswith(("--final_upsampling", "--interp_net",
                                       "--weights_est_net")))
    assert our_opts == ref_opts
    assert len(ref_opts) == 24


def test_flow_viz_bitexact_vs_reference():
    """flow_to_color / flow_to_image match the reference byte-for-byte on
    finite flows (our only change is defined behavior for non-finite
    values, which crash/garble the reference)."""
    import numpy as np

    sys.path.insert(0, REF_CORE)
    try:
        from utils import flow_viz as ref_viz
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.data import flow_viz as our_viz

    rng = np.random.default_rng(0)
    flow = (rng.standard_normal((16, 20, 2)) * 10).astype(np.float32)
    assert np.array_equal(our_viz.flow_to_color(flow.copy()),
                          ref_viz.flow_to_color(flow.copy()))
    assert np.array_equal(our_viz.flow_to_image(flow.copy()),
                          ref_viz.flow_to_image(flow.copy()))


def test_geometry_utils_bitexact_vs_reference():
    """InputPadder (both modes), bilinear_sampler, coords_grid, upflow8 and
    the scipy forward_interpolate warm start are bit-exact vs
    core/utils/utils.py."""
    sys.path.insert(0, REF_CORE)
    try:
        from utils import utils as ref_utils
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.utils import geometry as ours

    torch.manual_seed(0)
    for mode in ("sintel", "kitti"):
        x = torch.randn(2, 3, 437, 1021)
        pr = ref_utils.InputPadder(x.shape, mode=mode)
        po = ours.InputPadder(x.shape, mode=mode)
        a = pr.pad(x.clone())[0]
        b = po.pad(x.clone())[0]
        assert torch.equal(a, b)
        assert torch.equal(pr.unpad(a), po.unpad(b))

    img = torch.randn(4, 2, 17, 23)
    coords = torch.rand(4, 9, 11, 2) * 25 - 1  # includes out-of-bounds
    assert torch.equal(ref_utils.bilinear_sampler(img, coords),
                       ours.bilinear_sampler(img, coords))
    assert torch.equal(ref_utils.coords_grid(2, 7, 9), ours.coords_grid(2, 7, 9))
    f = torch.randn(1, 2, 6, 8)
    assert torch.equal(ref_utils.upflow8(f), ours.upflow8(f))
    fl = torch.randn(2, 12, 14) * 3
    assert torch.equal(ref_utils.forward_interpolate(fl.clone()),
                       ours.forward_interpolate(fl.clone()))


@pytest.mark.parametrize("norm", ["batch", "group", "instance", "none"])
def test_extractors_bitexact_vs_reference(ref_modules, norm):
    """Basic/Small encoders match the reference bit-for-bit under shared
    weights for every norm_fn."""
    sys.path.insert(0, REF_CORE)
    try:
        import extractor as ref_ext
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import extractor as our_ext

    x = torch.randn(2, 3, 64, 64)
    for our_cls, ref_cls, dim in ((our_ext.BasicEncoder, ref_ext.BasicEncoder, 64),
                                  (our_ext.SmallEncoder, ref_ext.SmallEncoder, 32)):
        torch.manual_seed(1)
        o = our_cls(output_dim=dim, norm_fn=norm, dropout=0.0)
        r = ref_cls(output_dim=dim, norm_fn=norm, dropout=0.0)
        r.load_state_dict(o.state_dict())
        o.eval()
        r.eval()
        assert torch.equal(o(x), r(x))


def test_update_blocks_bitexact_vs_reference(ref_modules):
    """Small/Basic update blocks (motion encoders + GRUs + flow/mask heads)
    match bit-for-bit under shared weights."""
    sys.path.insert(0, REF_CORE)
    try:
        import update as ref_upd
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.config.args import default_ncup_args
    from flowhip.nn import update as our_upd

    torch.manual_seed(0)

    args = default_ncup_args(model="raft", small=True)
    args.corr_levels, args.corr_radius = 4, 3  # set by RAFT.__init__ (raft.py:28-29)
    cor_planes = args.corr_levels * (2 * args.corr_radius + 1) ** 2
    o = our_upd.SmallUpdateBlock(args, hidden_dim=96)
    r = ref_upd.SmallUpdateBlock(args, hidden_dim=96)
    r.load_state_dict(o.state_dict())
    net = torch.randn(2, 96, 8, 8)
    inp = torch.randn(2, 64, 8, 8)
    corr = torch.randn(2, cor_planes, 8, 8)
    flow = torch.randn(2, 2, 8, 8)
    net_o, mask_o, df_o = o(net, inp, corr, flow)
    net_r, mask_r, df_r = r(net, inp, corr, flow)
    assert mask_o is None and mask_r is None
    assert torch.equal(net_o, net_r) and torch.equal(df_o, df_r)

    args = default_ncup_args(model="raft", small=False)
    args.corr_levels, args.corr_radius = 4, 4  # raft.py:33-34
    cor_planes = args.corr_levels * (2 * args.corr_radius + 1) ** 2
    o = our_upd.BasicUpdateBlock(args, hidden_dim=128)
    r = ref_upd.BasicUpdateBlock(args, hidden_dim=128)
    r.load_state_dict(o.state_dict())
    net = torch.randn(2, 128, 8, 8)
    inp = torch.randn(2, 128, 8, 8)
    corr = torch.randn(2, cor_planes, 8, 8)
    net_o, mask_o, df_o = o(net, inp, corr, flow)
    net_r, mask_r, df_r = r(net, inp, corr, flow)
    assert torch.equal(net_o, net_r)
    assert torch.equal(mask_o, mask_r)
    assert torch.equal(df_o, df_r)


def test_pac_gradients_match_reference_handwritten_backward():
    """the reference's hand-written PAC backwards
    (pac_modules.py:112-131,166-202) define the gradient contract. Our
    implementation matches them (and fp64 finite differences) through
    packernel2d -> pacconv2d."""
    sys.path.insert(0, REF_CORE)
    try:
        import pac_modules as ref_pac
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import pac as our_pac

    torch.manual_seed(0)
    guide = torch.randn(2, 4, 10, 10)
    x = torch.randn(2, 6, 10, 10)
    w = torch.randn(5, 6, 3, 3)
    b = torch.randn(5)

    def grads(mod):
        ins = [t.clone().requires_grad_(True) for t in (guide, x, w, b)]
        k, _ = mod.packernel2d(ins[0], kernel_size=3, stride=1, padding=1,
                               dilation=1)
        y = mod.pacconv2d(ins[1], k, ins[2], ins[3], stride=1, padding=1,
                          dilation=1)
        return torch.autograd.grad(y.square().sum(), ins)

    for g_o, g_r, name in zip(grads(our_pac), grads(ref_pac),
                              ["dguide", "dx", "dw", "db"]):
        rel = (g_o - g_r).norm().item() / (g_r.norm().item() + 1e-12)
        assert rel < 1e-5, f"{name}: rel {rel:.2e}"


def test_pac_modules_transpose_pool_bitexact_vs_reference(ref_modules):
    """PacConvTranspose2d and PacPool2d module classes match the reference
    bit-for-bit under shared weights."""
    sys.path.insert(0, REF_CORE)
    try:
        import pac_modules as ref_pac
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import pac as our_pac

    torch.manual_seed(0)
    o = our_pac.PacConvTranspose2d(4, 5, kernel_size=5, stride=2, padding=2,
                                   output_padding=1)
    r = ref_pac.PacConvTranspose2d(4, 5, kernel_size=5, stride=2, padding=2,
                                   output_padding=1)
    r.load_state_dict(o.state_dict())
    x = torch.randn(2, 4, 8, 8)
    guide = torch.randn(2, 3, 16, 16)
    assert torch.equal(o(x, guide), r(x, guide))

    o2 = our_pac.PacPool2d(kernel_size=3, stride=2, padding=1)
    r2 = ref_pac.PacPool2d(kernel_size=3, stride=2, padding=1)
    xp = torch.randn(2, 4, 16, 16)
    gp = torch.randn(2, 4, 16, 16)
    assert torch.equal(o2(xp, gp), r2(xp, gp))


def test_unet_weights_est_bitexact_vs_reference(ref_modules):
    sys.path.insert(0, REF_CORE)
    try:
        import interp_weights_est as ref_iwe
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import interp_weights_est as our_iwe

    torch.manual_seed(1)
    o = our_iwe.UNet(num_ch=[5, 8, 12, 16], out_ch=1)
    r = ref_iwe.UNet(num_ch=[5, 8, 12, 16], out_ch=1)
    r.load_state_dict(o.state_dict())
    o.eval()
    r.eval()
    x = torch.randn(2, 5, 32, 32)
    assert torch.equal(o(x), r(x))


@pytest.mark.parametrize("weights_in,kw", [
    (5, dict(use_data_for_guidance=True, channels_to_batch=True,
             use_residuals=False)),
    (5, dict(use_data_for_guidance=True, channels_to_batch=True,
             use_residuals=True)),
    (4, dict(use_data_for_guidance=False, channels_to_batch=False,
             use_residuals=False)),
    (0, dict(use_data_for_guidance=False)),  # binary-weights fallback
])
def test_nconv_upsampler_head_bitexact_vs_reference(ref_modules, weights_in, kw):
    """The full NCUP head (zero-injection + weights estimation + NConvUNet)
    matches the reference bit-for-bit across the guidance / folding /
    residual flag combinations, including the binary-weights fallback."""
    sys.path.insert(0, REF_CORE)
    try:
        import interp_weights_est as ref_iwe
        import nconv_modules as ref_nconv
        import upsampler as ref_up
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import upsampler as our_up
    from flowhip.nn.interp_weights_est import Simple
    from flowhip.nn.nconv import NConvUNet

    torch.manual_seed(3)
    inet_o = NConvUNet(in_ch=1, channels_multiplier=2, num_downsampling=1)
    inet_r = ref_nconv.NConvUNet(in_ch=1, channels_multiplier=2,
                                 num_downsampling=1)
    inet_r.load_state_dict(inet_o.state_dict())
    if weights_in:
        w_o = Simple(num_ch=[weights_in, 8], out_ch=1, filter_sz=[3, 1],
                     dilation=[1, 1], final_act=torch.sigmoid)
        w_r = ref_iwe.Simple(num_ch=[weights_in, 8], out_ch=1,
                             filter_sz=[3, 1], dilation=[1, 1],
                             final_act=torch.sigmoid)
        w_r.load_state_dict(w_o.state_dict())
    else:
        w_o = w_r = None

    o = our_up.NConvUpsampler(scale=4, interpolation_net=inet_o,
                              weights_est_net=w_o, **kw)
    r = ref_up.NConvUpsampler(scale=4, interpolation_net=inet_r,
                              weights_est_net=w_r, **kw)

    torch.manual_seed(4)
    lr = torch.rand(2, 1, 8, 8) * (torch.rand(2, 1, 8, 8) > 0.6)
    guide = (lr.new_ones(2, 1, 32, 32) if weights_in == 0
             else torch.randn(2, 4, 32, 32))
    assert torch.equal(o(lr.clone(), guide.clone()),
                       r(lr.clone(), guide.clone()))


def test_pac_upsampler_baseline_heads_bitexact_vs_reference(ref_modules):
    """The remaining guided-upsampling baseline heads (Lite/Wide/bilateral/
    bilinear) and the th_epe metric helper match the reference bit-for-bit."""
    sys.path.insert(0, REF_CORE)
    try:
        import pac_upsampler as ref_pup
    finally:
        sys.path.remove(REF_CORE)
    from flowhip.nn import pac_upsampler as our_pup

    torch.manual_seed(0)
    lr = torch.randn(2, 1, 8, 8)
    guide = torch.randn(2, 3, 32, 32)

    for name in ("PacJointUpsampleLite", "DJIFWide"):
        torch.manual_seed(5)
        o = getattr(our_pup, name)(factor=4, channels=1, guide_channels=3)
        r = getattr(ref_pup, name)(factor=4, channels=1, guide_channels=3)
        r.load_state_dict(o.state_dict())
        assert torch.equal(o(lr, guide), r(lr, guide)), name

    o = our_pup.JointBilateral(factor=4, channels=1, kernel_size=5,
                               scale_space=0.2, scale_color=5.0)
    r = ref_pup.JointBilateral(factor=4, channels=1, kernel_size=5,
                               scale_space=0.2, scale_color=5.0)
    r.load_state_dict(o.state_dict())
    assert torch.equal(o(lr, guide), r(lr, guide))

    assert torch.equal(our_pup.Bilinear(factor=4)(lr, guide),
                       ref_pup.Bilinear(factor=4)(lr, guide))

    a = torch.randn(2, 2, 16, 16)
    b = torch.randn(2, 2, 16, 16)
    assert torch.equal(our_pup.th_epe(a, b), ref_pup.th_epe(a, b))


def test_checkpoint_file_interchange_with_reference(ref_modules, tmp_path):
    """The literal user workflow: a checkpoint file written by this framework
    loads strict into the reference model (after its own module. strip), and
    a reference-style file (module.-prefixed raw state_dict, as its
    DataParallel training saves) loads into ours."""
    from flowhip.engine import checkpoints
    from flowhip.models import build_model
    _, ref_nc_dbl = ref_modules

    args = _ncup_args()
    ours = build_model(args)

    # ours -> reference
    path = os.path.join(tmp_path, "ours.pth")
    checkpoints.save_weights(ours, path)
    sd = torch.load(path, weights_only=True)
    ref = ref_nc_dbl.RAFT(args)
    ref.load_state_dict({k[len("module."):]: v for k, v in sd.items()},
                        strict=True)

    # reference-style file -> ours
    ref_path = os.path.join(tmp_path, "ref.pth")
    torch.save({"module." + k: v for k, v in ref.state_dict().items()},
               ref_path)
    ours2 = build_model(args)
    checkpoints.load_weights(ours2, ref_path, strict=True)
    assert torch.equal(ours2.state_dict()["fnet.conv1.weight"],
                       ours.state_dict()["fnet.conv1.weight"])


def test_enforce_pos_hook_bitexact_vs_reference(ref_modules):
    """The generic EnforcePos hook API vs the reference's
    (nconv_modules.py:218-283): identical weight_p initialization (the
    pos(weight) quirk), identical recomputed effective weight after a
    parameter update, identical remove() freeze."""
    import nconv_modules as ref_nc

    from flowhip.nn.nconv import EnforcePos, remove_weight_pos

    torch.manual_seed(31)
    w0 = torch.randn(3, 2, 3, 3)

    m_ref = torch.nn.Conv2d(2, 3, 3, bias=False)
    m_our = torch.nn.Conv2d(2, 3, 3, bias=False)
    with torch.no_grad():
        m_ref.weight.copy_(w0)
        m_our.weight.copy_(w0)

    ref_nc.EnforcePos.apply(m_ref, "weight", "softplus")
    EnforcePos.apply(m_our, "weight", "softplus")
    assert torch.equal(m_ref.weight_p, m_our.weight_p)

    with torch.no_grad():
        m_ref.weight_p.add_(0.25)
        m_our.weight_p.add_(0.25)
    x = torch.randn(1, 2, 8, 8)
    torch.testing.assert_close(m_ref(x), m_our(x), rtol=0, atol=0)
    assert torch.equal(m_ref.weight, m_our.weight)

    remove_weight_pos(m_our)
    assert torch.equal(m_our.weight, m_ref.weight)
    assert "weight_p" not in dict(m_our.named_parameters())
