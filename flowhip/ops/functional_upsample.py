"""Autograd binding for the fused convex-combination upsample kernel
(csrc/convex_upsample.hip; reference core/raft.py:73-84)."""

import torch

from . import _ext


class ConvexUpsampleFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, flow, mask, factor):
        flow = flow.contiguous()
        mask = mask.contiguous()
        out = _ext.ext().convex_up_fwd(flow, mask, factor)
        ctx.save_for_backward(flow, mask)
        ctx.factor = factor
        return out

    @staticmethod
    def backward(ctx, gout):
        flow, mask = ctx.saved_tensors
        gflow, gmask = _ext.ext().convex_up_bwd(gout.contiguous(), flow, mask,
                                                ctx.factor)
        return gflow, gmask, None


class ZeroInjectFn(torch.autograd.Function):
    """Sparse zero-injection (kernel #8; ref upsampler.py:179-210): one
    coalesced write pass forward, one strided gather backward."""

    @staticmethod
    def forward(ctx, inp, sH, sW, oh, ow):
        ctx.meta = (int(sH), int(sW), inp.shape[2], inp.shape[3])
        return _ext.ext().zero_inject_fwd(inp, int(sH), int(sW), int(oh),
                                          int(ow))

    @staticmethod
    def backward(ctx, gout):
        sH, sW, ih, iw = ctx.meta
        return _ext.ext().zero_inject_bwd(gout, sH, sW, ih, iw), None, None, None, None



class AreaUp2xFn(torch.autograd.Function):
    """Exact-2x area interpolation (== nearest duplication) with a
    gather-only backward (torch's is an atomic adaptive-pool scatter)."""

    @staticmethod
    def forward(ctx, x):
        return _ext.ext().area_up2x_fwd(x.contiguous())

    @staticmethod
    def backward(ctx, gout):
        return _ext.ext().area_up2x_bwd(gout)



class Up2xCatFn(torch.autograd.Function):
    """Fused nearest-2x upsample + channel concat (NConvUNet decoder skip;
    nconv_modules.py:128-134). Backward: the upsample grad is the area-2x
    gather kernel; the skip grad is a zero-copy channel slice."""

    @staticmethod
    def forward(ctx, low, skip):
        ctx.c1 = low.shape[1]
        ctx.low_hw = (low.shape[2], low.shape[3])
        return _ext.ext().up2x_cat_fwd(low.contiguous(), skip.contiguous())

    @staticmethod
    def backward(ctx, gout):
        c1 = ctx.c1
        g = gout.contiguous()
        dlow = _ext.ext().area_up2x_bwd(g[:, :c1].contiguous())
        return dlow, g[:, c1:]
