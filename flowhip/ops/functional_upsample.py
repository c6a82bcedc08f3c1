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
