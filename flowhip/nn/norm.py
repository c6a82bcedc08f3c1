"""Channels-last InstanceNorm2d backed by the fused HIP kernel.

torch lowers InstanceNorm to batch_norm on a (1, N*C, H, W) view, which on
channels_last tensors costs an uncoalesced layout copy both ways per call
(profiles/ tprof6). This module dispatches to a direct (N,P,C)-layout
reduction kernel on GPU and falls back to the stock implementation
elsewhere. affine/track_running_stats stay False (extractor config,
reference extractor.py:27), so the state dict is unchanged.
"""

import torch
import torch.nn as nn

from ..ops import _ext


class _InstNormCLFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, eps):
        y, mean, rstd = _ext.ext().instnorm_cl_fwd(x, eps)
        ctx.save_for_backward(x, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        return _ext.ext().instnorm_cl_bwd(x, dy, mean, rstd), None


class InstanceNorm2d(nn.InstanceNorm2d):
    def forward(self, x):
        if (x.is_cuda and not self.affine and not self.track_running_stats
                and x.is_contiguous(memory_format=torch.channels_last)
                and x.dtype in (torch.float32, torch.bfloat16)
                and x.shape[1] % 8 == 0 and x.shape[1] >= 8
                and _ext.ext() is not None and not _ext.force_ref()):
            return _InstNormCLFn.apply(x, self.eps)
        return super().forward(x)


class _FrozenBNFn(torch.autograd.Function):
    """Eval-mode BatchNorm y = (x - rm) * w/sqrt(rv+eps) + b with a fused
    backward: torch's native_batch_norm_backward spends ~105 us/call on the
    (sum g, sum g*xhat) reductions at encoder shapes; one col_sum2 pass +
    an elementwise scale replaces it (~3x). Frozen stats (freeze_bn after
    the chairs stage — reference train.py:185) make stats constants, so
        dx = g * s,  db = sum(g),  dw = (sum(g*x) - rm*sum(g)) * invstd.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, rm, rv, eps):
        invstd = (rv.float() + eps).rsqrt()
        s = (weight.float() * invstd).contiguous()
        t = (bias.float() - rm.float() * s).contiguous()
        # fp32 math, bf16 store: quantization lands exactly where the stock
        # autocast path (fp32 BN output -> bf16 cast at the next conv) puts
        # it, so frozen-BN introduces no extra drift vs the reference
        y = _ext.ext().frozen_bn_apply(x, s, t)
        ctx.save_for_backward(x, s, invstd, rm)
        return y

    @staticmethod
    def backward(ctx, g):
        x, s, invstd, rm = ctx.saved_tensors
        g = g.contiguous(memory_format=torch.channels_last)
        if g.dtype != torch.bfloat16:
            g = g.to(torch.bfloat16)
        sums = _ext.ext().col_sum2_bf16(
            g, x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16))
        db = sums[0]
        dw = (sums[1] - rm.float() * db) * invstd
        dx = _ext.ext().frozen_bn_apply(g, s, None)
        return dx, dw, db, None, None, None


class BatchNorm2d(nn.BatchNorm2d):
    """nn.BatchNorm2d whose FROZEN (eval-mode) GPU path runs the fused
    backward above. Training mode (chairs stage) and CPU use the stock
    implementation; state dict unchanged."""

    def forward(self, x):
        use = (not self.training and x.is_cuda and self.affine
               and self.track_running_stats
               and x.is_contiguous(memory_format=torch.channels_last)
               and x.dtype == torch.bfloat16
               and x.shape[1] % 8 == 0
               and _ext.ext() is not None and not _ext.force_ref())
        if use:
            return _FrozenBNFn.apply(x, self.weight, self.bias,
                                     self.running_mean, self.running_var,
                                     self.eps)
        return super().forward(x)
