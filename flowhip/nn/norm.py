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
