"""Autograd bindings for the PAC HIP kernels (#9/#10; reference
core/pac_modules.py). Cover the configuration space the guided-upsampling
baseline heads use (gaussian kernel, smooth none, no mask,
channel_wise=False, stride-1 core); everything else stays on the torch
implementation in flowhip/nn/pac.py.
"""

import torch

from . import _ext


class PacKernelGaussFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, f, K, dil, norm):
        f = f.contiguous()
        k = _ext.ext().packernel_fwd(f, K, dil, norm)  # (B, K2, H, W)
        ctx.save_for_backward(f, k)
        ctx.meta = (K, dil, norm)
        return k

    @staticmethod
    def backward(ctx, dk):
        f, k = ctx.saved_tensors
        K, dil, norm = ctx.meta
        df = _ext.ext().packernel_bwd(f, k, dk.contiguous(), K, dil, norm)
        return df, None, None, None


class PacConv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kr, weight, bias, pH, pW, dil, shared):
        x = x.contiguous()
        kr = kr.contiguous()
        w = weight.contiguous()
        b = bias.contiguous() if bias is not None else None
        out = _ext.ext().pacconv_fwd(x, kr, w, b, pH, pW, dil, shared)
        ctx.save_for_backward(x, kr, w)
        ctx.meta = (pH, pW, dil, shared, bias is not None)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, kr, w = ctx.saved_tensors
        pH, pW, dil, shared, has_bias = ctx.meta
        dy = dy.contiguous()
        dx, dk, dw = _ext.ext().pacconv_bwd(dy, x, kr, w, pH, pW, dil,
                                            shared)
        dbias = dy.sum(dim=(0, 2, 3)) if has_bias else None
        return dx, dk, dw, dbias, None, None, None, None


class PacPool2dFn(torch.autograd.Function):
    """Adaptive pooling (reference pac_modules.py:288-329): one gather
    kernel forward; backward = atomic-scatter dx + per-tap channel-gather
    dk (dx accumulation order is non-deterministic under overlapping
    windows — baseline-head op, not on the training hot path)."""

    @staticmethod
    def forward(ctx, x, kr, K, sH, sW, pH, pW, dil):
        x = x.contiguous()
        kr = kr.contiguous()
        out = _ext.ext().pacpool_fwd(x, kr, K, sH, sW, pH, pW, dil)
        ctx.save_for_backward(x, kr)
        ctx.meta = (K, sH, sW, pH, pW, dil)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, kr = ctx.saved_tensors
        K, sH, sW, pH, pW, dil = ctx.meta
        dx, dk = _ext.ext().pacpool_bwd(dy.contiguous(), x, kr, K, sH, sW,
                                        pH, pW, dil)
        return dx, dk, None, None, None, None, None, None


def pac_kernel_fusable(input, mask, kernel_type, smooth_kernel_type,
                       channel_wise, kernel_size, dilation, eff_stride,
                       eff_padding):
    K = kernel_size[0]
    d = dilation[0]
    return (input.is_cuda and input.dtype == torch.float32
            and _ext.ext() is not None and not _ext.force_ref()
            and mask is None
            and kernel_type == "gaussian" and smooth_kernel_type == "none"
            and not channel_wise
            and kernel_size[0] == kernel_size[1] and K in (3, 5, 7)
            and dilation[0] == dilation[1]
            and tuple(eff_stride) == (1, 1)
            and tuple(eff_padding) == ((K - 1) * d // 2, (K - 1) * d // 2))


def pac_conv_fusable(x, kernel, weight, stride, padding, dilation):
    K = weight.shape[-1]
    if not (x.is_cuda and x.dtype == torch.float32
            and _ext.ext() is not None and not _ext.force_ref()):
        return False
    from torch.nn.modules.utils import _pair
    stride, padding, dilation = _pair(stride), _pair(padding), _pair(dilation)
    if stride != (1, 1) or dilation[0] != dilation[1]:
        return False
    if weight.shape[-2] != K or K not in (3, 5, 7):
        return False
    nw = weight.numel()
    if nw * 4 > 65536:
        return False
    oh = x.shape[2] + 2 * padding[0] - (K - 1) * dilation[0]
    ow = x.shape[3] + 2 * padding[1] - (K - 1) * dilation[1]
    return kernel.shape[-2] == oh and kernel.shape[-1] == ow
