"""Pixel-adaptive convolution (PAC) ops — fresh implementation.

Capability parity with the reference `core/pac_modules.py` (kernels #9/#10 of
SURVEY.md §2.2; the API of Su et al., CVPR 2019): a content-adaptive kernel
K derived from guidance features multiplies the im2col window of the input
before the learned filter W is applied.

    packernel2d:  K[b, :, kh, kw, oy, ox] = exp(-1/2 ||g(window) - g(center)||^2)
                  ('gaussian'; 'inv_a_l[_asym][_fixed]' -> a + (d2+eps)^(l/2))
    pacconv2d:            out = sum_window (unfold(x) * K) . W  (+ bias)
    pacconv_transpose2d:  zero-stuffed stride expansion, then pacconv with
                          W^T (fractional-stride form)
    pacpool2d:            out = sum_window (unfold(x) * K)

Implementation notes (fresh, not a translation):
- forward is expressed with unfold + einsum and differentiated by torch
  autograd (the reference hand-writes backward Functions to save the im2col
  buffer; on MI355X's 288 GB the recompute/memory trade is not binding for
  these non-default upsampler baselines — revisit if PAC becomes a hot path);
- only `smooth_kernel_type` variants and fillers actually constructible via
  the reference CLI surface are supported; unsupported combos raise.
State-dict parameter names match the reference (weight / bias /
smooth_kernel / inv_alpha / inv_lambda).
"""

import math

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.nn.modules.utils import _pair


def nd2col(input_nd, kernel_size, stride=1, padding=0, output_padding=0,
           dilation=1, transposed=False):
    """im2col with a fractional-stride (transposed) mode.

    Out: (N, C, kh, kw, oH, oW). Transposed mode zero-stuffs the input by
    `stride` and pads by (k-1)*d - p (+output_padding on the far side),
    matching conv_transpose output geometry.
    """
    kernel_size = _pair(kernel_size)
    stride = _pair(stride)
    padding = _pair(padding)
    output_padding = _pair(output_padding)
    dilation = _pair(dilation)

    if transposed:
        w_one = input_nd.new_ones(1, 1, 1, 1)
        pad = [(k - 1) * d - p for (k, d, p) in zip(kernel_size, dilation, padding)]
        input_nd = F.conv_transpose2d(input_nd, w_one.expand(input_nd.shape[1], 1, 1, 1),
                                      stride=stride, groups=input_nd.shape[1])
        input_nd = F.pad(input_nd, (pad[1], pad[1] + output_padding[1],
                                    pad[0], pad[0] + output_padding[0]))
        stride = _pair(1)
        padding = _pair(0)

    bs, nch = input_nd.shape[:2]
    in_sz = input_nd.shape[2:]
    out_sz = tuple((i + 2 * p - d * (k - 1) - 1) // s + 1
                   for (i, k, d, p, s) in zip(in_sz, kernel_size, dilation,
                                              padding, stride))
    cols = F.unfold(input_nd, kernel_size, dilation, padding, stride)
    return cols.view(bs, nch, *kernel_size, *out_sz)


def packernel2d(input, mask=None, kernel_size=0, stride=1, padding=0,
                output_padding=0, dilation=1, kernel_type="gaussian",
                smooth_kernel_type="none", smooth_kernel=None, inv_alpha=None,
                inv_lambda=None, channel_wise=False, normalize_kernel=False,
                transposed=False):
    """Adapting kernel from guidance features (reference packernel2d
    :332-424). Returns (kernel, output_mask)."""
    kernel_size = _pair(kernel_size)
    stride_ = _pair(stride)
    padding_ = _pair(padding)
    output_padding_ = _pair(output_padding)
    dilation_ = _pair(dilation)
    output_mask = mask is not None
    norm = None

    if mask is not None and mask.dtype != input.dtype:
        mask = mask.to(dtype=input.dtype, device=input.device)

    if transposed:
        in_sz = tuple(int((o - op - 1 - (k - 1) * d + 2 * p) // s) + 1
                      for (o, k, s, p, op, d) in
                      zip(input.shape[-2:], kernel_size, stride_, padding_,
                          output_padding_, dilation_))
    else:
        in_sz = input.shape[-2:]

    # Fused HIP path (kernel #9): gaussian, no mask, smooth none, stride-1
    # effective unfold with center padding (both the direct stride-1 case
    # and the transposed case). Must dispatch BEFORE the ones-mask below
    # reassigns `mask`; the kernel replicates the mask's border-tap zeroing
    # for normalize_kernel.
    if transposed:
        _eff_stride = _pair(1)
        _eff_padding = tuple((k - 1) * d // 2
                             for (k, d) in zip(kernel_size, dilation_))
    else:
        _eff_stride, _eff_padding = stride_, padding_
    from ..ops.functional_pac import PacKernelGaussFn, pac_kernel_fusable
    if pac_kernel_fusable(input, mask, kernel_type, smooth_kernel_type,
                          channel_wise, kernel_size, dilation_,
                          _eff_stride, _eff_padding):
        k = PacKernelGaussFn.apply(input, kernel_size[0], dilation_[0],
                                   bool(normalize_kernel))
        bs_ = input.shape[0]
        out = k.view(bs_, 1, kernel_size[0], kernel_size[1],
                     input.shape[-2], input.shape[-1])
        return out, None

    if mask is not None or normalize_kernel:
        mask_pattern = input.new_ones(1, 1, *in_sz)
        mask_pattern = nd2col(mask_pattern, kernel_size, stride=stride_,
                              padding=padding_, output_padding=output_padding_,
                              dilation=dilation_, transposed=transposed)
        if mask is not None:
            mask = nd2col(mask, kernel_size, stride=stride_, padding=padding_,
                          output_padding=output_padding_, dilation=dilation_,
                          transposed=transposed)
            if not normalize_kernel:
                norm = (mask.sum(dim=2, keepdim=True).sum(dim=3, keepdim=True)
                        / mask_pattern.sum(dim=2, keepdim=True).sum(dim=3, keepdim=True))
        else:
            mask = mask_pattern

    if transposed:
        # guidance is already at the (high) output resolution: plain unfold
        # at stride 1 with center padding (reference packernel2d :363-365)
        eff_stride = _pair(1)
        eff_padding = tuple((k - 1) * d // 2
                            for (k, d) in zip(kernel_size, dilation_))
    else:
        eff_stride, eff_padding = stride_, padding_
    feat = nd2col(input, kernel_size, stride=eff_stride,
                  padding=eff_padding, dilation=dilation_)

    bs, k_ch = input.shape[:2]

    if smooth_kernel_type == "none":
        self_idx_h = kernel_size[0] // 2
        self_idx_w = kernel_size[1] // 2
        feat_0 = feat[:, :, self_idx_h:self_idx_h + 1,
                      self_idx_w:self_idx_w + 1, :, :]
    else:
        raise NotImplementedError(
            f"smooth_kernel_type={smooth_kernel_type!r} is not implemented "
            "in flowhip (not reachable from the reference CLI defaults)")

    d = feat - feat_0
    if kernel_type.find("_asym") >= 0:
        d = F.relu(d)
    d2 = d * d
    if not channel_wise:
        d2 = d2.sum(dim=1, keepdim=True)

    if kernel_type == "gaussian":
        out = torch.exp(-0.5 * d2)
    elif kernel_type.startswith("inv_"):
        epsilon = 1e-4
        out = (inv_alpha.view(1, -1, 1, 1, 1, 1)
               + torch.pow(d2 + epsilon,
                           0.5 * inv_lambda.view(1, -1, 1, 1, 1, 1)))
    else:
        raise ValueError(f"kernel_type {kernel_type!r}")

    if mask is not None:
        out = out * mask

    if normalize_kernel:
        norm = out.sum(dim=2, keepdim=True).sum(dim=3, keepdim=True)

    if norm is not None:
        empty_mask = (norm == 0).to(out.dtype)
        out = out / (norm + empty_mask)
        output_mask = (1 - empty_mask) if output_mask else None
    else:
        output_mask = None

    return out, output_mask


def pacconv2d(input, kernel, weight, bias=None, stride=1, padding=0,
              dilation=1, shared_filters=False):
    """out = sum over window of (unfold(input) * kernel) . weight."""
    kernel_size = tuple(weight.shape[-2:])

    from ..ops.functional_pac import PacConv2dFn, pac_conv_fusable
    if pac_conv_fusable(input, kernel, weight, stride, padding, dilation):
        p = _pair(padding)
        d = _pair(dilation)
        bs = input.shape[0]
        kr = kernel.reshape(bs, kernel_size[0] * kernel_size[1],
                            *kernel.shape[-2:])
        return PacConv2dFn.apply(input, kr, weight, bias, p[0], p[1], d[0],
                                 bool(shared_filters))

    cols = nd2col(input, kernel_size, stride=stride, padding=padding,
                  dilation=dilation)
    if shared_filters:
        out = torch.einsum("ijklmn,zykl->ijmn", cols * kernel, weight)
    else:
        out = torch.einsum("ijklmn,ojkl->iomn", cols * kernel, weight)
    if bias is not None:
        out = out + bias.view(1, -1, 1, 1)
    return out


def pacconv_transpose2d(input, kernel, weight, bias=None, stride=1, padding=0,
                        output_padding=0, dilation=1, shared_filters=False):
    """Transposed PAC conv via zero-stuffing + pacconv2d with W^T
    (reference pacconv_transpose2d :452-472)."""
    kernel_size = tuple(weight.shape[-2:])
    stride = _pair(stride)
    padding = _pair(padding)
    output_padding = _pair(output_padding)
    dilation = _pair(dilation)

    ch = input.shape[1]
    w = input.new_ones((ch, 1, 1, 1))
    x = F.conv_transpose2d(input, w, stride=stride, groups=ch)
    pad = [(kernel_size[i] - 1) * dilation[i] - padding[i] for i in range(2)]
    x = F.pad(x, (pad[1], pad[1] + output_padding[1],
                  pad[0], pad[0] + output_padding[0]))
    return pacconv2d(x, kernel, weight.permute(1, 0, 2, 3), bias,
                     dilation=dilation, shared_filters=shared_filters)


def pacpool2d(input, kernel, kernel_size, stride=1, padding=0, dilation=1):
    """Adaptive pooling: sum over window of (unfold(input) * kernel)."""
    kernel_size = _pair(kernel_size)
    bs, in_ch = input.shape[:2]

    from ..ops import _ext
    from ..ops.functional_pac import PacPool2dFn
    s, p, d = _pair(stride), _pair(padding), _pair(dilation)
    if (input.is_cuda and input.dtype == torch.float32
            and kernel.dtype == torch.float32
            and _ext.ext() is not None and not _ext.force_ref()
            and kernel_size[0] == kernel_size[1] and d[0] == d[1]
            and kernel.dim() >= 5):
        K = kernel_size[0]
        kr = kernel.reshape(bs, -1, K * K, *kernel.shape[-2:])
        if kr.shape[1] in (1, in_ch):
            return PacPool2dFn.apply(input, kr, K, s[0], s[1], p[0], p[1],
                                     d[0])

    cols = nd2col(input, kernel_size, stride=stride, padding=padding,
                  dilation=dilation)
    cols = cols * kernel
    return cols.view(bs, in_ch, -1, *cols.shape[-2:]).sum(dim=2)


class _PacConvNd(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride,
                 padding, dilation, transposed, output_padding, bias,
                 pool_only, kernel_type, smooth_kernel_type, channel_wise,
                 normalize_kernel, shared_filters, filler):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.dilation = dilation
        self.transposed = transposed
        self.output_padding = output_padding
        self.pool_only = pool_only
        self.kernel_type = kernel_type
        self.smooth_kernel_type = smooth_kernel_type
        self.channel_wise = channel_wise
        self.normalize_kernel = normalize_kernel
        self.shared_filters = shared_filters
        self.filler = filler

        if any(k % 2 != 1 for k in kernel_size):
            raise ValueError("kernel_size only accept odd numbers")
        if shared_filters:
            assert in_channels == out_channels

        if not pool_only:
            if filler in {"pool", "crf_pool"}:
                assert shared_filters
                self.register_buffer("weight", torch.ones(1, 1, *kernel_size))
                if filler == "crf_pool":
                    self.weight[(0, 0) + tuple(k // 2 for k in kernel_size)] = 0
            elif shared_filters:
                self.weight = nn.Parameter(torch.empty(1, 1, *kernel_size))
            elif transposed:
                self.weight = nn.Parameter(
                    torch.empty(in_channels, out_channels, *kernel_size))
            else:
                self.weight = nn.Parameter(
                    torch.empty(out_channels, in_channels, *kernel_size))
            if bias:
                self.bias = nn.Parameter(torch.empty(out_channels))
            else:
                self.register_parameter("bias", None)

        if kernel_type.startswith("inv_"):
            self.inv_alpha_init = float(kernel_type.split("_")[1])
            self.inv_lambda_init = float(kernel_type.split("_")[2])
            if channel_wise and kernel_type.find("_fixed") < 0:
                if out_channels <= 0:
                    raise ValueError("out_channels needed for channel_wise inv kernel")
                inv_alpha = self.inv_alpha_init * torch.ones(out_channels)
                inv_lambda = self.inv_lambda_init * torch.ones(out_channels)
            else:
                inv_alpha = torch.tensor(self.inv_alpha_init)
                inv_lambda = torch.tensor(self.inv_lambda_init)
            if kernel_type.find("_fixed") < 0:
                self.inv_alpha = nn.Parameter(inv_alpha)
                self.inv_lambda = nn.Parameter(inv_lambda)
            else:
                self.register_buffer("inv_alpha", inv_alpha)
                self.register_buffer("inv_lambda", inv_lambda)
        elif kernel_type != "gaussian":
            raise ValueError(f"kernel_type {kernel_type!r}")

        if smooth_kernel_type != "none":
            raise NotImplementedError(
                f"smooth_kernel_type={smooth_kernel_type!r} not implemented")

        self.reset_parameters()

    def reset_parameters(self):
        if not (self.pool_only or self.filler in {"pool", "crf_pool"}):
            if self.filler == "uniform":
                n = self.in_channels
                for k in self.kernel_size:
                    n *= k
                stdv = 1.0 / math.sqrt(n)
                if self.shared_filters:
                    stdv *= self.in_channels
                self.weight.data.uniform_(-stdv, stdv)
                if self.bias is not None:
                    self.bias.data.uniform_(-stdv, stdv)
            elif self.filler == "linear":
                # bilinear-interp initialization for transposed upsampling
                effective = tuple(2 * s - 1 for s in self.stride)
                pad = tuple(int((k - ek) // 2)
                            for k, ek in zip(self.kernel_size, effective))
                assert self.transposed and self.in_channels == self.out_channels
                w = 1.0
                for i, (p, s, k) in enumerate(zip(pad, self.stride, self.kernel_size)):
                    d = len(pad) - i - 1
                    row = (np.array((0.0,) * p + tuple(range(1, s))
                                    + tuple(range(s, 0, -1)) + (0,) * p) / s)
                    w = w * row.reshape((-1,) + (1,) * d)
                    if self.normalize_kernel:
                        w = w * np.array(
                            tuple(((k - j - 1) // s) + (j // s) + 1.0
                                  for j in range(k))).reshape((-1,) + (1,) * d)
                self.weight.data.fill_(0.0)
                for c in range(1 if self.shared_filters else self.in_channels):
                    self.weight.data[c, c, :] = torch.tensor(w, dtype=self.weight.dtype)
                if self.bias is not None:
                    self.bias.data.fill_(0.0)
            else:
                raise ValueError(f"filler {self.filler!r} not supported")
        if hasattr(self, "inv_alpha") and isinstance(self.inv_alpha, nn.Parameter):
            self.inv_alpha.data.fill_(self.inv_alpha_init)
            self.inv_lambda.data.fill_(self.inv_lambda_init)

    def _kernel_kwargs(self):
        return dict(
            kernel_type=self.kernel_type,
            smooth_kernel_type=self.smooth_kernel_type,
            smooth_kernel=getattr(self, "smooth_kernel", None),
            inv_alpha=getattr(self, "inv_alpha", None),
            inv_lambda=getattr(self, "inv_lambda", None),
            normalize_kernel=self.normalize_kernel,
        )


class PacConv2d(_PacConvNd):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, dilation=1, bias=True, kernel_type="gaussian",
                 smooth_kernel_type="none", normalize_kernel=False,
                 shared_filters=False, filler="uniform", native_impl=False):
        super().__init__(in_channels, out_channels, _pair(kernel_size),
                         _pair(stride), _pair(padding), _pair(dilation),
                         False, _pair(0), bias, False, kernel_type,
                         smooth_kernel_type, False, normalize_kernel,
                         shared_filters, filler)

    def compute_kernel(self, input_for_kernel, input_mask=None):
        return packernel2d(input_for_kernel, input_mask,
                           kernel_size=self.kernel_size, stride=self.stride,
                           padding=self.padding, dilation=self.dilation,
                           channel_wise=False, transposed=False,
                           **self._kernel_kwargs())

    def forward(self, input_2d, input_for_kernel, kernel=None, mask=None):
        output_mask = None
        if kernel is None:
            kernel, output_mask = self.compute_kernel(input_for_kernel, mask)
        out = pacconv2d(input_2d, kernel, self.weight, self.bias, self.stride,
                        self.padding, self.dilation, self.shared_filters)
        return out if output_mask is None else (out, output_mask)


class PacConvTranspose2d(_PacConvNd):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, output_padding=0, dilation=1, bias=True,
                 kernel_type="gaussian", smooth_kernel_type="none",
                 normalize_kernel=False, shared_filters=False,
                 filler="uniform", native_impl=False):
        super().__init__(in_channels, out_channels, _pair(kernel_size),
                         _pair(stride), _pair(padding), _pair(dilation),
                         True, _pair(output_padding), bias, False,
                         kernel_type, smooth_kernel_type, False,
                         normalize_kernel, shared_filters, filler)

    def compute_kernel(self, input_for_kernel, input_mask=None):
        return packernel2d(input_for_kernel, input_mask,
                           kernel_size=self.kernel_size, stride=self.stride,
                           padding=self.padding,
                           output_padding=self.output_padding,
                           dilation=self.dilation, channel_wise=False,
                           transposed=True, **self._kernel_kwargs())

    def forward(self, input_2d, input_for_kernel, kernel=None, mask=None):
        output_mask = None
        if kernel is None:
            kernel, output_mask = self.compute_kernel(input_for_kernel, mask)
        out = pacconv_transpose2d(input_2d, kernel, self.weight, self.bias,
                                  self.stride, self.padding,
                                  self.output_padding, self.dilation,
                                  self.shared_filters)
        return out if output_mask is None else (out, output_mask)


class PacPool2d(_PacConvNd):
    def __init__(self, kernel_size, stride=1, padding=0, dilation=1,
                 kernel_type="gaussian", smooth_kernel_type="none",
                 channel_wise=False, normalize_kernel=False, out_channels=-1,
                 native_impl=False):
        super().__init__(-1, out_channels, _pair(kernel_size), _pair(stride),
                         _pair(padding), _pair(dilation), False, _pair(0),
                         False, True, kernel_type, smooth_kernel_type,
                         channel_wise, normalize_kernel, False, None)

    def compute_kernel(self, input_for_kernel, input_mask=None):
        return packernel2d(input_for_kernel, input_mask,
                           kernel_size=self.kernel_size, stride=self.stride,
                           padding=self.padding, dilation=self.dilation,
                           channel_wise=self.channel_wise, transposed=False,
                           **self._kernel_kwargs())

    def forward(self, input_2d, input_for_kernel, kernel=None, mask=None):
        output_mask = None
        if kernel is None:
            kernel, output_mask = self.compute_kernel(input_for_kernel, mask)
        bs, in_ch = input_2d.shape[:2]
        if self.channel_wise and kernel.shape[1] != in_ch:
            raise ValueError("channel_wise kernel/input channel mismatch")
        assert self.out_channels <= 0 or self.out_channels == in_ch
        out = pacpool2d(input_2d, kernel, self.kernel_size, self.stride,
                        self.padding, self.dilation)
        return out if output_mask is None else (out, output_mask)


def np_gaussian_2d(width, sigma=-1):
    """Truncated, normalized 2D Gaussian filter (reference
    pac_modules.py:38-49 public utility; sigma defaults to width/4)."""
    assert width % 2 == 1
    if sigma <= 0:
        sigma = float(width) / 4
    r = np.arange(-(width // 2), width // 2 + 1, dtype=np.float32)
    g = np.exp(-0.5 * r * r / (sigma * sigma))
    g2 = g.reshape(-1, 1) * g
    return g2 / g2.sum()
