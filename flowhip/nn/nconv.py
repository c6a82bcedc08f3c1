"""Normalized-convolution modules (NCUP interpolation network).

State-dict compatible with the reference `core/nconv_modules.py`: parameter
names are `weight_p` (the softplus-reparameterized non-negative weight, ref
EnforcePos :218-265) and optional `bias`; module attributes nconv_in /
nconv_x2 / encoder / decoder / nconv_out with the shared-encoder aliasing.

Fresh-implementation differences (semantics preserved):
- EnforcePos's forward-pre-hook machinery is replaced by computing
  weight = softplus(weight_p, beta=10) functionally inside forward — same
  math, autograd-clean, and hipGraph/compile friendly.
- The compute goes through `flowhip.ops.nconv2d` (fused HIP kernel on GPU).
- NConvUNet skips its dead deepest-encoder branch: the reference's decoder
  index arithmetic (nconv_modules.py:128-134) overwrites x[nds+1] before ever
  reading it, so encoder stage `nds` (and its conf-pool input) contributes
  nothing to outputs or gradients. We compute the same dataflow without the
  dead nodes; outputs and gradients are bit-identical (with shared_encoder
  the stage's parameters stay trained through the full-res path).
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.nn.modules.utils import _pair

from .. import ops


def pos_transform(p, pos_fn):
    """Non-negativity map applied to the raw parameter (ref _pos :254-269)."""
    pos_fn = pos_fn.lower()
    if pos_fn == "softplus":
        return F.softplus(p, beta=10)
    if pos_fn == "exp":
        return torch.exp(p)
    if pos_fn == "sigmoid":
        return torch.sigmoid(p)
    if pos_fn == "softmax":
        p_sz = p.size()
        return F.softmax(p.view(p_sz[0], p_sz[1], -1), -1).view(p_sz)
    raise ValueError(f"Undefined positive function {pos_fn!r}")


class NConv2d(nn.Module):
    """Confidence-normalized 2D convolution with confidence propagation.

    forward((data, conf)) ->
        nconv = conv(data*conf, w) / (conv(conf, w) + 1e-20) [+ bias]
        cout  = conv(conf, w) / sum(w, per out channel)
    with w = softplus(weight_p, beta=10) >= 0.
    Reference: nconv_modules.py:140-216.
    """

    def __init__(self, in_channels, out_channels, kernel_size, stride=(1, 1),
                 padding=None, dilation=(1, 1), groups=1, bias=False,
                 pos_fn="softplus", prop_conf=True, init_method="n"):
        super().__init__()
        kernel_size = _pair(kernel_size)
        if padding is None:
            padding = _pair(kernel_size[0] // 2)
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = _pair(stride)
        self.padding = _pair(padding)
        self.dilation = _pair(dilation)
        self.groups = groups
        self.eps = 1e-20
        self.pos_fn = pos_fn
        self.init_method = init_method
        self.prop_conf = prop_conf

        w0 = torch.empty(out_channels, in_channels // groups, *kernel_size)
        self._init_weight(w0)
        if pos_fn is not None:
            # The reference initializes weight_p to pos(w0) (EnforcePos.apply
            # registers Parameter(_pos(weight).data) — nconv_modules.py:236).
            self.weight_p = nn.Parameter(pos_transform(w0, pos_fn).detach())
        else:
            self.weight_p = nn.Parameter(w0)

        if bias:
            b = torch.empty(out_channels)
            fan_in = in_channels // groups * kernel_size[0] * kernel_size[1]
            bound = 1 / math.sqrt(fan_in)
            nn.init.uniform_(b, -bound, bound)
            self.bias = nn.Parameter(b)
        else:
            self.register_parameter("bias", None)

    def _init_weight(self, w):
        if self.init_method == "x":
            nn.init.xavier_uniform_(w)
        elif self.init_method == "k":
            nn.init.kaiming_uniform_(w)
        elif self.init_method == "n":
            n = self.kernel_size[0] * self.kernel_size[1] * self.out_channels
            w.data.normal_(2, math.sqrt(2.0 / n))
        else:
            raise ValueError(f"unknown init_method {self.init_method!r}")

    @property
    def weight(self):
        if self.pos_fn is None:
            return self.weight_p
        return pos_transform(self.weight_p, self.pos_fn)

    def forward(self, inpt):
        data, conf = inpt[0], inpt[1]
        nconv, cout = ops.nconv2d(
            data, conf, self.weight, self.bias, self.stride, self.padding,
            self.dilation, self.groups, self.eps, self.prop_conf)
        return nconv, cout

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size={self.kernel_size}, pos_fn={self.pos_fn}")


class NConvUNet(nn.Module):
    """Tiny encoder/decoder U-Net over (data, conf) pairs (nconv_modules.py:25-136).

    Args mirror the reference constructor exactly (the reflective CLI system
    exposes them as --interp_net_* flags).
    """

    def __init__(self, in_ch=1, channels_multiplier=2, num_downsampling=3,
                 encoder_filter_sz=5, decoder_filter_sz=3, out_filter_sz=1,
                 pos_fn="SoftPlus", groups=1, use_bias=False,
                 data_pooling="conf_based", shared_encoder=True,
                 use_double_conv=True):
        super().__init__()
        self.__name__ = "NConvUNet"

        encoder_filter_sz = _pair(encoder_filter_sz)
        decoder_filter_sz = _pair(decoder_filter_sz)
        out_filter_sz = _pair(out_filter_sz)

        self.num_downsampling = num_downsampling
        self.data_pooling = data_pooling
        self.shared_encoder = shared_encoder
        self.use_double_conf = use_double_conv

        mid = in_ch * channels_multiplier
        self.nconv_in = NConv2d(in_ch, mid, encoder_filter_sz, stride=(1, 1),
                                pos_fn=pos_fn, groups=groups, bias=use_bias)

        mids = [NConv2d(mid, mid, encoder_filter_sz, stride=(1, 1), pos_fn=pos_fn,
                        groups=groups, bias=use_bias)
                for _ in range(2 if use_double_conv else 1)]
        self.nconv_x2 = nn.Sequential(*mids)

        self.encoder = nn.ModuleList([nn.Sequential(self.nconv_in, self.nconv_x2)])
        for _ in range(num_downsampling):
            if shared_encoder:
                # sparsity decreases after downsampling: reuse the first mid conv
                self.encoder.append(self.nconv_x2[0])
            else:
                self.encoder.append(NConv2d(mid, mid, encoder_filter_sz,
                                            stride=(1, 1), pos_fn=pos_fn,
                                            groups=groups, bias=use_bias))

        self.decoder = nn.ModuleList([
            NConv2d(2 * mid, mid, decoder_filter_sz, stride=(1, 1), pos_fn=pos_fn,
                    groups=groups, bias=use_bias)
            for _ in range(num_downsampling)])

        self.nconv_out = NConv2d(mid, in_ch, out_filter_sz, stride=(1, 1),
                                 pos_fn=pos_fn, groups=groups, bias=False)

    @staticmethod
    def downsample_data_conf(data, conf, ds_factor=2,
                             pooling_type="conf_based"):
        """Downsample a (data, conf) pair keeping data at the
        max-confidence positions (reference nconv_modules.py:94-104 public
        helper; routes through ops.conf_pool and the HIP kernel on GPU)."""
        return ops.conf_pool(data, conf, ds_factor, pooling_type)

    def forward(self, inpt):
        nds = self.num_downsampling
        x = [None] * (nds * 2 + 1)
        c = [None] * (nds * 2 + 1)
        x[0], c[0] = inpt[0], inpt[1]

        if nds == 0:
            x0, c0 = self.encoder[0]((x[0], c[0]))
            return self.nconv_out((x0, c0))

        # Encoder. Stage `nds` (the deepest) is dead in the reference's
        # decoder indexing (see module docstring) — not computed.
        for i in range(nds):
            if i == 0:
                x[i + 1], c[i + 1] = self.encoder[i]((x[i], c[i]))
            else:
                d_ds, c_ds = self.downsample_data_conf(
                    x[i], c[i], 2, self.data_pooling)
                x[i + 1], c[i + 1] = self.encoder[i]((d_ds, c_ds))

        # Decoder (reference index arithmetic: stage 0 pairs x[nds] with
        # itself at the same scale; later stages upsample and concat skips).
        for i in range(nds):
            # fused nearest-2x upsample + concat (ops.up2x_cat; falls back
            # to interpolate+cat off-GPU or at non-2x scales)
            xin = ops.up2x_cat(x[i + nds], x[nds - i])
            cin = ops.up2x_cat(c[i + nds], c[nds - i])
            x[i + nds + 1], c[i + nds + 1] = self.decoder[i]((xin, cin))

        return self.nconv_out((x[-1], c[-1]))


def retrieve_elements_from_indices(tensor, indices):
    """Gather per-pixel elements at pooling argmax indices (reference
    nconv_modules.py:19-22 public helper; the conf-based pooling itself
    runs through ops.conf_pool / the HIP kernel)."""
    flat = tensor.flatten(start_dim=2)
    return flat.gather(dim=2, index=indices.flatten(start_dim=2)) \
        .view_as(indices)


class EnforcePos:
    """Hook-based non-negativity reparameterization for ARBITRARY modules
    (reference nconv_modules.py:218-251 public API). ``apply(module,
    'weight', pos_fn)`` replaces ``weight`` with a ``weight_p`` parameter
    and recomputes ``weight = pos(weight_p)`` in a forward pre-hook.

    NConv2d applies the same reparameterization functionally inside its
    forward (identical ``weight_p`` state-dict surface); this class exists
    for user code that used the reference hook on other modules. The
    reference's observable initialization is preserved: ``weight_p`` starts
    as ``pos(weight)`` — NOT the inverse — so the first effective weight is
    ``pos(pos(weight))`` (a reference quirk checkpoints depend on).
    """

    def __init__(self, name, pos_fn):
        self.name = name
        self.pos_fn = pos_fn

    def compute_weight(self, module):
        return pos_transform(getattr(module, self.name + "_p"), self.pos_fn)

    @staticmethod
    def apply(module, name, pos_fn):
        fn = EnforcePos(name, pos_fn)
        weight = getattr(module, name)
        del module._parameters[name]
        module.register_parameter(
            name + "_p", nn.Parameter(pos_transform(weight, pos_fn).detach()))
        setattr(module, name, fn.compute_weight(module))
        module.register_forward_pre_hook(fn)
        return fn

    def remove(self, module):
        weight = self.compute_weight(module)
        delattr(module, self.name)
        del module._parameters[self.name + "_p"]
        module.register_parameter(self.name, nn.Parameter(weight.detach()))

    def __call__(self, module, inputs):
        setattr(module, self.name, self.compute_weight(module))


def remove_weight_pos(module, name="weight"):
    """Remove an EnforcePos reparameterization, freezing the effective
    weight back into a plain parameter (reference nconv_modules.py:272)."""
    for k, hook in list(module._forward_pre_hooks.items()):
        if isinstance(hook, EnforcePos) and hook.name == name:
            hook.remove(module)
            del module._forward_pre_hooks[k]
            return module
    raise ValueError(f"no EnforcePos reparameterization of {name!r} found")
