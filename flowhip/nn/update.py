"""Iterative update block: motion encoder + ConvGRU + flow head (+ mask head).

State-dict compatible with the reference `core/update.py` (parameter names
convz1/convr1/convq1/convz2/..., encoder.convc1..., flow_head.conv1/conv2,
mask.0/mask.2). Fresh implementation notes:

- SepConvGRU packs the z and r gate convolutions of each direction into ONE
  conv call by concatenating their weights at forward time (the parameters
  stay separate for checkpoint compatibility); q runs after r*h. This halves
  the conv launches per GRU pass at identical math.
- The mask head's 0.25 gradient-balance scale (update.py:140) is kept.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.functional_conv import (fused_conv2d, fused_conv2d_cat2,
                                   fused_gru_zr_conv)


class FusedConv2d(nn.Conv2d):
    """nn.Conv2d whose GPU bf16/NHWC stride-1 path runs the hand-written
    implicit-GEMM MFMA kernel (csrc/conv_gemm.hip, kernel #5). Parameter
    names/shapes are untouched — state-dict identical to nn.Conv2d."""

    def __init__(self, *a, **k):
        super().__init__(*a, **k)
        self._cg_cache = {}

    def forward(self, x):
        return fused_conv2d(x, self.weight, self.bias, self.stride,
                            self.padding, self.dilation, self.groups,
                            self._cg_cache)


class FlowHead(nn.Module):
    def __init__(self, input_dim=128, hidden_dim=256):
        super().__init__()
        self.conv1 = FusedConv2d(input_dim, hidden_dim, 3, padding=1)
        self.conv2 = FusedConv2d(hidden_dim, 2, 3, padding=1)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.conv2(self.relu(self.conv1(x)))


class ConvGRU(nn.Module):
    """Single 3x3 ConvGRU (small model — update.py:16-31)."""

    def __init__(self, hidden_dim=128, input_dim=192 + 128):
        super().__init__()
        self.convz = FusedConv2d(hidden_dim + input_dim, hidden_dim, 3, padding=1)
        self.convr = FusedConv2d(hidden_dim + input_dim, hidden_dim, 3, padding=1)
        self.convq = FusedConv2d(hidden_dim + input_dim, hidden_dim, 3, padding=1)

    def forward(self, h, x):
        hx = torch.cat([h, x], dim=1)
        if not hasattr(self, "_zr_cache"):
            self._zr_cache = {}
        zr = fused_conv2d(
            hx, torch.cat([self.convz.weight, self.convr.weight]),
            torch.cat([self.convz.bias, self.convr.bias]), 1, 1, 1, 1,
            self._zr_cache,
            key=(self.convz.weight.data_ptr(), self.convz.weight._version,
                 self.convr.weight.data_ptr(), self.convr.weight._version))
        if h.is_cuda:
            from ..ops.functional_gru import GruGate1Fn, GruGate2Fn
            z, rh = GruGate1Fn.apply(zr, h)
            qp = self.convq(torch.cat([rh, x], dim=1))
            return GruGate2Fn.apply(qp, z, h)
        z, r = torch.sigmoid(zr).chunk(2, dim=1)
        q = torch.tanh(self.convq(torch.cat([r * h, x], dim=1)))
        return (1 - z) * h + z * q


class SepConvGRU(nn.Module):
    """Separable 1x5 / 5x1 two-pass ConvGRU (update.py:33-60)."""

    def __init__(self, hidden_dim=128, input_dim=192 + 128):
        super().__init__()
        self.convz1 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (1, 5), padding=(0, 2))
        self.convr1 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (1, 5), padding=(0, 2))
        self.convq1 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (1, 5), padding=(0, 2))

        self.convz2 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (5, 1), padding=(2, 0))
        self.convr2 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (5, 1), padding=(2, 0))
        self.convq2 = FusedConv2d(hidden_dim + input_dim, hidden_dim, (5, 1), padding=(2, 0))

    def _pass(self, h, x, convz, convr, convq, padding, zr_cache, q_cache):
        # packed z+r conv over the virtually-concatenated input: weights,
        # bias and their packings cached per parameter version (no per-call
        # torch.cat of weights)
        zr = fused_gru_zr_conv(h, x, convz, convr, padding, zr_cache)
        if h.is_cuda:
            # fused gate kernels (ops/functional_gru): one kernel for
            # sigmoid/chunk/r*h, one for tanh + lerp, fused backwards
            from ..ops.functional_gru import GruGate1Fn, GruGate2Fn
            z, rh = GruGate1Fn.apply(zr, h)
            qp = fused_conv2d_cat2(rh, x, convq.weight, convq.bias, padding,
                                   q_cache, key=(convq.weight.data_ptr(),
                                                 convq.weight._version))
            return GruGate2Fn.apply(qp, z, h)
        z, r = torch.sigmoid(zr).chunk(2, dim=1)
        q = torch.tanh(convq(torch.cat([r * h, x], dim=1)))
        return (1 - z) * h + z * q

    def forward(self, h, x):
        if not hasattr(self, "_zr1_cache"):
            self._zr1_cache, self._zr2_cache = {}, {}
            self._q1_cache, self._q2_cache = {}, {}
        h = self._pass(h, x, self.convz1, self.convr1, self.convq1, (0, 2),
                       self._zr1_cache, self._q1_cache)  # horizontal
        h = self._pass(h, x, self.convz2, self.convr2, self.convq2, (2, 0),
                       self._zr2_cache, self._q2_cache)  # vertical
        return h


class SmallMotionEncoder(nn.Module):
    """Corr+flow -> 82ch motion features (update.py:62-77)."""

    def __init__(self, args):
        super().__init__()
        cor_planes = args.corr_levels * (2 * args.corr_radius + 1) ** 2
        self.convc1 = FusedConv2d(cor_planes, 96, 1, padding=0)
        self.convf1 = FusedConv2d(2, 64, 7, padding=3)
        self.convf2 = FusedConv2d(64, 32, 3, padding=1)
        self.conv = FusedConv2d(128, 80, 3, padding=1)

    def forward(self, flow, corr):
        if corr.is_cuda and corr.stride(1) == 1:
            flow = flow.contiguous(memory_format=torch.channels_last)
        cor = F.relu(self.convc1(corr))
        flo = F.relu(self.convf2(F.relu(self.convf1(flow))))
        out = F.relu(self.conv(torch.cat([cor, flo], dim=1)))
        return torch.cat([out, flow.to(out.dtype)], dim=1)


class BasicMotionEncoder(nn.Module):
    """Corr+flow -> 128ch motion features (update.py:79-97)."""

    def __init__(self, args):
        super().__init__()
        cor_planes = args.corr_levels * (2 * args.corr_radius + 1) ** 2
        self.convc1 = FusedConv2d(cor_planes, 256, 1, padding=0)
        self.convc2 = FusedConv2d(256, 192, 3, padding=1)
        self.convf1 = FusedConv2d(2, 128, 7, padding=3)
        self.convf2 = FusedConv2d(128, 64, 3, padding=1)
        self.conv = FusedConv2d(64 + 192, 128 - 2, 3, padding=1)

    def forward(self, flow, corr):
        if corr.is_cuda and corr.stride(1) == 1:
            # keep the whole motion-feature chain in one layout: a mixed
            # cat([cl, nchw]) falls back to NCHW and costs an uncoalesced
            # layout copy on every GRU conv input (profiles/, tprof6).
            # stride(1)==1, not is_contiguous(channels_last): the corr
            # lookup output is a channel-NARROWED channels-last view
            flow = flow.contiguous(memory_format=torch.channels_last)
        cor = F.relu(self.convc2(F.relu(self.convc1(corr))))
        flo = F.relu(self.convf2(F.relu(self.convf1(flow))))
        if cor.is_cuda and cor.dtype == torch.bfloat16:
            # virtually-concatenated conv input (192 | 64): no cat
            # materialization, no backward narrows
            if not hasattr(self, "_cv_cache"):
                self._cv_cache = {}
            out = F.relu(fused_conv2d_cat2(
                cor, flo, self.conv.weight, self.conv.bias, (1, 1),
                self._cv_cache,
                key=(self.conv.weight.data_ptr(),
                     self.conv.weight._version)))
        else:
            out = F.relu(self.conv(torch.cat([cor, flo], dim=1)))
        # keep the motion features in ONE dtype: cat([bf16, fp32]) promotes
        # the whole 128-ch feature map to fp32 and silently pushes all four
        # 384-ch GRU convs off the MFMA path (autocast would cast flow at
        # the conv anyway — same math, discovered via attr_profile r02)
        return torch.cat([out, flow.to(out.dtype)], dim=1)


class SmallUpdateBlock(nn.Module):
    def __init__(self, args, hidden_dim=96):
        super().__init__()
        self.encoder = SmallMotionEncoder(args)
        self.gru = ConvGRU(hidden_dim=hidden_dim, input_dim=82 + 64)
        self.flow_head = FlowHead(hidden_dim, hidden_dim=128)

    def forward(self, net, inp, corr, flow):
        motion_features = self.encoder(flow, corr)
        inp = torch.cat([inp, motion_features], dim=1)
        net = self.gru(net, inp)
        delta_flow = self.flow_head(net)
        return net, None, delta_flow


class BasicUpdateBlock(nn.Module):
    def __init__(self, args, hidden_dim=128, input_dim=128):
        super().__init__()
        self.args = args
        self.encoder = BasicMotionEncoder(args)
        self.gru = SepConvGRU(hidden_dim=hidden_dim, input_dim=128 + hidden_dim)
        self.flow_head = FlowHead(hidden_dim, hidden_dim=256)

        self.mask = nn.Sequential(
            nn.Conv2d(128, 256, 3, padding=1),
            nn.ReLU(inplace=True),
            nn.Conv2d(256, 64 * 9, 1, padding=0))

    def forward(self, net, inp, corr, flow, upsample=True):
        motion_features = self.encoder(flow, corr)
        inp = torch.cat([inp, motion_features], dim=1)

        net = self.gru(net, inp)
        delta_flow = self.flow_head(net)

        if len(self.mask) == 0:  # mask head removed (raft_nc_dbl)
            mask = None
        else:
            mask = .25 * self.mask(net)  # 0.25 balances gradient scale
        return net, mask, delta_flow
