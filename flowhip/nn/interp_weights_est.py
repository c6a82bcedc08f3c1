"""Confidence (interpolation-weights) estimation networks.

Maps guidance (+ optionally the low-res data) to a per-pixel confidence in
(0,1). State-dict compatible with the reference `core/interp_weights_est.py`:
`Simple` registers conv (ModuleList of Sequential(Conv2d[,BN],ReLU)) and out;
`UNet` registers encoder/decoder/out built from inconv/down/up/outconv.
Constructor signatures are mirrored because the reflective CLI system derives
the --weights_est_net_* flags from them.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.nn.modules.utils import _pair

from .update import FusedConv2d


class Simple(nn.Module):
    """Plain conv stack: num_ch[0] -> ... -> num_ch[-1] -> out_ch, sigmoid.

    Reference interp_weights_est.py:10-47. `num_ch[0]` is the input channel
    count (inserted by the upsampler factory); filter_sz/dilation have one
    extra entry for the output conv.
    """

    def __init__(self, num_ch, out_ch, filter_sz, dilation=None,
                 final_act=nn.Sigmoid(), use_bn=False):
        super().__init__()
        self.__name__ = "Simple"

        assert len(filter_sz) == len(num_ch)
        if dilation is None:
            dilation = [(1, 1)] * len(num_ch)

        self.in_ch = num_ch[0]
        self.num_layers = len(num_ch) - 1

        def pad_for(k, d):
            d = d if isinstance(d, int) else d[0]
            return _pair(int(k // 2 + ((k - 1) * (d - 1)) / 2))

        self.conv = nn.ModuleList()
        for i in range(self.num_layers):
            layers = [FusedConv2d(num_ch[i], num_ch[i + 1], filter_sz[i],
                                  padding=pad_for(filter_sz[i], dilation[i]),
                                  dilation=dilation[i], stride=1)]
            if use_bn:
                from .norm import BatchNorm2d
                layers.append(BatchNorm2d(num_ch[i + 1]))
            layers.append(nn.ReLU(inplace=True))
            self.conv.append(nn.Sequential(*layers))

        self.out = FusedConv2d(num_ch[-1], out_ch, filter_sz[-1],
                               padding=pad_for(filter_sz[-1], dilation[-1]),
                               dilation=dilation[-1], stride=1)

        self.final_act = nn.Sequential() if final_act is None else final_act

    def forward(self, x):
        for i in range(self.num_layers):
            x = self.conv[i](x)
        return self.final_act(self.out(x))


class double_conv(nn.Module):
    """(conv => BN => ReLU) * 2 (reference interp_weights_est.py:85-100)."""

    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(in_ch, out_ch, 3, padding=1),
            nn.BatchNorm2d(out_ch),
            nn.ReLU(inplace=True),
            nn.Conv2d(out_ch, out_ch, 3, padding=1),
            nn.BatchNorm2d(out_ch),
            nn.ReLU(inplace=True),
        )

    def forward(self, x):
        return self.conv(x)


class inconv(nn.Module):
    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.conv = double_conv(in_ch, out_ch)

    def forward(self, x):
        return self.conv(x)


class down(nn.Module):
    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.mpconv = nn.Sequential(nn.MaxPool2d(2), double_conv(in_ch, out_ch))

    def forward(self, x):
        return self.mpconv(x)


class up(nn.Module):
    def __init__(self, in_ch1, in_ch2, out_ch, bilinear=False):
        super().__init__()
        if bilinear:
            self.up = nn.Upsample(scale_factor=2, mode="bilinear", align_corners=True)
        else:
            self.up = nn.ConvTranspose2d(in_ch1, in_ch1, 2, stride=2)
        self.conv = double_conv(in_ch1 + in_ch2, out_ch)

    def forward(self, x1, x2):
        x1 = self.up(x1)
        diffY = x2.size()[2] - x1.size()[2]
        diffX = x2.size()[3] - x1.size()[3]
        x1 = F.pad(x1, (diffX // 2, diffX - diffX // 2,
                        diffY // 2, diffY - diffY // 2))
        return self.conv(torch.cat([x2, x1], dim=1))


class outconv(nn.Module):
    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, 1)

    def forward(self, x):
        return self.conv(x)


class UNet(nn.Module):
    """Conv U-Net weights estimator (reference interp_weights_est.py:50-83)."""

    def __init__(self, num_ch, out_ch, final_act=torch.sigmoid):
        super().__init__()
        self.__name__ = "UNet"

        self.in_ch = num_ch[0]
        self.final_act = final_act
        self.num_downsampling = len(num_ch) - 2

        self.encoder = nn.ModuleList([inconv(num_ch[0], num_ch[1])])
        for i in range(1, self.num_downsampling + 1):
            self.encoder.append(down(in_ch=num_ch[i], out_ch=num_ch[i + 1]))

        self.decoder = nn.ModuleList([
            up(in_ch1=num_ch[-i - 1], in_ch2=num_ch[-i - 2], out_ch=num_ch[-i - 2],
               bilinear=False)
            for i in range(self.num_downsampling)])

        self.out = outconv(num_ch[1], out_ch)

    def forward(self, x0):
        x_encoder = [x0]
        for i in range(self.num_downsampling + 1):
            x_encoder.append(self.encoder[i](x_encoder[i]))

        x_decoder = [x_encoder[-1]]
        for i in range(self.num_downsampling):
            x_decoder.append(self.decoder[i](x_decoder[-1], x_encoder[-i - 2]))

        return self.final_act(self.out(x_decoder[-1]))
