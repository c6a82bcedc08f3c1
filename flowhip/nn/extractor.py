"""Feature / context encoders (stride-8 ResNet-style).

State-dict compatible with the reference `core/extractor.py` (attribute names
conv1/norm1/relu1/layer{1,2,3}/conv2 and per-block conv1/conv2/conv3/norm*/
downsample — see SURVEY.md §2.7). The implementation is fresh: a norm factory
replaces the per-branch boilerplate, and the two input frames are processed
as one 2B batch (reference extractor.py:168-191 behavior).

Convolutions execute through torch (MIOpen on ROCm) under bf16 autocast on
GPU; per-shape fused HIP conv kernels are an ops-level optimization that
slots in underneath without touching this module.
"""

import os

import torch
import torch.nn as nn

from .update import FusedConv2d


def _enc_conv(*a, **k):
    """Encoder convs run on the hand-written MFMA kernel by DEFAULT since
    round 2: the BM=128 8-wave tile + tr_b16 wrw beat MIOpen in the
    same-box A/B (34.69 vs 34.38 pairs/s at flip time, wider after the
    later eager-tax work). FLOWHIP_FUSED_ENCODER=0 restores the library
    path (A/B runs)."""
    if os.environ.get("FLOWHIP_FUSED_ENCODER", "1") == "1":
        return FusedConv2d(*a, **k)
    return nn.Conv2d(*a, **k)


def _norm(norm_fn, planes, groups_planes=None):
    """Build one normalization module. `group` uses planes//8 groups unless an
    explicit count is given (reference uses 8 for the stem, planes//8 in
    blocks — extractor.py:14-38, 124-131)."""
    if norm_fn == "group":
        ng = groups_planes if groups_planes is not None else planes // 8
        return nn.GroupNorm(num_groups=ng, num_channels=planes)
    if norm_fn == "batch":
        from .norm import BatchNorm2d
        return BatchNorm2d(planes)
    if norm_fn == "instance":
        from .norm import InstanceNorm2d
        return InstanceNorm2d(planes)
    if norm_fn == "none":
        return nn.Sequential()
    raise ValueError(f"unknown norm_fn {norm_fn!r}")


def _kaiming_init(module):
    for m in module.modules():
        if isinstance(m, nn.Conv2d):
            nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
        elif isinstance(m, (nn.BatchNorm2d, nn.InstanceNorm2d, nn.GroupNorm)):
            if m.weight is not None:
                nn.init.constant_(m.weight, 1)
            if m.bias is not None:
                nn.init.constant_(m.bias, 0)


class ResidualBlock(nn.Module):
    """3x3-3x3 residual block (reference extractor.py:6-57)."""

    def __init__(self, in_planes, planes, norm_fn="group", stride=1):
        super().__init__()
        self.conv1 = _enc_conv(in_planes, planes, kernel_size=3, padding=1, stride=stride)
        self.conv2 = _enc_conv(planes, planes, kernel_size=3, padding=1)
        self.relu = nn.ReLU(inplace=True)

        self.norm1 = _norm(norm_fn, planes)
        self.norm2 = _norm(norm_fn, planes)
        if stride == 1:
            self.downsample = None
        else:
            self.norm3 = _norm(norm_fn, planes)
            self.downsample = nn.Sequential(
                _enc_conv(in_planes, planes, kernel_size=1, stride=stride), self.norm3)

    def forward(self, x):
        y = self.relu(self.norm1(self.conv1(x)))
        y = self.relu(self.norm2(self.conv2(y)))
        if self.downsample is not None:
            x = self.downsample(x)
        return self.relu(x + y)


class BottleneckBlock(nn.Module):
    """1x1-3x3-1x1 bottleneck block (reference extractor.py:60-116).

    Note the reference quirk kept for weight compatibility: group-norm uses
    planes//8 groups for ALL three norms, including the planes//4-channel
    ones (extractor.py:70-74)."""

    def __init__(self, in_planes, planes, norm_fn="group", stride=1):
        super().__init__()
        self.conv1 = _enc_conv(in_planes, planes // 4, kernel_size=1, padding=0)
        self.conv2 = _enc_conv(planes // 4, planes // 4, kernel_size=3, padding=1, stride=stride)
        self.conv3 = _enc_conv(planes // 4, planes, kernel_size=1, padding=0)
        self.relu = nn.ReLU(inplace=True)

        ng = planes // 8
        self.norm1 = _norm(norm_fn, planes // 4, groups_planes=ng if norm_fn == "group" else None)
        self.norm2 = _norm(norm_fn, planes // 4, groups_planes=ng if norm_fn == "group" else None)
        self.norm3 = _norm(norm_fn, planes)
        if stride == 1:
            self.downsample = None
        else:
            self.norm4 = _norm(norm_fn, planes)
            self.downsample = nn.Sequential(
                _enc_conv(in_planes, planes, kernel_size=1, stride=stride), self.norm4)

    def forward(self, x):
        y = self.relu(self.norm1(self.conv1(x)))
        y = self.relu(self.norm2(self.conv2(y)))
        y = self.relu(self.norm3(self.conv3(y)))
        if self.downsample is not None:
            x = self.downsample(x)
        return self.relu(x + y)


class _Encoder(nn.Module):
    """Shared stem/stage/output scaffolding for both encoder sizes."""

    block_cls = None
    stem_planes = None
    stage_planes = ()

    def __init__(self, output_dim=128, norm_fn="batch", dropout=0.0):
        super().__init__()
        self.norm_fn = norm_fn

        self.norm1 = _norm(norm_fn, self.stem_planes,
                           groups_planes=8 if norm_fn == "group" else None)
        # the 7x7 s2 stem runs on the MFMA kernel too (3->8 channel pad,
        # zero-page K tail): the last library conv out of the hot path
        self.conv1 = _enc_conv(3, self.stem_planes, kernel_size=7, stride=2,
                               padding=3)
        self.relu1 = nn.ReLU(inplace=True)

        self.in_planes = self.stem_planes
        self.layer1 = self._make_layer(self.stage_planes[0], stride=1)
        self.layer2 = self._make_layer(self.stage_planes[1], stride=2)
        self.layer3 = self._make_layer(self.stage_planes[2], stride=2)

        self.conv2 = _enc_conv(self.stage_planes[2], output_dim, kernel_size=1)

        self.dropout = nn.Dropout2d(p=dropout) if dropout > 0 else None
        _kaiming_init(self)

    def _make_layer(self, dim, stride=1):
        blocks = (self.block_cls(self.in_planes, dim, self.norm_fn, stride=stride),
                  self.block_cls(dim, dim, self.norm_fn, stride=1))
        self.in_planes = dim
        return nn.Sequential(*blocks)

    def forward(self, x):
        # Both frames in one pass: callers may hand a [img1, img2] list which
        # is run as a 2B batch and split back (reference extractor.py:168-191).
        is_list = isinstance(x, (tuple, list))
        if is_list:
            batch_dim = x[0].shape[0]
            x = torch.cat(x, dim=0)

        x = self.relu1(self.norm1(self.conv1(x)))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.conv2(x)

        if self.training and self.dropout is not None:
            x = self.dropout(x)

        if is_list:
            x = torch.split(x, [batch_dim, batch_dim], dim=0)
        return x


class BasicEncoder(_Encoder):
    """64-stem, stages 64/96/128, residual blocks (extractor.py:118-193)."""
    block_cls = ResidualBlock
    stem_planes = 64
    stage_planes = (64, 96, 128)


class SmallEncoder(_Encoder):
    """32-stem, stages 32/64/96, bottleneck blocks (extractor.py:195-267)."""
    block_cls = BottleneckBlock
    stem_planes = 32
    stage_planes = (32, 64, 96)
