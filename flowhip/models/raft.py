"""RAFT baseline model (learned convex-combination x8 upsampling).

State-dict compatible with the reference `core/raft.py` (fnet / cnet /
update_block attribute names; small and basic variants). Mixed precision is
bf16 autocast on ROCm (the reference used fp16 AMP on CUDA); the correlation
volume runs in bf16-in/fp32-accumulate MFMA on GPU.
"""

import torch
import torch.nn as nn

from ..nn.corr import CorrBlock
from ..nn.extractor import BasicEncoder, SmallEncoder
from ..nn.update import BasicUpdateBlock, SmallUpdateBlock
from ..utils.amp import autocast_ctx, autocast_off_ctx
from ..utils.geometry import coords_grid, upflow8
from .. import ops


class RAFT(nn.Module):
    def __init__(self, args):
        super().__init__()
        self.args = args

        if args.small:
            self.hidden_dim = hdim = 96
            self.context_dim = cdim = 64
            args.corr_levels = 4
            args.corr_radius = 3
        else:
            self.hidden_dim = hdim = 128
            self.context_dim = cdim = 128
            args.corr_levels = 4
            args.corr_radius = 4

        if not hasattr(args, "dropout"):
            args.dropout = 0

        if args.small:
            self.fnet = SmallEncoder(output_dim=128, norm_fn="instance", dropout=args.dropout)
            self.cnet = SmallEncoder(output_dim=hdim + cdim, norm_fn="none", dropout=args.dropout)
            self.update_block = SmallUpdateBlock(self.args, hidden_dim=hdim)
        else:
            self.fnet = BasicEncoder(output_dim=256, norm_fn="instance", dropout=args.dropout)
            self.cnet = BasicEncoder(output_dim=hdim + cdim, norm_fn="batch", dropout=args.dropout)
            self.update_block = BasicUpdateBlock(self.args, hidden_dim=hdim)

    def freeze_bn(self):
        for m in self.modules():
            if isinstance(m, nn.BatchNorm2d):
                m.eval()

    def initialize_flow(self, img):
        """flow = coords1 - coords0; both start at the identity grid (H/8)."""
        N, C, H, W = img.shape
        coords0 = coords_grid(N, H // 8, W // 8, device=img.device)
        coords1 = coords_grid(N, H // 8, W // 8, device=img.device)
        return coords0, coords1

    def upsample_flow(self, flow, mask):
        """x8 convex-combination upsample (ops kernel #11; raft.py:73-84)."""
        return ops.convex_upsample(flow, mask, factor=8)

    def _features(self, image1, image2):
        """Normalize images and run feature + context networks."""
        from ..utils.layout import to_model_layout
        image1 = 2 * (image1 / 255.0) - 1.0
        image2 = 2 * (image2 / 255.0) - 1.0
        image1 = to_model_layout(image1)
        image2 = to_model_layout(image2)

        fmap1, fmap2 = self.fnet([image1, image2])

        with autocast_off_ctx(image1):
            fmap1 = fmap1.float()
            fmap2 = fmap2.float()
            corr_fn = CorrBlock(fmap1, fmap2, radius=self.args.corr_radius)

        cnet = self.cnet(image1)
        net, inp = torch.split(cnet, [self.hidden_dim, self.context_dim], dim=1)
        # tanh/relu on the narrowed channels-last views materialize plain
        # NCHW tensors; pin both back to the model layout ONCE here or the
        # GRU input cat goes NCHW and every 384-ch GRU conv leaves the
        # MFMA path (attr_profile r02)
        net = to_model_layout(torch.tanh(net))
        inp = to_model_layout(torch.relu(inp))
        return image1, corr_fn, net, inp

    def forward(self, image1, image2, iters=12, flow_init=None, upsample=True,
                test_mode=False):
        """Estimate optical flow between a pair of frames.

        `upsample` is accepted for API compatibility and ignored, as in the
        reference (raft.py:87 — SURVEY.md §2.9 quirk 4).
        """
        with autocast_ctx(image1, enabled=self.args.mixed_precision):
            image1, corr_fn, net, inp = self._features(image1, image2)

            coords0, coords1 = self.initialize_flow(image1)
            if flow_init is not None:
                coords1 = coords1 + flow_init

            flow_predictions = []
            flow_up = None
            for _ in range(iters):
                coords1 = coords1.detach()
                with autocast_off_ctx(image1):
                    corr = corr_fn(coords1)

                flow = coords1 - coords0
                net, up_mask, delta_flow = self.update_block(net, inp, corr,
                                                             flow)

                coords1 = coords1 + delta_flow.float()

                with autocast_off_ctx(image1):
                    if up_mask is None:
                        flow_up = upflow8(coords1 - coords0)
                    else:
                        flow_up = self.upsample_flow(
                            (coords1 - coords0).float(), up_mask.float())
                flow_predictions.append(flow_up)

        if test_mode:
            return coords1 - coords0, flow_up
        return flow_predictions
