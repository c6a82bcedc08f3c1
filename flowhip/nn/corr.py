"""Correlation block: volume build + pyramid + per-iteration window lookup.

Reference behavior: core/corr.py (CorrBlock). The compute goes through
`flowhip.ops` so the HIP kernels (MFMA GEMM #1, pyramid #2, fused lookup #3)
run on GPU and the torch path on CPU. Unlike the reference, the window offset
grid lives on-device inside the lookup kernel (no per-iteration host->device
`delta` transfer — SURVEY.md §2.9 quirk 8), which also makes the iteration
loop hipGraph-capturable.
"""

from .. import ops


class CorrBlock:
    def __init__(self, fmap1, fmap2, num_levels=4, radius=4):
        self.num_levels = num_levels
        self.radius = radius
        corr = ops.corr_volume(fmap1, fmap2)  # (B*H*W, 1, H, W)
        self.corr_pyramid = ops.corr_pyramid(corr, num_levels)

    def __call__(self, coords):
        out, self.corr_pyramid = ops.corr_lookup_chained(
            self.corr_pyramid, coords, self.radius)
        return out
