"""flowhip — MI355X-native RAFT + NCUP optical-flow engine.

A brand-new AMD CDNA4 (gfx950) implementation of the RAFT optical-flow
architecture with the Normalized Convolution Upsampler (NCUP), matching the
capabilities of the reference `abdo-eldesokey/RAFT-NCUP` while
being designed MI355X-first:

- hot ops (all-pairs correlation volume, pyramid build, 4-level window lookup,
  normalized convolution, SepConvGRU, convex upsampling) are hand-written HIP
  kernels for gfx950 (MFMA, LDS tiling) exposed through `flowhip.ops`;
- data-parallel training is one process per GPU over RCCL/xGMI
  (`flowhip.engine.distributed`);
- model state-dict layout stays compatible with the reference checkpoints
  (`module.`-prefixed DataParallel keys, `weight_p` nconv reparameterization).

Layers (bottom-up): utils -> ops -> nn -> models -> engine.
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
