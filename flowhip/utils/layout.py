"""Memory-format policy for the GPU path.

MIOpen's bf16 conv kernels are NHWC-native: running the model NCHW costs a
batched_transpose pair around every conv call (~1,470 transpose launches per
training step measured in profiles/r01). The GPU path therefore runs the
whole network channels_last:

- conv modules + activations carry torch.channels_last strides;
- the corr GEMM (ops.functional.CorrVolumeFn) reads (B,H,W,D)-contiguous
  feature maps directly as its (B,P,D) row-major operands — the explicit
  transpose of the NCHW path disappears;
- the fused corr-lookup kernel writes its (B, L*K*K, H, W) output with
  channels_last strides (per-pixel-contiguous taps, fully coalesced stores)
  so the motion-encoder convs consume it natively.

CPU keeps NCHW (the torch reference path; layout has no semantic effect).
"""

import os

import torch

_CHANNELS_LAST = True

# HBM residency dtype for the correlation volume + pyramid (north star /
# BASELINE config 5: full-res corr volume resident in bf16). The GEMM
# emits bf16 directly, the pyramid/lookup kernels read bf16, the bilinear
# blend and the lookup output stay fp32. FLOWHIP_CORR_DTYPE=fp32 restores
# the round-1 fp32-resident pyramid (numerics ablation / parity runs).
_CORR_BF16 = os.environ.get("FLOWHIP_CORR_DTYPE", "bf16").lower() != "fp32"


def set_corr_bf16(enabled):
    global _CORR_BF16
    _CORR_BF16 = bool(enabled)


def corr_bf16_enabled():
    return _CORR_BF16


def set_channels_last(enabled):
    global _CHANNELS_LAST
    _CHANNELS_LAST = bool(enabled)


def channels_last_enabled():
    return _CHANNELS_LAST


def apply_channels_last(model):
    """Convert conv weights to channels_last — except the NCUP upsampler.

    The upsampler subtree (nconv kernels + the tiny fp32 weights-est CNN)
    runs NCHW: the fused nconv kernels are NCHW-native, and MIOpen's fp32
    NCHW Winograd (Sp3AsmConv) beats its NHWC fp32 igemm ~3x at the
    weights-est shapes (profiles/r01 trace3 vs trace1). Keeping the subtree
    NCHW also removes the mixed-layout cat/copy storm at the boundary."""
    if _CHANNELS_LAST:
        model.to(memory_format=torch.channels_last)
        ups = getattr(model, "upsampler", None)
        if ups is not None:
            ups.to(memory_format=torch.contiguous_format)
            # ...except the confidence net, which runs bf16/NHWC on the
            # MFMA conv kernel (see NConvUpsampler.forward)
            wen = getattr(ups, "weights_est_net", None)
            if isinstance(wen, torch.nn.Module):
                wen.to(memory_format=torch.channels_last)
    return model


def to_model_layout(x):
    """Bring a CUDA input batch into the model's memory format."""
    if _CHANNELS_LAST and x.is_cuda:
        return x.contiguous(memory_format=torch.channels_last)
    return x.contiguous()
