"""Mixed-precision policy.

On MI355X the right mixed-precision dtype is bf16 (MFMA bf16 peak ~2.5 PF
dense; no fp16 advantage, no loss-scaling needed). The reference used fp16
AMP + GradScaler on CUDA (train.py:25-39,192); here `--mixed_precision`
means bf16 autocast and the GradScaler becomes a no-op shim kept only for
API compatibility.
"""

import contextlib

import torch


def autocast_ctx(ref_tensor, enabled=True, dtype=torch.bfloat16):
    """Autocast scoped to the device of `ref_tensor` (cuda==ROCm here)."""
    if not enabled:
        return contextlib.nullcontext()
    device_type = "cuda" if ref_tensor.is_cuda else "cpu"
    return torch.autocast(device_type=device_type, dtype=dtype, enabled=True)


def autocast_off_ctx(ref_tensor):
    """fp32 island inside an enclosing autocast region.

    The model forward runs under ONE autocast region (keeping autocast's
    bf16 weight-cast cache alive across the 12-32 refinement iterations —
    per-iteration regions re-cast every conv weight every iteration, ~430
    extra cast kernels/step in profiles/r01). The correlation volume and
    the NCUP upsampler stay fp32 through this context, mirroring the
    reference's fp32 sections (raft.py:103-104, raft_nc_dbl.py:161).
    """
    device_type = "cuda" if ref_tensor.is_cuda else "cpu"
    return torch.autocast(device_type=device_type, enabled=False)


class NoOpGradScaler:
    """API-compatible stand-in for torch.cuda.amp.GradScaler under bf16
    (bf16 has fp32's exponent range — no scaling required)."""

    def scale(self, loss):
        return loss

    def unscale_(self, optimizer):
        pass

    def step(self, optimizer):
        optimizer.step()

    def update(self):
        pass

    def state_dict(self):
        return {}

    def load_state_dict(self, state):
        pass
