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
