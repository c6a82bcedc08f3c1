"""Autograd bindings for the fused ConvGRU gate kernels (kernel #4 support).

The GRU convolutions stay on MIOpen; these two Functions replace the ~6
eager elementwise kernels around them per GRU pass (sigmoid/chunk/mul,
tanh/lerp) with one fused kernel each way. Activations are recomputed from
the saved pre-activations in backward. Reference math: update.py:16-60.
"""

import torch

from . import _ext


def _match_layout(g, ref):
    if ref.is_contiguous(memory_format=torch.channels_last):
        return g.contiguous(memory_format=torch.channels_last)
    return g.contiguous()


class GruGate1Fn(torch.autograd.Function):
    """(zr_preact, h) -> (z, r*h) with z = sigmoid(zr[:, :C]), r = sigmoid(zr[:, C:])."""

    @staticmethod
    def forward(ctx, zr, h):
        z, rh = _ext.ext().gru_gate1_fwd(zr, h)
        ctx.save_for_backward(zr, h)
        return z, rh

    @staticmethod
    def backward(ctx, dz, drh):
        zr, h = ctx.saved_tensors
        dz = _match_layout(dz, zr) if dz is not None else None
        if drh is None:  # rh unused downstream (never in the GRU path)
            drh = torch.zeros_like(h)
        drh = _match_layout(drh, zr)
        dzr, dh = _ext.ext().gru_gate1_bwd(dz, drh, zr, h)
        return dzr, dh


class GruGate2Fn(torch.autograd.Function):
    """(q_preact, z, h) -> (1-z)*h + z*tanh(q_preact)."""

    @staticmethod
    def forward(ctx, qp, z, h):
        hnew = _ext.ext().gru_gate2_fwd(qp, z, h)
        ctx.save_for_backward(qp, z, h)
        return hnew

    @staticmethod
    def backward(ctx, dhnew):
        qp, z, h = ctx.saved_tensors
        dhnew = _match_layout(dhnew, qp)
        dqp, dz, dh = _ext.ext().gru_gate2_bwd(dhnew, qp, z, h)
        return dqp, dz, dh


def gru_gates_available(t):
    return (t.is_cuda and t.dtype in (torch.float32, torch.bfloat16)
            and _ext.ext() is not None and not _ext.force_ref())
