"""Autograd binding for the fused sequence loss (kernel #12).

Forward: one masked-reduction kernel over all n predictions + a finalize
kernel -> [loss, epe, 1px, 3px, 5px] on device (the eager chain was ~5
kernels per prediction per direction). Backward: ONE kernel writes every
prediction's gradient. Math contract: torch_ref.sequence_loss
(reference train.py:46-71).
"""

import torch

from . import _ext


class SequenceLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gt, valid, gamma, max_flow, *preds):
        preds = [p.contiguous() for p in preds]
        gt = gt.contiguous()
        valid = valid.contiguous().float()
        out = _ext.ext().seq_loss_fwd(list(preds), gt, valid, gamma, max_flow)
        ctx.save_for_backward(gt, valid, *preds)
        ctx.gamma = gamma
        ctx.max_flow = max_flow
        ctx.mark_non_differentiable(out[1:])
        return out[0], out[1:]

    @staticmethod
    def backward(ctx, gloss, _gmetrics):
        gt, valid, *preds = ctx.saved_tensors
        grads = _ext.ext().seq_loss_bwd(list(preds), gt, valid,
                                        gloss.contiguous(), ctx.gamma,
                                        ctx.max_flow)
        return (None, None, None, None, *grads)


def sequence_loss_fused(flow_preds, flow_gt, valid, gamma, max_flow):
    loss, m = SequenceLossFn.apply(flow_gt, valid, gamma, max_flow,
                                   *flow_preds)
    mv = m.tolist()  # single D2H sync for all four metrics
    metrics = {"epe": mv[0], "1px": mv[1], "3px": mv[2], "5px": mv[3]}
    return loss, metrics
