"""Autograd binding for the fused normalized-convolution kernel.

Forward: one fused HIP kernel (csrc/nconv.hip) producing (nconv, cout).
Backward: two fused HIP kernels (bwd-data + weight-grad); only the tiny
elementwise dnomin/ddenom pre-computation stays in torch. denom is
reconstructed from the saved cout (denom = cout * sum(w)) so no extra
forward tensor is stored. (The original torch/MIOpen composition fell into
naive_conv wrw fallbacks at ~240 ms/call for these tiny-channel full-res
shapes — see profiles/.)

Math (d denotes upstream grads):
  nomin = (out − bias) · (denom + eps)
  dnomin = dout / (denom + eps)
  ddenom = −dout · nomin / (denom+eps)² + dcout / s
  ddata  = conf · convT(dnomin, w)
  dconf  = data · convT(dnomin, w) + convT(ddenom, w)
  dw     = ∇w[conv(data·conf) vs dnomin] + ∇w[conv(conf) vs ddenom]
           − Σ(cout·dcout)/s   (through s = Σw per out-channel)
  dbias  = Σ dout
Reference math: nconv_modules.py:164-199.
"""

import torch

from . import _ext


class NConv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, data, conf, weight, bias, padding, eps, prop_conf):
        data = data.contiguous()
        conf = conf.contiguous()
        weight = weight.contiguous()
        b = bias.contiguous() if bias is not None else None
        out, cout = _ext.ext().nconv_fwd(data, conf, weight, b)
        if b is not None:
            ctx.save_for_backward(data, conf, weight, out, cout, b)
        else:
            ctx.save_for_backward(data, conf, weight, out, cout)
        ctx.has_bias = bias is not None
        ctx.padding = padding
        ctx.eps = eps
        return out, cout

    @staticmethod
    def backward(ctx, gout, gcout):
        data, conf, weight, out, cout, *maybe_bias = ctx.saved_tensors
        pad = ctx.padding if isinstance(ctx.padding, int) else ctx.padding[0]
        eps = ctx.eps

        s = weight.sum(dim=(1, 2, 3)).contiguous()         # (Co)

        gout = gout.contiguous()
        gcout = gcout.contiguous() if gcout is not None else None

        # fused elementwise preamble (one kernel): denom reconstructed from
        # cout; dnomin = gout/de, ddenom = -gout*(out-bias)/de + gcout/s
        dnomin, ddenom = _ext.ext().nconv_bwd_prep(
            gout, gcout, out, cout, s,
            maybe_bias[0] if ctx.has_bias else None, eps)

        ddata, dconf, dweight = _ext.ext().nconv_bwd(dnomin, ddenom, data,
                                                     conf, weight)
        if not ctx.needs_input_grad[2]:
            dweight = None
        elif gcout is not None:
            # cout = denom/s depends on s = sum(w) per out channel; one
            # fused per-channel plane reduction instead of the full-res
            # mul + sum pair (~48 pairs/step at (B*2, Co, H, W))
            ds = -_ext.ext().plane_sum_nchw(cout, gcout) / s
            dweight = dweight + ds.view(-1, 1, 1, 1)

        if ctx.has_bias:
            dbias = _ext.ext().plane_sum_nchw(gout.contiguous())
        else:
            dbias = None
        return ddata, dconf, dweight, dbias, None, None, None



class ConfPoolFn(torch.autograd.Function):
    """Confidence-based 2x pooling (kernel #7; nconv_modules.py:94-104):
    one fused kernel each way, full-write backward (no scatter)."""

    @staticmethod
    def forward(ctx, data, conf):
        dds, cds, code = _ext.ext().conf_pool_fwd(data.contiguous(),
                                                  conf.contiguous())
        ctx.save_for_backward(code)
        ctx.in_shape = list(data.shape)
        return dds, cds

    @staticmethod
    def backward(ctx, gdds, gcds):
        (code,) = ctx.saved_tensors
        gdds = gdds.contiguous() if gdds is not None else None
        gcds = gcds.contiguous() if gcds is not None else None
        gdata, gconf = _ext.ext().conf_pool_bwd(gdds, gcds, code,
                                                ctx.in_shape)
        return gdata, gconf
