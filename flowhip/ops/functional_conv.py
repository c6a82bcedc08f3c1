"""Implicit-GEMM conv (kernel #5) autograd binding for the update block.

Routes the stride-1 same-padding NHWC bf16 convolutions of the motion
encoder / SepConvGRU / flow head (reference update.py:6-146) onto the
hand-written MFMA kernel in csrc/conv_gemm.hip. The packed bf16 weights
([kyx][o][c_pad] for forward, flipped/transposed for backward-data) are
cached per weight version so they are rebuilt only after optimizer steps.

  forward:   out = conv(x, w) + b          (one kernel)
  bwd-data:  dx  = conv(dy, flipT(w))      (same kernel, swapped packing)
  bwd-wrw:   dW  = split-M MFMA partials + reduce (fp32)
  bwd-bias:  dy.sum((0,2,3))               (torch reduce)
"""

import torch
import torch.nn.functional as F

from . import _ext


def _pad64(n):
    return (n + 63) // 64 * 64


def _pad_c8(x, weight):
    """Channel-pad input AND weight to a multiple of 8 via differentiable
    cats. The staging kernel reads whole 16-B pieces; with Cin % 8 != 0 the
    tail piece of the last pixel would read past the tensor allocation
    (fault, allocator-layout dependent). Zero input channels x zero weight
    columns = identical math, all reads in-bounds."""
    O, I, KH, KW = weight.shape
    pad = (-I) % 8
    if pad == 0:
        return x, weight
    z = x.new_zeros(x.shape[0], pad, x.shape[2], x.shape[3])         .contiguous(memory_format=torch.channels_last)
    xp = torch.cat([x, z], dim=1)
    wz = weight.new_zeros(O, pad, KH, KW)
    wp = torch.cat([weight, wz], dim=1)
    return xp, wp


def _pad_dy8(dy):
    """Channel-pad a backward gradient to a multiple of 8 (same OOB-tail
    argument as _pad_c8; the packed-bwd weight k-columns beyond Cout are
    zero, so padded dy channels contribute nothing)."""
    O = dy.shape[1]
    pad = (-O) % 8
    if pad == 0:
        return dy, O
    z = dy.new_zeros(dy.shape[0], pad, dy.shape[2], dy.shape[3])         .contiguous(memory_format=torch.channels_last)
    return torch.cat([dy, z], dim=1), O


def _pack_fwd(weight):
    """(O, I, KH, KW) fp32 -> (KYX, O, pad64(I)) bf16 contiguous."""
    w = weight.detach()
    if w.is_cuda and w.dtype == torch.float32:
        return _ext.ext().conv_gemm_pack(w.contiguous(), False)
    O, I, KH, KW = w.shape
    w = w.to(torch.bfloat16).permute(2, 3, 0, 1).reshape(KH * KW, O, I)
    return F.pad(w, (0, _pad64(I) - I)).contiguous()


def _pack_bwd(weight):
    """flip + transpose: (O, I, KH, KW) -> (KYX, I, pad64(O)) bf16."""
    w = weight.detach()
    if w.is_cuda and w.dtype == torch.float32:
        return _ext.ext().conv_gemm_pack(w.contiguous(), True)
    O, I, KH, KW = w.shape
    w = w.to(torch.bfloat16).flip(2, 3).permute(2, 3, 1, 0)
    w = w.reshape(KH * KW, I, O)
    return F.pad(w, (0, _pad64(O) - O)).contiguous()


def _packs(weight, cache, key=None):
    if key is None:
        key = (weight._version, weight.data_ptr())
    if cache.get("k") != key:
        cache["k"] = key
        cache["fwd"] = _pack_fwd(weight)
        cache["bwd"] = _pack_bwd(weight)
    return cache["fwd"], cache["bwd"]


class ConvGemmFn(torch.autograd.Function):
    """Stride-1 (sH=sW=1) or strided (smode 1 fwd / smode 2 bwd-data)
    same-padding conv on the MFMA kernel."""

    @staticmethod
    def forward(ctx, x, weight, bias, wpk_fwd, wpk_bwd, sH, sW):
        O, I, KH, KW = weight.shape
        b = bias.detach().float().contiguous() if bias is not None else None
        if sH == 1 and sW == 1:
            out = _ext.ext().conv_gemm_fwd(x, wpk_fwd, b, O, KH, KW, 0)
        else:
            out = _ext.ext().conv_gemm_fwd2(x, None, wpk_fwd, b, O, KH, KW,
                                            0, 0, sH, sW, 1)[0]
        ctx.save_for_backward(x, wpk_bwd)
        ctx.meta = (O, I, KH, KW, bias is not None, sH, sW)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, wpk_bwd = ctx.saved_tensors
        O, I, KH, KW, has_bias, sH, sW = ctx.meta
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dyp, O_real = _pad_dy8(dy)
        dx = None
        if ctx.needs_input_grad[0]:  # leaf inputs (the stem images) skip dx
            if sH == 1 and sW == 1:
                dx = _ext.ext().conv_gemm_fwd(dyp, wpk_bwd, None, I, KH, KW,
                                              0)
            else:
                dx = _ext.ext().conv_gemm_fwd2(dyp, None, wpk_bwd, None, I,
                                               KH, KW, 0, 0, sH, sW, 2,
                                               x.shape[2], x.shape[3])[0]
        dw, db = _ext.ext().conv_gemm_wrw(dyp, x, None, KH, KW, sH, sW,
                                          has_bias)
        if dw.shape[0] != O_real:
            dw = dw[:O_real].contiguous()
        dbias = db[:O_real] if has_bias else None
        return dx, dw, dbias, None, None, None, None


def can_fuse_conv(x, weight, stride, padding, dilation, groups):
    if not (x.is_cuda and x.dim() == 4 and x.dtype == torch.bfloat16):
        return False
    if x.stride(1) != 1:  # channels-last-like only
        return False
    if _ext.ext() is None or _ext.force_ref():
        return False
    from torch.nn.modules.utils import _pair
    if (_pair(stride) not in ((1, 1), (2, 2))
            or _pair(dilation) != (1, 1) or groups != 1):
        return False
    O, I, KH, KW = weight.shape
    if _pair(padding) != (KH // 2, KW // 2):
        return False
    if I != x.shape[1] or O > 512 or KH * KW > 49:
        return False
    if x.stride(3) < x.size(1):
        return False
    if I % 8 != 0 and x.stride(3) != x.size(1):
        # narrowed view with ragged Cin: the 16-B tail piece must stay
        # inside the row (the padded-allocation case has ld % 8 == 0)
        if x.stride(3) % 8 != 0 or x.stride(3) < I + ((-I) % 8):
            return False
    return True


def fused_conv2d(x, weight, bias, stride, padding, dilation, groups, cache,
                 key=None):
    """F.conv2d drop-in that routes supported shapes to the MFMA kernel."""
    # autocast does NOT auto-cast custom autograd Functions: an fp32 input
    # under an active bf16 autocast region (convc1's corr features, convf1's
    # flow) would silently fall back to MIOpen — do autocast's cast here
    if (x.is_cuda and x.dtype == torch.float32
            and torch.is_autocast_enabled("cuda")
            and torch.get_autocast_dtype("cuda") == torch.bfloat16):
        xc = x.to(torch.bfloat16)
        if xc.stride(1) != 1 or not xc.is_contiguous(
                memory_format=torch.channels_last):
            xc = xc.contiguous(memory_format=torch.channels_last)
        if can_fuse_conv(xc, weight, stride, padding, dilation, groups):
            x = xc
    if can_fuse_conv(x, weight, stride, padding, dilation, groups):
        if key is None:
            key = (weight._version, weight.data_ptr())
        if weight.shape[1] % 8 != 0 and x.stride(3) == x.size(1):
            # ragged Cin on a tight row: pad both (16-B staging tail).
            # A channel-narrowed view with padded ld (the corr-lookup
            # output) needs NO pad — its tail bytes are in-row and the
            # packed weight's zero columns null them (saves a full-tensor
            # cat copy per GRU iteration).
            x, weight = _pad_c8(x, weight)
        wf, wb = _packs(weight, cache, key)
        from torch.nn.modules.utils import _pair
        sH, sW = _pair(stride)
        return ConvGemmFn.apply(x, weight, bias, wf, wb, sH, sW)
    import os
    if x.is_cuda and os.environ.get("FLOWHIP_LOG_FALLBACK", "0") == "1":
        print(f"[conv fallback] w={tuple(weight.shape)} stride={stride} "
              f"pad={padding} dil={dilation} g={groups} x={tuple(x.shape)} "
              f"xs={x.stride()} dt={x.dtype}", flush=True)
    return F.conv2d(x, weight, bias, stride, padding, dilation, groups)



class ConvGemmCat2Fn(torch.autograd.Function):
    """conv over a virtually-concatenated input pair cat([x1, x2], 1) —
    removes the per-call cat materialization and its backward narrows
    (4 per GRU iteration). x1's channel count must be a multiple of 64."""

    @staticmethod
    def forward(ctx, x1, x2, weight, bias, wpk_fwd, wpk_bwd):
        O, I, KH, KW = weight.shape
        b = bias.detach().float().contiguous() if bias is not None else None
        out = _ext.ext().conv_gemm_fwd2(x1, x2, wpk_fwd, b, O, KH, KW, 0,
                                        0)[0]
        ctx.save_for_backward(x1, x2, wpk_bwd)
        ctx.meta = (O, I, KH, KW, bias is not None, x1.shape[1])
        return out

    @staticmethod
    def backward(ctx, dy):
        x1, x2, wpk_bwd = ctx.saved_tensors
        O, I, KH, KW, has_bias, C1 = ctx.meta
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dyp, O_real = _pad_dy8(dy)
        dx1, dx2 = _ext.ext().conv_gemm_fwd2(dyp, None, wpk_bwd, None, I, KH,
                                             KW, C1, 0)
        dw, db = _ext.ext().conv_gemm_wrw(dyp, x1, x2, KH, KW, 1, 1,
                                          has_bias)
        if dw.shape[0] != O_real:
            dw = dw[:O_real].contiguous()
        dbias = db[:O_real] if has_bias else None
        return dx1, dx2, dw, dbias, None, None


class ConvGemmCat2ZrFn(torch.autograd.Function):
    """Packed z+r gate convolution over the virtually-concatenated GRU
    input: ONE kernel computes conv(cat([h, x]), cat([wz, wr])) without
    materializing either cat. The packed weights/bias are cached per
    parameter version (rebuilt after optimizer steps only); gradients are
    routed back to the four separate parameters by slicing — replaces the
    per-call torch.cat of weights AND its CatBackward (4 cat launches per
    GRU pass)."""

    @staticmethod
    def forward(ctx, x1, x2, wz, wr, bz, br, bias_f, wpk_fwd, wpk_bwd):
        Oz, I, KH, KW = wz.shape
        O = Oz + wr.shape[0]
        out = _ext.ext().conv_gemm_fwd2(x1, x2, wpk_fwd, bias_f, O, KH, KW,
                                        0, 0)[0]
        ctx.save_for_backward(x1, x2, wpk_bwd)
        ctx.meta = (O, Oz, I, KH, KW, x1.shape[1])
        return out

    @staticmethod
    def backward(ctx, dy):
        x1, x2, wpk_bwd = ctx.saved_tensors
        O, Oz, I, KH, KW, C1 = ctx.meta
        dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dx1, dx2 = _ext.ext().conv_gemm_fwd2(dy, None, wpk_bwd, None, I, KH,
                                             KW, C1, 0)
        dw, db = _ext.ext().conv_gemm_wrw(dy, x1, x2, KH, KW, 1, 1, True)
        return (dx1, dx2, dw[:Oz], dw[Oz:].contiguous(), db[:Oz],
                db[Oz:].contiguous(), None, None, None)


def fused_gru_zr_conv(h, x, convz, convr, padding, cache):
    """conv2d(cat([h, x]), cat([wz, wr])) + cat([bz, br]) in one fused
    kernel with version-cached packing. Falls back to the eager
    composition off-GPU."""
    key = (convz.weight.data_ptr(), convz.weight._version,
           convr.weight.data_ptr(), convr.weight._version,
           convz.bias._version, convr.bias._version)
    O1, I, KH, KW = convz.weight.shape
    fusable = (h.is_cuda and h.dtype == torch.bfloat16
               and x.dtype == torch.bfloat16
               and h.shape[1] % 64 == 0
               and (h.shape[1] + x.shape[1]) % 8 == 0
               and h.stride(1) == 1 and x.stride(1) == 1
               and _ext.ext() is not None and not _ext.force_ref())
    from torch.nn.modules.utils import _pair
    if fusable and _pair(padding) == (KH // 2, KW // 2):
        if cache.get("k") != key:
            with torch.no_grad():
                w = torch.cat([convz.weight, convr.weight])
                cache["k"] = key
                cache["fwd"] = _pack_fwd(w)
                cache["bwd"] = _pack_bwd(w)
                cache["bias"] = torch.cat([convz.bias, convr.bias]) \
                    .float().contiguous()
        return ConvGemmCat2ZrFn.apply(h, x, convz.weight, convr.weight,
                                      convz.bias, convr.bias, cache["bias"],
                                      cache["fwd"], cache["bwd"])
    import os
    if h.is_cuda and os.environ.get("FLOWHIP_LOG_FALLBACK", "0") == "1":
        print(f"[zr fallback] h={tuple(h.shape)} hd={h.dtype} "
              f"hs={h.stride()} x={tuple(x.shape)} xd={x.dtype} "
              f"xs={x.stride()} pad={padding} K=({KH},{KW})", flush=True)
    zr_w = torch.cat([convz.weight, convr.weight])
    zr_b = torch.cat([convz.bias, convr.bias])
    hx = torch.cat([h, x], dim=1)
    return F.conv2d(hx, zr_w, zr_b, padding=padding)


def fused_conv2d_cat2(x1, x2, weight, bias, padding, cache, key):
    """conv2d(cat([x1, x2], 1), weight) without materializing the cat.
    Falls back to the eager cat + F.conv2d outside the fused envelope."""
    O, I, KH, KW = weight.shape
    fusable = (x1.is_cuda and x1.dtype == torch.bfloat16
               and x2.dtype == torch.bfloat16
               and x1.shape[1] % 64 == 0
               and (x1.shape[1] + x2.shape[1]) % 8 == 0  # 16-B staging tail
               and x1.stride(1) == 1 and x2.stride(1) == 1
               and _ext.ext() is not None and not _ext.force_ref())
    from torch.nn.modules.utils import _pair
    if fusable and _pair(padding) == (KH // 2, KW // 2):
        wf, wb = _packs(weight, cache, key)
        return ConvGemmCat2Fn.apply(x1, x2, weight, bias, wf, wb)
    import os
    if x1.is_cuda and os.environ.get("FLOWHIP_LOG_FALLBACK", "0") == "1":
        print(f"[cat2 fallback] x1={tuple(x1.shape)} d={x1.dtype} "
              f"s={x1.stride()} x2={tuple(x2.shape)} d2={x2.dtype} "
              f"s2={x2.stride()} w={tuple(weight.shape)} pad={padding}",
              flush=True)
    hx = torch.cat([x1, x2], dim=1)
    return F.conv2d(hx, weight, bias, padding=padding)
