"""Autograd bindings for the HIP kernels.

Each Function pairs a hand-written gfx950 forward kernel with an explicit
backward (HIP kernels where hot, torch/MIOpen composition where not). The
math contracts live in `torch_ref.py`; GPU unit tests compare both paths.

Layout conventions for the correlation stack (chosen for MFMA/LDS on CDNA4,
not inherited from the reference):
  - feature maps enter as (B, D, H, W) fp32 and are staged as k-contiguous
    bf16 (P, D) operands so both GEMM tiles are ds_read_b128-friendly;
  - the batched GEMM computes C = alpha * A @ B^T with A (M,K), B (N,K) both
    row-major bf16, C (M,N) fp32 — forward and both backward products all map
    onto this one kernel (see CorrVolumeFn docstring).
"""

import math

import torch

from . import _ext


def _pad_k(t):
    """Zero-pad the trailing (K) dim to a multiple of 64 (kernel tile depth).
    Zero columns contribute nothing to the dot products."""
    k = t.shape[-1]
    pad = (-k) % 64
    if pad == 0:
        return t
    return torch.nn.functional.pad(t, (0, pad))


def _bgemm_nt(a, b, alpha, out_bf16=False):
    """C[bat] = alpha * A[bat] @ B[bat]^T ; a (Bt,M,K) bf16, b (Bt,N,K) bf16
    -> fp32 (default) or bf16 (resident corr volume)."""
    return _ext.ext().bgemm_nt(_pad_k(a).contiguous(), _pad_k(b).contiguous(),
                               alpha, out_bf16)


class CorrVolumeFn(torch.autograd.Function):
    """All-pairs correlation C[b,i,j] = <f1[:,i], f2[:,j]>/sqrt(D).

    forward:  C (B,P,P) = 1/sqrt(D) * F1t (B,P,D) @ F2t (B,P,D)^T
    backward: dF1t = 1/sqrt(D) * dC @ F2  -> gemm_nt(dC, F2 (D,P) as (N=D,K=P)? )
    Both backward products reuse the same gemm_nt kernel:
      dF1t (P,D) = gemm_nt(dC  (P,P), F2 (D,P))   [B operand = f2 k-major over P]
      dF2t (P,D) = gemm_nt(dC^T(P,P), F1 (D,P))
    Reference math: core/corr.py:47-55. Inputs are cast to bf16 (fp32
    accumulate in MFMA) — documented deviation from the reference's fp32
    matmul; tolerance covered by tests/test_gpu_corr.py.
    """

    @staticmethod
    def forward(ctx, fmap1, fmap2):
        B, D, H, W = fmap1.shape
        P = H * W

        def as_pd(f):
            # channels_last (B,H,W,D) memory IS the (B,P,D) operand — no
            # transpose kernel, just the bf16 cast
            if f.is_contiguous(memory_format=torch.channels_last):
                return f.permute(0, 2, 3, 1).reshape(B, P, D).to(torch.bfloat16)
            return f.reshape(B, D, P).transpose(1, 2).contiguous().to(
                torch.bfloat16)

        from ..utils.layout import corr_bf16_enabled
        f1t = as_pd(fmap1)  # (B,P,D)
        f2t = as_pd(fmap2)
        # (B,P,P); bf16 residency halves HBM footprint + all pyramid/lookup
        # read traffic (fp32 accumulate inside the MFMA either way)
        corr = _bgemm_nt(f1t, f2t, 1.0 / math.sqrt(D),
                         out_bf16=corr_bf16_enabled())
        ctx.save_for_backward(f1t, f2t)
        ctx.shape = (B, D, H, W)
        ctx.cl = fmap1.is_contiguous(memory_format=torch.channels_last)
        return corr.reshape(B * P, 1, H, W)

    @staticmethod
    def backward(ctx, grad):
        f1t, f2t = ctx.saved_tensors
        B, D, H, W = ctx.shape
        P = H * W
        alpha = 1.0 / math.sqrt(D)
        g3 = grad.reshape(B, P, P).contiguous()
        dc = g3 if g3.dtype == torch.bfloat16 else g3.to(torch.bfloat16)
        # tiled transpose(+cast) kernel (eager transpose().to(bf16) is an
        # uncoalesced ~300us elementwise op at P=7168); accepts fp32 or bf16
        dct = _ext.ext().transpose_cast_bf16(g3)
        # dF1t[i,d] = sum_j dC[i,j] * F2t[j,d]: A = dC (M=P, K=P); the B
        # operand must be (N=D, K=P) row-major = f2 in (D, P) layout with P
        # contiguous.
        f2_kn = f2t.transpose(1, 2).contiguous()  # (B,D,P) bf16, P contiguous
        f1_kn = f1t.transpose(1, 2).contiguous()
        df1t = _bgemm_nt(dc, f2_kn, alpha)   # (B,P,D) fp32
        df2t = _bgemm_nt(dct, f1_kn, alpha)  # (B,P,D) fp32
        if ctx.cl:
            # (B,P,D) memory == channels_last (B,D,H,W): zero-copy view
            df1 = df1t.reshape(B, H, W, D).permute(0, 3, 1, 2)
            df2 = df2t.reshape(B, H, W, D).permute(0, 3, 1, 2)
        else:
            df1 = df1t.transpose(1, 2).reshape(B, D, H, W)
            df2 = df2t.transpose(1, 2).reshape(B, D, H, W)
        return df1, df2


class CorrLookupFn(torch.autograd.Function):
    """Fused 4-level (2r+1)^2-tap bilinear window lookup (core/corr.py:23-44).

    coords never require grad in the RAFT iteration loop (coords1 is detached
    right before every lookup — raft.py:122, raft_nc_dbl.py:149), so backward
    produces gradients for the pyramid levels only and asserts that contract.

    Returns (out, *levels): the pyramid is passed THROUGH so CorrBlock can
    thread it along the iteration loop — the autograd graph then forms a
    CHAIN over the 12 lookups instead of a 12-way fan-out, and each
    backward ACCUMULATES its patch contribution into the incoming level
    grads in-kernel (no per-iteration zero-fill, no autograd add_ fan-in).
    Callers that drop the passthrough (tests, single lookups) still get
    correct fan-out semantics: missing chain grads select the fresh-buffer
    path.
    """

    @staticmethod
    def forward(ctx, coords, radius, *pyramid):
        from ..utils.layout import channels_last_enabled
        B, _, H1, W1 = coords.shape
        assert not coords.requires_grad, (
            "corr_lookup: coords must be detached (reference contract)")
        coords_c = coords.contiguous()
        cl = channels_last_enabled()
        out = _ext.ext().corr_lookup_fwd(list(pyramid), coords_c, int(radius),
                                         cl)
        ctx.save_for_backward(coords_c)
        ctx.radius = int(radius)
        ctx.cl = cl
        ctx.level_shapes = [tuple(p.shape) for p in pyramid]
        ctx.levels_bf16 = pyramid[0].dtype == torch.bfloat16
        return (out, *pyramid)

    @staticmethod
    def backward(ctx, grad, *glevels):
        (coords,) = ctx.saved_tensors
        dt = torch.bfloat16 if ctx.levels_bf16 else torch.float32
        if grad.dtype != dt:
            grad = grad.to(dt)
        if ctx.cl:
            grad = grad.contiguous(memory_format=torch.channels_last)
        else:
            grad = grad.contiguous()
        have = [g is not None for g in glevels]
        prev = None
        if all(have):
            prev = [g.contiguous() for g in glevels]
        else:
            assert not any(have), "corr_lookup chain: partial level grads"
        grads = _ext.ext().corr_lookup_bwd(
            grad, coords, ctx.radius,
            [list(s) for s in ctx.level_shapes], ctx.cl, ctx.levels_bf16,
            prev)
        return (None, None, *grads)


class CorrPyramidFn(torch.autograd.Function):
    """Fused avg-pool pyramid (reference corr.py:19-21).

    forward: one LDS-staged kernel emits levels 1..n-1 (level 0 = input,
    passed through). backward: one gather kernel combines the per-level
    grads, dcorr[y,x] = g0 + g1[y/2,x/2]/4 + g2[..]/16 + g3[..]/64 —
    replacing the 3-deep avg_pool2d-backward chain.
    """

    @staticmethod
    def forward(ctx, corr, num_levels):
        corr = corr.contiguous()
        levels = _ext.ext().corr_pyramid_fwd(corr, int(num_levels))
        ctx.corr_shape = list(corr.shape)
        return (corr, *levels)

    @staticmethod
    def backward(ctx, *grads):
        gs = [g.contiguous() if g is not None else None for g in grads]
        dcorr = _ext.ext().corr_pyramid_bwd(gs, ctx.corr_shape)
        return dcorr, None
