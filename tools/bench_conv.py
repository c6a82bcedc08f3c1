#!/usr/bin/env python3
"""Microbenchmark: hand-written implicit-GEMM conv (csrc/conv_gemm.hip) vs
MIOpen (F.conv2d) at the update-block shapes, batch 3, H/8 grid 56x128.
Prints one JSON line per shape: fwd/bwd times for both paths.

Run on the GPU box:  python tools/bench_conv.py
"""

import json
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
torch.backends.cudnn.benchmark = True

SHAPES = [
    ("convc1", 324, 256, 1, 1),
    ("convc2", 256, 192, 3, 3),
    ("convf2", 128, 64, 3, 3),
    ("conv", 256, 126, 3, 3),
    ("gru_zr_1x5", 384, 256, 1, 5),
    ("gru_q_5x1", 384, 128, 5, 1),
    ("flowhead1", 128, 256, 3, 3),
]


def timeit(fn, warmup=5, iters=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from flowhip.ops.functional_conv import fused_conv2d

    dev = torch.device("cuda:0")
    B, H, W = 3, 56, 128
    for name, ci, co, kh, kw in SHAPES:
        pad = (kh // 2, kw // 2)
        x = (torch.randn(B, ci, H, W, device=dev) / 8).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last).requires_grad_(True)
        w = (torch.randn(co, ci, kh, kw, device=dev) /
             (ci * kh * kw) ** 0.5).requires_grad_(True)
        b = torch.randn(co, device=dev).requires_grad_(True)
        cache = {}

        out = fused_conv2d(x, w, b, 1, pad, 1, 1, cache)
        g = torch.randn_like(out)
        t_fwd = timeit(lambda: fused_conv2d(x, w, b, 1, pad, 1, 1, cache))

        def bwd_ours():
            o = fused_conv2d(x, w, b, 1, pad, 1, 1, cache)
            o.backward(g)
        t_fb = timeit(bwd_ours)

        wb = w.detach().to(torch.bfloat16).requires_grad_(True)
        bb = b.detach().to(torch.bfloat16).requires_grad_(True)
        t_mi_fwd = timeit(lambda: F.conv2d(x, wb, bb, padding=pad))

        def bwd_mi():
            o = F.conv2d(x, wb, bb, padding=pad)
            o.backward(g)
        t_mi_fb = timeit(bwd_mi)

        flops = 2.0 * B * H * W * ci * co * kh * kw
        print(json.dumps({
            "shape": name, "ci": ci, "co": co, "k": [kh, kw],
            "ours_fwd_ms": round(t_fwd, 4),
            "ours_fwd_tflops": round(flops / (t_fwd / 1e3) / 1e12, 1),
            "ours_fwd+bwd_ms": round(t_fb, 4),
            "miopen_fwd_ms": round(t_mi_fwd, 4),
            "miopen_fwd+bwd_ms": round(t_mi_fb, 4),
            "fwd_speedup": round(t_mi_fwd / t_fwd, 2),
            "fwdbwd_speedup": round(t_mi_fb / t_fb, 2),
        }))


if __name__ == "__main__":
    main()
