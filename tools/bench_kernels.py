#!/usr/bin/env python3
"""Micro-benchmarks of the flowhip HIP kernels at the flagship bench shapes
(448x1024, basic model: P=7168, D=256, r=4). Prints one JSON line per probe.

Run on the GPU box:  python tools/bench_kernels.py [--batch 3]
"""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=3)
    ap.add_argument("--h8", type=int, default=56)
    ap.add_argument("--w8", type=int, default=128)
    ap.add_argument("--dim", type=int, default=256)
    args = ap.parse_args()

    import flowhip._C as C
    from flowhip.ops import torch_ref

    dev = torch.device("cuda:0")
    B, D = args.batch, args.dim
    H, W = args.h8, args.w8
    P = H * W

    # --- corr GEMM (kernel #1): (B,P,D) @ (B,P,D)^T ---
    a = torch.randn(B, P, D, device=dev).to(torch.bfloat16)
    b = torch.randn(B, P, D, device=dev).to(torch.bfloat16)
    t = timeit(lambda: C.bgemm_nt(a, b, 1.0))
    flops = 2.0 * B * P * P * D
    print(json.dumps({"probe": "bgemm_nt_fwd", "B": B, "P": P, "D": D,
                      "ms": t * 1e3, "tflops": flops / t / 1e12}))

    # torch.matmul comparison (hipBLASLt) on the same operands
    af, bf = a.float(), b.float()
    t = timeit(lambda: torch.matmul(af, bf.transpose(1, 2)))
    print(json.dumps({"probe": "torch_matmul_fp32", "ms": t * 1e3,
                      "tflops": flops / t / 1e12}))
    t = timeit(lambda: torch.matmul(a, b.transpose(1, 2)))
    print(json.dumps({"probe": "torch_matmul_bf16", "ms": t * 1e3,
                      "tflops": flops / t / 1e12}))

    # backward-shaped GEMM: (B,P,P) @ (B,D,P)^T, K=P
    dc = torch.randn(B, P, P, device=dev).to(torch.bfloat16)
    f2 = torch.randn(B, D, P, device=dev).to(torch.bfloat16)
    t = timeit(lambda: C.bgemm_nt(dc, f2, 1.0))
    flops = 2.0 * B * P * D * P
    print(json.dumps({"probe": "bgemm_nt_bwd", "ms": t * 1e3,
                      "tflops": flops / t / 1e12}))
    del dc

    # --- pyramid build (torch avg_pool chain, kernel #2 site) ---
    corr = torch.randn(B * P, 1, H, W, device=dev)
    t = timeit(lambda: torch_ref.corr_pyramid(corr, 4))
    bytes_ = corr.numel() * 4 * (1 + 0.25 + 0.25 * 0.25 * 2)
    print(json.dumps({"probe": "corr_pyramid_torch", "ms": t * 1e3,
                      "gbps": bytes_ / t / 1e9}))

    # --- lookup (kernel #3) ---
    pyramid = [p.contiguous() for p in torch_ref.corr_pyramid(corr, 4)]
    coords = (torch.rand(B, 2, H, W, device=dev) *
              torch.tensor([W, H], device=dev).view(1, 2, 1, 1)).contiguous()
    t = timeit(lambda: C.corr_lookup_fwd(pyramid, coords, 4, False))
    out_bytes = B * 4 * 81 * P * 4
    print(json.dumps({"probe": "corr_lookup_fwd", "ms": t * 1e3,
                      "out_gbps": out_bytes / t / 1e9}))

    tr = timeit(lambda: torch_ref.corr_lookup(pyramid, coords, 4))
    print(json.dumps({"probe": "corr_lookup_torch_ref", "ms": tr * 1e3,
                      "speedup_vs_ref": tr / t}))

    g = torch.randn(B, 4 * 81, H, W, device=dev)
    shapes = [list(p.shape) for p in pyramid]
    t = timeit(lambda: C.corr_lookup_bwd(g, coords, 4, shapes, False))
    print(json.dumps({"probe": "corr_lookup_bwd", "ms": t * 1e3}))
    del corr, pyramid, g

    # --- nconv family (kernels #6) at the NCUP bench shapes:
    # channels_to_batch full-res grids, N = 2*batch, H,W = 8*h8, 8*w8 ---
    Hf, Wf = 8 * H, 8 * W
    Nf = 2 * B
    for (ci, co, k) in [(1, 2, 5), (2, 2, 5), (4, 2, 3), (2, 1, 1)]:
        data = torch.randn(Nf, ci, Hf, Wf, device=dev)
        conf = torch.rand(Nf, ci, Hf, Wf, device=dev)
        wt = torch.rand(co, ci, k, k, device=dev) + 0.05
        out, cout = C.nconv_fwd(data, conf, wt, None)
        ref_o, ref_c = torch_ref.nconv2d(data, conf, wt, padding=k // 2,
                                         prop_conf=True)
        err_o = (out - ref_o).abs().max().item()
        err_c = (cout - ref_c).abs().max().item()
        t = timeit(lambda: C.nconv_fwd(data, conf, wt, None))
        gn = torch.randn_like(out)
        gd = torch.randn_like(out)
        tb = timeit(lambda: C.nconv_bwd(gn, gd, data, conf, wt))
        print(json.dumps({"probe": f"nconv_k{k}_ci{ci}_co{co}",
                          "shape": [Nf, ci, Hf, Wf], "fwd_ms": t * 1e3,
                          "bwd_ms": tb * 1e3, "fwd_err": err_o,
                          "cout_err": err_c}))


if __name__ == "__main__":
    main()
