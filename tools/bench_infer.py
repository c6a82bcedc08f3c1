#!/usr/bin/env python3
"""Inference benchmark — BASELINE config 4: RAFT-NCUP, KITTI shape 288x960,
24 refinement iterations, hipGraph-captured, 1 GPU. Prints eager vs graphed
pairs/sec as JSON lines.

    python tools/bench_infer.py [--iters 24] [--height 288] [--width 960]
"""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")

from flowhip.config.args import default_ncup_args
from flowhip.engine.graph import GraphedInference
from flowhip.models import build_model


def timeit(fn, warmup=3, iters=20):
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
    ap.add_argument("--iters", type=int, default=24)
    ap.add_argument("--height", type=int, default=288)
    ap.add_argument("--width", type=int, default=960)
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--model", default="raft_nc_dbl")
    args = ap.parse_args()

    dev = torch.device("cuda:0")
    margs = default_ncup_args(model=args.model, mixed_precision=True)
    torch.manual_seed(1234)
    model = build_model(margs).to(dev).eval()

    shape = (args.batch, 3, args.height, args.width)
    img1 = torch.rand(shape, device=dev) * 255
    img2 = torch.rand(shape, device=dev) * 255

    with torch.no_grad():
        t_eager = timeit(lambda: model(img1, img2, iters=args.iters,
                                       test_mode=True))
    print(json.dumps({"probe": "infer_eager", "ms": t_eager * 1e3,
                      "pairs_per_sec": args.batch / t_eager,
                      "iters": args.iters, "hw": [args.height, args.width]}))

    g = GraphedInference(model, shape, args.iters)
    low_e, up_e = model(img1, img2, iters=args.iters, test_mode=True)
    low_g, up_g = g(img1, img2)
    err = (up_g - up_e).abs().max().item()
    t_graph = timeit(lambda: g(img1, img2))
    print(json.dumps({"probe": "infer_hipgraph", "ms": t_graph * 1e3,
                      "pairs_per_sec": args.batch / t_graph,
                      "speedup_vs_eager": t_eager / t_graph,
                      "max_abs_diff_vs_eager": err}))


if __name__ == "__main__":
    main()
