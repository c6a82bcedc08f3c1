"""Attribute eager-mode overhead (copies, casts, grad-accumulation add_,
zero fills) to source lines: one flagship training step under
torch.profiler with python stacks, grouped by op + call site.

Run on the GPU box:  python tools/attr_profile.py [--steps 2]
Writes gpurun_out/attr_profile.txt
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--iters", type=int, default=12)
    ap.add_argument("--size", type=int, nargs=2, default=[448, 1024])
    ap.add_argument("--batch", type=int, default=3)
    ap.add_argument("--out", default="gpurun_out/attr_profile.txt")
    args = ap.parse_args()

    from flowhip import ops
    from flowhip.config.args import default_ncup_args
    from flowhip.models import build_model
    from flowhip.utils.layout import apply_channels_last, to_model_layout

    torch.manual_seed(1234)
    margs = default_ncup_args(model="raft_nc_dbl", mixed_precision=True,
                              dataset="sintel")
    model = build_model(margs).cuda()
    model.train()
    model.freeze_bn()
    apply_channels_last(model)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4, weight_decay=1e-5)

    h, w = args.size
    img1 = torch.rand(args.batch, 3, h, w, device="cuda") * 255
    img2 = torch.rand(args.batch, 3, h, w, device="cuda") * 255
    flow = torch.randn(args.batch, 2, h, w, device="cuda")
    valid = torch.ones(args.batch, h, w, device="cuda")

    def step():
        opt.zero_grad(set_to_none=True)
        preds = model(to_model_layout(img1), to_model_layout(img2),
                      iters=args.iters)
        loss, _ = ops.sequence_loss(preds, flow, valid, 0.85)
        loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=True) as prof:
        for _ in range(args.steps):
            step()
        torch.cuda.synchronize()

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    keys = prof.key_averages(group_by_stack_n=6)
    interesting = ("copy_", "to", "contiguous", "add_", "zero_", "fill_",
                   "zeros", "cat", "clone", "sum", "norm", "mul_", "div_",
                   "batch_norm")
    rows = []
    for e in keys:
        cuda_us = float(e.self_device_time_total)
        if cuda_us <= 0:
            continue
        base = e.key.split("::")[-1]
        if not any(t in base for t in interesting):
            continue
        stack = [f for f in (e.stack or [])
                 if "flowhip" in f or "bench" in f or "torch/autograd" in f
                 or "raft" in f]
        rows.append((cuda_us, e.key, int(e.count), stack[:4]))
    rows.sort(reverse=True)

    with open(args.out, "w") as f:
        tot = sum(r[0] for r in rows)
        f.write(f"eager-overhead ops total: {tot/1e3/args.steps:.2f} ms/step "
                f"over {args.steps} steps\n\n")
        for us, key, cnt, stack in rows[:60]:
            f.write(f"{us/1e3/args.steps:9.3f} ms/step x{cnt//args.steps:5d} "
                    f" {key}\n")
            for s in stack:
                f.write(f"            {s.strip()[:150]}\n")
        f.write("\n\n==== full table (top 50 by self CUDA) ====\n")
        for e in sorted(keys, key=lambda e: -e.self_device_time_total)[:50]:
            f.write(f"{e.self_device_time_total/1e3/args.steps:9.3f} ms/step "
                    f"x{e.count//args.steps:5d}  {e.key[:120]}\n")

    # identify any convs still on the torch/MIOpen path by input shape
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof2:
        step()
        torch.cuda.synchronize()
    with open(args.out, "a") as f:
        f.write("\n\n==== library conv calls by input shape ====\n")
        for e in prof2.key_averages(group_by_input_shape=True):
            if ("miopen_convolution" in e.key or "convolution_backward"
                    in e.key or "batch_norm" in e.key):
                f.write(f"{e.device_time_total/1e3:9.3f} ms x{e.count:4d}  "
                        f"{e.key[:60]}  shapes={e.input_shapes}\n")
    print("wrote", args.out)


if __name__ == "__main__":
    main()
