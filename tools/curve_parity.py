"""Short-horizon training-curve parity: the HIP kernel path vs the torch
reference path (FLOWHIP_FORCE_REF=1), seed-locked, same synthetic data
stream — converts per-op oracle correctness into end-to-end TRAINING
correctness (reference loop semantics train.py:201-255).

Run on the GPU box:
    python tools/curve_parity.py --steps 300
Writes gpurun_out/curve_parity.json with both loss/EPE trajectories and
the agreement stats.
"""

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_curve(force_ref, steps, h, w, batch, iters, lr):
    from flowhip import ops
    from flowhip.config.args import default_ncup_args
    from flowhip.engine.train import fetch_optimizer
    from flowhip.models import build_model
    from flowhip.utils import layout
    from flowhip.utils.layout import apply_channels_last, to_model_layout

    os.environ["FLOWHIP_FORCE_REF"] = "1" if force_ref else "0"
    layout.set_corr_bf16(not force_ref and layout.corr_bf16_enabled())

    args = default_ncup_args(model="raft_nc_dbl", mixed_precision=True,
                             dataset="sintel", iters=iters)
    args.optimizer = "adamw"
    args.scheduler = "cyclic"
    args.lr = lr
    args.num_steps = steps
    args.wdecay = 1e-5
    args.epsilon = 1e-8
    args.gamma = 0.85

    torch.manual_seed(1234)
    model = build_model(args).cuda()
    model.train()
    model.freeze_bn()
    if not force_ref:
        apply_channels_last(model)
    optimizer, scheduler = fetch_optimizer(args, model)

    losses, epes = [], []
    for step in range(steps):
        g = torch.Generator().manual_seed(77 + step)
        base = torch.rand(batch, 3, h, w, generator=g) * 255
        img1 = base.cuda()
        img2 = (base + torch.randn(batch, 3, h, w, generator=g) * 8) \
            .clamp(0, 255).cuda()
        coarse = (torch.rand(batch, 2, h // 32 + 1, w // 32 + 1,
                             generator=g) * 2 - 1) * 8
        flow_gt = torch.nn.functional.interpolate(
            coarse, size=(h, w), mode="bilinear", align_corners=False).cuda()
        valid = torch.ones(batch, h, w, device="cuda")

        optimizer.zero_grad(set_to_none=True)
        i1 = to_model_layout(img1) if not force_ref else img1
        i2 = to_model_layout(img2) if not force_ref else img2
        preds = model(i1, i2, iters=iters)
        loss, metrics = ops.sequence_loss(preds, flow_gt, valid, args.gamma)
        loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        optimizer.step()
        scheduler.step()
        losses.append(float(loss))
        epes.append(float(metrics["epe"]))
    torch.cuda.synchronize()
    return losses, epes


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--size", type=int, nargs=2, default=[256, 512])
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--iters", type=int, default=12)
    ap.add_argument("--lr", type=float, default=1.25e-4)
    ap.add_argument("--out", default="gpurun_out/curve_parity.json")
    args = ap.parse_args()

    h, w = args.size
    hip_l, hip_e = run_curve(False, args.steps, h, w, args.batch, args.iters,
                             args.lr)
    ref_l, ref_e = run_curve(True, args.steps, h, w, args.batch, args.iters,
                             args.lr)

    import numpy as np
    hl, rl = np.array(hip_l), np.array(ref_l)
    he, re_ = np.array(hip_e), np.array(ref_e)
    tail = args.steps // 3

    def window_mean(a, k=20):
        return np.convolve(a, np.ones(k) / k, mode="valid")

    wl_h, wl_r = window_mean(hl), window_mean(rl)
    rel_loss = np.abs(wl_h - wl_r) / np.maximum(np.abs(wl_r), 1e-6)
    rec = {
        "steps": args.steps,
        "size": [h, w],
        "hip_loss_first10_mean": float(hl[:10].mean()),
        "ref_loss_first10_mean": float(rl[:10].mean()),
        "hip_loss_tail_mean": float(hl[-tail:].mean()),
        "ref_loss_tail_mean": float(rl[-tail:].mean()),
        "hip_epe_tail_mean": float(he[-tail:].mean()),
        "ref_epe_tail_mean": float(re_[-tail:].mean()),
        "max_rel_loss_window20": float(rel_loss.max()),
        "mean_rel_loss_window20": float(rel_loss.mean()),
        "hip_decreased": bool(hl[-tail:].mean() < hl[:10].mean()),
        "ref_decreased": bool(rl[-tail:].mean() < rl[:10].mean()),
        "hip_loss": [round(float(x), 5) for x in hip_l],
        "ref_loss": [round(float(x), 5) for x in ref_l],
        "hip_epe": [round(float(x), 5) for x in hip_e],
        "ref_epe": [round(float(x), 5) for x in ref_e],
    }
    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(rec, f, indent=1)
    print(json.dumps({k: v for k, v in rec.items()
                      if not isinstance(v, list)}, indent=1))


if __name__ == "__main__":
    main()
