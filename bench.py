#!/usr/bin/env python3
"""Benchmark harness — the driver contract.

Measures the flagship training step: RAFT-NCUP (raft_nc_dbl, basic) on
Sintel-shape 448x1024 synthetic pairs, 12 refinement iterations, bf16
autocast, full forward + sequence loss + backward + AdamW step per step.

    python bench.py --gpus N --steps K --warmup W
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: per-GPU batch is fixed (--batch-per-gpu); the reported value
is WHOLE-JOB image-pairs/sec (max step time over ranks). Rank 0 prints one
JSON line. Without a GPU the plumbing config (RAFT-small NCUP, 2 iters,
128x128 — BASELINE config 1) runs on CPU so the harness stays testable here.
"""

import argparse
import json
import os
import time

import numpy as np
import torch

import os as _os

if torch.cuda.is_available() and _os.environ.get("FLOWHIP_MIOPEN_FIND", "0") == "1":
    # MIOpen benchmark/find mode, off by default: the A/B measured no
    # steady-state difference (bench10: 30.80 vs 30.74 pairs/s), and with
    # one process per GPU the concurrent find's user-db file locking only
    # adds warmup wall time and trace noise
    torch.backends.cudnn.benchmark = True

from flowhip.config.args import default_ncup_args
from flowhip.engine import distributed
from flowhip.engine.train import fetch_optimizer
from flowhip.models import build_model
from flowhip import ops


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch-per-gpu", type=int, default=3,
                   help="per-GPU batch (reference schedule: batch 6 on 2 GPUs)")
    p.add_argument("--height", type=int, default=448)
    p.add_argument("--width", type=int, default=1024)
    p.add_argument("--iters", type=int, default=12)
    p.add_argument("--model", default="raft_nc_dbl")
    p.add_argument("--profile-steps", type=int, default=0,
                   help="if >0, run only this many timed steps per rank "
                   "without JSON (for rocprofv3 kernel capture)")
    p.add_argument("--torch-profile", default=None,
                   help="write torch.profiler kernel table for steady-state "
                   "steps to this path (after warmup; excludes MIOpen tuning)")
    return p.parse_args()


def main():
    cli = parse_args()
    rank, world_size, device = distributed.init_distributed()
    on_gpu = device.type == "cuda"

    if on_gpu:
        h, w, iters, small = cli.height, cli.width, cli.iters, False
        batch = cli.batch_per_gpu
        dtype_name = "bf16"
        model_name = cli.model
    else:  # BASELINE config 1: CPU plumbing
        h, w, iters, small = 128, 128, 2, True
        batch = 1
        dtype_name = "fp32"
        model_name = cli.model

    args = default_ncup_args(model=model_name, small=small,
                             mixed_precision=on_gpu, iters=iters,
                             dataset="sintel")
    args.name = "bench"
    args.optimizer = "adamw"
    args.scheduler = "cyclic"
    args.lr = 1.25e-4
    args.num_steps = max(cli.steps + cli.warmup, 1)
    args.wdecay = 1e-5
    args.epsilon = 1e-8
    args.stage = "sintel"  # freeze_bn as in fine-tuning stages

    torch.manual_seed(1234 + rank)
    np.random.seed(1234 + rank)

    model = build_model(args).to(device)
    model.train()
    model.freeze_bn()

    optimizer, scheduler = fetch_optimizer(args, model)
    ddp_model = distributed.wrap_ddp(model, device)

    # synthetic data of the benchmark shape, staged on device
    g = torch.Generator().manual_seed(4321 + rank)
    image1 = (torch.rand(batch, 3, h, w, generator=g) * 255).to(device)
    image2 = (torch.rand(batch, 3, h, w, generator=g) * 255).to(device)
    coarse = (torch.rand(batch, 2, h // 32 + 1, w // 32 + 1, generator=g) * 2 - 1) * 16
    flow_gt = torch.nn.functional.interpolate(
        coarse, size=(h, w), mode="bilinear", align_corners=False).to(device)
    valid = torch.ones(batch, h, w, device=device)

    def step():
        optimizer.zero_grad(set_to_none=True)
        preds = ddp_model(image1, image2, iters=iters)
        loss, metrics = ops.sequence_loss(preds, flow_gt, valid, args.gamma)
        loss.backward()
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        optimizer.step()
        scheduler.step()
        return metrics

    for _ in range(cli.warmup):
        step()

    if cli.torch_profile:
        from torch.profiler import ProfilerActivity, profile
        if on_gpu:
            torch.cuda.synchronize()
        with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                     record_shapes=True) as prof:
            for _ in range(max(cli.steps, 2)):
                step()
            if on_gpu:
                torch.cuda.synchronize()
        sort = "self_cuda_time_total" if on_gpu else "self_cpu_time_total"
        table = prof.key_averages().table(sort_by=sort, row_limit=60)
        shapes = prof.key_averages(group_by_input_shape=True).table(
            sort_by=sort, row_limit=80)
        with open(cli.torch_profile, "w") as f:
            f.write(table)
            f.write("\n\n==== grouped by input shape ====\n\n")
            f.write(shapes)
        if rank == 0:
            print(table[:4000])
        return

    if cli.profile_steps > 0:
        if on_gpu:
            torch.cuda.synchronize()
        for _ in range(cli.profile_steps):
            step()
        if on_gpu:
            torch.cuda.synchronize()
        return

    distributed.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(cli.steps):
        step()
    distributed.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks = whole-job wall time
    if world_size > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if on_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    global_batch = batch * world_size
    ms_per_step = elapsed / cli.steps * 1000.0
    pairs_per_sec = global_batch * cli.steps / elapsed
    peak_gb = (torch.cuda.max_memory_allocated() / 2**30) if on_gpu else None

    if rank == 0:
        print(json.dumps({
            "metric": "train_image_pairs_per_sec",
            "value": pairs_per_sec,
            "unit": "pairs/s",
            "n_gpus": world_size if on_gpu else 0,
            "steps": cli.steps,
            "warmup": cli.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype_name,
            "data": "synthetic",
            "config": {
                "model": model_name + ("-small" if small else ""),
                "global_batch": global_batch,
                "image_size": [h, w],
                "refinement_iters": iters,
                "parallelism": f"dp{world_size}" if on_gpu else "cpu",
                "peak_mem_gb": round(peak_gb, 2) if peak_gb else None,
            },
        }))

    distributed.cleanup()


if __name__ == "__main__":
    main()
