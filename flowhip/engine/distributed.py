"""Distributed runtime: one process per GPU over RCCL/xGMI.

Replaces the reference's single-process nn.DataParallel (train.py:169-175 —
SURVEY.md §2.3) with torch.distributed DDP. On ROCm the "nccl" backend IS
RCCL; gradient all-reduce rides the 7 point-to-point xGMI links per GPU.

Bucket policy: RAFT-NCUP has ~5.3M params (~21 MB fp32
grads) — latency-dominated, so DDP is configured with ONE large bucket
(bucket_cap_mb=64 > total grad bytes) and gradient_as_bucket_view to skip
the copy; the single all-reduce still overlaps with the long multi-iteration
backward since it fires once the last-used parameters are ready.
"""

import datetime
import os

import torch
import torch.distributed as dist


def env_rank():
    return int(os.environ.get("RANK", "0"))


def env_world_size():
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank():
    return int(os.environ.get("LOCAL_RANK", "0"))


def init_distributed(backend=None, timeout_s=600):
    """Initialize the process group from torchrun env vars.

    Returns (rank, world_size, device). Single-process (WORLD_SIZE absent or
    1) needs no process group. Backend defaults to nccl (=RCCL) on GPU,
    gloo on CPU.
    """
    world_size = env_world_size()
    use_cuda = torch.cuda.is_available()

    if use_cuda:
        device = torch.device("cuda", env_local_rank() % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_cuda else "gloo"
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_s))
    return env_rank(), world_size, device


def is_initialized():
    return dist.is_available() and dist.is_initialized()


def is_main():
    return (not is_initialized()) or dist.get_rank() == 0


def barrier():
    if is_initialized():
        dist.barrier()


def cleanup():
    if is_initialized():
        dist.destroy_process_group()


def wrap_ddp(model, device, static_iters=True):
    """Wrap for data-parallel training (one bucket, see module docstring).

    CONTRACT: static_graph=True assumes the unrolled autograd
    graph is IDENTICAL every step — fixed --iters, no activation
    checkpointing, no conditionally-skipped branches. The training driver
    satisfies this (iters is constant for a run). Pass static_iters=False
    if a variable-schedule caller is ever added, or DDP will fail loudly
    mid-training.
    """
    if not is_initialized():
        return model
    # ~21 MB of fp32 grads (5.3M params): one flat bucket — a ring stage is
    # bound by one xGMI link either way, and a single all-reduce minimizes
    # latency. static_graph: the unrolled iteration graph is identical every
    # step (fixed --iters), letting DDP skip graph re-discovery.
    kwargs = dict(bucket_cap_mb=64, gradient_as_bucket_view=True,
                  static_graph=static_iters)
    if device.type == "cuda":
        kwargs["device_ids"] = [device.index]
    return torch.nn.parallel.DistributedDataParallel(model, **kwargs)


def allreduce_mean_scalar(value, device):
    """Mean of a python float across ranks (for logging)."""
    if not is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.AVG)
    return t.item()
