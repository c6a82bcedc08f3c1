"""Distributed-correctness tests on gloo (CPU, world_size=2):
DDP gradient all-reduce equivalence — a world_size=2 step with per-rank
batch 1 must produce the same gradients as a single-process batch-2 step
(SURVEY.md §4.2 item 4)."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from flowhip.config.args import default_ncup_args
from flowhip.models import build_model
from flowhip import ops


def _make_data(seed=7, b=2, h=128, w=128):
    g = torch.Generator().manual_seed(seed)
    img1 = torch.rand(b, 3, h, w, generator=g) * 255
    img2 = torch.rand(b, 3, h, w, generator=g) * 255
    flow = torch.randn(b, 2, h, w, generator=g)
    valid = torch.ones(b, h, w)
    return img1, img2, flow, valid


def _build_model(seed=11):
    torch.manual_seed(seed)
    args = default_ncup_args(model="raft_nc_dbl", small=False)
    model = build_model(args)
    # freeze BN (as in all fine-tuning stages): per-batch BN stats would
    # differ between batch-2 single-process and per-rank batch-1 otherwise.
    model.freeze_bn()
    return model


def _single_process_grads():
    model = _build_model()
    img1, img2, flow, valid = _make_data()
    preds = model(img1, img2, iters=2)
    loss, _ = ops.sequence_loss(preds, flow, valid, 0.85)
    loss.backward()
    return {n: p.grad.clone() for n, p in model.named_parameters()
            if p.grad is not None}


def _ddp_worker(rank, world_size, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        model = _build_model()
        ddp = torch.nn.parallel.DistributedDataParallel(
            model, bucket_cap_mb=64, gradient_as_bucket_view=True)

        img1, img2, flow, valid = _make_data()
        sl = slice(rank, rank + 1)  # shard the batch across ranks
        preds = ddp(img1[sl], img2[sl], iters=2)
        loss, _ = ops.sequence_loss(preds, flow[sl], valid[sl], 0.85)
        loss.backward()

        if rank == 0:
            grads = {n: p.grad.clone() for n, p in model.named_parameters()
                     if p.grad is not None}
            torch.save(grads, os.path.join(out_dir, "ddp_grads.pth"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_allreduce_equals_batched(tmp_path):
    port = int(np.random.default_rng(os.getpid()).integers(20000, 40000))
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    ddp_grads = torch.load(os.path.join(tmp_path, "ddp_grads.pth"),
                           weights_only=True)
    ref_grads = _single_process_grads()

    assert set(ddp_grads.keys()) == set(ref_grads.keys())
    for name in ref_grads:
        # DDP averages over ranks; the single-process loss already averages
        # over the batch -> equal up to numeric noise.
        assert torch.allclose(ddp_grads[name], ref_grads[name],
                              atol=1e-5, rtol=1e-4), name


def test_distributed_helpers_single_process():
    from flowhip.engine import distributed
    rank, world, device = distributed.init_distributed()
    assert rank == 0 and world == 1
    assert distributed.is_main()
    distributed.barrier()  # no-op


@pytest.mark.timeout(900)
def test_bench_contract_world_size_2(tmp_path):
    """Launch bench.py exactly the way the driver does (torchrun, nnodes=1,
    2 ranks — gloo on CPU) and validate the one-JSON-line contract: whole-job
    value, max-over-ranks timing, n_gpus/steps/warmup/config fields."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29713", os.path.join(repo, "bench.py"),
           "--gpus", "2", "--steps", "2", "--warmup", "1"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=840,
                         cwd=repo).stdout
    lines = [l for l in out.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {out!r}"
    rec = json.loads(lines[0])
    assert rec["metric"] == "train_image_pairs_per_sec"
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["higher_is_better"] is True and rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    # CPU plumbing config: n_gpus 0, world-size-2 global batch
    assert rec["n_gpus"] == 0
    assert rec["config"]["global_batch"] == 2


@pytest.mark.timeout(900)
def test_bench_default_invocation(tmp_path):
    """`python bench.py` with no flags (the driver's fallback invocation)
    emits exactly one valid JSON line and finishes quickly on CPU."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"),
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=840, cwd=repo).stdout
    lines = [l for l in out.splitlines() if l.startswith("{")]
    assert len(lines) == 1
    rec = json.loads(lines[0])
    assert rec["metric"] == "train_image_pairs_per_sec"
    assert rec["value"] > 0
    assert rec["config"]["global_batch"] == 1


@pytest.mark.timeout(900)
def test_train_entry_world_size_2(tmp_path):
    """The full train.py CLI under torchrun with 2 CPU ranks (gloo):
    global --batch_size 2 shards to 1/rank, rank 0 writes the checkpoint
    and log, both ranks exit 0 — the same launch shape the driver uses for
    the round-end SCALE pass (with nccl/RCCL on GPUs)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29719", os.path.join(repo, "train.py"),
           "--name", "ddp2", "--model", "raft_nc_dbl", "--stage", "synthetic",
           "--small", "--num_steps", "2", "--batch_size", "2",
           "--image_size", "64", "64", "--iters", "2", "--lr", "1e-4",
           "--num_workers", "0"]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=840,
                         cwd=str(tmp_path))
    assert res.returncode == 0, res.stderr[-2000:]
    ckpt = tmp_path / "checkpoints" / "ddp2" / "final_model.pth"
    assert ckpt.exists()
    log = (tmp_path / "checkpoints" / "ddp2" / "log.txt").read_text()
    assert "Parameter Count" in log
