"""RCCL-on-hardware tests (SURVEY §5.8 / VERDICT r01 missing #1).

The builder's lease is one GPU, so the multi-rank path is exercised as two
torchrun ranks SHARING cuda:0 over the real nccl(=RCCL) backend: process
group init, a bare all_reduce, and one DDP flat-bucket training step whose
all-reduced gradients must equal the single-process batched step's.
"""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _single_process_grads():
    from flowhip import ops
    from flowhip.config.args import default_ncup_args
    from flowhip.models import build_model
    from flowhip.utils.layout import apply_channels_last, to_model_layout

    torch.manual_seed(11)
    margs = default_ncup_args(model="raft_nc_dbl", mixed_precision=True,
                              dataset="sintel")
    model = build_model(margs).cuda()
    model.freeze_bn()
    apply_channels_last(model)
    g = torch.Generator().manual_seed(7)
    b, h, w = 2, 128, 128
    img1 = (torch.rand(b, 3, h, w, generator=g) * 255).cuda()
    img2 = (torch.rand(b, 3, h, w, generator=g) * 255).cuda()
    flow = torch.randn(b, 2, h, w, generator=g).cuda()
    valid = torch.ones(b, h, w).cuda()
    preds = model(to_model_layout(img1), to_model_layout(img2), iters=2)
    loss, _ = ops.sequence_loss(preds, flow, valid, 0.85)
    loss.backward()
    return {n: p.grad.float().cpu() for n, p in model.named_parameters()
            if p.grad is not None}


@pytest.mark.timeout(900)
def test_two_ranks_one_gpu_ddp(tmp_path):
    """Two torchrun ranks sharing cuda:0 over nccl(=RCCL). RCCL
    categorically refuses two ranks on one device — verified on this stack:
    ncclInvalidUsage "Duplicate GPU detected" (see profiles/README.md) — so
    on a 1-GPU box this documents the limitation as a skip; on a multi-GPU
    node (driver SCALE run environment) the same launch would bind distinct
    devices and must pass. Single-rank RCCL collectives are covered by
    test_nccl_world1_process_group; DDP all-reduce semantics by the gloo
    world-size-2 CPU tests."""
    env = dict(os.environ, HSA_ENABLE_IPC_MODE_LEGACY="0")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29741",
           os.path.join(REPO, "tools", "rccl_smoke.py"),
           "--out", str(tmp_path), "--profile"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=840,
                       cwd=REPO, env=env)

    skip_marker = os.path.join(str(tmp_path), "skip.json")
    if os.path.exists(skip_marker):
        with open(skip_marker) as f:
            why = json.load(f)["error"]
        # known shared-device refusal signatures (RCCL wording varies by
        # build: "Duplicate GPU detected" / ncclInvalidUsage); anything
        # else is a real failure and must fail, not skip
        assert ("Duplicate GPU" in why or "invalid usage" in why.lower()
                or "ncclInvalidUsage" in why), why
        pytest.skip(f"RCCL refuses 2 ranks on one GPU on this stack: {why}")

    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    with open(os.path.join(str(tmp_path), "rccl_smoke.json")) as f:
        rec = json.load(f)
    assert rec["rccl_allreduce_ok"] and rec["world_size"] == 2
    assert rec["nccl_backend"] == "nccl"

    # DDP-averaged grads == single-process batched grads (the gloo/CPU
    # equivalence test, now over real RCCL kernels)
    ddp_grads = torch.load(os.path.join(str(tmp_path), "rccl_grads.pth"),
                           weights_only=True)
    ref = _single_process_grads()
    assert set(ddp_grads) == set(ref)
    bad = [n for n in ref
           if not torch.allclose(ddp_grads[n], ref[n], atol=2e-2, rtol=2e-2)]
    assert not bad, bad[:8]


@pytest.mark.timeout(600)
def test_nccl_world1_process_group():
    """Single-rank nccl init + collective through engine.distributed —
    the world_size=1 path bench.py takes on the driver's 1-GPU run."""
    import torch.distributed as dist
    if dist.is_initialized():
        pytest.skip("process group already active")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29742")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        t = torch.ones(8, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert t.allclose(torch.ones_like(t))
    finally:
        dist.destroy_process_group()
        os.environ.pop("RANK", None)
        os.environ.pop("WORLD_SIZE", None)
