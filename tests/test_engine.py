"""Engine tests (CPU): full train() smoke on the synthetic stage, synthetic
validation, logger file outputs."""

import json
import os

import pytest
import torch

from flowhip.config import build_train_parser, finalize_args
from flowhip.config.args import default_ncup_args
from flowhip.engine.evaluate import validate_synthetic
from flowhip.models import build_model


@pytest.mark.timeout(600)
def test_train_synthetic_two_steps(tmp_path, monkeypatch):
    """train() runs end-to-end for 2 steps on CPU (BASELINE config 1
    plumbing: small model, 2 iters, 128x128 synthetic pairs) and writes the
    reference checkpoint layout."""
    monkeypatch.chdir(tmp_path)

    argv = ["--name", "smoke", "--model", "raft_nc_dbl", "--stage", "synthetic",
            "--small", "--num_steps", "2", "--batch_size", "1",
            "--image_size", "128", "128", "--iters", "2", "--lr", "1e-4",
            "--num_workers", "0"]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))

    from flowhip.engine.train import train
    path = train(args)
    assert path == "checkpoints/smoke/final_model.pth"
    assert os.path.exists(path)

    sd = torch.load(path, weights_only=True)
    assert all(k.startswith("module.") for k in sd)

    log = open("checkpoints/smoke/log.txt").read()
    assert "Parameter Count" in log


def test_validate_synthetic_runs():
    args = default_ncup_args(small=True)
    model = build_model(args)
    res = validate_synthetic(model, iters=2, n_samples=1, image_size=(128, 128))
    assert "synthetic" in res and res["synthetic"] > 0


def test_logger_files(tmp_path):
    import argparse
    from flowhip.engine.logger import Logger

    args = argparse.Namespace(name="logtest")
    logger = Logger(None, args, sum_freq=2, run_dir=str(tmp_path / "run"))
    logger.scheduler = type("S", (), {"get_last_lr": lambda self: [1e-4]})()
    for i in range(4):
        logger.push({"loss": float(i)}, n_imgs=2)
    logger.write_dict({"synthetic": 1.5})
    logger.close()

    jsonl = [json.loads(l) for l in open(tmp_path / "run" / "metrics.jsonl")]
    assert any("loss" in r for r in jsonl)
    assert any("validation" in r for r in jsonl)
    assert (tmp_path / "run" / "log.txt").exists()


def test_sintel_submission_writer(tmp_path, monkeypatch):
    """End-to-end submission path on CPU: fake test split -> .flo files,
    warm-started second frame (reference evaluate.py:23-57)."""
    import numpy as np
    from PIL import Image

    from flowhip.config.args import default_ncup_args
    from flowhip.data import frame_utils
    from flowhip.engine.evaluate import create_sintel_submission
    from flowhip.models import build_model

    for dstype in ("clean", "final"):
        scene = tmp_path / "datasets" / "Sintel" / "test" / dstype / "seq_1"
        scene.mkdir(parents=True)
        for i in range(3):
            # >=128 px per side: below that the 4-level pyramid has a 1-px
            # level and the reference-parity lookup path produces NaNs
            # (PARITY.md deviation 6)
            arr = (np.random.rand(128, 136, 3) * 255).astype(np.uint8)
            Image.fromarray(arr).save(scene / f"frame_{i:04d}.png")

    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    args = default_ncup_args(model="raft_nc_dbl", small=True)
    model = build_model(args)
    create_sintel_submission(model, iters=2, warm_start=True,
                             output_path=str(tmp_path / "out"))

    flo = tmp_path / "out" / "clean" / "seq_1" / "frame0001.flo"
    assert flo.exists()
    flow = frame_utils.readFlow(str(flo))
    assert flow.shape == (128, 136, 2)
    assert (tmp_path / "out" / "final" / "seq_1" / "frame0002.flo").exists()


def test_kitti_submission_writer(tmp_path, monkeypatch):
    import numpy as np
    from PIL import Image

    from flowhip.config.args import default_ncup_args
    from flowhip.data import frame_utils
    from flowhip.engine.evaluate import create_kitti_submission
    from flowhip.models import build_model

    img_dir = tmp_path / "datasets" / "KITTI" / "testing" / "image_2"
    img_dir.mkdir(parents=True)
    for suffix in ("10", "11"):
        arr = (np.random.rand(128, 136, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(img_dir / f"000000_{suffix}.png")

    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    args = default_ncup_args(model="raft_nc_dbl", small=True)
    model = build_model(args)
    create_kitti_submission(model, iters=2,
                            output_path=str(tmp_path / "kout"))

    out = tmp_path / "kout" / "000000_10.png"
    assert out.exists()
    flow, valid = frame_utils.readFlowKITTI(str(out))
    assert flow.shape == (128, 136, 2)
    assert valid.min() == 1  # submission marks everything valid


def test_demo_cli_end_to_end(tmp_path):
    """demo.py over a directory of frames writes flow visualizations."""
    import subprocess
    import sys

    import numpy as np
    from PIL import Image

    from flowhip.config.args import default_ncup_args
    from flowhip.engine import checkpoints
    from flowhip.models import build_model

    frames = tmp_path / "frames"
    frames.mkdir()
    # >=128 px: the 4-level pyramid needs level 3 >= 2 px (the reference's
    # grid_sample(align_corners=True) divides by size-1 on a 1x1 level)
    for i in range(3):
        arr = (np.random.rand(128, 128, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(frames / f"f{i:03d}.png")

    torch.manual_seed(0)
    args = default_ncup_args(model="raft_nc_dbl", small=True)
    model = build_model(args)
    ckpt = tmp_path / "model.pth"
    checkpoints.save_weights(model, str(ckpt))

    out = tmp_path / "viz"
    r = subprocess.run(
        [sys.executable, "demo.py", "--model", str(ckpt), "--path",
         str(frames), "--out", str(out), "--arch", "raft_nc_dbl",
         "--small"],
        capture_output=True, text=True, timeout=500)
    assert r.returncode == 0, r.stderr[-2000:]
    pngs = list(out.glob("*.png"))
    assert len(pngs) == 2  # consecutive pairs


def test_pmc_summarize_tool(tmp_path):
    """tools/pmc_summarize.py aggregates rocprofv3 counter rows per kernel."""
    import subprocess
    import sys

    csv_path = tmp_path / "counters.csv"
    csv_path.write_text(
        "Kernel_Name,Counter_Name,Counter_Value,Dispatch_Id\n"
        "kA,SQ_WAVE_CYCLES,100,1\n"
        "kA,SQ_WAVE_CYCLES,50,2\n"
        "kA,SQ_WAIT_ANY,30,1\n"
        "kB,SQ_WAVE_CYCLES,10,3\n")
    r = subprocess.run([sys.executable, "tools/pmc_summarize.py",
                        str(csv_path)], capture_output=True, text=True,
                       timeout=120)
    assert r.returncode == 0, r.stderr
    lines = r.stdout.strip().splitlines()
    assert lines[0].startswith("Kernel,Dispatches")
    assert any(row.startswith("kA,2,") and ",150" in row for row in lines)


@pytest.mark.timeout(600)
def test_train_resume_full_state(tmp_path, monkeypatch):
    """Full-state resume through the train() entry point:
    a run checkpoints train_state.pth at the VAL_FREQ cadence; a second run
    with --resume_full continues from the restored step counter instead of
    restarting, and re-checkpoints at a later step."""
    monkeypatch.chdir(tmp_path)

    from flowhip.engine import train as train_mod
    monkeypatch.setattr(train_mod, "VAL_FREQ", 2)

    base = ["--name", "resume", "--model", "raft_nc_dbl", "--stage",
            "synthetic", "--small", "--batch_size", "1",
            "--image_size", "128", "128", "--iters", "2", "--lr", "1e-4",
            "--num_workers", "0"]

    argv = base + ["--num_steps", "3"]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))
    train_mod.train(args)

    state_path = "checkpoints/resume/train_state.pth"
    assert os.path.exists(state_path)
    state = torch.load(state_path, weights_only=False)
    first_ckpt_step = state["total_steps"]
    # fires when total_steps % VAL_FREQ == 1: at steps 1 and 3 of this run,
    # so the surviving file carries total_steps == 3
    assert first_ckpt_step == 3

    argv = base + ["--num_steps", "5", "--resume_full", state_path]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))
    train_mod.train(args)

    state2 = torch.load(state_path, weights_only=False)
    # resumed run continued from step 1 and re-checkpointed later (3 or 5),
    # proving the counter (and opt/sched state with it) was restored
    assert state2["total_steps"] > first_ckpt_step
    assert state2["scheduler"]["last_epoch"] > state["scheduler"]["last_epoch"]


@pytest.mark.timeout(900)
def test_validators_on_fake_datasets(tmp_path, monkeypatch):
    """validate_chairs / validate_sintel / validate_kitti run end-to-end on
    CPU against tiny fake dataset trees and return the reference's metric
    dict keys (reference evaluate.py:91-181)."""
    import numpy as np
    from PIL import Image

    from flowhip.data import frame_utils
    from flowhip.engine import evaluate as ev

    monkeypatch.chdir(tmp_path)
    h, w = 160, 160  # >=128: 4-level pyramid constraint (PARITY.md)

    def put_img(path):
        path.parent.mkdir(parents=True, exist_ok=True)
        arr = (np.random.rand(h, w, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(path)

    # FlyingChairs: validation rows (xid==2) + split table in cwd
    chairs = tmp_path / "datasets" / "FlyingChairs_release" / "data"
    chairs.mkdir(parents=True)
    for i in (1, 2):
        put_img(chairs / f"{i:05d}_img1.png")
        put_img(chairs / f"{i:05d}_img2.png")
        frame_utils.writeFlow(str(chairs / f"{i:05d}_flow.flo"),
                              np.random.randn(h, w, 2).astype(np.float32))
    (tmp_path / "chairs_split.txt").write_text("2\n2\n")

    # Sintel training: clean/final scenes + flow
    for dstype in ("clean", "final"):
        scene = tmp_path / "datasets" / "Sintel" / "training" / dstype / "s1"
        for i in range(3):
            put_img(scene / f"frame_{i:04d}.png")
    fdir = tmp_path / "datasets" / "Sintel" / "training" / "flow" / "s1"
    fdir.mkdir(parents=True)
    for i in range(2):
        frame_utils.writeFlow(str(fdir / f"frame_{i:04d}.flo"),
                              np.random.randn(h, w, 2).astype(np.float32))

    # KITTI training: image pairs + sparse flow_occ 16-bit pngs
    kroot = tmp_path / "datasets" / "KITTI" / "training"
    for i in range(2):
        put_img(kroot / "image_2" / f"{i:06d}_10.png")
        put_img(kroot / "image_2" / f"{i:06d}_11.png")
    (kroot / "flow_occ").mkdir(parents=True)
    for i in range(2):
        frame_utils.writeFlowKITTI(
            str(kroot / "flow_occ" / f"{i:06d}_10.png"),
            np.random.randn(h, w, 2).astype(np.float32) * 4)

    args = default_ncup_args(model="raft_nc_dbl", small=True)
    model = build_model(args)

    res = ev.validate_chairs(model, iters=2)
    assert set(res) == {"chairs"} and np.isfinite(res["chairs"])

    res = ev.validate_sintel(model, iters=2)
    assert set(res) == {"clean", "final"}
    assert all(np.isfinite(v) for v in res.values())

    res = ev.validate_kitti(model, iters=2)
    assert set(res) == {"kitti-epe", "kitti-f1"}
    assert np.isfinite(res["kitti-epe"]) and 0 <= res["kitti-f1"] <= 100


@pytest.mark.timeout(600)
def test_evaluate_cli_synthetic(tmp_path):
    """Top-level evaluate.py CLI end-to-end on the synthetic validator."""
    import subprocess
    import sys

    from flowhip.engine import checkpoints

    torch.manual_seed(0)
    # dataset controls BatchNorm in the weights-est net (BN only for
    # sintel): the saved model must match the CLI's --dataset synthetic
    args = default_ncup_args(model="raft_nc_dbl", small=True,
                             dataset="synthetic")
    model = build_model(args)
    ckpt = tmp_path / "m.pth"
    checkpoints.save_weights(model, str(ckpt))

    r = subprocess.run(
        [sys.executable, "evaluate.py", "--model", "raft_nc_dbl", "--small",
         "--restore_ckpt", str(ckpt), "--dataset", "synthetic",
         "--iters", "2"],
        capture_output=True, text=True, timeout=500)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "synthetic" in r.stdout.lower()


@pytest.mark.timeout(600)
def test_train_determinism_same_seed(tmp_path, monkeypatch):
    """Two identical-seed runs produce byte-identical final weights
    (SURVEY §2.8 RNG surface: torch/np seeding + deterministic synthetic
    loader with num_workers=0)."""
    weights = []
    for run in ("a", "b"):
        rundir = tmp_path / run
        rundir.mkdir()
        monkeypatch.chdir(rundir)
        argv = ["--name", "det", "--model", "raft_nc_dbl", "--stage",
                "synthetic", "--small", "--num_steps", "2", "--batch_size",
                "1", "--image_size", "128", "128", "--iters", "2",
                "--lr", "1e-4", "--num_workers", "0", "--seed", "7"]
        parser = build_train_parser(argv=argv)
        args = finalize_args(parser.parse_args(argv))
        from flowhip.engine.train import train
        path = train(args)
        weights.append(torch.load(path, weights_only=True))

    a, b = weights
    assert a.keys() == b.keys()
    for k in a:
        assert torch.equal(a[k], b[k]), k


def test_graphed_inference_requires_gpu():
    """GraphedInference fails loudly off-GPU (capture is hipGraph-only)."""
    from flowhip.engine.graph import GraphedInference

    args = default_ncup_args(model="raft_nc_dbl", small=True)
    model = build_model(args)
    if not torch.cuda.is_available():
        with pytest.raises(AssertionError):
            GraphedInference(model, (1, 3, 128, 128), iters=2)


def test_logger_accepts_numpy_scalars(tmp_path):
    """Validators return np.float64 means — the jsonl stream must accept
    them (np.float64 subclasses float; a regression to np.float32 would
    break json serialization)."""
    import argparse

    import numpy as np

    from flowhip.engine.logger import Logger

    logger = Logger(None, argparse.Namespace(name="nplog"),
                    run_dir=str(tmp_path / "run"))
    logger.write_dict({"kitti-epe": np.float64(1.25),
                       "kitti-f1": np.float64(7.5)})
    logger.close()
    rec = json.loads(open(tmp_path / "run" / "metrics.jsonl").read())
    assert rec["validation"]["kitti-epe"] == 1.25


def test_train_profile_dir_writes_trace(tmp_path, monkeypatch):
    """--profile_dir captures one torch.profiler cycle inside the train loop
    (SURVEY §5.1): wait 1 / warmup 2 / active 3 over 7 steps, chrome trace +
    op table written, training otherwise unaffected."""
    monkeypatch.chdir(tmp_path)

    argv = ["--name", "prof", "--model", "raft_nc_dbl", "--stage", "synthetic",
            "--small", "--num_steps", "7", "--batch_size", "1",
            "--image_size", "64", "64", "--iters", "2", "--lr", "1e-4",
            "--num_workers", "0", "--profile_dir", str(tmp_path / "prof_out")]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))

    from flowhip.engine.train import train
    path = train(args)
    assert os.path.exists(path)
    trace = tmp_path / "prof_out" / "train_trace.json"
    table = tmp_path / "prof_out" / "train_ops.txt"
    assert trace.exists() and trace.stat().st_size > 0
    assert "Self CPU" in table.read_text()


def test_train_add_noise_branch(tmp_path, monkeypatch):
    """--add_noise applies per-step gaussian noise (stdv ~ U[0,5], clamped to
    [0,255] — reference train.py:206-209) without breaking the step."""
    monkeypatch.chdir(tmp_path)

    argv = ["--name", "noise", "--model", "raft_nc_dbl", "--stage",
            "synthetic", "--small", "--num_steps", "2", "--batch_size", "1",
            "--image_size", "64", "64", "--iters", "2", "--lr", "1e-4",
            "--num_workers", "0", "--add_noise"]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))

    from flowhip.engine.train import train
    path = train(args)
    assert os.path.exists(path)


def test_tools_compile_and_guard_gpu():
    """Every tools/*.py compiles and its module-level code does not require
    a GPU (each tool guards its CUDA work behind main()/run() so CPU boxes
    can parse-check them)."""
    import py_compile

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tools = sorted(f for f in os.listdir(os.path.join(repo, "tools"))
                   if f.endswith(".py"))
    assert len(tools) >= 7
    for f in tools:
        py_compile.compile(os.path.join(repo, "tools", f), doraise=True)


def test_train_raft_baseline_model(tmp_path, monkeypatch):
    """--model raft (convex-upsample mask head, reference train.py flow)
    trains end-to-end — the baseline family, not just the NCUP one."""
    monkeypatch.chdir(tmp_path)

    argv = ["--name", "raftb", "--model", "raft", "--stage", "synthetic",
            "--small", "--num_steps", "2", "--batch_size", "1",
            "--image_size", "128", "128", "--iters", "2", "--lr", "1e-4",
            "--num_workers", "0"]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))

    from flowhip.engine.train import train
    path = train(args)
    assert os.path.exists(path)
    sd = torch.load(path, weights_only=True)
    assert all(k.startswith("module.") for k in sd)
