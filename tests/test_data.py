"""Data-layer tests: synthetic dataset, file formats, augmentor invariants."""

import numpy as np
import torch

from flowhip.data import SyntheticFlowDataset, frame_utils
from flowhip.data.augmentor import FlowAugmentor, SparseFlowAugmentor


def test_synthetic_dataset():
    ds = SyntheticFlowDataset(image_size=(64, 96), length=5)
    img1, img2, flow, valid = ds[0]
    assert img1.shape == (3, 64, 96) and img2.shape == (3, 64, 96)
    assert flow.shape == (2, 64, 96) and valid.shape == (64, 96)
    assert img1.max() <= 255 and img1.min() >= 0
    # deterministic per index
    img1b = ds[0][0]
    assert torch.equal(img1, img1b)
    assert not torch.equal(img1, ds[1][0])


def test_flo_roundtrip(tmp_path):
    flow = np.random.randn(13, 17, 2).astype(np.float32)
    path = str(tmp_path / "t.flo")
    frame_utils.writeFlow(path, flow)
    back = frame_utils.readFlow(path)
    assert back.shape == (13, 17, 2)
    assert np.allclose(back, flow, atol=1e-6)


def test_kitti_png_roundtrip(tmp_path):
    flow = (np.random.rand(16, 24, 2).astype(np.float32) - 0.5) * 100
    path = str(tmp_path / "k.png")
    frame_utils.writeFlowKITTI(path, flow)
    back, valid = frame_utils.readFlowKITTI(path)
    assert back.shape == (16, 24, 2)
    assert np.allclose(back, flow, atol=1 / 64.0 + 1e-5)
    assert np.all(valid == 1)


def test_read_gen_dispatch(tmp_path):
    flow = np.random.randn(4, 6, 2).astype(np.float32)
    path = str(tmp_path / "x.flo")
    frame_utils.writeFlow(path, flow)
    out = frame_utils.read_gen(path)
    assert out.dtype == np.float32 and out.shape == (4, 6, 2)


def test_flow_augmentor_shapes():
    np.random.seed(0)
    aug = FlowAugmentor(crop_size=(64, 64), min_scale=-0.2, max_scale=0.5)
    img1 = np.random.randint(0, 255, (100, 120, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (100, 120, 3), dtype=np.uint8)
    flow = np.random.randn(100, 120, 2).astype(np.float32)
    for _ in range(5):
        a, b, f = aug(img1.copy(), img2.copy(), flow.copy())
        assert a.shape == (64, 64, 3) and b.shape == (64, 64, 3)
        assert f.shape == (64, 64, 2)
        assert a.dtype == np.uint8 and f.dtype == np.float32


def test_sparse_augmentor_shapes_and_valid():
    np.random.seed(1)
    aug = SparseFlowAugmentor(crop_size=(64, 64))
    img1 = np.random.randint(0, 255, (120, 160, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (120, 160, 3), dtype=np.uint8)
    flow = np.random.randn(120, 160, 2).astype(np.float32)
    valid = (np.random.rand(120, 160) > 0.5).astype(np.float32)
    for _ in range(5):
        a, b, f, v = aug(img1.copy(), img2.copy(), flow.copy(), valid.copy())
        assert a.shape == (64, 64, 3)
        assert f.shape == (64, 64, 2)
        assert v.shape == (64, 64)
        assert set(np.unique(v)).issubset({0, 1})


def test_flow_viz():
    from flowhip.data import flow_viz
    flow = np.random.randn(8, 8, 2).astype(np.float32)
    img = flow_viz.flow_to_color(flow)
    assert img.shape == (8, 8, 3) and img.dtype == np.uint8
    img2 = flow_viz.flow_to_image(flow.copy())
    assert img2.shape == (8, 8, 3)


def _write_png(path, h=64, w=96):
    from PIL import Image
    arr = (np.random.rand(h, w, 3) * 255).astype(np.uint8)
    Image.fromarray(arr).save(path)


def test_mpisintel_dataset_from_disk(tmp_path):
    """File-list scanning + dense .flo ground truth (datasets.py:102-119)."""
    from flowhip.data import frame_utils
    from flowhip.data.datasets import MpiSintel

    root = tmp_path / "Sintel"
    img_dir = root / "training" / "clean" / "alley_1"
    flow_dir = root / "training" / "flow" / "alley_1"
    img_dir.mkdir(parents=True)
    flow_dir.mkdir(parents=True)
    for i in range(3):
        _write_png(str(img_dir / f"frame_{i:04d}.png"))
    for i in range(2):
        frame_utils.writeFlow(str(flow_dir / f"frame_{i:04d}.flo"),
                              np.random.randn(64, 96, 2).astype(np.float32))

    ds = MpiSintel(aug_params=None, root=str(root), dstype="clean")
    assert len(ds) == 2  # consecutive pairs
    img1, img2, flow, valid = ds[0]
    assert img1.shape == (3, 64, 96) and flow.shape == (2, 64, 96)
    assert valid.shape == (64, 96) and valid.all()  # dense GT: all valid

    ds3 = 3 * ds  # __rmul__ oversampling (datasets.py:93)
    assert len(ds3) == 6


def test_kitti_dataset_sparse_from_disk(tmp_path):
    """Sparse KITTI GT: 16-bit png flow + valid mask (datasets.py:169-186)."""
    from flowhip.data import frame_utils
    from flowhip.data.datasets import KITTI

    root = tmp_path / "KITTI" / "training"
    (root / "image_2").mkdir(parents=True)
    (root / "flow_occ").mkdir(parents=True)
    _write_png(str(root / "image_2" / "000000_10.png"), 60, 80)
    _write_png(str(root / "image_2" / "000000_11.png"), 60, 80)
    # craft the 16-bit png directly so the VALID channel is sparse (the
    # writer itself marks everything valid — frame_utils.py:116-120 parity)
    flow = np.random.randn(60, 80, 2).astype(np.float32) * 10
    valid = (np.random.rand(60, 80) > 0.5).astype(np.uint16)
    enc = np.zeros((60, 80, 3), dtype=np.uint16)
    enc[..., :2] = (flow * 64.0 + 2 ** 15).astype(np.uint16)
    enc[..., 2] = valid
    import flowhip.data.frame_utils as fu
    fu._write_png16_rgb(str(root / "flow_occ" / "000000_10.png"), enc)

    ds = KITTI(aug_params=None, root=str(tmp_path / "KITTI"))
    assert len(ds) == 1
    img1, img2, f, v = ds[0]
    assert f.shape == (2, 60, 80)
    # sparse: the valid mask reflects the crafted third channel
    assert v.shape == (60, 80)
    assert 0 < v.sum() < 60 * 80
    assert abs(v.numpy().sum() - valid.sum()) == 0


def test_flyingthings_compressed_from_disk(tmp_path):
    """Compressed FT3D (webp images + npz flow; datasets.py:138-167).
    Also covers the reference defect fix: BOTH flow directions must be
    listed in compressed mode (the reference's in-loop `dstype += '_webp'`
    emptied the into_past glob)."""
    from PIL import Image
    from flowhip.data.datasets import FlyingThings3D

    root = tmp_path / "FT3D"
    img_dir = root / "frames_cleanpass_webp" / "TRAIN" / "A" / "0000" / "left"
    img_dir.mkdir(parents=True)
    for i in range(3):
        arr = (np.random.rand(32, 48, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(img_dir / f"{i:04d}.webp")
    for direction in ("into_future", "into_past"):
        fdir = (root / "optical_flow" / "TRAIN" / "A" / "0000" / direction
                / "left")
        fdir.mkdir(parents=True)
        for i in range(3):
            # compressed-FT3D npz layout: key 'optical_flow', CHW
            # (frame_utils read_gen transposes to HWC — ref :137-139)
            np.savez(fdir / f"{i:04d}.npz",
                     optical_flow=np.random.randn(2, 32, 48)
                     .astype(np.float32))

    ds = FlyingThings3D(aug_params=None, root=str(root),
                        load_compressed=True)
    # 2 consecutive pairs per direction
    assert len(ds) == 4
    img1, img2, flow, valid = ds[0]
    assert img1.shape == (3, 32, 48) and flow.shape == (2, 32, 48)


def test_flying_chairs_split_table(tmp_path):
    """chairs_split.txt drives the canonical train/val split
    (datasets.py:121-137; the table ships with the repo)."""
    import os
    from PIL import Image
    from flowhip.data import frame_utils
    from flowhip.data.datasets import FlyingChairs

    assert os.path.exists("chairs_split.txt")
    root = tmp_path / "chairs"
    root.mkdir()
    # first 4 entries of the real table are 1 1 1 1 (training)
    for i in range(1, 5):
        for k in (1, 2):
            arr = (np.random.rand(24, 32, 3) * 255).astype(np.uint8)
            Image.fromarray(arr).save(root / f"{i:05d}_img{k}.png")
        frame_utils.writeFlow(str(root / f"{i:05d}_flow.flo"),
                              np.random.randn(24, 32, 2).astype(np.float32))

    tr = FlyingChairs(aug_params=None, split="training", root=str(root))
    va = FlyingChairs(aug_params=None, split="validation", root=str(root))
    assert len(tr) + len(va) == 4
    assert len(tr) == 4  # table rows 0-3 are all split 1


def test_augmentor_hflip_negates_u():
    """h-flip mirrors columns and negates the u component (reference
    augmentor.py:91-95). Probabilities pinned to make the path
    deterministic; input = crop+1 so the crop offset draw is always 0."""
    np.random.seed(3)
    aug = FlowAugmentor(crop_size=(32, 32), do_flip=True)
    aug.spatial_aug_prob = 0.0   # no resize
    aug.stretch_prob = 0.0
    aug.h_flip_prob = 1.0
    aug.v_flip_prob = 0.0
    h, w = 33, 33
    img1 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    flow = np.random.randn(h, w, 2).astype(np.float32)
    _, _, f = aug.spatial_transform(img1.copy(), img2.copy(), flow.copy())
    exp = (flow[:, ::-1] * [-1.0, 1.0])[:32, :32]
    np.testing.assert_allclose(f, exp)


def test_augmentor_vflip_negates_v():
    np.random.seed(4)
    aug = FlowAugmentor(crop_size=(32, 32), do_flip=True)
    aug.spatial_aug_prob = 0.0
    aug.stretch_prob = 0.0
    aug.h_flip_prob = 0.0
    aug.v_flip_prob = 1.0
    h, w = 33, 33
    img1 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    flow = np.random.randn(h, w, 2).astype(np.float32)
    _, _, f = aug.spatial_transform(img1.copy(), img2.copy(), flow.copy())
    exp = (flow[::-1, :] * [1.0, -1.0])[:32, :32]
    np.testing.assert_allclose(f, exp)


def test_sparse_resize_identity_keeps_valid_flow():
    """resize_sparse_flow_map at scale 1 re-rasterizes losslessly except
    row/col 0, which the validity test (x>0, y>0 — reference
    augmentor.py:180) intentionally drops."""
    np.random.seed(5)
    aug = SparseFlowAugmentor(crop_size=(8, 8))
    h, w = 20, 24
    flow = np.random.randn(h, w, 2).astype(np.float32)
    valid = (np.random.rand(h, w) > 0.4).astype(np.float32)
    f1, v1 = aug.resize_sparse_flow_map(flow, valid, fx=1.0, fy=1.0)
    assert f1.shape == (h, w, 2) and v1.shape == (h, w)
    np.testing.assert_array_equal(v1[0, :], 0)
    np.testing.assert_array_equal(v1[:, 0], 0)
    inner_valid = valid[1:, 1:].astype(bool)
    np.testing.assert_array_equal(v1[1:, 1:].astype(bool), inner_valid)
    np.testing.assert_allclose(f1[1:, 1:][inner_valid], flow[1:, 1:][inner_valid])


def test_sparse_resize_scales_flow_values():
    """Scaling the grid by f scales flow vectors by f (pixel units)."""
    aug = SparseFlowAugmentor(crop_size=(8, 8))
    h, w = 10, 12
    flow = np.ones((h, w, 2), dtype=np.float32) * [3.0, -2.0]
    valid = np.ones((h, w), dtype=np.float32)
    f2, v2 = aug.resize_sparse_flow_map(flow, valid, fx=2.0, fy=2.0)
    assert f2.shape == (2 * h, 2 * w, 2)
    got = f2[v2.astype(bool)]
    np.testing.assert_allclose(got, np.broadcast_to([6.0, -4.0], got.shape))


def test_eraser_touches_only_img2():
    np.random.seed(6)
    aug = FlowAugmentor(crop_size=(32, 32))
    aug.eraser_aug_prob = 1.0
    h, w = 120, 140
    img1 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (h, w, 3), dtype=np.uint8)
    a, b = aug.eraser_transform(img1.copy(), img2.copy())
    np.testing.assert_array_equal(a, img1)          # img1 untouched
    changed = (b != img2).any(axis=-1)
    assert changed.any()                            # a rectangle was erased
    mean_color = np.mean(img2.reshape(-1, 3), axis=0)
    np.testing.assert_allclose(
        b[changed], np.broadcast_to(mean_color, b[changed].shape), atol=1.0)


def test_hd1k_dataset(tmp_path):
    """HD1K sequence discovery (reference datasets.py:181-197): per-sequence
    %06d_* globs, sparse 16-bit flow, last frame of each sequence unpaired."""
    from flowhip.data.datasets import HD1K

    root = tmp_path / "hd1k"
    img_dir = root / "hd1k_input" / "image_2"
    flo_dir = root / "hd1k_flow_gt" / "flow_occ"
    img_dir.mkdir(parents=True)
    flo_dir.mkdir(parents=True)
    from PIL import Image
    for seq in range(2):
        for i in range(3):
            arr = (np.random.rand(32, 48, 3) * 255).astype(np.uint8)
            Image.fromarray(arr).save(img_dir / f"{seq:06d}_{i:04d}.png")
            frame_utils.writeFlowKITTI(
                str(flo_dir / f"{seq:06d}_{i:04d}.png"),
                np.random.randn(32, 48, 2).astype(np.float32))

    ds = HD1K(aug_params=None, root=str(root))
    assert len(ds) == 4  # 2 sequences x (3 flows - 1)
    img1, img2, flow, valid = ds[0]
    assert img1.shape == (3, 32, 48) and flow.shape == (2, 32, 48)
    assert set(valid.unique().tolist()).issubset({0.0, 1.0})


def test_fetch_dataloader_chairs_stage(tmp_path, monkeypatch):
    """The chairs stage recipe builds a crop-augmented loader from the split
    table (reference datasets.py:210-213)."""
    import argparse

    from PIL import Image
    from flowhip.data.datasets import fetch_dataloader

    monkeypatch.chdir(tmp_path)
    root = tmp_path / "datasets" / "FlyingChairs_release" / "data"
    root.mkdir(parents=True)
    for i in (1, 2):
        for k in (1, 2):
            arr = (np.random.rand(96, 128, 3) * 255).astype(np.uint8)
            Image.fromarray(arr).save(root / f"{i:05d}_img{k}.png")
        frame_utils.writeFlow(str(root / f"{i:05d}_flow.flo"),
                              np.random.randn(96, 128, 2).astype(np.float32))
    (tmp_path / "chairs_split.txt").write_text("1\n1\n")

    args = argparse.Namespace(stage="chairs", image_size=[64, 64],
                              batch_size=2, num_workers=0)
    loader = fetch_dataloader(args)
    img1, img2, flow, valid = next(iter(loader))
    assert img1.shape == (2, 3, 64, 64)
    assert flow.shape == (2, 2, 64, 64) and valid.shape == (2, 64, 64)


def _write_pfm(path, arr):
    """Minimal color-PFM writer for fixtures (matches frame_utils.readPFM:
    'PF', dims, negative scale = little-endian, rows bottom-to-top)."""
    h, w, c = arr.shape
    assert c == 3
    with open(path, "wb") as f:
        f.write(b"PF\n")
        f.write(f"{w} {h}\n".encode())
        f.write(b"-1.0\n")
        np.flipud(arr).astype("<f4").tofile(f)


def test_sintel_stage_recipe_composition(tmp_path, monkeypatch):
    """The sintel fine-tune stage mixes 100*clean + 100*final + 200*kitti +
    5*hd1k + things (reference datasets.py:222-232) — dataset lengths
    compose exactly and a batch loads through the augmentors."""
    import argparse

    from PIL import Image
    from flowhip.data.datasets import fetch_dataloader

    monkeypatch.chdir(tmp_path)
    h, w = 96, 128

    def put_img(path):
        path.parent.mkdir(parents=True, exist_ok=True)
        arr = (np.random.rand(h, w, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(path)

    # Sintel training: 1 scene, 3 frames -> 2 pairs per dstype
    for dstype in ("clean", "final"):
        for i in range(3):
            put_img(tmp_path / "datasets" / "Sintel" / "training" / dstype
                    / "s1" / f"frame_{i:04d}.png")
    fdir = tmp_path / "datasets" / "Sintel" / "training" / "flow" / "s1"
    fdir.mkdir(parents=True)
    for i in range(2):
        frame_utils.writeFlow(str(fdir / f"frame_{i:04d}.flo"),
                              np.random.randn(h, w, 2).astype(np.float32))

    # KITTI training: 1 pair
    kroot = tmp_path / "datasets" / "KITTI" / "training"
    put_img(kroot / "image_2" / "000000_10.png")
    put_img(kroot / "image_2" / "000000_11.png")
    (kroot / "flow_occ").mkdir(parents=True)
    frame_utils.writeFlowKITTI(str(kroot / "flow_occ" / "000000_10.png"),
                               np.random.randn(h, w, 2).astype(np.float32))

    # HD1K: 1 sequence, 2 frames -> 1 pair
    hroot = tmp_path / "datasets" / "HD1k"
    for i in range(2):
        put_img(hroot / "hd1k_input" / "image_2" / f"000000_{i:04d}.png")
        (hroot / "hd1k_flow_gt" / "flow_occ").mkdir(parents=True, exist_ok=True)
        frame_utils.writeFlowKITTI(
            str(hroot / "hd1k_flow_gt" / "flow_occ" / f"000000_{i:04d}.png"),
            np.random.randn(h, w, 2).astype(np.float32))

    # FlyingThings3D cleanpass: 1 dir, 3 frames/pfm -> 2+2 pairs (2 dirs)
    idir = (tmp_path / "datasets" / "FlyingThings3D" / "frames_cleanpass"
            / "TRAIN" / "A" / "0000" / "left")
    fdir3 = (tmp_path / "datasets" / "FlyingThings3D" / "optical_flow"
             / "TRAIN" / "A" / "0000")
    for i in range(3):
        put_img(idir / f"{i:04d}.png")
    for direction in ("into_future", "into_past"):
        d = fdir3 / direction / "left"
        d.mkdir(parents=True)
        for i in range(3):
            _write_pfm(d / f"{i:04d}.pfm",
                       np.random.randn(h, w, 3).astype(np.float32))

    args = argparse.Namespace(stage="sintel", image_size=[64, 64],
                              batch_size=2, num_workers=0)
    loader = fetch_dataloader(args)
    # 100*2 + 100*2 + 200*1 + 5*1 + (2 into_future + 2 into_past)
    assert len(loader.dataset) == 100 * 2 + 100 * 2 + 200 * 1 + 5 * 1 + 4
    img1, img2, flow, valid = next(iter(loader))
    assert img1.shape == (2, 3, 64, 64) and flow.shape == (2, 2, 64, 64)


def test_things_and_kitti_stage_recipes(tmp_path, monkeypatch):
    """things stage = cleanpass + finalpass concat; kitti stage = sparse
    KITTI with no flips (reference datasets.py:214-236)."""
    import argparse

    from PIL import Image
    from flowhip.data.datasets import fetch_dataloader

    monkeypatch.chdir(tmp_path)
    h, w = 96, 128

    def put_img(path):
        path.parent.mkdir(parents=True, exist_ok=True)
        arr = (np.random.rand(h, w, 3) * 255).astype(np.uint8)
        Image.fromarray(arr).save(path)

    for dstype in ("frames_cleanpass", "frames_finalpass"):
        idir = (tmp_path / "datasets" / "FlyingThings3D" / dstype
                / "TRAIN" / "A" / "0000" / "left")
        for i in range(3):
            put_img(idir / f"{i:04d}.png")
    for direction in ("into_future", "into_past"):
        d = (tmp_path / "datasets" / "FlyingThings3D" / "optical_flow"
             / "TRAIN" / "A" / "0000" / direction / "left")
        d.mkdir(parents=True)
        for i in range(3):
            _write_pfm(d / f"{i:04d}.pfm",
                       np.random.randn(h, w, 3).astype(np.float32))

    args = argparse.Namespace(stage="things", image_size=[64, 64],
                              batch_size=2, num_workers=0,
                              compressed_ft=False)
    loader = fetch_dataloader(args)
    # (2 into_future + 2 into_past) per pass, both passes share the flow tree
    assert len(loader.dataset) == 8
    img1, _, flow, _ = next(iter(loader))
    assert img1.shape == (2, 3, 64, 64) and flow.shape == (2, 2, 64, 64)

    kroot = tmp_path / "datasets" / "KITTI" / "training"
    for i in range(2):
        put_img(kroot / "image_2" / f"{i:06d}_10.png")
        put_img(kroot / "image_2" / f"{i:06d}_11.png")
    (kroot / "flow_occ").mkdir(parents=True, exist_ok=True)
    for i in range(2):
        frame_utils.writeFlowKITTI(
            str(kroot / "flow_occ" / f"{i:06d}_10.png"),
            np.random.randn(h, w, 2).astype(np.float32))

    args = argparse.Namespace(stage="kitti", image_size=[64, 64],
                              batch_size=2, num_workers=0)
    loader = fetch_dataloader(args)
    assert len(loader.dataset) == 2
    img1, _, flow, valid = next(iter(loader))
    assert valid.shape == (2, 64, 64)


def test_read_disp_kitti(tmp_path):
    """16-bit grayscale disparity png: disp = px/256, valid = disp>0, flow =
    (-disp, 0) (reference frame_utils.py:109-113; PIL's 16-bit read
    replaces cv2.IMREAD_ANYDEPTH)."""
    import struct
    import zlib

    disp = (np.arange(12, dtype=np.uint16).reshape(3, 4)) * 128
    # write the 16-bit grayscale png manually (PIL deprecated I-mode saves)
    raw = b"".join(b"\x00" + row.astype(">u2").tobytes() for row in disp)

    def chunk(tag, payload):
        return (struct.pack(">I", len(payload)) + tag + payload
                + struct.pack(">I", zlib.crc32(tag + payload)))

    png = (b"\x89PNG\r\n\x1a\n"
           + chunk(b"IHDR", struct.pack(">IIBBBBB", 4, 3, 16, 0, 0, 0, 0))
           + chunk(b"IDAT", zlib.compress(raw)) + chunk(b"IEND", b""))
    (tmp_path / "d.png").write_bytes(png)

    flow, valid = frame_utils.readDispKITTI(str(tmp_path / "d.png"))
    assert flow.shape == (3, 4, 2)
    np.testing.assert_allclose(flow[..., 0], -disp.astype(np.float32) / 256.0)
    np.testing.assert_array_equal(flow[..., 1], 0)
    np.testing.assert_array_equal(valid, disp > 0)


def test_read_pfm_big_endian_and_gray(tmp_path):
    """readPFM's scale-sign endianness branch (scale > 0 => big-endian) and
    the single-channel 'Pf' header, both used by FlyingThings disparity/GT
    variants (reference frame_utils.py readPFM)."""
    from flowhip.data import frame_utils

    rng = np.random.default_rng(3)
    arr = rng.standard_normal((5, 7)).astype(np.float32)
    p = tmp_path / "be.pfm"
    with open(p, "wb") as f:
        f.write(b"Pf\n")
        f.write(b"7 5\n")
        f.write(b"1.0\n")  # positive scale = big-endian
        arr[::-1].astype(">f4").tofile(f)
    out = frame_utils.readPFM(str(p))
    assert out.shape == (5, 7)
    np.testing.assert_allclose(out, arr, rtol=0, atol=0)


def test_flo_roundtrip_extreme_values(tmp_path):
    """.flo round-trip preserves exact fp32 bits for extreme magnitudes
    (large displacements near the MAX_FLOW=400 exclusion boundary and
    subnormal-small values)."""
    from flowhip.data import frame_utils

    flow = np.array(
        [[[399.9, -399.9], [1e-30, -1e-30]],
         [[65504.0, -65504.0], [0.0, -0.0]]], dtype=np.float32)
    p = tmp_path / "x.flo"
    frame_utils.writeFlow(str(p), flow)
    back = frame_utils.readFlow(str(p))
    np.testing.assert_array_equal(back, flow)


def test_flow_augmentor_arbitrary_input_sizes():
    """The augmentor crops to crop_size for a RANGE of input geometries
    (taller/wider/near-crop-size inputs), with outputs always uint8 images
    + fp32 flow — the invariant train-time batching depends on."""
    from flowhip.data.augmentor import FlowAugmentor

    for seed, (H, W) in enumerate([(65, 65), (64, 200), (200, 64),
                                   (97, 131), (300, 70)]):
        np.random.seed(seed)
        aug = FlowAugmentor(crop_size=(64, 64), min_scale=-0.1, max_scale=0.4)
        img1 = np.random.randint(0, 255, (H, W, 3), dtype=np.uint8)
        img2 = np.random.randint(0, 255, (H, W, 3), dtype=np.uint8)
        flow = np.random.randn(H, W, 2).astype(np.float32)
        a, b, f = aug(img1, img2, flow)
        assert a.shape == (64, 64, 3) and b.shape == (64, 64, 3)
        assert f.shape == (64, 64, 2)
        assert a.dtype == np.uint8 and f.dtype == np.float32
        assert np.isfinite(f).all()


def test_flow_viz_invariants():
    """flow_to_color: uint8 HxWx3; zero flow renders (near-)white center
    colors; scaling the whole field by the max preserves hue structure
    (Middlebury wheel convention)."""
    from flowhip.data.flow_viz import flow_to_color

    z = np.zeros((8, 10, 2), dtype=np.float32)
    img = flow_to_color(z)
    assert img.shape == (8, 10, 3) and img.dtype == np.uint8
    assert (img > 220).all()  # zero displacement = unsaturated (white-ish)

    rng = np.random.default_rng(5)
    f = rng.standard_normal((16, 16, 2)).astype(np.float32) * 3
    img1 = flow_to_color(f)
    assert img1.shape == (16, 16, 3) and img1.dtype == np.uint8
    # the rendering normalizes by the max magnitude: scaling the whole
    # field leaves the image identical
    img2 = flow_to_color(f * 4.0)
    np.testing.assert_array_equal(img1, img2)


def test_flow_viz_name_compat_helpers():
    """VCN-naming aliases (reference flow_viz.py:158/199) agree with the
    canonical variants."""
    from flowhip.data.flow_viz import (compute_color, flow_compute_color,
                                       make_color_wheel, make_colorwheel)

    np.testing.assert_array_equal(make_color_wheel(), make_colorwheel())
    rng = np.random.default_rng(2)
    u = rng.standard_normal((5, 6)) * 0.4
    v = rng.standard_normal((5, 6)) * 0.4
    c = compute_color(u, v)
    assert c.dtype == np.float64
    np.testing.assert_array_equal(np.uint8(c), flow_compute_color(u, v))
