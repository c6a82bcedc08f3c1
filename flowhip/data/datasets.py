"""Flow datasets + the training dataloader recipe.

Matches the observable behavior of the reference `core/datasets.py`
(directory layouts, train/val splits, test mode, oversampling, per-worker
RNG reseeding, stage mixture recipes) with a record-based design: each
dataset is a discovery function producing a flat list of `FlowSample`
records, consumed by one generic loading pipeline.

Additions for the MI355X framework:

- `SyntheticFlowDataset`: random image pairs + random smooth flow of a given
  shape — powers bench.py and GPU tests (no network for real datasets).
- `fetch_dataloader` takes a `distributed` flag: under DDP it shards with
  DistributedSampler (drop_last=True mirrors the reference loader's
  drop_last — datasets.py:240-241).
"""

import os.path as osp
import random
from dataclasses import dataclass, field
from glob import glob
from itertools import groupby
from pathlib import Path
from typing import Optional, Tuple

import numpy as np
import torch
import torch.utils.data as data

from . import frame_utils
from .augmentor import FlowAugmentor, SparseFlowAugmentor


@dataclass(frozen=True)
class FlowSample:
    """One training/eval example: a frame pair, its GT flow (if any), and
    the metadata the submission writers need (scene/frame id)."""

    frame_a: str
    frame_b: str
    flow: Optional[str] = None
    meta: Tuple = field(default_factory=tuple)


def _seed_worker_once():
    """First-touch per-worker RNG reseed (reference datasets.py:45-51):
    each DataLoader worker reseeds all three RNG streams from its worker id
    so augmentation draws differ across workers but are reproducible."""
    info = torch.utils.data.get_worker_info()
    if info is not None:
        torch.manual_seed(info.id)
        np.random.seed(info.id)
        random.seed(info.id)
    return info is not None


def _load_frame(path):
    """Decode one image to uint8 HxWx3: alpha dropped, grayscale
    replicated across channels (reference datasets.py:63-73)."""
    arr = np.asarray(frame_utils.read_gen(path)).astype(np.uint8)
    if arr.ndim == 2:
        arr = np.repeat(arr[:, :, None], 3, axis=2)
    return arr[:, :, :3]


def _chw_float(arr):
    return torch.from_numpy(np.ascontiguousarray(arr.transpose(2, 0, 1))).float()


class FlowDataset(data.Dataset):
    """Generic pipeline over a list of FlowSample records.

    `sparse` selects the KITTI 16-bit GT reader + the sparse augmentor;
    `test_mode` (set by test splits) makes __getitem__ return
    (img1, img2, meta) instead of (img1, img2, flow, valid).
    """

    def __init__(self, aug_params=None, sparse=False):
        self.sparse = sparse
        self.test_mode = False
        self.samples = []
        self._worker_seeded = False
        cls = SparseFlowAugmentor if sparse else FlowAugmentor
        self.augmentor = cls(**aug_params) if aug_params is not None else None

    def __len__(self):
        return len(self.samples)

    def __rmul__(self, times):
        """`k * ds` oversamples by repeating the record list (reference
        datasets.py:93-96; used by the sintel mixture weights)."""
        self.samples = times * self.samples
        return self

    def _read_gt(self, sample):
        """Returns (flow HxWx2 float32, valid HxW or None)."""
        if self.sparse:
            return frame_utils.readFlowKITTI(sample.flow)
        return np.asarray(frame_utils.read_gen(sample.flow),
                          dtype=np.float32), None

    def __getitem__(self, index):
        sample = self.samples[index % len(self.samples)]
        img1 = _load_frame(sample.frame_a)
        img2 = _load_frame(sample.frame_b)

        if self.test_mode:
            return _chw_float(img1), _chw_float(img2), sample.meta

        if not self._worker_seeded:
            self._worker_seeded = _seed_worker_once()

        flow, valid = self._read_gt(sample)

        if self.augmentor is not None:
            if self.sparse:
                img1, img2, flow, valid = self.augmentor(img1, img2, flow,
                                                         valid)
            else:
                img1, img2, flow = self.augmentor(img1, img2, flow)

        flow = _chw_float(flow)
        if valid is None:
            # dense GT: anything below the reference's sentinel magnitude
            # counts as valid (datasets.py:88)
            valid = (flow.abs() < 1000).all(dim=0)
        else:
            valid = torch.from_numpy(np.ascontiguousarray(valid))
        return _chw_float(img1), _chw_float(img2), flow, valid.float()


class MpiSintel(FlowDataset):
    """Scene-structured Sintel layout: <root>/<split>/<dstype>/<scene>/*.png
    with per-scene flow dirs (reference datasets.py:102-119)."""

    def __init__(self, aug_params=None, split="training",
                 root="datasets/Sintel", dstype="clean"):
        super().__init__(aug_params)
        self.test_mode = split == "test"

        scene_root = Path(root) / split / dstype
        scenes = sorted(p for p in scene_root.iterdir() if p.is_dir()) \
            if scene_root.is_dir() else []
        for scene in scenes:
            frames = sorted(str(p) for p in scene.glob("*.png"))
            flows = [] if self.test_mode else sorted(
                str(p) for p in (Path(root) / split / "flow"
                                 / scene.name).glob("*.flo"))
            for i, (a, b) in enumerate(zip(frames, frames[1:])):
                self.samples.append(FlowSample(
                    a, b, flows[i] if flows else None,
                    meta=(scene.name, i)))


class FlyingChairs(FlowDataset):
    """Flat-directory chairs layout: NNNNN_img{1,2}.png + NNNNN_flow.flo,
    train/val membership from the shipped split table (reference
    datasets.py:121-137; chairs_split.txt rows: 1=train, 2=val)."""

    def __init__(self, aug_params=None, split="train",
                 root="datasets/FlyingChairs_release/data",
                 split_file="chairs_split.txt"):
        super().__init__(aug_params)

        flows = sorted(glob(osp.join(root, "*_flow.flo")))
        if not flows:
            return
        table = np.loadtxt(split_file, dtype=np.int32)
        want = 1 if split == "training" else 2
        for i, flow_path in enumerate(flows):
            if table[i] != want:
                continue
            stem = flow_path[:-len("_flow.flo")]
            self.samples.append(FlowSample(
                stem + "_img1.png", stem + "_img2.png", flow_path))


class FlyingThings3D(FlowDataset):
    """FlyingThings3D: per-scene frame dirs paired with per-direction flow
    dirs; each scene contributes forward pairs (into_future) and reversed
    pairs (into_past) (reference datasets.py:138-167).

    Reference defect fixed (core/datasets.py:144-146): the reference appends
    "_webp" to dstype INSIDE the direction loop, so the second pass globs
    "<dstype>_webp_webp" and the into_past pairs silently vanish in
    compressed mode. Appended once here (divergence noted in PARITY.md).
    """

    def __init__(self, aug_params=None, root="datasets/FlyingThings3D",
                 dstype="frames_cleanpass", load_compressed=False):
        super().__init__(aug_params)

        img_ext, flo_ext = ((".webp", ".npz") if load_compressed
                            else (".png", ".pfm"))
        if load_compressed:
            dstype = dstype + "_webp"

        scene_dirs = sorted(glob(osp.join(root, dstype, "TRAIN/*/*")))
        for scene in scene_dirs:
            rel = osp.join(*scene.split("/")[-2:])  # e.g. A/0000
            frames = sorted(glob(osp.join(scene, "left", "*" + img_ext)))
            for direction in ("into_future", "into_past"):
                flows = sorted(glob(osp.join(
                    root, "optical_flow/TRAIN", rel, direction, "left",
                    "*" + flo_ext)))
                if direction == "into_future":
                    for a, b, fl in zip(frames, frames[1:], flows):
                        self.samples.append(FlowSample(a, b, fl))
                else:
                    # into_past: the pair is reversed and the flow belongs
                    # to the LATER frame
                    for a, b, fl in zip(frames, frames[1:], flows[1:]):
                        self.samples.append(FlowSample(b, a, fl))


class KITTI(FlowDataset):
    """KITTI-2015 layout: image_2/<id>_10.png + <id>_11.png frame pairs with
    sparse flow_occ 16-bit GT named after the first frame (reference
    datasets.py:169-186)."""

    def __init__(self, aug_params=None, split="training",
                 root="datasets/KITTI"):
        super().__init__(aug_params, sparse=True)
        self.test_mode = split == "testing"

        base = Path(root) / split
        for first in sorted((base / "image_2").glob("*_10.png")):
            second = first.with_name(first.name.replace("_10.", "_11."))
            gt = base / "flow_occ" / first.name
            self.samples.append(FlowSample(
                str(first), str(second),
                None if self.test_mode else str(gt),
                meta=(first.name,)))


class HD1K(FlowDataset):
    """HD1K: frames named <seq>_<frame>.png; consecutive frames within each
    sequence form pairs, the last frame of a sequence is unpaired
    (reference datasets.py:188-204)."""

    def __init__(self, aug_params=None, root="datasets/HD1k"):
        super().__init__(aug_params, sparse=True)

        def frame_of(flow_path):
            return osp.join(root, "hd1k_input/image_2",
                            osp.basename(flow_path))

        flows = sorted(glob(osp.join(root, "hd1k_flow_gt/flow_occ/*.png")))
        for _, group in groupby(flows, key=lambda p: osp.basename(p).split("_")[0]):
            group = list(group)
            for fl, nxt in zip(group, group[1:]):
                self.samples.append(FlowSample(frame_of(fl), frame_of(nxt), fl))


class SyntheticFlowDataset(data.Dataset):
    """Random image pairs + smooth random flow of a fixed shape.

    Used by bench.py and the GPU tests (BASELINE protocol: synthetic random
    pairs, random-init weights — there is no network for real datasets).
    Samples are generated deterministically from (seed, index).
    """

    def __init__(self, image_size=(448, 1024), length=10000, seed=1234,
                 max_mag=16.0):
        self.image_size = tuple(image_size)
        self.length = length
        self.seed = seed
        self.max_mag = max_mag

    def __len__(self):
        return self.length

    def __getitem__(self, index):
        h, w = self.image_size
        g = torch.Generator().manual_seed(self.seed * 100003 + index)
        img1 = torch.randint(0, 256, (3, h, w), generator=g).float()
        img2 = torch.randint(0, 256, (3, h, w), generator=g).float()
        # smooth flow: random coarse grid, bilinearly upsampled
        coarse = (torch.rand(2, h // 32 + 1, w // 32 + 1, generator=g) * 2 - 1) * self.max_mag
        flow = torch.nn.functional.interpolate(
            coarse[None], size=(h, w), mode="bilinear", align_corners=False)[0]
        valid = torch.ones(h, w)
        return img1, img2, flow, valid


# Per-stage augmentation envelopes (scale range is log2; flips per stage).
# Values are the reference's published training schedules (datasets.py:210-236).
_STAGE_AUG = {
    "chairs": dict(min_scale=-0.1, max_scale=1.0, do_flip=True),
    "things": dict(min_scale=-0.4, max_scale=0.8, do_flip=True),
    "sintel": dict(min_scale=-0.2, max_scale=0.6, do_flip=True),
    "sintel_kitti": dict(min_scale=-0.3, max_scale=0.5, do_flip=True),
    "sintel_hd1k": dict(min_scale=-0.5, max_scale=0.2, do_flip=True),
    "kitti": dict(min_scale=-0.2, max_scale=0.4, do_flip=False),
}


def _stage_dataset(args, TRAIN_DS):
    """Assemble the training mixture for a stage.

    Oversampling weights for the sintel fine-tune mixture follow the
    reference exactly: 100x each Sintel pass + 200x KITTI + 5x HD1K + one
    pass of FlyingThings (datasets.py:231).
    """
    def aug(key):
        return {"crop_size": args.image_size, **_STAGE_AUG[key]}

    stage = args.stage
    if stage == "chairs":
        return FlyingChairs(aug("chairs"), split="training")

    if stage == "things":
        clean, final = (FlyingThings3D(aug("things"), dstype=pas,
                                       load_compressed=args.compressed_ft)
                        for pas in ("frames_cleanpass", "frames_finalpass"))
        return clean + final

    if stage == "sintel":
        base = (100 * MpiSintel(aug("sintel"), split="training", dstype="clean")
                + 100 * MpiSintel(aug("sintel"), split="training",
                                  dstype="final"))
        things = FlyingThings3D(aug("sintel"), dstype="frames_cleanpass")
        if TRAIN_DS == "C+T+K+S+H":
            return (base + 200 * KITTI(aug("sintel_kitti"))
                    + 5 * HD1K(aug("sintel_hd1k")) + things)
        if TRAIN_DS == "C+T+K/S":
            return base + things
        raise ValueError(f"unknown TRAIN_DS {TRAIN_DS!r}")

    if stage == "kitti":
        return KITTI(aug("kitti"), split="training")

    if stage == "synthetic":
        return SyntheticFlowDataset(image_size=args.image_size)

    raise ValueError(f"unknown stage {args.stage!r}")


def fetch_dataloader(args, TRAIN_DS="C+T+K+S+H", distributed=False, rank=0,
                     world_size=1):
    """Create the training loader for a stage (reference datasets.py:207-243).

    Under DDP each rank gets a DistributedSampler shard; batch_size is the
    PER-PROCESS batch (the reference's single-process batch was split across
    GPUs by DataParallel — keep global batch = batch_size * world_size in
    mind when reproducing reference schedules).
    """
    train_dataset = _stage_dataset(args, TRAIN_DS)

    sampler = None
    shuffle = True
    if distributed:
        sampler = data.distributed.DistributedSampler(
            train_dataset, num_replicas=world_size, rank=rank, shuffle=True,
            drop_last=True)
        shuffle = False

    train_loader = data.DataLoader(
        train_dataset, batch_size=args.batch_size,
        pin_memory=torch.cuda.is_available(),
        shuffle=shuffle, sampler=sampler,
        num_workers=getattr(args, "num_workers", 4), drop_last=True,
        persistent_workers=getattr(args, "num_workers", 4) > 0)

    if rank == 0:
        print("Training with %d image pairs" % len(train_dataset))
    return train_loader
