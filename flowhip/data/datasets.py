"""Flow datasets + the training dataloader recipe.

Behavioral parity with the reference `core/datasets.py` (file-list layouts,
per-worker RNG reseeding, test mode, oversampling via __rmul__, stage
mixture recipes). Additions for the MI355X framework:

- `SyntheticFlowDataset`: random image pairs + random smooth flow of a given
  shape — powers bench.py and GPU tests (no network for real datasets).
- `fetch_dataloader` takes a `distributed` flag: under DDP it shards with
  DistributedSampler (drop_last=True mirrors the reference loader's
  drop_last — datasets.py:240-241).
"""

import os
import os.path as osp
import random
from glob import glob

import numpy as np
import torch
import torch.utils.data as data

from . import frame_utils
from .augmentor import FlowAugmentor, SparseFlowAugmentor


class FlowDataset(data.Dataset):
    def __init__(self, aug_params=None, sparse=False):
        self.augmentor = None
        self.sparse = sparse
        if aug_params is not None:
            if sparse:
                self.augmentor = SparseFlowAugmentor(**aug_params)
            else:
                self.augmentor = FlowAugmentor(**aug_params)

        self.is_test = False
        self.init_seed = False
        self.flow_list = []
        self.image_list = []
        self.extra_info = []

    def __getitem__(self, index):
        if self.is_test:
            img1 = frame_utils.read_gen(self.image_list[index][0])
            img2 = frame_utils.read_gen(self.image_list[index][1])
            img1 = np.array(img1).astype(np.uint8)[..., :3]
            img2 = np.array(img2).astype(np.uint8)[..., :3]
            img1 = torch.from_numpy(img1).permute(2, 0, 1).float()
            img2 = torch.from_numpy(img2).permute(2, 0, 1).float()
            return img1, img2, self.extra_info[index]

        if not self.init_seed:
            worker_info = torch.utils.data.get_worker_info()
            if worker_info is not None:
                torch.manual_seed(worker_info.id)
                np.random.seed(worker_info.id)
                random.seed(worker_info.id)
                self.init_seed = True

        index = index % len(self.image_list)
        valid = None
        if self.sparse:
            flow, valid = frame_utils.readFlowKITTI(self.flow_list[index])
        else:
            flow = frame_utils.read_gen(self.flow_list[index])

        img1 = frame_utils.read_gen(self.image_list[index][0])
        img2 = frame_utils.read_gen(self.image_list[index][1])

        flow = np.array(flow).astype(np.float32)
        img1 = np.array(img1).astype(np.uint8)
        img2 = np.array(img2).astype(np.uint8)

        if len(img1.shape) == 2:  # grayscale
            img1 = np.tile(img1[..., None], (1, 1, 3))
            img2 = np.tile(img2[..., None], (1, 1, 3))
        else:
            img1 = img1[..., :3]
            img2 = img2[..., :3]

        if self.augmentor is not None:
            if self.sparse:
                img1, img2, flow, valid = self.augmentor(img1, img2, flow, valid)
            else:
                img1, img2, flow = self.augmentor(img1, img2, flow)

        img1 = torch.from_numpy(img1).permute(2, 0, 1).float()
        img2 = torch.from_numpy(img2).permute(2, 0, 1).float()
        flow = torch.from_numpy(flow).permute(2, 0, 1).float()

        if valid is not None:
            valid = torch.from_numpy(np.ascontiguousarray(valid))
        else:
            valid = (flow[0].abs() < 1000) & (flow[1].abs() < 1000)

        return img1, img2, flow, valid.float()

    def __rmul__(self, v):
        self.flow_list = v * self.flow_list
        self.image_list = v * self.image_list
        return self

    def __len__(self):
        return len(self.image_list)


class MpiSintel(FlowDataset):
    def __init__(self, aug_params=None, split="training", root="datasets/Sintel",
                 dstype="clean"):
        super().__init__(aug_params)
        flow_root = osp.join(root, split, "flow")
        image_root = osp.join(root, split, dstype)

        if split == "test":
            self.is_test = True

        for scene in sorted(os.listdir(image_root)) if osp.isdir(image_root) else []:
            image_list = sorted(glob(osp.join(image_root, scene, "*.png")))
            for i in range(len(image_list) - 1):
                self.image_list += [[image_list[i], image_list[i + 1]]]
                self.extra_info += [(scene, i)]
            if split != "test":
                self.flow_list += sorted(glob(osp.join(flow_root, scene, "*.flo")))


class FlyingChairs(FlowDataset):
    def __init__(self, aug_params=None, split="train",
                 root="datasets/FlyingChairs_release/data",
                 split_file="chairs_split.txt"):
        super().__init__(aug_params)

        images = sorted(glob(osp.join(root, "*_img*.png")))
        flows = sorted(glob(osp.join(root, "*_flow.flo")))
        if not flows:
            return
        assert len(images) // 2 == len(flows)

        split_list = np.loadtxt(split_file, dtype=np.int32)
        for i in range(len(flows)):
            xid = split_list[i]
            if (split == "training" and xid == 1) or (split == "validation" and xid == 2):
                self.flow_list += [flows[i]]
                self.image_list += [[images[2 * i], images[2 * i + 1]]]


class FlyingThings3D(FlowDataset):
    def __init__(self, aug_params=None, root="datasets/FlyingThings3D",
                 dstype="frames_cleanpass", load_compressed=False):
        super().__init__(aug_params)

        # Reference defect fixed (core/datasets.py:144-146): the reference
        # appends "_webp" to dstype INSIDE the direction loop, so the second
        # pass globs "<dstype>_webp_webp" and the into_past pairs silently
        # vanish in compressed mode. Appended once here.
        if load_compressed:
            dstype += "_webp"

        for cam in ["left"]:
            for direction in ["into_future", "into_past"]:
                image_dirs = sorted(glob(osp.join(root, dstype, "TRAIN/*/*")))
                image_dirs = sorted([osp.join(f, cam) for f in image_dirs])

                flow_dirs = sorted(glob(osp.join(root, "optical_flow/TRAIN/*/*")))
                flow_dirs = sorted([osp.join(f, direction, cam) for f in flow_dirs])

                for idir, fdir in zip(image_dirs, flow_dirs):
                    if load_compressed:
                        images = sorted(glob(osp.join(idir, "*.webp")))
                        flows = sorted(glob(osp.join(fdir, "*.npz")))
                    else:
                        images = sorted(glob(osp.join(idir, "*.png")))
                        flows = sorted(glob(osp.join(fdir, "*.pfm")))
                    for i in range(len(flows) - 1):
                        if direction == "into_future":
                            self.image_list += [[images[i], images[i + 1]]]
                            self.flow_list += [flows[i]]
                        elif direction == "into_past":
                            self.image_list += [[images[i + 1], images[i]]]
                            self.flow_list += [flows[i + 1]]


class KITTI(FlowDataset):
    def __init__(self, aug_params=None, split="training", root="datasets/KITTI"):
        super().__init__(aug_params, sparse=True)
        if split == "testing":
            self.is_test = True

        root = osp.join(root, split)
        images1 = sorted(glob(osp.join(root, "image_2/*_10.png")))
        images2 = sorted(glob(osp.join(root, "image_2/*_11.png")))

        for img1, img2 in zip(images1, images2):
            frame_id = img1.split("/")[-1]
            self.extra_info += [[frame_id]]
            self.image_list += [[img1, img2]]

        if split == "training":
            self.flow_list = sorted(glob(osp.join(root, "flow_occ/*_10.png")))


class HD1K(FlowDataset):
    def __init__(self, aug_params=None, root="datasets/HD1k"):
        super().__init__(aug_params, sparse=True)

        seq_ix = 0
        while 1:
            flows = sorted(glob(osp.join(root, "hd1k_flow_gt", "flow_occ/%06d_*.png" % seq_ix)))
            images = sorted(glob(osp.join(root, "hd1k_input", "image_2/%06d_*.png" % seq_ix)))
            if len(flows) == 0:
                break
            for i in range(len(flows) - 1):
                self.flow_list += [flows[i]]
                self.image_list += [[images[i], images[i + 1]]]
            seq_ix += 1


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
