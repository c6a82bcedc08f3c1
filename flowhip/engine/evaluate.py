"""Validation + leaderboard-submission drivers.

Behavioral parity with the reference `evaluate.py` (metric definitions,
per-dataset iteration counts, warm-started Sintel submission, KITTI 16-bit
png writer). Fresh additions: device-agnostic (CPU or GPU), and
`validate_synthetic` for environments without the real datasets.
"""

import os

import numpy as np
import torch

from ..data import datasets, flow_viz, frame_utils
from ..utils.geometry import InputPadder, forward_interpolate


def _device_of(model):
    return next(model.parameters()).device


@torch.no_grad()
def create_sintel_submission(model, iters=32, warm_start=False,
                             output_path="sintel_submission", write_png=False):
    """Write Sintel leaderboard .flo files (reference evaluate.py:23-57),
    optionally warm-starting each sequence from the previous frame's flow."""
    model.eval()
    device = _device_of(model)
    for dstype in ["clean", "final"]:
        test_dataset = datasets.MpiSintel(split="test", aug_params=None, dstype=dstype)

        flow_prev, sequence_prev = None, None
        for test_id in range(len(test_dataset)):
            image1, image2, (sequence, frame) = test_dataset[test_id]
            if sequence != sequence_prev:
                flow_prev = None

            padder = InputPadder(image1.shape)
            image1, image2 = padder.pad(image1[None].to(device), image2[None].to(device))

            flow_low, flow_pr = model(image1, image2, iters=iters,
                                      flow_init=flow_prev, test_mode=True)
            flow = padder.unpad(flow_pr[0]).permute(1, 2, 0).cpu().numpy()

            if warm_start:
                flow_prev = forward_interpolate(flow_low[0])[None].to(device)

            output_dir = os.path.join(output_path, dstype, sequence)
            os.makedirs(output_dir, exist_ok=True)
            if write_png:
                from PIL import Image
                png_dir = os.path.join(output_path + "_png", dstype, sequence)
                os.makedirs(png_dir, exist_ok=True)
                Image.fromarray(flow_viz.flow_to_image(flow)).save(
                    os.path.join(png_dir, "frame%04d.png" % (frame + 1)))

            frame_utils.writeFlow(
                os.path.join(output_dir, "frame%04d.flo" % (frame + 1)), flow)
            sequence_prev = sequence


@torch.no_grad()
def create_kitti_submission(model, iters=24, output_path="kitti_submission",
                            write_png=False):
    """Write KITTI leaderboard 16-bit pngs (reference evaluate.py:61-87)."""
    model.eval()
    device = _device_of(model)
    test_dataset = datasets.KITTI(split="testing", aug_params=None)
    os.makedirs(output_path, exist_ok=True)
    if write_png:
        os.makedirs(output_path + "_png", exist_ok=True)

    for test_id in range(len(test_dataset)):
        image1, image2, (frame_id,) = test_dataset[test_id]
        padder = InputPadder(image1.shape, mode="kitti")
        image1, image2 = padder.pad(image1[None].to(device), image2[None].to(device))

        _, flow_pr = model(image1, image2, iters=iters, test_mode=True)
        flow = padder.unpad(flow_pr[0]).permute(1, 2, 0).cpu().numpy()

        if write_png:
            from PIL import Image
            Image.fromarray(flow_viz.flow_to_image(flow)).save(
                os.path.join(output_path + "_png", frame_id + ".png"))
        frame_utils.writeFlowKITTI(os.path.join(output_path, frame_id), flow)


@torch.no_grad()
def validate_chairs(model, iters=24):
    """FlyingChairs validation EPE (reference evaluate.py:91-108)."""
    model.eval()
    device = _device_of(model)
    epe_list = []

    val_dataset = datasets.FlyingChairs(split="validation")
    for val_id in range(len(val_dataset)):
        image1, image2, flow_gt, _ = val_dataset[val_id]
        image1 = image1[None].to(device)
        image2 = image2[None].to(device)

        _, flow_pr = model(image1, image2, iters=iters, test_mode=True)
        epe = torch.sum((flow_pr[0].cpu() - flow_gt) ** 2, dim=0).sqrt()
        epe_list.append(epe.view(-1).numpy())

    epe = np.mean(np.concatenate(epe_list))
    print("Validation Chairs EPE: %f" % epe)
    return {"chairs": epe}


@torch.no_grad()
def validate_sintel(model, iters=32):
    """Sintel train-split EPE + inlier rates (reference evaluate.py:112-143)."""
    model.eval()
    device = _device_of(model)
    results = {}
    for dstype in ["clean", "final"]:
        val_dataset = datasets.MpiSintel(split="training", dstype=dstype)
        epe_list = []

        for val_id in range(len(val_dataset)):
            image1, image2, flow_gt, _ = val_dataset[val_id]
            image1 = image1[None].to(device)
            image2 = image2[None].to(device)

            padder = InputPadder(image1.shape)
            image1, image2 = padder.pad(image1, image2)

            flow_low, flow_pr = model(image1, image2, iters=iters, test_mode=True)
            flow = padder.unpad(flow_pr[0]).cpu()

            epe = torch.sum((flow - flow_gt) ** 2, dim=0).sqrt()
            epe_list.append(epe.view(-1).numpy())

        epe_all = np.concatenate(epe_list)
        epe = np.mean(epe_all)
        px1 = np.mean(epe_all < 1)
        px3 = np.mean(epe_all < 3)
        px5 = np.mean(epe_all < 5)

        print("Validation (%s) EPE: %f, 1px: %f, 3px: %f, 5px: %f"
              % (dstype, epe, px1, px3, px5))
        results[dstype] = np.mean(epe_list)
    return results


@torch.no_grad()
def validate_kitti(model, iters=24):
    """KITTI-2015 train-split EPE + F1 (reference evaluate.py:147-181)."""
    model.eval()
    device = _device_of(model)
    val_dataset = datasets.KITTI(split="training")

    out_list, epe_list = [], []
    for val_id in range(len(val_dataset)):
        image1, image2, flow_gt, valid_gt = val_dataset[val_id]
        image1 = image1[None].to(device)
        image2 = image2[None].to(device)

        padder = InputPadder(image1.shape, mode="kitti")
        image1, image2 = padder.pad(image1, image2)

        flow_low, flow_pr = model(image1, image2, iters=iters, test_mode=True)
        flow = padder.unpad(flow_pr[0]).cpu()

        epe = torch.sum((flow - flow_gt) ** 2, dim=0).sqrt()
        mag = torch.sum(flow_gt ** 2, dim=0).sqrt()

        epe = epe.view(-1)
        mag = mag.view(-1)
        val = valid_gt.view(-1) >= 0.5

        out = ((epe > 3.0) & ((epe / mag) > 0.05)).float()
        epe_list.append(epe[val].mean().item())
        out_list.append(out[val].cpu().numpy())

    epe = np.mean(np.array(epe_list))
    f1 = 100 * np.mean(np.concatenate(out_list))
    print("Validation KITTI: %f, %f" % (epe, f1))
    return {"kitti-epe": epe, "kitti-f1": f1}


@torch.no_grad()
def validate_synthetic(model, iters=12, n_samples=4, image_size=(128, 128)):
    """Synthetic-pair smoke validation (framework addition; no datasets on
    disk). Reports EPE against the generated ground truth."""
    model.eval()
    device = _device_of(model)
    ds = datasets.SyntheticFlowDataset(image_size=image_size, length=n_samples)
    epe_list = []
    for i in range(n_samples):
        image1, image2, flow_gt, _ = ds[i]
        _, flow_pr = model(image1[None].to(device), image2[None].to(device),
                           iters=iters, test_mode=True)
        epe = torch.sum((flow_pr[0].cpu() - flow_gt) ** 2, dim=0).sqrt()
        epe_list.append(epe.view(-1).numpy())
    epe = float(np.mean(np.concatenate(epe_list)))
    print("Validation Synthetic EPE: %f" % epe)
    return {"synthetic": epe}
