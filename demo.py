#!/usr/bin/env python3
"""Inference demo on a directory of frames (reference demo.py:50-79).

Runs the model over consecutive frame pairs and writes flow visualizations
as PNGs (this image has no display; the reference used cv2.imshow).
As in the reference, --model is the CHECKPOINT path; the architecture is
selected with --arch (reference hardcoded RAFT).

    python demo.py --model ckpt.pth --path frames_dir [--out demo_out]
"""

import argparse
import glob
import os

import numpy as np
import torch
from PIL import Image

from flowhip.config import add_ncup_module_flags, finalize_args
from flowhip.data import flow_viz
from flowhip.engine import checkpoints
from flowhip.models import build_model
from flowhip.utils.geometry import InputPadder
from flowhip.utils.geometry import InputPadder

DEVICE = "cuda" if torch.cuda.is_available() else "cpu"


def load_image(imfile):
    img = np.array(Image.open(imfile)).astype(np.uint8)
    img = torch.from_numpy(img).permute(2, 0, 1).float()
    return img[None].to(DEVICE)


def load_image_list(image_files):
    """Batch-load + pad a sorted frame list (reference demo.py:26-35)."""
    images = torch.cat([load_image(f) for f in sorted(image_files)], dim=0)
    padder = InputPadder(images.shape)
    return padder.pad(images)[0]


def viz(img, flo, out_path):
    img = img[0].permute(1, 2, 0).cpu().numpy()
    flo = flo[0].permute(1, 2, 0).cpu().numpy()
    flo = flow_viz.flow_to_image(flo)
    img_flo = np.concatenate([img, flo], axis=0).astype(np.uint8)
    Image.fromarray(img_flo).save(out_path)


def demo(args):
    model_args = finalize_args(args)
    model_args.model = args.arch
    model = build_model(model_args)
    if args.model and os.path.exists(args.model):
        checkpoints.load_weights(model, args.model)
    model.to(DEVICE)
    model.eval()

    os.makedirs(args.out, exist_ok=True)
    with torch.no_grad():
        files = sorted(glob.glob(os.path.join(args.path, "*.png"))
                       + glob.glob(os.path.join(args.path, "*.jpg")))
        images = [load_image(f) for f in files]

        for i, (image1, image2) in enumerate(zip(images[:-1], images[1:])):
            padder = InputPadder(image1.shape)
            image1, image2 = padder.pad(image1, image2)
            flow_low, flow_up = model(image1, image2, iters=20, test_mode=True)
            viz(image1, flow_up, os.path.join(args.out, "flow_%04d.png" % i))


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", help="checkpoint path")
    parser.add_argument("--arch", default="raft", help="model architecture")
    parser.add_argument("--path", help="directory of frames")
    parser.add_argument("--out", default="demo_out")
    parser.add_argument("--small", action="store_true")
    parser.add_argument("--mixed_precision", action="store_true")
    add_ncup_module_flags(parser)
    demo(parser.parse_args())
