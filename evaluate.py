#!/usr/bin/env python3
"""Evaluation CLI — reference-compatible (see reference evaluate.py:185-272).

    python evaluate.py --model raft_nc_dbl --restore_ckpt ckpt.pth --dataset sintel ...
"""

import torch

from flowhip.config import build_eval_parser, finalize_args
from flowhip.engine import checkpoints
from flowhip.engine.evaluate import (
    create_kitti_submission,
    create_sintel_submission,
    validate_chairs,
    validate_kitti,
    validate_sintel,
    validate_synthetic,
)
from flowhip.models import build_model

if __name__ == "__main__":
    args = finalize_args(build_eval_parser().parse_args())

    model = build_model(args)
    if args.restore_ckpt is not None:
        checkpoints.load_weights(model, args.restore_ckpt)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    model.to(device)
    model.eval()

    with torch.no_grad():
        if args.dataset == "chairs":
            validate_chairs(model, args.iters or 24)
        elif args.dataset == "sintel":
            validate_sintel(model, args.iters or 32)
        elif args.dataset == "kitti":
            validate_kitti(model, args.iters or 24)
        elif args.dataset == "synthetic":
            validate_synthetic(model, args.iters or 12)
        elif args.dataset == "sintel_submission":
            # the reference ships these as commented-out lines the user must
            # edit in (evaluate.py:267,271); defined behavior: CLI choices
            create_sintel_submission(model, warm_start=True)
        elif args.dataset == "kitti_submission":
            create_kitti_submission(model)
        else:
            raise SystemExit(f"unknown --dataset {args.dataset!r}")
