#!/usr/bin/env python3
"""Training CLI — reference-compatible flag surface (see reference
train.py:264-343), DDP execution.

Single GPU / CPU:
    python train.py --name exp --model raft_nc_dbl --stage sintel ...
Multi-GPU (one rank per GPU over RCCL):
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 train.py ...
"""

import os

import numpy as np
import torch

from flowhip.config import build_train_parser, finalize_args
from flowhip.engine.train import count_parameters, fetch_optimizer, train
from flowhip.engine.logger import Logger
from flowhip.ops import sequence_loss

if __name__ == "__main__":
    args = finalize_args(build_train_parser().parse_args())

    torch.manual_seed(args.seed)
    np.random.seed(args.seed)

    os.makedirs("checkpoints", exist_ok=True)
    train(args)
