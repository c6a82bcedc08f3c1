"""Checkpoint-layout compatibility with the reference:
`module.`-prefixed DataParallel keys, `weight_p` nconv parameters, exact
attribute paths. Plus full train-state round trip."""

import argparse
import os

import torch

from flowhip.config.args import default_ncup_args
from flowhip.engine import checkpoints
from flowhip.models import RAFT, RAFT_NC_DBL

# Key names a reference raft_nc_dbl (sintel config) checkpoint contains —
# derived from the reference module structure (raft_nc_dbl.py, update.py,
# extractor.py, upsampler.py, nconv_modules.py, interp_weights_est.py).
REFERENCE_NCUP_KEYS = [
    "module.fnet.conv1.weight",
    "module.fnet.layer1.0.conv1.weight",
    "module.fnet.layer2.0.downsample.0.weight",
    "module.fnet.layer3.1.conv2.bias",
    "module.fnet.conv2.weight",
    "module.cnet.conv1.weight",
    "module.cnet.norm1.running_mean",
    "module.update_block.encoder.convc1.weight",
    "module.update_block.encoder.convc2.weight",
    "module.update_block.encoder.convf1.weight",
    "module.update_block.encoder.convf2.weight",
    "module.update_block.encoder.conv.weight",
    "module.update_block.gru.convz1.weight",
    "module.update_block.gru.convr1.bias",
    "module.update_block.gru.convq1.weight",
    "module.update_block.gru.convz2.weight",
    "module.update_block.gru.convr2.weight",
    "module.update_block.gru.convq2.bias",
    "module.update_block.flow_head.conv1.weight",
    "module.update_block.flow_head.conv2.bias",
    "module.upsampler.interpolation_net.nconv_in.weight_p",
    "module.upsampler.interpolation_net.nconv_x2.0.weight_p",
    "module.upsampler.interpolation_net.encoder.0.0.weight_p",
    "module.upsampler.interpolation_net.encoder.1.weight_p",
    "module.upsampler.interpolation_net.decoder.0.weight_p",
    "module.upsampler.interpolation_net.nconv_out.weight_p",
    "module.upsampler.weights_est_net.conv.0.0.weight",
    "module.upsampler.weights_est_net.conv.0.1.weight",  # BN (sintel)
    "module.upsampler.weights_est_net.conv.1.0.weight",
    "module.upsampler.weights_est_net.out.weight",
]

RAFT_BASIC_KEYS = [
    "module.update_block.mask.0.weight",
    "module.update_block.mask.2.weight",
]


def test_ncup_reference_keys_present():
    model = RAFT_NC_DBL(default_ncup_args(dataset="sintel"))
    sd = checkpoints.reference_state_dict(model)
    for k in REFERENCE_NCUP_KEYS:
        assert k in sd, f"missing reference key {k}"


def test_raft_basic_mask_keys_present():
    args = argparse.Namespace(model="raft", small=False, dropout=0.0,
                              mixed_precision=False)
    sd = checkpoints.reference_state_dict(RAFT(args))
    for k in RAFT_BASIC_KEYS:
        assert k in sd


def test_kitti_config_disables_bn():
    model = RAFT_NC_DBL(default_ncup_args(dataset="kitti"))
    sd = model.state_dict()
    assert not any("weights_est_net.conv.0.1" in k for k in sd)


def test_weight_shapes():
    model = RAFT_NC_DBL(default_ncup_args(dataset="sintel"))
    sd = model.state_dict()
    assert sd["upsampler.interpolation_net.nconv_in.weight_p"].shape == (2, 1, 5, 5)
    assert sd["upsampler.interpolation_net.decoder.0.weight_p"].shape == (2, 4, 3, 3)
    assert sd["upsampler.interpolation_net.nconv_out.weight_p"].shape == (1, 2, 1, 1)
    assert sd["upsampler.weights_est_net.conv.0.0.weight"].shape == (64, 130, 3, 3)
    assert sd["update_block.gru.convz1.weight"].shape == (128, 256 + 128, 1, 5)


def test_save_load_roundtrip(tmp_path):
    model = RAFT_NC_DBL(default_ncup_args())
    path = os.path.join(tmp_path, "ckpt", "m.pth")
    checkpoints.save_weights(model, path)

    model2 = RAFT_NC_DBL(default_ncup_args())
    checkpoints.load_weights(model2, path)
    for (k1, v1), (k2, v2) in zip(model.state_dict().items(),
                                  model2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(v1, v2)


def test_load_pretrained_raft_into_ncup(tmp_path):
    """raft_nc_dbl --load_pretrained consumes a module.-prefixed RAFT basic
    checkpoint (ref raft_nc_dbl.py:57-66)."""
    raft_args = argparse.Namespace(model="raft", small=False, dropout=0.0,
                                   mixed_precision=False)
    raft = RAFT(raft_args)
    path = os.path.join(tmp_path, "raft.pth")
    torch.save(checkpoints.reference_state_dict(raft), path)

    ncup = RAFT_NC_DBL(default_ncup_args(load_pretrained=path))
    assert torch.equal(ncup.fnet.conv1.weight, raft.fnet.conv1.weight)


def test_train_state_roundtrip(tmp_path):
    import torch.optim as optim
    model = RAFT_NC_DBL(default_ncup_args())
    opt = optim.AdamW(model.parameters(), lr=1e-4)
    sched = optim.lr_scheduler.OneCycleLR(opt, 1e-4, 100)

    # one step so optimizer has state
    loss = sum(p.sum() for p in model.parameters())
    loss.backward()
    opt.step()
    sched.step()

    path = os.path.join(tmp_path, "state.pth")
    checkpoints.save_train_state(path, model, opt, sched, total_steps=7)

    model2 = RAFT_NC_DBL(default_ncup_args())
    opt2 = optim.AdamW(model2.parameters(), lr=1e-4)
    sched2 = optim.lr_scheduler.OneCycleLR(opt2, 1e-4, 100)
    steps = checkpoints.load_train_state(path, model2, opt2, sched2)
    assert steps == 7
    assert sched2.last_epoch == sched.last_epoch
    assert torch.equal(model2.state_dict()["fnet.conv1.weight"],
                       model.state_dict()["fnet.conv1.weight"])
