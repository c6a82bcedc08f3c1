"""Config-surface tests: the CLI accepts the exact flag set of the reference
shell scripts and reproduces the shipped NCUP config."""

import shlex

from flowhip.config import build_eval_parser, build_train_parser, finalize_args

# the canonical invocation from reference train_raft_nc_sintel.sh
SINTEL_SCRIPT_ARGS = shlex.split("""
--name raft_nc_sintel_ft --model raft_nc_dbl --stage sintel
--validation sintel --gpus 0 1 --num_steps 50000 --batch_size 6
--lr 0.000125 --image_size 368 768 --optimizer adamW --scheduler cyclic
--gamma=0.85
--final_upsampling=NConvUpsampler --final_upsampling_scale=4
--final_upsampling_use_data_for_guidance=True
--final_upsampling_channels_to_batch=True
--final_upsampling_use_residuals=False
--final_upsampling_est_on_high_res=False
--interp_net=NConvUNet --interp_net_channels_multiplier=2
--interp_net_num_downsampling=1 --interp_net_data_pooling=conf_based
--interp_net_encoder_filter_sz=5 --interp_net_decoder_filter_sz=3
--interp_net_out_filter_sz=1 --interp_net_shared_encoder=True
--interp_net_use_double_conv=False --interp_net_use_bias=False
--weights_est_net=Simple --weights_est_net_num_ch=[64,32]
--weights_est_net_filter_sz=[3,3,1] --weights_est_net_dilation=[1,1,1]
""")


def test_train_parser_accepts_reference_script():
    parser = build_train_parser(argv=SINTEL_SCRIPT_ARGS)
    args = finalize_args(parser.parse_args(SINTEL_SCRIPT_ARGS))

    assert args.model == "raft_nc_dbl"
    assert args.stage == "sintel"
    assert args.dataset == "sintel"  # derived (reference defect fix)
    assert args.gamma == 0.85
    assert args.lr == 0.000125
    assert args.image_size == [368, 768]
    assert args.final_upsampling == "NConvUpsampler"
    assert args.final_upsampling_scale == 4
    assert args.final_upsampling_use_data_for_guidance is True
    assert args.final_upsampling_use_residuals is False
    assert args.interp_net == "NConvUNet"
    assert args.interp_net_channels_multiplier == 2
    assert args.interp_net_num_downsampling == 1
    assert args.interp_net_data_pooling == "conf_based"
    assert args.interp_net_use_double_conv is False
    assert args.weights_est_net == "Simple"
    assert args.weights_est_net_num_ch == [64, 32]
    assert args.weights_est_net_filter_sz == [3, 3, 1]
    assert args.weights_est_net_dilation == [1, 1, 1]


def test_train_parser_defaults():
    parser = build_train_parser(argv=[])
    args = parser.parse_args([])
    assert args.lr == 0.00002
    assert args.batch_size == 6
    assert args.gamma == 0.8
    assert args.clip == 1.0
    assert args.iters == 12
    assert args.image_size == [384, 512]


def test_model_built_from_parsed_args():
    from flowhip.models import build_model
    parser = build_train_parser(argv=SINTEL_SCRIPT_ARGS)
    args = finalize_args(parser.parse_args(SINTEL_SCRIPT_ARGS))
    args.small = False
    model = build_model(args)
    # shipped config: weights est net input = 128 guidance + 2 data = 130
    assert model.upsampler.weights_est_net.in_ch == 130
    assert model.upsampler.scaleH == 4.0


def test_eval_parser():
    argv = shlex.split("--model raft_nc_dbl --dataset sintel "
                       "--weights_est_net Simple --weights_est_net_num_ch [64,32] "
                       "--weights_est_net_filter_sz [3,3,1] "
                       "--weights_est_net_dilation [1,1,1]")
    parser = build_eval_parser(argv=argv)
    args = parser.parse_args(argv)
    assert args.dataset == "sintel"
    assert args.weights_est_net_num_ch == [64, 32]


def test_eval_parser_accepts_reference_eval_script():
    """The canonical eval invocation (eval_raft_nc_sintel.sh flags)."""
    import shlex
    argv = shlex.split("""
    --model checkpoints/raft_nc_sintel/final_model.pth
    --dataset sintel
    --final_upsampling=NConvUpsampler --final_upsampling_scale=4
    --final_upsampling_use_data_for_guidance=True
    --final_upsampling_channels_to_batch=True
    --interp_net=NConvUNet --interp_net_channels_multiplier=2
    --interp_net_num_downsampling=1
    --weights_est_net=Simple --weights_est_net_num_ch=[64,32]
    --weights_est_net_filter_sz=[3,3,1]
    """)
    parser = build_eval_parser(argv=argv)
    args = finalize_args(parser.parse_args(argv))
    assert args.dataset == "sintel"
    assert args.interp_net_channels_multiplier == 2
    assert args.weights_est_net_num_ch == [64, 32]


def test_weights_est_choices_are_module_classes_only():
    """The reflective --weights_est_net choices expose only classes defined
    in interp_weights_est (imported helpers like the conv wrapper must not
    leak into the CLI surface)."""
    parser = build_train_parser(argv=[])
    for action in parser._actions:
        if action.dest == "weights_est_net":
            assert "FusedConv2d" not in (action.choices or [])
            assert "Simple" in action.choices and "UNet" in action.choices
            break
    else:
        raise AssertionError("--weights_est_net flag missing")


def test_intlist_and_bool_flag_parsing():
    """str2intlist / str2bool semantics of the reflective system
    (reference utils/args.py:166-175)."""
    argv = ["--weights_est_net_num_ch", "[8,4]",
            "--interp_net_use_bias", "False",
            "--interp_net_shared_encoder", "true"]
    parser = build_train_parser(argv=argv)
    args = finalize_args(parser.parse_args(
        ["--name", "t", "--model", "raft_nc_dbl"] + argv))
    assert args.weights_est_net_num_ch == [8, 4]
    assert args.interp_net_use_bias is False
    assert args.interp_net_shared_encoder is True


def test_filter_list_of_strings():
    """Reference args.py:152-156 helper API."""
    from flowhip.config.args import filter_list_of_strings

    lst = ["alpha", "beta", "alpine", "gamma"]
    out = filter_list_of_strings(lst, include=("al*",), exclude=("alpine",))
    assert sorted(out) == ["alpha"]
    assert sorted(filter_list_of_strings(lst)) == sorted(lst)
