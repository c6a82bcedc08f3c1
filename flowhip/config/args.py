"""CLI flag surface — reproduces the reference's flag names exactly.

The reference generates `--<name>_<param>` flags by reflecting over a chosen
class's __init__ signature (`core/utils/args.py:8-114`), invoked for three
module families: final_upsampling (upsampler classes), interp_net (nconv
classes), weights_est_net (Simple/UNet). This module reimplements that
mechanism on `inspect.signature` (getargspec is long gone) and assembles the
full train/eval parsers, including the plain trainer knobs of the reference
train.py:264-297 plus the framework's DDP/bench additions.

Defect fixes defined here:
- `--dataset` exists on the TRAIN parser too (default derived from --stage)
  so the BN-off-for-KITTI behavior works without the reference's crash.
"""

import argparse
import fnmatch
import inspect
import itertools
import sys


def str2bool(v):
    if isinstance(v, bool):
        return v
    if v.lower() in ("yes", "true", "t", "y", "1"):
        return True
    if v.lower() in ("no", "false", "f", "n", "0"):
        return False
    raise argparse.ArgumentTypeError("Boolean value expected.")


def str2intlist(v):
    return [int(x.strip()) for x in v.strip()[1:-1].split(",")]


def module_classes_to_dict(module, include_classes="*", exclude_classes=()):
    if isinstance(include_classes, str):
        include_classes = [include_classes]
    if isinstance(exclude_classes, str):
        exclude_classes = [exclude_classes]

    # only classes DEFINED in the module: imported helpers (e.g. conv
    # wrappers) must not leak into the CLI choice surface
    items = {name: getattr(module, name) for name in dir(module)
             if inspect.isclass(getattr(module, name))
             and getattr(module, name).__module__ == module.__name__}

    matched = set(itertools.chain.from_iterable(
        fnmatch.filter(items.keys(), pat) for pat in include_classes))
    excluded = set(itertools.chain.from_iterable(
        fnmatch.filter(items.keys(), pat) for pat in exclude_classes))
    return {name: items[name] for name in matched - excluded}


def _type_from_default(value):
    if isinstance(value, bool):
        return str2bool
    if isinstance(value, list):
        return str2intlist
    return type(value)


def add_arguments_for_module(parser, module, name, default_class,
                             add_class_argument=True, include_classes="*",
                             exclude_classes=(), exclude_params=("self", "args"),
                             param_defaults=None, forced_default_types=None,
                             unknown_default_types=None, argv=None):
    """Add `--<name>` (class choice) and `--<name>_<param>` flags derived
    from the chosen class's constructor signature.

    Fresh implementation of the reference's reflective generator
    (core/utils/args.py:8-114) on inspect.signature.
    """
    param_defaults = param_defaults or {}
    forced_default_types = forced_default_types or {}
    unknown_default_types = unknown_default_types or {}
    argv = sys.argv[1:] if argv is None else argv

    module_dict = module_classes_to_dict(module, include_classes, exclude_classes)

    if add_class_argument:
        parser.add_argument("--%s" % name, type=str, default=default_class,
                            choices=sorted(module_dict.keys()))
        known_args = parser.parse_known_args(argv)[0]
    else:
        tmp = argparse.ArgumentParser()
        tmp.add_argument("--%s" % name, type=str, default=default_class,
                         choices=sorted(module_dict.keys()))
        known_args = tmp.parse_known_args(argv)[0]

    class_name = vars(known_args)[name]
    if class_name is None:
        return

    sig = inspect.signature(module_dict[class_name].__init__)
    for argname, param in sig.parameters.items():
        if argname in exclude_params or param.kind in (
                inspect.Parameter.VAR_POSITIONAL, inspect.Parameter.VAR_KEYWORD):
            continue

        sub = "%s_%s" % (name, argname)
        if argname in param_defaults:
            parser.add_argument("--%s" % sub,
                                type=_type_from_default(param_defaults[argname]),
                                default=param_defaults[argname])
        elif param.default is not inspect.Parameter.empty:
            argtype = forced_default_types.get(argname,
                                               _type_from_default(param.default))
            parser.add_argument("--%s" % sub, type=argtype, default=param.default)
        elif argname in unknown_default_types:
            parser.add_argument("--%s" % sub, type=unknown_default_types[argname])
        else:
            raise ValueError(
                "Do not know how to handle argument '%s' for class '%s'"
                % (argname, name))


def add_ncup_module_flags(parser, argv=None):
    """The three reflective families of the reference CLI
    (train.py:299-342, evaluate.py:198-241).

    Unlike the reference (which crashes on a bare `--model raft_nc_dbl` run
    because no flags exist until a class is chosen), the families default to
    the shipped NCUP configuration — the canonical scripts
    pass every flag explicitly, so their behavior is unchanged."""
    from ..nn import interp_weights_est as interp_weights_est_mod
    from ..nn import nconv as nconv_mod
    from ..nn import upsampler as upsampler_mod

    add_arguments_for_module(
        parser, upsampler_mod, name="final_upsampling",
        default_class="NConvUpsampler",
        exclude_classes=["_*"],
        exclude_params=["self", "args", "interpolation_net", "weights_est_net",
                        "size"],
        param_defaults={"scale": 4},
        forced_default_types={"scale": int,
                              "use_data_for_guidance": str2bool,
                              "channels_to_batch": str2bool,
                              "use_residuals": str2bool,
                              "est_on_high_res": str2bool},
        argv=argv)

    add_arguments_for_module(
        parser, nconv_mod, name="interp_net", default_class="NConvUNet",
        exclude_classes=["_*"],
        exclude_params=["self", "args"],
        param_defaults={"num_downsampling": 1, "use_double_conv": False},
        forced_default_types={"encoder_filter_sz": int,
                              "decoder_filter_sz": int,
                              "out_filter_sz": int,
                              "use_double_conv": str2bool,
                              "use_bias": str2bool},
        argv=argv)

    add_arguments_for_module(
        parser, interp_weights_est_mod, name="weights_est_net",
        default_class="Simple", exclude_classes=["_*"],
        exclude_params=["self", "args", "out_ch", "final_act"],
        param_defaults={"num_ch": [64, 32], "filter_sz": [3, 3, 1],
                        "dilation": [1, 1, 1]},
        unknown_default_types={"num_ch": str2intlist,
                               "filter_sz": str2intlist},
        forced_default_types={"dilation": str2intlist},
        argv=argv)


def build_train_parser(argv=None):
    """Full training CLI (reference train.py:264-343 + framework additions)."""
    parser = argparse.ArgumentParser()
    parser.add_argument("--name", default="raft", help="name your experiment")
    parser.add_argument("--model", default="raft", help="model to train")
    parser.add_argument("--stage", help="determines which dataset to use for training")
    parser.add_argument("--restore_ckpt", help="restore checkpoint")
    parser.add_argument("--small", action="store_true", help="use small model")
    parser.add_argument("--validation", type=str, nargs="+", default=[])

    parser.add_argument("--lr", type=float, default=0.00002)
    parser.add_argument("--num_steps", type=int, default=100000)
    parser.add_argument("--batch_size", type=int, default=6,
                        help="GLOBAL batch size (split across DDP ranks)")
    parser.add_argument("--image_size", type=int, nargs="+", default=[384, 512])
    parser.add_argument("--gpus", type=int, nargs="+", default=[0, 1],
                        help="accepted for reference-CLI compatibility; "
                        "device placement is per-rank under DDP")
    parser.add_argument("--mixed_precision", action="store_true",
                        help="bf16 autocast")

    parser.add_argument("--iters", type=int, default=12)
    parser.add_argument("--wdecay", type=float, default=.00005)
    parser.add_argument("--epsilon", type=float, default=1e-8)
    parser.add_argument("--clip", type=float, default=1.0)
    parser.add_argument("--dropout", type=float, default=0.0)
    parser.add_argument("--add_noise", action="store_true")
    parser.add_argument("--gamma", type=float, default=0.8, help="exponential weighting")

    parser.add_argument("--optimizer", default="adamw")
    parser.add_argument("--scheduler", default="cyclic")
    parser.add_argument("--scheduler_step", type=int, default=20000)
    parser.add_argument("--upsampler_bi", action="store_true")
    parser.add_argument("--align_corners", action="store_true")
    parser.add_argument("--freeze_raft", action="store_true")
    parser.add_argument("--load_pretrained", default=None)
    parser.add_argument("--compressed_ft", action="store_true")

    # framework additions (not in the reference CLI)
    parser.add_argument("--dataset", default=None,
                        help="dataset flavor for model config (BN on for "
                        "sintel); derived from --stage when omitted")
    parser.add_argument("--num_workers", type=int, default=4)
    parser.add_argument("--resume_full", default=None,
                        help="resume full train state (model/opt/sched/step)")
    parser.add_argument("--seed", type=int, default=1234)
    parser.add_argument("--profile_dir", default=None,
                        help="rank-0 torch.profiler capture: wait 1 / warmup "
                        "2 / active 3 steps, chrome trace + op table written "
                        "here (SURVEY.md §5.1)")

    add_ncup_module_flags(parser, argv=argv)
    return parser


def build_eval_parser(argv=None):
    """Evaluation CLI (reference evaluate.py:185-241)."""
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", help="model name")
    parser.add_argument("--restore_ckpt", help="restore checkpoint")
    parser.add_argument("--dataset", help="dataset for evaluation")
    parser.add_argument("--small", action="store_true")
    parser.add_argument("--mixed_precision", action="store_true")
    parser.add_argument("--upsampler_bi", action="store_true")
    parser.add_argument("--align_corners", action="store_true")
    parser.add_argument("--load_pretrained", default=None)
    parser.add_argument("--freeze_raft", action="store_true")
    parser.add_argument("--compressed_ft", action="store_true")
    parser.add_argument("--iters", type=int, default=None,
                        help="override per-dataset default refinement iters")

    add_ncup_module_flags(parser, argv=argv)
    return parser


def finalize_args(args):
    """Post-parse fixups: derive --dataset from --stage (keeps the reference's
    BN-for-sintel behavior without its crash — SURVEY.md §2.9 quirk 2)."""
    if getattr(args, "dataset", None) is None:
        args.dataset = getattr(args, "stage", None)
    return args


def default_ncup_args(**overrides):
    """The shipped NCUP configuration (SURVEY.md §2.5 — identical across all
    five reference shell scripts) as a Namespace, for programmatic model
    construction (tests, bench)."""
    ns = argparse.Namespace(
        model="raft_nc_dbl", small=False, dropout=0.0, mixed_precision=False,
        dataset="sintel", gamma=0.85, iters=12,
        load_pretrained=None, freeze_raft=False,
        final_upsampling="NConvUpsampler",
        final_upsampling_scale=4,
        final_upsampling_use_data_for_guidance=True,
        final_upsampling_channels_to_batch=True,
        final_upsampling_use_residuals=False,
        final_upsampling_est_on_high_res=False,
        interp_net="NConvUNet",
        interp_net_channels_multiplier=2,
        interp_net_num_downsampling=1,
        interp_net_data_pooling="conf_based",
        interp_net_encoder_filter_sz=5,
        interp_net_decoder_filter_sz=3,
        interp_net_out_filter_sz=1,
        interp_net_shared_encoder=True,
        interp_net_use_double_conv=False,
        interp_net_use_bias=False,
        weights_est_net="Simple",
        weights_est_net_num_ch=[64, 32],
        weights_est_net_filter_sz=[3, 3, 1],
        weights_est_net_dilation=[1, 1, 1],
    )
    for k, v in overrides.items():
        setattr(ns, k, v)
    return ns


def filter_list_of_strings(lst, include=("*",), exclude=()):
    """fnmatch include/exclude filter (reference args.py:152-156 API)."""
    import fnmatch
    import itertools

    matches = set(itertools.chain.from_iterable(
        fnmatch.filter(lst, pat) for pat in include))
    nomatch = set(itertools.chain.from_iterable(
        fnmatch.filter(lst, pat) for pat in exclude))
    return list(matches - nomatch)
