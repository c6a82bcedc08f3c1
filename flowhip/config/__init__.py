from .args import (
    add_arguments_for_module,
    add_ncup_module_flags,
    build_eval_parser,
    build_train_parser,
    finalize_args,
    str2bool,
    str2intlist,
)

__all__ = [
    "add_arguments_for_module", "add_ncup_module_flags", "build_train_parser",
    "build_eval_parser", "finalize_args", "str2bool", "str2intlist",
]
