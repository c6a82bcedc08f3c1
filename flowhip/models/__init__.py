"""Model registry.

The reference CLI accepts four --model names (train.py:170-180). Two of the
model files (`raft_nc`, `raft_nc_sep`) are missing from the reference
snapshot and crash its entry points at import; this
framework defines their behavior: both resolve to the working NCUP variant
`raft_nc_dbl` (a warning is emitted) so every reference invocation runs.
"""

import warnings

from .raft import RAFT
from .raft_nc_dbl import RAFT_NC_DBL
from ..utils.layout import apply_channels_last

MODEL_NAMES = ("raft", "raft_nc", "raft_nc_sep", "raft_nc_dbl")


def build_model(args):
    name = getattr(args, "model", "raft")
    if name == "raft":
        model = RAFT(args)
    elif name == "raft_nc_dbl":
        model = RAFT_NC_DBL(args)
    elif name in ("raft_nc", "raft_nc_sep"):
        warnings.warn(
            f"model {name!r} is missing from the reference snapshot; "
            "resolving to raft_nc_dbl (see SURVEY.md §2.9).")
        model = RAFT_NC_DBL(args)
    else:
        raise NotImplementedError(f"Model {name!r} not found!")
    # GPU path runs NHWC (see utils/layout.py); layout has no effect on
    # state-dict values or CPU semantics.
    return apply_channels_last(model)


__all__ = ["RAFT", "RAFT_NC_DBL", "build_model", "MODEL_NAMES"]
