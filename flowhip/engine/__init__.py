from . import checkpoints, distributed, evaluate, logger, train  # noqa: F401
