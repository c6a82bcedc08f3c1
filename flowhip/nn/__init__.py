from .corr import CorrBlock
from .extractor import BasicEncoder, BottleneckBlock, ResidualBlock, SmallEncoder
from .update import (
    BasicMotionEncoder,
    BasicUpdateBlock,
    ConvGRU,
    FlowHead,
    SepConvGRU,
    SmallMotionEncoder,
    SmallUpdateBlock,
)

__all__ = [
    "CorrBlock",
    "BasicEncoder", "SmallEncoder", "ResidualBlock", "BottleneckBlock",
    "FlowHead", "ConvGRU", "SepConvGRU", "SmallMotionEncoder",
    "BasicMotionEncoder", "SmallUpdateBlock", "BasicUpdateBlock",
]
