from .datasets import (
    HD1K,
    KITTI,
    FlowDataset,
    FlyingChairs,
    FlyingThings3D,
    MpiSintel,
    SyntheticFlowDataset,
    fetch_dataloader,
)

__all__ = [
    "FlowDataset", "MpiSintel", "FlyingChairs", "FlyingThings3D", "KITTI",
    "HD1K", "SyntheticFlowDataset", "fetch_dataloader",
]
