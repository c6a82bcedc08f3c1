from .geometry import (
    InputPadder,
    bilinear_sampler,
    coords_grid,
    forward_interpolate,
    upflow8,
)

__all__ = [
    "InputPadder",
    "bilinear_sampler",
    "coords_grid",
    "forward_interpolate",
    "upflow8",
]
