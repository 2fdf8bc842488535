from .params import (
    MiningRegion,
    MiningMethod,
    NPairLossConfig,
    SolverConfig,
    parse_net_prototxt,
    parse_solver_prototxt,
)
from .prototxt import parse_prototxt, format_prototxt

__all__ = [
    "MiningRegion",
    "MiningMethod",
    "NPairLossConfig",
    "SolverConfig",
    "parse_prototxt",
    "format_prototxt",
    "parse_net_prototxt",
    "parse_solver_prototxt",
]
