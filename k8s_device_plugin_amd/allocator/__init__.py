from .besteffort import AllocationError, BestEffortPolicy
from .weights import PAIR_WEIGHTS, compute_pair_weights, pair_weight

__all__ = [
    "AllocationError",
    "BestEffortPolicy",
    "PAIR_WEIGHTS",
    "compute_pair_weights",
    "pair_weight",
]
