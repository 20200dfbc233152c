"""Coarsening strategies.

Coarsening concept (parity: amgcl/coarsening/smoothed_aggregation.hpp:130-242):
an object with .defaults(), .transfer_operators(A) -> (P, R) and
.coarse_operator(A, P, R) -> Ac.
"""
from .aggregation import Aggregation
from .smoothed_aggregation import SmoothedAggregation

REGISTRY = {
    "aggregation": Aggregation,
    "smoothed_aggregation": SmoothedAggregation,
}


def make_coarsening(prm=None):
    prm = dict(prm or {})
    kind = prm.pop("type", "smoothed_aggregation")
    try:
        from .ruge_stuben import RugeStuben  # optional, added later

        REGISTRY.setdefault("ruge_stuben", RugeStuben)
    except ImportError:
        pass
    if kind not in REGISTRY:
        raise ValueError(f"unknown coarsening '{kind}'")
    return REGISTRY[kind](prm)
