"""Coarsening strategies.

Coarsening concept (parity: amgcl/coarsening/smoothed_aggregation.hpp:130-242):
an object with .defaults(), .transfer_operators(A) -> (P, R) and
.coarse_operator(A, P, R) -> Ac.
"""
from .aggregation import Aggregation
from .ruge_stuben import RugeStuben
from .smoothed_aggregation import SmoothedAggregation
from .smoothed_aggr_emin import SmoothedAggrEMin

REGISTRY = {
    "aggregation": Aggregation,
    "ruge_stuben": RugeStuben,
    "smoothed_aggregation": SmoothedAggregation,
    "smoothed_aggr_emin": SmoothedAggrEMin,
}


def make_coarsening(prm=None):
    prm = dict(prm or {})
    kind = prm.pop("type", "smoothed_aggregation")
    if kind == "as_scalar":
        from .as_scalar import AsScalar

        return AsScalar(prm)
    if kind not in REGISTRY:
        raise ValueError(f"unknown coarsening '{kind}'")
    return REGISTRY[kind](prm)
