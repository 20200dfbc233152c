"""Nested parameter handling with unknown-key detection.

Mirrors the reference's two-tier config system (amgcl/util.hpp:103-183,
AMGCL_PARAMS_IMPORT_VALUE/CHILD + check_params): every component declares a
defaults() dict; user-supplied dicts (possibly parsed from JSON) are merged
onto the defaults and any unknown key raises immediately.
"""
import copy
import json


class UnknownParameter(ValueError):
    pass


def merge_params(defaults, prm, path="", opaque=()):
    """Merge user dict `prm` onto `defaults`, raising on unknown keys.
    Keys in `opaque` (nested component configs, e.g. a sub-solver tree)
    are taken verbatim without recursive validation — the sub-component
    validates its own tree when constructed."""
    out = copy.deepcopy(defaults)
    if prm is None:
        return out
    if not isinstance(prm, dict):
        raise TypeError(f"params at '{path or '<root>'}' must be a dict")
    for key, value in prm.items():
        if key not in out:
            raise UnknownParameter(f"unknown parameter '{path + key}'")
        if key in opaque or key.endswith("_raw") or not isinstance(out[key], dict):
            out[key] = value
        else:
            out[key] = merge_params(out[key], value, path + key + ".")
    return out


def from_json(defaults, text_or_path):
    """Load params from a JSON string or file path and merge onto defaults."""
    try:
        prm = json.loads(text_or_path)
    except (ValueError, TypeError):
        with open(text_or_path) as f:
            prm = json.load(f)
    return merge_params(defaults, prm)


def set_kv(prm, key, value):
    """Apply a 'a.b.c=value' style CLI override onto a nested dict in place."""
    parts = key.split(".")
    d = prm
    for p in parts[:-1]:
        d = d.setdefault(p, {})
    for cast in (int, float):
        try:
            value = cast(value)
            break
        except (TypeError, ValueError):
            continue
    if value in ("true", "false"):
        value = value == "true"
    d[parts[-1]] = value
    return prm
