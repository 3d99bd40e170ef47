"""numpy equivalents of the fastremap calls on the MeshTask host path
(/root/reference/igneous/tasks/mesh/mesh.py:201-206,318-320,368-369).

The reference also calls fastremap.renumber (mesh.py:206) purely to shrink
the dtype and give the mesher small ids, inverting the map afterwards
(mesh.py:207, 371-383). Our engine hashes raw uint32/uint64 labels directly,
so the renumber+invert round trip is the identity composition and is
intentionally omitted from the product path (documented in DESIGN.md).
"""
from __future__ import annotations

import numpy as np


def mask(data: np.ndarray, labels, in_place: bool = True) -> np.ndarray:
    """Zero out the given labels (fastremap.mask, mesh.py:204,320)."""
    if len(labels) == 0:
        return data
    sel = np.isin(data, np.asarray(list(labels), dtype=data.dtype))
    out = data if in_place else data.copy()
    out[sel] = 0
    return out


def mask_except(data: np.ndarray, labels, in_place: bool = True) -> np.ndarray:
    """Zero out everything except the given labels
    (fastremap.mask_except, mesh.py:201,355,368)."""
    sel = np.isin(data, np.asarray(list(labels), dtype=data.dtype))
    out = data if in_place else data.copy()
    out[~sel] = 0
    return out


def remap(data: np.ndarray, table: dict, in_place: bool = True) -> np.ndarray:
    """Apply {orig: new} label mapping (fastremap.remap, mesh.py:369).
    Labels absent from the table are left unchanged (the reference masks
    them away first, mesh.py:368)."""
    out = data if in_place else data.copy()
    if not table:
        return out
    keys = np.fromiter(table.keys(), dtype=data.dtype, count=len(table))
    vals = np.fromiter(table.values(), dtype=data.dtype, count=len(table))
    order = np.argsort(keys)
    keys, vals = keys[order], vals[order]
    idx = np.searchsorted(keys, out)
    idx[idx >= len(keys)] = 0
    hit = keys[idx] == out
    out[hit] = vals[idx[hit]]
    return out


def unique(data: np.ndarray, return_counts: bool = False):
    """fastremap.unique equivalent (mesh.py:318)."""
    return np.unique(data, return_counts=return_counts)
