"""Mesher — GPU-backed drop-in for the zmesh.Mesher API surface the
reference uses (re-exported from the package root like the reference's
`from zmesh import Mesher`, /root/reference/igneous/__init__.py:1):

    mesher = Mesher(resolution)            # mesh.py:151
    mesher.mesh(data, preserve_order=...)  # mesh.py:245
    for id in mesher.ids():                # mesh.py:374
        m = mesher.get(id, reduction_factor=, max_error=,
                       voxel_centered=True)  # mesh.py:376-381
    mesher.erase(id); mesher.clear()

Implementation note: the engine meshes and (optionally) simplifies every
label in ONE device pass, so `get` lazily triggers a single batched
mg_mesh_chunk with the first call's simplification parameters and serves
the rest from that result (re-running only if parameters change).
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from .meshes import Mesh


class Mesher:
    def __init__(self, voxel_res):
        self.voxel_res = tuple(float(v) for v in voxel_res)
        self._data = None
        self._raw = None          # unsimplified {label: (verts, faces)}
        self._cache = None        # result at self._params
        self._params = None
        self._erased = set()

    # -- zmesh API ---------------------------------------------------------
    def mesh(self, data: np.ndarray, close: bool = False,
             preserve_order: bool = False) -> None:
        if data.ndim == 4:
            data = data[..., 0]
        if close:
            padded = np.zeros(
                tuple(s + 2 for s in data.shape), dtype=data.dtype, order="F")
            padded[1:-1, 1:-1, 1:-1] = data
            data = padded
        self._data = np.asfortranarray(data)
        self._raw = None
        self._cache = None
        self._params = None
        self._erased = set()

    def ids(self):
        self._require_meshed()
        if self._cache is not None:
            labels = self._cache.keys()
        else:
            self._ensure_raw()
            labels = self._raw.keys()
        return sorted(l for l in labels if l not in self._erased)

    def get(self, label: int, reduction_factor: int = 0,
            max_error: Optional[float] = None,
            voxel_centered: bool = True) -> Mesh:
        self._require_meshed()
        if max_error is None:
            max_error = 40.0
        params = (int(reduction_factor), float(max_error), bool(voxel_centered))
        if self._params != params:
            self._cache = self._run(
                reduction_factor=params[0], max_error=params[1],
                voxel_centered=params[2])
            self._params = params
        verts, faces = self._cache[int(label)]
        return Mesh(verts.copy(), faces.copy(), id=int(label))

    def get_mesh(self, label: int, simplification_factor: int = 0,
                 max_simplification_error: Optional[float] = None,
                 voxel_centered: bool = False) -> Mesh:
        """Legacy API variant (reference mesh.py:605-614)."""
        return self.get(label, reduction_factor=simplification_factor,
                        max_error=max_simplification_error,
                        voxel_centered=voxel_centered)

    def erase(self, label: int) -> None:
        self._erased.add(int(label))

    def clear(self) -> None:
        self._data = None
        self._raw = None
        self._cache = None
        self._params = None
        self._erased = set()

    # -- internals ---------------------------------------------------------
    def _require_meshed(self):
        if self._data is None and self._raw is None:
            raise ValueError("call .mesh(data) first")

    def _ensure_raw(self):
        if self._raw is None:
            self._raw = self._run(reduction_factor=0, max_error=0.0,
                                  voxel_centered=True)

    def _run(self, reduction_factor: int, max_error: float,
             voxel_centered: bool) -> dict:
        from .tasks.mesh import _get_mesher
        mesher = _get_mesher()
        return mesher(
            self._data, resolution=self.voxel_res,
            reduction_factor=reduction_factor, max_error=max_error,
            voxel_centered=voxel_centered)


# ---------------------------------------------------------------------------
# zmesh module-level function compat (the reference imports these from
# zmesh in the multires merge: multires.py:321 zmesh.Mesh, :342
# zmesh.simplify_fqmr, :550/:574 zmesh.chunk_mesh).

def simplify_fqmr(mesh, target_count: int, aggressiveness: float = 7.0,
                  preserve_border: bool = False,
                  return_iterations: bool = False,
                  max_iterations: int = 0, K: int = 3, alpha: float = None,
                  update_rate: int = None):
    """zmesh.simplify_fqmr-shaped wrapper over the GPU quadric
    simplifier (mg_simplify_mesh). The pyfqmr aggressiveness/K error
    schedule is NOT replicated: this repo's
    deterministic contract (reduction_factor = ntris//target, unbounded
    max_error) drives the reduction instead; the schedule arguments are
    accepted for signature compatibility."""
    from . import engine
    out = engine.simplify_mesh(mesh, target_count)
    if return_iterations:
        return out, 1
    return out


def chunk_mesh(mesh, scale, offset):
    """zmesh.chunk_mesh equivalent: grid-partition with triangle
    clipping at cell boundaries (igneous_amd.meshops)."""
    from . import meshops
    return meshops.chunk_mesh(mesh, scale, offset)
