"""Minimal Vec/Bbox matching the subset of cloudvolume.lib semantics the
reference's MeshTask path uses (construction, arithmetic, clamp, clone,
to_filename) — see /root/reference/igneous/tasks/mesh/mesh.py:146-160,409.

Written from scratch; cloud-volume is not a dependency of this package.
"""
from __future__ import annotations

import numpy as np


class Vec(np.ndarray):
    """Small integer/float vector with .x/.y/.z accessors."""

    def __new__(cls, *args, dtype=None):
        if len(args) == 1 and hasattr(args[0], "__len__"):
            args = tuple(args[0])
        if dtype is None:
            dtype = np.float64 if any(isinstance(a, float) for a in args) else np.int64
        return np.asarray(args, dtype=dtype).view(cls)

    def clone(self) -> "Vec":
        return np.copy(self).view(Vec)

    @property
    def x(self): return self[0]
    @property
    def y(self): return self[1]
    @property
    def z(self): return self[2]

    def __setattr__(self, name, value):
        if name in ("x", "y", "z"):
            self["xyz".index(name)] = value
        else:
            super().__setattr__(name, value)


class Bbox:
    """Axis-aligned integer bounding box [minpt, maxpt)."""

    def __init__(self, minpt, maxpt):
        self.minpt = Vec(*minpt).clone()
        self.maxpt = Vec(*maxpt).clone()

    @classmethod
    def from_filename(cls, fname: str) -> "Bbox":
        parts = fname.split("_")
        mins, maxs = [], []
        for p in parts[:3]:
            a, b = p.split("-")
            mins.append(int(a)); maxs.append(int(b))
        return cls(mins, maxs)

    def clone(self) -> "Bbox":
        return Bbox(self.minpt, self.maxpt)

    def size3(self) -> Vec:
        return Vec(*(self.maxpt - self.minpt))

    @classmethod
    def clamp(cls, bbx: "Bbox", bounds: "Bbox") -> "Bbox":
        return cls(
            np.minimum(np.maximum(bbx.minpt, bounds.minpt), bounds.maxpt),
            np.minimum(np.maximum(bbx.maxpt, bounds.minpt), bounds.maxpt),
        )

    def to_filename(self, precision=None) -> str:
        # reference naming: "x0-x1_y0-y1_z0-z1" (mesh.py:409 via Bbox.to_filename).
        # precision mirrors cloudvolume Bbox.to_filename(precision) as used by
        # the spatial index (mesh.py:454,460): None → plain integer repr;
        # an int → floats formatted to that many decimals.
        if precision is None:
            return "_".join(
                f"{int(self.minpt[i])}-{int(self.maxpt[i])}" for i in range(3)
            )
        return "_".join(
            f"{float(self.minpt[i]):.{int(precision)}f}-"
            f"{float(self.maxpt[i]):.{int(precision)}f}" for i in range(3)
        )

    def contains_bbox(self, other: "Bbox") -> bool:
        return bool(np.all(other.minpt >= self.minpt) and np.all(other.maxpt <= self.maxpt))

    def volume(self) -> int:
        return int(np.prod(np.maximum(self.maxpt - self.minpt, 0)))

    def __eq__(self, other):
        return (np.array_equal(self.minpt, other.minpt)
                and np.array_equal(self.maxpt, other.maxpt))

    def __repr__(self):
        return f"Bbox({list(self.minpt)}, {list(self.maxpt)})"
