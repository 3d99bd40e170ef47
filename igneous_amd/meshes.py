"""Mesh container matching the zmesh.Mesh subset the reference path uses:
.vertices / .faces arrays, to_precomputed() (mesh.py:448), and
Mesh.concatenate (mesh.py:240)."""
from __future__ import annotations

import struct
from typing import Optional

import numpy as np


class Mesh:
    def __init__(self, vertices: np.ndarray, faces: np.ndarray,
                 normals: Optional[np.ndarray] = None, id: Optional[int] = None):
        self.vertices = np.ascontiguousarray(vertices, dtype=np.float32)
        self.faces = np.ascontiguousarray(faces, dtype=np.uint32)
        self.normals = normals
        self.id = id

    def __len__(self):
        return self.vertices.shape[0]

    def __eq__(self, other):
        return (np.array_equal(self.vertices, other.vertices)
                and np.array_equal(self.faces, other.faces))

    def to_precomputed(self) -> bytes:
        """Neuroglancer legacy ('precomputed') mesh fragment byte layout:
        [uint32 num_vertices][float32 x 3V vertex xyz][uint32 x 3F indices],
        little-endian — the format mesh.to_precomputed() emits for the
        reference at mesh.py:448."""
        nv = self.vertices.shape[0]
        return (struct.pack("<I", nv)
                + self.vertices.astype("<f4", copy=False).tobytes()
                + self.faces.astype("<u4", copy=False).tobytes())

    @classmethod
    def from_precomputed(cls, binary: bytes) -> "Mesh":
        nv = struct.unpack("<I", binary[:4])[0]
        voff = 4
        foff = voff + 12 * nv
        verts = np.frombuffer(binary[voff:foff], dtype="<f4").reshape(nv, 3)
        faces = np.frombuffer(binary[foff:], dtype="<u4").reshape(-1, 3)
        return cls(verts.copy(), faces.copy())

    @classmethod
    def concatenate(cls, *meshes: "Mesh", id: Optional[int] = None) -> "Mesh":
        offs = 0
        vs, fs = [], []
        for m in meshes:
            vs.append(m.vertices)
            fs.append(m.faces + offs)
            offs += m.vertices.shape[0]
        return cls(np.concatenate(vs), np.concatenate(fs), id=id)


# cloud-volume Mesh API conveniences used by the reference merge code
# (multires.py:239 mesh.consolidate(), :551 merge_close_vertices)
def _consolidate(self) -> "Mesh":
    from . import meshops
    return meshops.consolidate(self)


def _merge_close_vertices(self, radius: float = 1e-5) -> "Mesh":
    from . import meshops
    return meshops.merge_close_vertices(self, radius=radius)


def _empty(self) -> bool:
    return self.vertices.shape[0] == 0 or self.faces.shape[0] == 0


Mesh.consolidate = _consolidate
Mesh.merge_close_vertices = _merge_close_vertices
Mesh.empty = _empty
