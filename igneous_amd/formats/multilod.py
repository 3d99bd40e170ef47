"""Neuroglancer multi-resolution mesh manifest (neuroglancer_multilod_draco).

Restates the PUBLISHED neuroglancer multiscale mesh-format manifest that
the reference gets from cloud-volume's MultiLevelPrecomputedMeshManifest
/ to_stored_model_space (/root/reference/igneous/tasks/mesh/multires.py:
24-25,131-141,152-158).

Manifest binary layout (little-endian):
    3 x float32   chunk_shape
    3 x float32   grid_origin
    uint32        num_lods
    num_lods x float32        lod_scales
    num_lods x 3 x float32    vertex_offsets
    num_lods x uint32         num_fragments_per_lod
    for each lod:
        num_fragments x 3 x uint32   fragment octree positions
                                     (rows of x,y,z; C order)
        num_fragments x uint32       fragment byte sizes

Decoded model-space position of a stored vertex v (quantized to
2^vertex_quantization_bits - 1 steps) in fragment f of a given lod:
    grid_origin + vertex_offsets[lod]
      + chunk_shape * lod_scale[lod] * (frag_position[f] + v / (2^vqb-1))
`to_stored_model_space` inverts that mapping.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import List

import numpy as np


@dataclass
class MultiLevelPrecomputedMeshManifest:
    segment_id: int
    chunk_shape: np.ndarray          # (3,) float
    grid_origin: np.ndarray          # (3,) float
    num_lods: int
    lod_scales: List[float]
    vertex_offsets: List[List[float]]
    num_fragments_per_lod: List[int]
    fragment_positions: List         # per lod: sequence of (x,y,z)
    fragment_offsets: List[int] = field(default_factory=list)  # byte sizes

    def __len__(self) -> int:
        return len(self.to_binary())

    def to_binary(self) -> bytes:
        out = bytearray()
        out += np.asarray(self.chunk_shape, dtype="<f4").tobytes()
        out += np.asarray(self.grid_origin, dtype="<f4").tobytes()
        out += struct.pack("<I", int(self.num_lods))
        out += np.asarray(self.lod_scales, dtype="<f4").tobytes()
        out += np.asarray(self.vertex_offsets, dtype="<f4").tobytes()
        out += np.asarray(self.num_fragments_per_lod,
                          dtype="<u4").tobytes()
        k = 0
        for lod in range(int(self.num_lods)):
            n = int(self.num_fragments_per_lod[lod])
            pos = np.asarray(self.fragment_positions[lod],
                             dtype="<u4").reshape(n, 3)
            out += pos.tobytes()
            offs = self.fragment_offsets[k:k + n]
            if len(offs) != n:
                raise ValueError("fragment_offsets incomplete")
            out += np.asarray(offs, dtype="<u4").tobytes()
            k += n
        return bytes(out)

    @classmethod
    def from_binary(cls, binary: bytes,
                    segment_id: int = 0) -> "MultiLevelPrecomputedMeshManifest":
        off = 0
        chunk_shape = np.frombuffer(binary, dtype="<f4", count=3, offset=off)
        off += 12
        grid_origin = np.frombuffer(binary, dtype="<f4", count=3, offset=off)
        off += 12
        (num_lods,) = struct.unpack_from("<I", binary, off)
        off += 4
        lod_scales = np.frombuffer(binary, dtype="<f4", count=num_lods,
                                   offset=off)
        off += 4 * num_lods
        vertex_offsets = np.frombuffer(
            binary, dtype="<f4", count=3 * num_lods,
            offset=off).reshape(num_lods, 3)
        off += 12 * num_lods
        nfrags = np.frombuffer(binary, dtype="<u4", count=num_lods,
                               offset=off)
        off += 4 * num_lods
        fragment_positions = []
        fragment_offsets = []
        for lod in range(num_lods):
            n = int(nfrags[lod])
            pos = np.frombuffer(binary, dtype="<u4", count=3 * n,
                                offset=off).reshape(n, 3)
            off += 12 * n
            offs = np.frombuffer(binary, dtype="<u4", count=n, offset=off)
            off += 4 * n
            fragment_positions.append([tuple(int(x) for x in p)
                                       for p in pos])
            fragment_offsets.extend(int(x) for x in offs)
        return cls(
            segment_id=segment_id,
            chunk_shape=np.asarray(chunk_shape, dtype=np.float32),
            grid_origin=np.asarray(grid_origin, dtype=np.float32),
            num_lods=int(num_lods),
            lod_scales=[float(x) for x in lod_scales],
            vertex_offsets=[list(map(float, vo)) for vo in vertex_offsets],
            num_fragments_per_lod=[int(x) for x in nfrags],
            fragment_positions=fragment_positions,
            fragment_offsets=fragment_offsets,
        )


def to_stored_model_space(vertices: np.ndarray,
                          manifest: MultiLevelPrecomputedMeshManifest,
                          lod: int,
                          vertex_quantization_bits: int,
                          frag: int) -> np.ndarray:
    """Model space -> quantized stored ints for one fragment (inverse of
    the decode mapping in the module docstring). Returns uint32 (N,3)
    clipped to [0, 2^vqb - 1]."""
    vqb = int(vertex_quantization_bits)
    quant_max = float((1 << vqb) - 1)
    chunk = np.asarray(manifest.chunk_shape, dtype=np.float64)
    origin = np.asarray(manifest.grid_origin, dtype=np.float64)
    voffset = np.asarray(manifest.vertex_offsets[lod], dtype=np.float64)
    scale = float(manifest.lod_scales[lod])
    fpos = np.asarray(manifest.fragment_positions[lod][frag],
                      dtype=np.float64)
    v = np.asarray(vertices, dtype=np.float64)
    rel = (v - origin - voffset) / (chunk * scale) - fpos
    q = np.rint(rel * quant_max)
    q = np.clip(q, 0, quant_max)
    return q.astype(np.uint32)
