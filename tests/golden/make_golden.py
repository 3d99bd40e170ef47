#!/usr/bin/env python3
"""Generate the committed golden parity fixtures: small seeded synthetic
chunks + the CPU oracle's meshes for them. The GPU parity tests compare the
HIP engine against these WITHOUT touching /root/reference or re-running the
oracle (both are also cross-checked in the same run when available).

Run from the repo root:  python tests/golden/make_golden.py
"""
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

import oracle  # noqa: E402
from igneous_amd.synth import voronoi_labels  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

CASES = [
    # (name, shape, K, seed, dtype, resolution)
    ("vor32_u32", (32, 32, 32), 8, 7, np.uint32, (16, 16, 40)),
    ("vor48_u64", (48, 48, 48), 40, 11, np.uint64, (16, 16, 40)),
    ("box64_u32", None, None, None, np.uint32, (1, 1, 1)),  # reference fixture
]


def make_case(name, shape, K, seed, dtype, resolution):
    if name == "box64_u32":
        data = np.zeros((65, 65, 65), dtype=np.uint32, order="F")
        data[1:63, 1:63, 1:63] = 1
    else:
        data = voronoi_labels(shape, K, seed, dtype=dtype)
    meshes = oracle.mesh_chunk(data, resolution=resolution)
    out = {"labels": np.asfortranarray(data),
           "resolution": np.asarray(resolution, dtype=np.float32)}
    for label, (v, f) in meshes.items():
        out[f"verts_{label}"] = v
        out[f"faces_{label}"] = f
    path = os.path.join(HERE, f"{name}.npz")
    np.savez_compressed(path, **out)
    nv = sum(v.shape[0] for v, _ in meshes.values())
    nf = sum(f.shape[0] for _, f in meshes.values())
    print(f"{name}: {len(meshes)} labels, {nv} verts, {nf} tris "
          f"-> {os.path.getsize(path) / 1024:.0f} KiB")


if __name__ == "__main__":
    for case in CASES:
        make_case(*case)
