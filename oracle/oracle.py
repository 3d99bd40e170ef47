"""ctypes wrapper around oracle/liboracle.so (the CPU oracle).

TEST INFRASTRUCTURE ONLY — may be used by tests/, __graft_entry__.smoke()
(as the checker) and bench.py's cpu_baseline leg; never by the product path.
See oracle/mc_oracle.c header for the parity statement (parity vs zmesh is
unpinned; the HIP engine is pinned bit-exact against this oracle).
"""
import ctypes
import os
import subprocess

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "liboracle.so")
_lib = None


def build(force: bool = False) -> str:
    """Compile liboracle.so with gcc (idempotent)."""
    srcs = [os.path.join(_HERE, "mc_oracle.c"), os.path.join(_HERE, "simplify.c")]
    hdr = os.path.join(_HERE, "..", "igneous_amd", "csrc", "mc_table.h")
    if not force and os.path.exists(_SO):
        newest = max(os.path.getmtime(p) for p in srcs + [hdr])
        if os.path.getmtime(_SO) >= newest:
            return _SO
    cmd = [
        "gcc", "-O2", "-shared", "-fPIC",
        "-I", os.path.dirname(hdr),
        *srcs, "-o", _SO, "-lm",
    ]
    subprocess.run(cmd, check=True)
    return _SO


class _Mesh(ctypes.Structure):
    _fields_ = [
        ("label", ctypes.c_uint64),
        ("nverts", ctypes.c_uint32),
        ("ntris", ctypes.c_uint32),
        ("verts", ctypes.POINTER(ctypes.c_float)),
        ("faces", ctypes.POINTER(ctypes.c_uint32)),
    ]


class _MeshSet(ctypes.Structure):
    _fields_ = [
        ("nmeshes", ctypes.c_uint32),
        ("meshes", ctypes.POINTER(_Mesh)),
    ]


def _get_lib():
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_SO)
        _lib.omc_mesh_chunk.restype = ctypes.c_int
        _lib.omc_mesh_chunk.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, ctypes.c_float, ctypes.c_float, ctypes.c_float,
            ctypes.c_uint32, ctypes.c_float, ctypes.c_int,
            ctypes.POINTER(ctypes.POINTER(_MeshSet)),
        ]
        _lib.omc_meshset_free.argtypes = [ctypes.POINTER(_MeshSet)]
        _lib.omc_simplify_mesh.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32),
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32),
            ctypes.c_uint32, ctypes.c_float,
        ]
    return _lib


def mesh_chunk(labels: np.ndarray, resolution=(1.0, 1.0, 1.0),
               reduction_factor: int = 0, max_error: float = 40.0,
               voxel_centered: bool = True) -> dict:
    """Run the oracle on an F-order (sx,sy,sz) uint32/uint64 label array.

    Returns {label: (verts float32 (V,3) chunk-local nm, faces uint32 (F,3))},
    labels in ascending order. Mirrors the reference call sequence
    zmesh.Mesher.mesh + .get for every id (mesh.py:245, 374-381).
    """
    lib = _get_lib()
    labels = np.asfortranarray(labels)
    if labels.dtype == np.uint32:
        dtype = 0
    elif labels.dtype == np.uint64:
        dtype = 1
    else:
        raise ValueError(f"unsupported dtype {labels.dtype}")
    sx, sy, sz = labels.shape
    out = ctypes.POINTER(_MeshSet)()
    rc = lib.omc_mesh_chunk(
        labels.ctypes.data_as(ctypes.c_void_p), sx, sy, sz, dtype,
        float(resolution[0]), float(resolution[1]), float(resolution[2]),
        int(reduction_factor), float(max_error), int(bool(voxel_centered)),
        ctypes.byref(out))
    if rc != 0:
        raise RuntimeError(f"omc_mesh_chunk failed rc={rc}")
    result = {}
    try:
        ms = out.contents
        for i in range(ms.nmeshes):
            m = ms.meshes[i]
            v = np.ctypeslib.as_array(m.verts, shape=(m.nverts, 3)).copy()
            f = np.ctypeslib.as_array(m.faces, shape=(m.ntris, 3)).copy()
            result[int(m.label)] = (v, f)
    finally:
        lib.omc_meshset_free(out)
    return result


def simplify_mesh(verts: np.ndarray, faces: np.ndarray,
                  reduction_factor: int, max_error: float = 1e30):
    """Standalone quadric edge-collapse on one mesh (the same
    omc_simplify_mesh the per-label path runs; checker for the engine's
    mg_simplify_mesh used by the multires LOD chain)."""
    lib = _get_lib()
    v = np.ascontiguousarray(verts, dtype=np.float32).copy()
    f = np.ascontiguousarray(faces, dtype=np.uint32).copy()
    nv = ctypes.c_uint32(v.shape[0])
    nt = ctypes.c_uint32(f.shape[0])
    lib.omc_simplify_mesh(
        v.ctypes.data_as(ctypes.c_void_p), ctypes.byref(nv),
        f.ctypes.data_as(ctypes.c_void_p), ctypes.byref(nt),
        int(reduction_factor), float(max_error))
    return v[:nv.value].copy(), f[:nt.value].copy()
