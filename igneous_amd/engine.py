"""ctypes host over libmeshgine.so — the MI355X product compute path.

This is the only mesher the product path uses. If the HIP extension is
missing, fails to load, or no GPU is present, every compute call raises
loudly — there is NO CPU fallback (the CPU oracle under oracle/ is test
infrastructure and must never be routed here).

C ABI: include/meshgine.h. Built by __graft_entry__.build() into
igneous_amd/csrc/libmeshgine.so (in-tree so it travels to GPU hosts).
"""
from __future__ import annotations

import ctypes
import os
import threading
from typing import Optional

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(_HERE, "csrc", "libmeshgine.so")

MG_U32, MG_U64 = 0, 1
MG_FLAG_DEVICE_ONLY = 1
MG_FLAG_SKIP_H2D = 2

_EXPORTED_SYMBOLS = [
    "mg_init", "mg_destroy", "mg_mesh_chunk", "mg_meshset_free",
    "mg_get_stats", "mg_last_error", "mg_device_count", "mg_version",
]


class _MgMesh(ctypes.Structure):
    _fields_ = [
        ("label", ctypes.c_uint64),
        ("nverts", ctypes.c_uint32),
        ("ntris", ctypes.c_uint32),
        ("verts", ctypes.POINTER(ctypes.c_float)),
        ("faces", ctypes.POINTER(ctypes.c_uint32)),
    ]


class _MgMeshSet(ctypes.Structure):
    _fields_ = [
        ("nmeshes", ctypes.c_uint32),
        ("meshes", ctypes.POINTER(_MgMesh)),
        ("verts_base", ctypes.POINTER(ctypes.c_float)),
        ("faces_base", ctypes.POINTER(ctypes.c_uint32)),
        ("total_verts", ctypes.c_uint64),
        ("total_tris", ctypes.c_uint64),
        ("labels_arr", ctypes.POINTER(ctypes.c_uint64)),
        ("voff_arr", ctypes.POINTER(ctypes.c_uint32)),
        ("nv_arr", ctypes.POINTER(ctypes.c_uint32)),
        ("foff_arr", ctypes.POINTER(ctypes.c_uint32)),
        ("nf_arr", ctypes.POINTER(ctypes.c_uint32)),
    ]


class MgStats(ctypes.Structure):
    _fields_ = [
        ("ms_h2d", ctypes.c_double),
        ("ms_count", ctypes.c_double),
        ("ms_scan", ctypes.c_double),
        ("ms_emit", ctypes.c_double),
        ("ms_partition", ctypes.c_double),
        ("ms_weld", ctypes.c_double),
        ("ms_simplify", ctypes.c_double),
        ("ms_d2h", ctypes.c_double),
        ("ms_total", ctypes.c_double),
        ("total_tris", ctypes.c_uint64),
        ("total_verts", ctypes.c_uint64),
        ("n_labels", ctypes.c_uint64),
        ("bytes_read_algorithmic", ctypes.c_uint64),
    ]

    def as_dict(self) -> dict:
        return {name: getattr(self, name) for name, _t in self._fields_}


_lib = None
_lib_lock = threading.Lock()


def load_library() -> ctypes.CDLL:
    """dlopen the engine; raises with a clear message if absent."""
    global _lib
    with _lib_lock:
        if _lib is not None:
            return _lib
        if not os.path.exists(SO_PATH):
            raise RuntimeError(
                f"meshgine HIP engine not built: {SO_PATH} missing. "
                f"Run __graft_entry__.build() (hipcc --offload-arch=gfx950). "
                f"There is no CPU fallback.")
        lib = ctypes.CDLL(SO_PATH)
        for sym in _EXPORTED_SYMBOLS:
            if not hasattr(lib, sym):
                raise RuntimeError(f"{SO_PATH} missing symbol {sym}")
        lib.mg_init.restype = ctypes.c_void_p
        lib.mg_init.argtypes = [ctypes.c_int]
        lib.mg_destroy.argtypes = [ctypes.c_void_p]
        lib.mg_mesh_chunk.restype = ctypes.c_int
        lib.mg_mesh_chunk.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_float, ctypes.c_float, ctypes.c_float,
            ctypes.c_uint32, ctypes.c_float, ctypes.c_int,
            ctypes.c_uint64, ctypes.c_uint32,
            ctypes.POINTER(ctypes.POINTER(_MgMeshSet)),
        ]
        lib.mg_meshset_free.argtypes = [ctypes.POINTER(_MgMeshSet)]
        lib.mg_get_stats.restype = ctypes.c_int
        lib.mg_get_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(MgStats)]
        lib.mg_last_error.restype = ctypes.c_char_p
        lib.mg_last_error.argtypes = [ctypes.c_void_p]
        lib.mg_device_count.restype = ctypes.c_int
        lib.mg_version.restype = ctypes.c_char_p
        lib.mg_simplify_mesh.restype = ctypes.c_int
        lib.mg_simplify_mesh.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_uint32,
            ctypes.c_void_p, ctypes.c_uint32,
            ctypes.c_uint32, ctypes.c_float,
            ctypes.POINTER(ctypes.POINTER(ctypes.c_float)),
            ctypes.POINTER(ctypes.c_uint32),
            ctypes.POINTER(ctypes.POINTER(ctypes.c_uint32)),
            ctypes.POINTER(ctypes.c_uint32),
        ]
        _lib = lib
        return lib


# When True, Engine.get() returns one context (and HIP stream) per
# (device, thread): concurrent MeshTasks on one GPU overlap their
# H2D / kernels / D2H — the reference's "one HIP stream per in-flight
# chunk" fan-out. Set by igneous_amd.dispatch.
PER_THREAD_CTX = False


class Engine:
    """One mg_ctx on one HIP device."""

    _per_device: dict = {}
    _cls_lock = threading.Lock()

    def __init__(self, device_id: int = 0):
        lib = load_library()
        self.lib = lib
        self.device_id = int(device_id)
        self._lock = threading.Lock()
        self.ctx = lib.mg_init(self.device_id)
        if not self.ctx:
            raise RuntimeError(
                f"mg_init({device_id}) failed — no usable HIP device. "
                f"The meshgine product path requires an MI355X GPU.")

    @classmethod
    def get(cls, device_id: int = 0) -> "Engine":
        key = (device_id, threading.get_ident()) if PER_THREAD_CTX \
            else device_id
        with cls._cls_lock:
            eng = cls._per_device.get(key)
            if eng is None:
                eng = cls(device_id)
                cls._per_device[key] = eng
            return eng

    def mesh_chunk(self, labels: np.ndarray, resolution=(1.0, 1.0, 1.0),
                   reduction_factor: int = 0, max_error: float = 40.0,
                   voxel_centered: bool = True,
                   device_only: bool = False,
                   skip_h2d: bool = False,
                   dust_threshold: int = 0,
                   copy: bool = True) -> dict:
        """GPU counterpart of oracle.mesh_chunk: F-order (sx,sy,sz)
        uint32/uint64 labels -> {label: (verts (V,3) f32 nm, faces (F,3) u32)},
        labels ascending. Under device_only returns {} (stats still filled).

        copy=False returns VIEWS into this context's pinned staging
        buffers — zero host copies, but the arrays are only valid until
        the next mesh_chunk on this Engine. Use when results are consumed
        (encoded/written) before the next call, as MeshTask does."""
        labels = np.asfortranarray(labels)
        if labels.dtype == np.uint32:
            dtype = MG_U32
        elif labels.dtype == np.uint64:
            dtype = MG_U64
        else:
            raise ValueError(f"unsupported label dtype {labels.dtype}")
        sx, sy, sz = labels.shape
        flags = (MG_FLAG_DEVICE_ONLY if device_only else 0) | \
                (MG_FLAG_SKIP_H2D if skip_h2d else 0)
        out = ctypes.POINTER(_MgMeshSet)()
        with self._lock:
            rc = self.lib.mg_mesh_chunk(
                self.ctx, labels.ctypes.data_as(ctypes.c_void_p),
                sx, sy, sz, dtype,
                float(resolution[0]), float(resolution[1]), float(resolution[2]),
                int(reduction_factor), float(max_error),
                int(bool(voxel_centered)), int(dust_threshold), flags,
                ctypes.byref(out))
            if rc != 0:
                err = self.lib.mg_last_error(self.ctx)
                raise RuntimeError(
                    f"mg_mesh_chunk failed rc={rc}: "
                    f"{err.decode() if err else 'unknown'}")
            result = {}
            try:
                ms = out.contents
                n = int(ms.nmeshes)
                if n:
                    # one GIL-releasing memmove per flat buffer (the flat
                    # buffers are ctx-owned pinned staging, valid until
                    # the next call on this ctx), then per-label views
                    tv, tt = int(ms.total_verts), int(ms.total_tris)
                    if copy:
                        all_v = np.empty((tv, 3), dtype=np.float32)
                        all_f = np.empty((tt, 3), dtype=np.uint32)
                        if tv:
                            ctypes.memmove(all_v.ctypes.data, ms.verts_base,
                                           tv * 12)
                        if tt:
                            ctypes.memmove(all_f.ctypes.data, ms.faces_base,
                                           tt * 12)
                    else:
                        all_v = np.ctypeslib.as_array(
                            ms.verts_base, shape=(max(tv, 1), 3))[:tv]
                        all_f = np.ctypeslib.as_array(
                            ms.faces_base, shape=(max(tt, 1), 3))[:tt]
                    labels = np.ctypeslib.as_array(
                        ms.labels_arr, shape=(n,)).tolist()
                    voff = np.ctypeslib.as_array(
                        ms.voff_arr, shape=(n,)).tolist()
                    nv = np.ctypeslib.as_array(
                        ms.nv_arr, shape=(n,)).tolist()
                    foff = np.ctypeslib.as_array(
                        ms.foff_arr, shape=(n,)).tolist()
                    nf = np.ctypeslib.as_array(
                        ms.nf_arr, shape=(n,)).tolist()
                    result = {
                        lab: (all_v[vo:vo + kv], all_f[fo:fo + kf])
                        for lab, vo, kv, fo, kf
                        in zip(labels, voff, nv, foff, nf)
                    }
            finally:
                self.lib.mg_meshset_free(out)
        return result

    def simplify_mesh(self, verts: np.ndarray, faces: np.ndarray,
                      reduction_factor: int,
                      max_error: float = 1e30):
        """Standalone quadric simplification of one mesh on the GPU
        (mg_simplify_mesh; the multires LOD chain's simplify_fqmr
        replacement, multires.py:342). Returns owned (verts, faces)."""
        v = np.ascontiguousarray(verts, dtype=np.float32)
        f = np.ascontiguousarray(faces, dtype=np.uint32)
        ov = ctypes.POINTER(ctypes.c_float)()
        of = ctypes.POINTER(ctypes.c_uint32)()
        onv = ctypes.c_uint32()
        ont = ctypes.c_uint32()
        rc = self.lib.mg_simplify_mesh(
            self.ctx,
            v.ctypes.data_as(ctypes.c_void_p), v.shape[0],
            f.ctypes.data_as(ctypes.c_void_p), f.shape[0],
            int(reduction_factor), float(max_error),
            ctypes.byref(ov), ctypes.byref(onv),
            ctypes.byref(of), ctypes.byref(ont))
        if rc != 0:
            raise RuntimeError(
                f"mg_simplify_mesh failed rc={rc}: "
                f"{self.lib.mg_last_error(self.ctx).decode()}")
        out_v = np.ctypeslib.as_array(ov, shape=(onv.value, 3)).copy()
        out_f = np.ctypeslib.as_array(of, shape=(ont.value, 3)).copy()
        return out_v, out_f

    def stats(self) -> dict:
        s = MgStats()
        rc = self.lib.mg_get_stats(self.ctx, ctypes.byref(s))
        if rc != 0:
            raise RuntimeError("mg_get_stats failed")
        return s.as_dict()


def mesh_chunk(labels: np.ndarray, resolution=(1.0, 1.0, 1.0),
               reduction_factor: int = 0, max_error: float = 40.0,
               voxel_centered: bool = True, device_id: Optional[int] = None,
               dust_threshold: int = 0,
               copy: bool = True) -> dict:
    """Module-level product mesher (the default MeshTask path)."""
    if device_id is None:
        device_id = int(os.environ.get("MESHGINE_DEVICE",
                                       os.environ.get("LOCAL_RANK", "0")))
    return Engine.get(device_id).mesh_chunk(
        labels, resolution, reduction_factor, max_error, voxel_centered,
        dust_threshold=dust_threshold, copy=copy)


# MeshTask checks this to delegate dust_threshold preprocessing to the
# device (three HIP volume passes) instead of a host numpy unique/mask
mesh_chunk.handles_dust = True


def simplify_mesh(mesh, target_count: int, max_error: float = 1e30,
                  device_id: Optional[int] = None):
    """Module-level LOD simplifier: Mesh -> Mesh at ~target_count faces
    (reduction_factor = ntris // target; <=1 returns the mesh as-is)."""
    from .meshes import Mesh
    nt = int(mesh.faces.shape[0])
    target = max(int(target_count), 1)
    if nt <= target:
        return Mesh(mesh.vertices.copy(), mesh.faces.copy(), id=mesh.id)
    rf = max(nt // target, 2)
    if device_id is None:
        device_id = int(os.environ.get("MESHGINE_DEVICE",
                                       os.environ.get("LOCAL_RANK", "0")))
    v, f = Engine.get(device_id).simplify_mesh(
        mesh.vertices, mesh.faces, rf, max_error)
    return Mesh(v, f, id=mesh.id)
