"""Minimal Neuroglancer-precomputed volume over file:// — the subset of
CloudVolume the reference's MeshTask path uses (download at
/root/reference/igneous/tasks/mesh/mesh.py:141-182: bounded=False,
fill_missing, F-order (x,y,z,1) output; info/resolution/bounds metadata;
commit_info; from_numpy-style creation for the test harness, mirroring
test/layer_harness.py:31-55).

Precomputed 'raw' chunk encoding stores voxels x-fastest ([c][z][y][x] in
C-order terms == F-order in (x,y,z,c)); chunk files are named
"x0-x1_y0-y1_z0-z1" under the scale key directory.

Written from scratch; cloud-volume is not a dependency of this package.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from .lib import Bbox, Vec
from .storage import CloudFiles


class PrecomputedVolume:
    def __init__(self, cloudpath: str, mip: int = 0, bounded: bool = True,
                 fill_missing: bool = False, parallel: int = 1, **kw):
        self.cloudpath = cloudpath.rstrip("/")
        self.cf = CloudFiles(self.cloudpath)
        self.mip = int(mip)
        self.bounded = bounded
        self.fill_missing = fill_missing
        info = self.cf.get_json("info")
        if info is None:
            raise FileNotFoundError(f"no info file at {cloudpath}")
        self.info = info

    # -- metadata ---------------------------------------------------------
    @property
    def scale(self) -> dict:
        return self.info["scales"][self.mip]

    @property
    def resolution(self) -> Vec:
        return Vec(*self.scale["resolution"])

    @property
    def voxel_offset(self) -> Vec:
        return Vec(*self.scale.get("voxel_offset", [0, 0, 0]))

    @property
    def volume_size(self) -> Vec:
        return Vec(*self.scale["size"])

    @property
    def bounds(self) -> Bbox:
        return Bbox(self.voxel_offset, self.voxel_offset + self.volume_size)

    @property
    def chunk_size(self) -> Vec:
        return Vec(*self.scale["chunk_sizes"][0])

    @property
    def dtype(self):
        return np.dtype(self.info["data_type"])

    @property
    def key(self) -> str:
        return self.scale["key"]

    def mip_bounds(self, mip: int) -> Bbox:
        s = self.info["scales"][mip]
        off = Vec(*s.get("voxel_offset", [0, 0, 0]))
        return Bbox(off, off + Vec(*s["size"]))

    def commit_info(self) -> None:
        self.cf.put_json("info", self.info)

    # -- provenance (reference: vol.provenance.processing.append + commit) --
    @property
    def provenance(self):
        return _Provenance(self)

    def commit_provenance(self) -> None:
        pass  # _Provenance writes through on append

    # -- IO ----------------------------------------------------------------
    def _chunk_name(self, cb: Bbox) -> str:
        return f"{self.key}/{cb.to_filename()}"

    def _grid_chunks(self, bbox: Bbox):
        cs = self.chunk_size
        off = self.voxel_offset
        vs = self.bounds
        lo = (bbox.minpt - off) // cs
        hi = -(-(bbox.maxpt - off) // cs)  # ceil div
        for gz in range(int(lo[2]), int(hi[2])):
            for gy in range(int(lo[1]), int(hi[1])):
                for gx in range(int(lo[0]), int(hi[0])):
                    cmin = off + Vec(gx, gy, gz) * cs
                    cmax = np.minimum(cmin + cs, vs.maxpt)
                    if np.any(cmin >= vs.maxpt):
                        continue
                    yield Bbox(cmin, cmax)

    def download(self, bbox: Bbox, **kw) -> np.ndarray:
        """F-order (sx,sy,sz,1) array covering bbox; voxels outside the
        volume bounds (bounded=False) or in missing chunks (fill_missing)
        read as 0."""
        size = bbox.size3()
        out = np.zeros((int(size[0]), int(size[1]), int(size[2]), 1),
                       dtype=self.dtype, order="F")
        inner = Bbox.clamp(bbox, self.bounds)
        if inner.volume() == 0:
            if self.bounded:
                raise ValueError(f"{bbox} outside volume bounds {self.bounds}")
            return out
        if self.bounded and not (self.bounds.contains_bbox(bbox)):
            raise ValueError(f"{bbox} outside volume bounds {self.bounds}")
        for cb in self._grid_chunks(inner):
            data = self.cf.get(self._chunk_name(cb))
            csz = cb.size3()
            if data is None:
                if not self.fill_missing:
                    raise FileNotFoundError(
                        f"missing chunk {self._chunk_name(cb)} "
                        f"(pass fill_missing=True to zero-fill)")
                continue
            arr = np.frombuffer(data, dtype=self.dtype).reshape(
                (int(csz[0]), int(csz[1]), int(csz[2]), 1), order="F")
            isect = Bbox(np.maximum(cb.minpt, inner.minpt),
                         np.minimum(cb.maxpt, inner.maxpt))
            src = tuple(
                slice(int(isect.minpt[i] - cb.minpt[i]),
                      int(isect.maxpt[i] - cb.minpt[i])) for i in range(3))
            dst = tuple(
                slice(int(isect.minpt[i] - bbox.minpt[i]),
                      int(isect.maxpt[i] - bbox.minpt[i])) for i in range(3))
            out[dst[0], dst[1], dst[2], :] = arr[src[0], src[1], src[2], :]
        return out

    def upload(self, bbox: Bbox, data: np.ndarray) -> None:
        """Write data (F-order (sx,sy,sz[,1])) at bbox, chunk-aligned."""
        if data.ndim == 3:
            data = data[..., np.newaxis]
        data = np.asfortranarray(data.astype(self.dtype, copy=False))
        for cb in self._grid_chunks(bbox):
            src = tuple(
                slice(int(cb.minpt[i] - bbox.minpt[i]),
                      int(cb.maxpt[i] - bbox.minpt[i])) for i in range(3))
            chunk = np.asfortranarray(data[src[0], src[1], src[2], :])
            self.cf.put(self._chunk_name(cb), chunk.tobytes(order="F"),
                        compress="gzip")

    # -- creation ----------------------------------------------------------
    @classmethod
    def from_numpy(cls, data: np.ndarray, cloudpath: str,
                   resolution=(1, 1, 1), voxel_offset=(0, 0, 0),
                   chunk_size=(64, 64, 64), layer_type: Optional[str] = None,
                   mesh_dir: Optional[str] = None) -> "PrecomputedVolume":
        if data.ndim == 3:
            data = data[..., np.newaxis]
        if layer_type is None:
            layer_type = ("segmentation"
                          if np.issubdtype(data.dtype, np.integer) else "image")
        key = "_".join(str(int(r)) for r in resolution)
        info = {
            "type": layer_type,
            "data_type": str(data.dtype),
            "num_channels": int(data.shape[3]),
            "scales": [{
                "key": key,
                "resolution": [int(r) for r in resolution],
                "size": [int(s) for s in data.shape[:3]],
                "voxel_offset": [int(v) for v in voxel_offset],
                "chunk_sizes": [[int(c) for c in chunk_size]],
                "encoding": "raw",
            }],
        }
        if mesh_dir:
            info["mesh"] = mesh_dir
        cf = CloudFiles(cloudpath)
        cf.put_json("info", info)
        vol = cls(cloudpath, 0)
        off = Vec(*voxel_offset)
        vol.upload(Bbox(off, off + Vec(*data.shape[:3])), data)
        return vol


class _Provenance:
    """Append-only provenance JSON beside the info file, mirroring the
    reference's vol.provenance.processing.append + commit_provenance
    (/root/reference/igneous/task_creation/mesh.py:237-265)."""

    def __init__(self, vol: PrecomputedVolume):
        self.vol = vol
        self.processing = _ProvenanceList(vol)


class _ProvenanceList:
    def __init__(self, vol: PrecomputedVolume):
        self.vol = vol

    def append(self, entry: dict) -> None:
        cf = self.vol.cf
        prov = cf.get_json("provenance") or {
            "description": "", "owners": [], "processing": [], "sources": []}
        prov["processing"].append(entry)
        cf.put_json("provenance", prov)
