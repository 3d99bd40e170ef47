"""Spatial index reader — the subset of cloud-volume's
`cv.mesh.spatial_index` the multires merge uses
(/root/reference/igneous/tasks/mesh/multires.py:471-482,497-508):
  query(bbox)                    -> all labels in the index
  file_locations_per_label(...)  -> {label: [".spatial" filenames]}

The index is the per-chunk ".spatial" JSON files MeshTask uploads
(mesh.py:452-464 / our tasks/mesh.py): filename = chunk bbox in nm,
value = {label: [minx,miny,minz,maxx,maxy,maxz] in nm}.
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Optional

from .storage import CloudFiles


class SpatialIndex:
    def __init__(self, layer_path: str, mesh_dir: str):
        self.layer_path = layer_path
        self.mesh_dir = mesh_dir
        self._cache: Optional[Dict[str, dict]] = None

    def _load(self) -> Dict[str, dict]:
        if self._cache is None:
            cf = CloudFiles(self.layer_path)
            out = {}
            for name in cf.list(prefix=f"{self.mesh_dir}/"):
                if not name.endswith(".spatial"):
                    continue
                fname = name.split("/")[-1]
                out[fname] = cf.get_json(name) or {}
            self._cache = out
        return self._cache

    def query(self, bbox=None) -> List[int]:
        """All labels in the index (bbox filtering unused by the mesh
        merge path, which always queries the full dataset bounds)."""
        labels = set()
        for content in self._load().values():
            labels.update(int(k) for k in content)
        return sorted(labels)

    def file_locations_per_label(self, labels: Iterable[int]
                                 ) -> Dict[int, List[str]]:
        labels = set(int(l) for l in labels)
        out: Dict[int, List[str]] = {l: [] for l in labels}
        for fname, content in sorted(self._load().items()):
            for k in content:
                k = int(k)
                if k in labels:
                    out[k].append(fname)
        return {k: v for k, v in out.items() if v}
