"""Minimal file:// object store with the CloudFiles semantics the reference's
MeshTask output path relies on (cloudfiles.CloudFiles at
/root/reference/igneous/tasks/mesh/mesh.py:399-430,452-464):

 - put(name, content, compress='gzip'|None): gzip-compressed objects are
   stored as "<name>.gz" on disk;
 - get(name): transparently decompresses, trying "<name>" then "<name>.gz";
 - list(prefix): yields logical names (compression extension stripped);
 - put_json / get_json helpers.

Written from scratch; cloud-files is not a dependency of this package.
"""
from __future__ import annotations

import gzip
import json
import os
from typing import Iterable, Optional, Tuple, Union


def asfilepath(cloudpath: str) -> str:
    if cloudpath.startswith("file://"):
        return cloudpath[len("file://"):]
    if "://" in cloudpath:
        raise ValueError(f"only file:// paths are supported, got {cloudpath}")
    return cloudpath


class CloudFiles:
    def __init__(self, cloudpath: str, progress: bool = False):
        self.cloudpath = cloudpath.rstrip("/")
        self.base = asfilepath(self.cloudpath)

    def join(self, *paths: str) -> str:
        return "/".join(p.strip("/") for p in paths)

    def _diskpath(self, name: str) -> str:
        return os.path.join(self.base, name)

    def put(self, name: str, content: bytes,
            compress: Optional[str] = None, **kw) -> None:
        if isinstance(content, str):
            content = content.encode("utf-8")
        path = self._diskpath(name)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        if compress in ("gzip", True):
            path += ".gz"
            content = gzip.compress(content)
        elif compress not in (None, False, ""):
            raise ValueError(f"unsupported compression {compress!r}")
        # atomic publish: concurrent readers (multi-rank dispatch workers
        # sharing one layer, e.g. the info/provenance JSON both ranks
        # rewrite in create_meshing_tasks) must never see a torn file
        tmp = f"{path}.tmp.{os.getpid()}"
        with open(tmp, "wb") as f:
            f.write(content)
        os.replace(tmp, path)

    def puts(self, items: Iterable[Tuple[str, bytes]],
             compress: Optional[str] = None, **kw) -> None:
        for name, content in items:
            self.put(name, content, compress=compress)

    def put_json(self, name: str, obj, compress: Optional[str] = None, **kw) -> None:
        self.put(name, json.dumps(obj).encode("utf-8"), compress=compress)

    def put_jsons(self, items, compress: Optional[str] = None, **kw) -> None:
        for name, obj in items:
            self.put_json(name, obj, compress=compress)

    def get(self, name: str) -> Optional[bytes]:
        path = self._diskpath(name)
        if os.path.exists(path):
            with open(path, "rb") as f:
                data = f.read()
            # objects stored pre-compressed by a gzip put keep raw bytes here
            return data
        if os.path.exists(path + ".gz"):
            with open(path + ".gz", "rb") as f:
                return gzip.decompress(f.read())
        return None

    def get_json(self, name: str):
        data = self.get(name)
        return None if data is None else json.loads(data.decode("utf-8"))

    def exists(self, name: str) -> bool:
        path = self._diskpath(name)
        return os.path.exists(path) or os.path.exists(path + ".gz")

    def list(self, prefix: str = "") -> Iterable[str]:
        """Yield logical object names under this store matching prefix,
        compression extension stripped, sorted."""
        results = []
        root = self.base
        for dirpath, _dirnames, filenames in os.walk(root):
            rel = os.path.relpath(dirpath, root)
            rel = "" if rel == "." else rel + "/"
            for fn in filenames:
                name = rel + fn
                if name.endswith(".gz"):
                    name = name[:-3]
                if name.startswith(prefix):
                    results.append(name)
        return sorted(results)

    def delete(self, names: Union[str, Iterable[str]]) -> None:
        if isinstance(names, str):
            names = [names]
        for name in names:
            for path in (self._diskpath(name), self._diskpath(name) + ".gz"):
                if os.path.exists(path):
                    os.remove(path)
