"""Grid decomposition for task fan-out — mirrors the subset of
/root/reference/igneous/task_creation/common.py the meshing path uses:
num_tasks (:57), FinelyDividedTaskIterator (:60-104, linear index -> grid
coordinate at :91-98), operator_contact (:11-24).
"""
from __future__ import annotations

import copy
import os
import subprocess

import numpy as np

from ..lib import Bbox, Vec


def operator_contact() -> str:
    try:
        contact = subprocess.check_output(
            "git config user.email", shell=True,
            stderr=subprocess.DEVNULL)
        return str(contact.rstrip().decode("utf-8", "replace"))
    except Exception:
        return os.environ.get("USER", "")


def num_tasks(bounds: Bbox, shape) -> int:
    shape = Vec(*shape)
    return int(np.prod(np.ceil(bounds.size3() / shape)))


class FinelyDividedTaskIterator:
    """Splits `bounds` into `shape`-sized tasks; linear index maps to a grid
    coordinate x-fastest, matching the reference's to_coord (:91-98)."""

    def __init__(self, bounds: Bbox, shape):
        self.bounds = bounds
        self.shape = Vec(*shape)
        self.start = 0
        self.end = num_tasks(bounds, shape)

    def __len__(self):
        return self.end - self.start

    def __getitem__(self, slc: slice):
        itr = copy.deepcopy(self)
        itr.start = max(self.start + slc.start, self.start)
        itr.end = min(self.start + slc.stop, self.end)
        return itr

    def __iter__(self):
        for i in range(self.start, self.end):
            pt = self.to_coord(i)
            offset = pt * self.shape + self.bounds.minpt
            yield self.task(self.shape.clone(), offset.clone())
        # Under multi-rank dispatch every rank drains the iterator
        # (dispatch.shard_tasks), so side effects like provenance writes
        # must run once: rank 0 only (lost-update race otherwise).
        if int(os.environ.get("RANK", "0")) == 0:
            self.on_finish()

    def to_coord(self, index: int) -> Vec:
        gx, gy, gz = np.ceil(self.bounds.size3() / self.shape).astype(int)
        gxy = gx * gy
        z = index // gxy
        y = (index - (z * gxy)) // gx
        x = index - gx * (y + z * gy)
        return Vec(x, y, z)

    def task(self, shape, offset):
        raise NotImplementedError

    def on_finish(self):
        pass
