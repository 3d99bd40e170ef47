"""SpatialIndexTask — mirror of /root/reference/igneous/tasks/
spatial_index.py:22-76: (re)build the per-chunk ".spatial" JSON of
label -> physical bbox for datasets whose MeshTask run predates the
index (or used a different chunking). scipy.ndimage.find_objects does
the per-label bbox scan, C-order for speed (spatial_index.py:10-20)."""
from __future__ import annotations

from typing import Optional, Tuple, Union

import numpy as np
import scipy.ndimage

from ..lib import Bbox, Vec
from ..storage import CloudFiles
from ..volume import PrecomputedVolume


def find_objects(labels):
    """find_objects runs ~7-8x faster on C-order arrays; transpose
    F-order input and flip the slices back (spatial_index.py:10-20)."""
    if labels.flags.c_contiguous:
        return scipy.ndimage.find_objects(labels)
    all_slices = scipy.ndimage.find_objects(labels.T)
    return [(slcs and slcs[::-1]) for slcs in all_slices]


def SpatialIndexTask(
        cloudpath: str,
        shape: Tuple[int, int, int],
        offset: Tuple[int, int, int],
        subdir: str,
        precision: int,
        mip: int = 0,
        fill_missing: bool = False,
        compress: Optional[Union[str, bool]] = 'gzip') -> None:
    cv = PrecomputedVolume(cloudpath, mip=mip, bounded=False,
                           fill_missing=fill_missing)
    cf = CloudFiles(cloudpath)

    bounds = Bbox(Vec(*offset), Vec(*shape) + Vec(*offset))
    bounds = Bbox.clamp(bounds, cv.bounds)

    data_bounds = bounds.clone()
    data_bounds.maxpt += 1  # match typical Marching Cubes overlap

    resolution = np.asarray(cv.resolution, dtype=np.int64)

    img = cv.download(data_bounds)[..., 0]
    # renumber for find_objects (it scans up to max label): dense 1..N
    uniq = np.unique(img)
    uniq = uniq[uniq != 0]
    dense = np.searchsorted(uniq, img).astype(np.int32) + 1
    dense[img == 0] = 0
    slcs = find_objects(dense)
    del img

    bboxes = {}
    for idx, slc in enumerate(slcs):
        if slc is None:
            continue
        lo = [s.start for s in slc]
        hi = [s.stop for s in slc]
        obj = Bbox(Vec(*lo) + Vec(*offset), Vec(*hi) + Vec(*offset))
        obj = Bbox(obj.minpt * resolution, obj.maxpt * resolution)
        bboxes[str(int(uniq[idx]))] = (obj.minpt.tolist()
                                       + obj.maxpt.tolist())

    nm_bounds = Bbox(bounds.minpt * resolution, bounds.maxpt * resolution)
    cf.put_json(
        f"{subdir}/{nm_bounds.to_filename(precision)}.spatial",
        bboxes,
        compress=compress,
    )
