"""Seeded synthetic segmentation chunks: nearest-seed
(Voronoi) labeling of K uniformly random seed points, labels drawn from a
shuffled sparse uint64 id set (defeats renumber-free shortcuts), with ~3%
of voxels relabeled 0 (background) past a distance threshold so background
surface exists. Deterministic per (shape, K, seed, dtype).
"""
from __future__ import annotations

import numpy as np


def voronoi_labels(shape, K: int, seed: int, dtype=np.uint64,
                   background_frac: float = 0.03,
                   cache: bool = True) -> np.ndarray:
    """F-order (sx,sy,sz) label volume. Deterministic; large volumes are
    cached on local disk (generation of 512^3/K=50k takes ~30 s)."""
    import os
    cache_path = None
    if cache and int(np.prod(shape)) >= 64 ** 3:
        cdir = os.environ.get("MESHGINE_CACHE", "/tmp/meshgine_cache")
        os.makedirs(cdir, exist_ok=True)
        tag = "x".join(str(int(s)) for s in shape)
        cache_path = os.path.join(
            cdir, f"vor_{tag}_K{K}_s{seed}_{np.dtype(dtype).name}"
                  f"_b{background_frac}.npy")
        if os.path.exists(cache_path):
            return np.asfortranarray(np.load(cache_path))
    rng = np.random.default_rng(seed)
    shape = tuple(int(s) for s in shape)
    seeds = np.stack([
        rng.integers(0, s, size=K) for s in shape], axis=1).astype(np.float32)
    if dtype == np.uint64 or np.dtype(dtype) == np.uint64:
        ids = rng.choice(np.uint64(1) << np.uint64(40), size=K, replace=False)
        ids = ids.astype(np.uint64) + np.uint64(1)
    else:
        ids = (rng.choice(np.uint32(1) << np.uint32(28), size=K, replace=False)
               .astype(np.uint32) + np.uint32(1))

    from scipy.spatial import cKDTree
    tree = cKDTree(seeds)

    out = np.empty(shape, dtype=dtype, order="F")
    dist_thresh = None
    # process in z-slabs to bound memory
    zs = max(1, min(shape[2], int(64 * (256 ** 3) / (shape[0] * shape[1] * 256))))
    xs = np.arange(shape[0], dtype=np.float32)
    ys = np.arange(shape[1], dtype=np.float32)
    gx, gy = np.meshgrid(xs, ys, indexing="ij")
    # first pass: sample distances to pick the background threshold
    sample_pts = np.stack([
        rng.uniform(0, s, size=200_000).astype(np.float32) for s in shape],
        axis=1)
    sdist, _ = tree.query(sample_pts, workers=-1)
    dist_thresh = np.quantile(sdist, 1.0 - background_frac)

    for z0 in range(0, shape[2], zs):
        z1 = min(z0 + zs, shape[2])
        nz = z1 - z0
        pts = np.empty((shape[0] * shape[1] * nz, 3), dtype=np.float32)
        for i, z in enumerate(range(z0, z1)):
            base = i * shape[0] * shape[1]
            pts[base:base + shape[0] * shape[1], 0] = gx.ravel(order="F")
            pts[base:base + shape[0] * shape[1], 1] = gy.ravel(order="F")
            pts[base:base + shape[0] * shape[1], 2] = z
        dist, idx = tree.query(pts, workers=-1)
        lab = ids[idx]
        lab[dist > dist_thresh] = 0
        out[:, :, z0:z1] = lab.reshape(
            (shape[0], shape[1], nz), order="F")
    if cache_path is not None:
        # pid-unique tmp: concurrent ranks generating the same chunk
        # (driver 8-GPU scale runs) must not tear each other's writes
        tmp = f"{cache_path}.tmp{os.getpid()}.npy"
        np.save(tmp, out)
        os.replace(tmp, cache_path)
    return out
