"""Precomputed file:// volume + storage round trips (the reference relies
on CloudVolume/CloudFiles here; these pin our from-scratch equivalents)."""
import gzip
import os

import numpy as np
import pytest

from igneous_amd.lib import Bbox, Vec
from igneous_amd.storage import CloudFiles
from igneous_amd.volume import PrecomputedVolume


def test_storage_gzip_roundtrip(tmp_path):
    cf = CloudFiles(f"file://{tmp_path}/store")
    cf.put("a/b", b"hello", compress="gzip")
    assert cf.get("a/b") == b"hello"
    # stored with .gz extension on disk, listed without
    assert os.path.exists(f"{tmp_path}/store/a/b.gz")
    assert list(cf.list("a/")) == ["a/b"]
    cf.put("a/c", b"raw", compress=None)
    assert sorted(cf.list("a/")) == ["a/b", "a/c"]
    cf.delete("a/b")
    assert list(cf.list("a/")) == ["a/c"]


def test_storage_json(tmp_path):
    cf = CloudFiles(f"file://{tmp_path}/store")
    cf.put_json("info", {"x": 1})
    assert cf.get_json("info") == {"x": 1}
    assert cf.get_json("missing") is None


def test_volume_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    data = rng.integers(0, 100, size=(70, 60, 50)).astype(np.uint64)
    path = f"file://{tmp_path}/vol"
    vol = PrecomputedVolume.from_numpy(
        data, path, resolution=(16, 16, 40), chunk_size=(64, 64, 64),
        mesh_dir="mesh")
    assert vol.info["mesh"] == "mesh"
    assert np.array_equal(vol.resolution, [16, 16, 40])
    back = vol.download(Bbox((0, 0, 0), (70, 60, 50)))
    assert np.array_equal(back[..., 0], data)
    # F-order layout like CloudVolume (mesh.py:177-182)
    assert back.flags["F_CONTIGUOUS"]


def test_volume_bounded_false_zero_fill(tmp_path):
    data = np.ones((10, 10, 10), dtype=np.uint32)
    path = f"file://{tmp_path}/vol"
    PrecomputedVolume.from_numpy(data, path, chunk_size=(16, 16, 16))
    vol = PrecomputedVolume(path, bounded=False)
    out = vol.download(Bbox((0, 0, 0), (11, 11, 11)))
    assert out.shape == (11, 11, 11, 1)
    assert np.all(out[:10, :10, :10, 0] == 1)
    assert np.all(out[10, :, :, 0] == 0)
    assert np.all(out[:, :, 10, 0] == 0)


def test_volume_voxel_offset(tmp_path):
    data = np.full((8, 8, 8), 3, dtype=np.uint32)
    path = f"file://{tmp_path}/vol"
    PrecomputedVolume.from_numpy(
        data, path, voxel_offset=(16, 32, 48), chunk_size=(8, 8, 8))
    vol = PrecomputedVolume(path, bounded=False)
    out = vol.download(Bbox((16, 32, 48), (24, 40, 56)))
    assert np.all(out[..., 0] == 3)
    out2 = vol.download(Bbox((8, 24, 40), (24, 40, 56)))
    assert np.all(out2[:8, :, :, 0] == 0) and np.all(out2[8:, 8:, 8:, 0] == 3)


def test_missing_chunk_behavior(tmp_path):
    data = np.ones((64, 64, 64), dtype=np.uint32)
    path = f"file://{tmp_path}/vol"
    vol = PrecomputedVolume.from_numpy(data, path, chunk_size=(32, 32, 32))
    os.remove(f"{tmp_path}/vol/1_1_1/32-64_0-32_0-32.gz")
    vol = PrecomputedVolume(path, fill_missing=False)
    with pytest.raises(FileNotFoundError):
        vol.download(Bbox((0, 0, 0), (64, 64, 64)))
    vol = PrecomputedVolume(path, fill_missing=True)
    out = vol.download(Bbox((0, 0, 0), (64, 64, 64)))
    assert np.all(out[32:, :32, :32, 0] == 0)
    assert np.all(out[:32, :, :, 0] == 1)


def test_bbox_filename():
    assert Bbox((0, 0, 0), (64, 64, 64)).to_filename() == "0-64_0-64_0-64"
    b = Bbox.from_filename("0-64_0-64_0-64")
    assert b == Bbox((0, 0, 0), (64, 64, 64))
