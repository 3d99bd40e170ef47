"""Round-trip and known-answer tests for the sharded-pipeline formats
(mapbuffer stand-in, neuroglancer sharded format, multilod manifest)
and the mesh geometry ops. All CPU."""
import gzip

import numpy as np
import pytest

from igneous_amd.formats.mapbuffer import MapBuffer
from igneous_amd.formats import sharding
from igneous_amd.formats.multilod import (
    MultiLevelPrecomputedMeshManifest, to_stored_model_space)
from igneous_amd.meshes import Mesh
from igneous_amd import meshops


# ---------------------------------------------------------------- mapbuffer

def test_mapbuffer_roundtrip():
    data = {7: b"seven", 1: b"one", 2 ** 40 + 3: b"big", 0: b""}
    buf = MapBuffer(data, compress=None).tobytes()
    mb = MapBuffer(buf)
    assert mb.validate()
    assert len(mb) == 4
    assert sorted(mb.keys()) == sorted(data.keys())
    for k, v in data.items():
        assert mb[k] == v
        assert k in mb
    assert 999 not in mb
    with pytest.raises(KeyError):
        mb[999]
    assert mb.get(999) is None


def test_mapbuffer_gzip_and_br_downgrade():
    data = {5: b"x" * 1000, 6: b"y"}
    for codec in ("gzip", "br"):
        buf = MapBuffer(data, compress=codec).tobytes()
        assert len(buf) < 1000  # value actually compressed
        mb = MapBuffer(buf)
        assert mb[5] == b"x" * 1000
        assert mb[6] == b"y"


def test_mapbuffer_frombytesfn():
    m = Mesh(np.zeros((3, 3), np.float32),
             np.array([[0, 1, 2]], np.uint32))
    buf = MapBuffer({42: m.to_precomputed()}, compress="br").tobytes()
    mb = MapBuffer(buf, frombytesfn=Mesh.from_precomputed)
    out = mb[42]
    assert isinstance(out, Mesh)
    assert np.array_equal(out.faces, m.faces)


def test_mapbuffer_deterministic():
    data = {i: bytes([i % 251]) * (i % 97) for i in range(200)}
    assert (MapBuffer(data, compress="gzip").tobytes()
            == MapBuffer(dict(reversed(list(data.items()))),
                         compress="gzip").tobytes())


# ---------------------------------------------------------------- sharding

def test_murmur_vectorized_matches_scalar():
    keys = np.array([0, 1, 2, 12345, 2 ** 63 + 11], dtype=np.uint64)
    vec = sharding.murmurhash3_x86_128_low64(keys)
    for k, h in zip(keys, vec):
        assert int(sharding.murmurhash3_x86_128_low64(int(k))) == int(h)
    # distinct inputs scatter (sanity that the mixer does something)
    assert len(set(int(x) for x in vec)) == len(keys)


def test_murmur_known_answer():
    """Pin against an independent pure-python scalar restatement of
    MurmurHash3_x86_128 (tail-only, len=8), so the vectorized form is
    cross-checked structurally."""
    def rotl(x, r):
        return ((x << r) | (x >> (32 - r))) & 0xFFFFFFFF

    def fmix(h):
        h ^= h >> 16
        h = (h * 0x85EBCA6B) & 0xFFFFFFFF
        h ^= h >> 13
        h = (h * 0xC2B2AE35) & 0xFFFFFFFF
        h ^= h >> 16
        return h

    def ref(key):
        c1, c2, c3 = 0x239B961B, 0xAB0E9789, 0x38B34AE5
        h1 = h2 = h3 = h4 = 0
        k1 = key & 0xFFFFFFFF
        k2 = (key >> 32) & 0xFFFFFFFF
        k2 = (k2 * c2) & 0xFFFFFFFF
        k2 = rotl(k2, 16)
        k2 = (k2 * c3) & 0xFFFFFFFF
        h2 ^= k2
        k1 = (k1 * c1) & 0xFFFFFFFF
        k1 = rotl(k1, 15)
        k1 = (k1 * c2) & 0xFFFFFFFF
        h1 ^= k1
        h1 ^= 8; h2 ^= 8; h3 ^= 8; h4 ^= 8
        h1 = (h1 + h2 + h3 + h4) & 0xFFFFFFFF
        h2 = (h2 + h1) & 0xFFFFFFFF
        h3 = (h3 + h1) & 0xFFFFFFFF
        h4 = (h4 + h1) & 0xFFFFFFFF
        h1, h2, h3, h4 = fmix(h1), fmix(h2), fmix(h3), fmix(h4)
        h1 = (h1 + h2 + h3 + h4) & 0xFFFFFFFF
        h2 = (h2 + h1) & 0xFFFFFFFF
        return h1 | (h2 << 32)

    for key in (0, 1, 0xDEADBEEF, 2 ** 64 - 1, 987654321987654321):
        assert int(sharding.murmurhash3_x86_128_low64(key)) == ref(key)


@pytest.mark.parametrize("hashfn", ["identity", "murmurhash3_x86_128"])
@pytest.mark.parametrize("mini_enc", ["raw", "gzip"])
def test_shard_synthesize_read_roundtrip(hashfn, mini_enc):
    spec = sharding.ShardingSpecification(
        preshift_bits=0, hash=hashfn, minishard_bits=3, shard_bits=2,
        minishard_index_encoding=mini_enc, data_encoding="raw")
    rng = np.random.default_rng(11)
    labels = rng.choice(2 ** 40, size=100, replace=False)
    data = {int(k): bytes(rng.integers(0, 256, size=int(k) % 50 + 1,
                                       dtype=np.uint8)) for k in labels}
    files = sharding.synthesize_shard_files(spec, data)
    assert all(name.endswith(".shard") for name in files)
    reader = sharding.ShardReader(spec, lambda n: files.get(n))
    for k, v in data.items():
        assert reader.get(k) == v
    assert reader.get(999999999999) is None
    # every label is listed in exactly the shard the spec routes it to
    listed = []
    for name in files:
        listed.extend(reader.list_labels_in_shard(name))
    assert sorted(listed) == sorted(data.keys())


def test_shard_data_offset_trailing_manifest():
    spec = sharding.ShardingSpecification(
        hash="identity", minishard_bits=1, shard_bits=0)
    payload = b"FRAGDATA" + b"MANIFEST"
    files = sharding.synthesize_shard_files(
        spec, {3: payload}, data_offset={3: 8})
    reader = sharding.ShardReader(spec, lambda n: files.get(n))
    # recorded range covers the trailing manifest only
    assert reader.get(3) == b"MANIFEST"
    blob, start, size = reader.byte_range(3)
    # the fragment bytes sit immediately before the manifest
    assert blob[start - 8:start] == b"FRAGDATA"


def test_assign_labels_to_shards_consistent_with_spec():
    labels = np.arange(1, 500, dtype=np.uint64)
    out = sharding.assign_labels_to_shards(labels, 0, 3, 2)
    spec = sharding.ShardingSpecification(
        preshift_bits=0, shard_bits=3, minishard_bits=2)
    for name, ls in out.items():
        for l in ls:
            assert spec.compute_shard_location(l) == name
    assert sum(len(v) for v in out.values()) == len(labels)


def test_compute_shard_params_matches_reference_examples():
    # small label counts collapse to (0, small, 0)
    assert sharding.compute_shard_params_for_hashed(0) == (0, 0, 0)
    sb, mb, pb = sharding.compute_shard_params_for_hashed(10 ** 6)
    assert pb == 0 and sb >= 0 and mb >= 0
    # capacity must cover the labels with reasonable index sizes
    assert (2 ** sb) * (2 ** mb) * (2 ** 15 / 24) >= 10 ** 6 * 0.4


# ---------------------------------------------------------------- multilod

def test_manifest_binary_roundtrip():
    man = MultiLevelPrecomputedMeshManifest(
        segment_id=9,
        chunk_shape=np.array([64, 64, 40], np.float32),
        grid_origin=np.array([0, 128, 64], np.float32),
        num_lods=2,
        lod_scales=[1.0, 2.0],
        vertex_offsets=[[0, 0, 0], [0, 0, 0]],
        num_fragments_per_lod=[2, 1],
        fragment_positions=[[(0, 0, 0), (1, 0, 0)], [(0, 0, 0)]],
        fragment_offsets=[100, 200, 50],
    )
    binary = man.to_binary()
    man2 = MultiLevelPrecomputedMeshManifest.from_binary(binary,
                                                         segment_id=9)
    assert man2.num_lods == 2
    assert man2.num_fragments_per_lod == [2, 1]
    assert man2.fragment_positions[0] == [(0, 0, 0), (1, 0, 0)]
    assert man2.fragment_offsets == [100, 200, 50]
    assert np.allclose(man2.chunk_shape, man.chunk_shape)
    assert man2.to_binary() == binary
    assert len(man) == len(binary)


def test_to_stored_model_space_roundtrip():
    man = MultiLevelPrecomputedMeshManifest(
        segment_id=1,
        chunk_shape=np.array([32, 32, 32], np.float32),
        grid_origin=np.array([10, 20, 30], np.float32),
        num_lods=1, lod_scales=[1.0], vertex_offsets=[[0, 0, 0]],
        num_fragments_per_lod=[2],
        fragment_positions=[[(0, 0, 0), (1, 1, 0)]],
        fragment_offsets=[0, 0])
    rng = np.random.default_rng(3)
    v = rng.uniform(0, 32, size=(50, 3)) + np.array([10, 20, 30])
    stored = to_stored_model_space(v, man, lod=0,
                                   vertex_quantization_bits=16, frag=0)
    assert stored.dtype == np.uint32
    assert stored.max() <= 2 ** 16 - 1
    # invert the decode mapping: error bounded by one quantization step
    decoded = (np.array([10, 20, 30])
               + np.array([32, 32, 32]) * (0 + stored / (2 ** 16 - 1)))
    assert np.abs(decoded - v).max() < 32 / (2 ** 16 - 1) * 0.51 + 1e-6


# ---------------------------------------------------------------- meshops

def _box_mesh(lo, hi):
    """12-triangle axis box."""
    lo, hi = np.asarray(lo, float), np.asarray(hi, float)
    corners = np.array([[lo[0], lo[1], lo[2]], [hi[0], lo[1], lo[2]],
                        [lo[0], hi[1], lo[2]], [hi[0], hi[1], lo[2]],
                        [lo[0], lo[1], hi[2]], [hi[0], lo[1], hi[2]],
                        [lo[0], hi[1], hi[2]], [hi[0], hi[1], hi[2]]],
                       dtype=np.float32)
    faces = np.array([
        [0, 2, 1], [1, 2, 3], [4, 5, 6], [5, 7, 6],
        [0, 1, 4], [1, 5, 4], [2, 6, 3], [3, 6, 7],
        [0, 4, 2], [2, 4, 6], [1, 3, 5], [3, 7, 5]], dtype=np.uint32)
    return Mesh(corners, faces)


def _area(mesh):
    t = mesh.vertices.astype(np.float64)[mesh.faces.astype(np.int64)]
    return 0.5 * np.linalg.norm(
        np.cross(t[:, 1] - t[:, 0], t[:, 2] - t[:, 0]), axis=1).sum()


def test_consolidate_dedupes_and_drops_degenerates():
    v = np.array([[0, 0, 0], [1, 0, 0], [0, 1, 0], [0, 0, 0]], np.float32)
    f = np.array([[0, 1, 2], [3, 1, 2], [0, 0, 1]], np.uint32)
    m = meshops.consolidate(Mesh(v, f))
    assert len(m.vertices) == 3            # duplicate [0,0,0] welded
    assert len(m.faces) == 2               # degenerate [0,0,1] dropped
    # welded faces now identical index triples
    assert np.array_equal(m.faces[0], m.faces[1])


def test_merge_close_vertices_stitches():
    v = np.array([[0, 0, 0], [1, 0, 0], [0, 1, 0],
                  [1e-7, 0, 0], [1, 0, 0], [0, -1, 0]], np.float32)
    f = np.array([[0, 1, 2], [3, 5, 4]], np.uint32)
    m = meshops.merge_close_vertices(Mesh(v, f), radius=1e-5)
    assert len(m.vertices) == 4  # (0,0,0)~(1e-7,0,0) and the two (1,0,0)


def test_chunk_mesh_partitions_and_preserves_area():
    box = _box_mesh([1, 1, 1], [63, 63, 63])
    chunks = meshops.chunk_mesh(box, scale=(32, 32, 32), offset=(0, 0, 0))
    assert len(chunks) == 8  # box spans all 8 octants
    total = sum(_area(m) for m in chunks.values())
    assert abs(total - _area(box)) < 1e-6 * _area(box)
    # every chunk's geometry stays inside its cell (+epsilon)
    for (cx, cy, cz), m in chunks.items():
        lo = np.array([cx, cy, cz]) * 32.0
        assert np.all(m.vertices >= lo - 1e-4)
        assert np.all(m.vertices <= lo + 32 + 1e-4)


def test_chunk_mesh_interior_only():
    box = _box_mesh([1, 1, 1], [10, 10, 10])
    chunks = meshops.chunk_mesh(box, scale=(64, 64, 64), offset=(0, 0, 0))
    assert list(chunks.keys()) == [(0, 0, 0)]
    assert abs(_area(chunks[(0, 0, 0)]) - _area(box)) < 1e-9


def test_draco_varint_index_width():
    """Index-width tiers: u8 (<256 points), u16 (<65536) are fuzz
    covered; pin the varint tier (>=2^16 points) explicitly."""
    from igneous_amd.formats import draco as draco_fmt
    nv = 70000
    rng = np.random.default_rng(9)
    v = rng.integers(0, 2 ** 20, size=(nv, 3)).astype(np.uint32)
    f = np.stack([np.arange(nv - 2), np.arange(1, nv - 1),
                  np.arange(2, nv)], axis=1).astype(np.uint32)[:500]
    blob = draco_fmt.encode(v, f)
    v2, f2 = draco_fmt.decode(blob)
    assert np.array_equal(v, v2)
    assert np.array_equal(f, f2)
