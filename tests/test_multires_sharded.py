"""End-to-end host tests of the sharded mesh pipeline (SURVEY §8f rows
2-3, reference mesh.py:385-397 + multires.py + task_creation/mesh.py):
MeshTask(sharded=True) -> .frags MapBuffer files + spatial index ->
create_sharded_multires_mesh_tasks -> MultiResShardedMeshMergeTask ->
one neuroglancer shard file per shard, read back and geometry-checked.

CPU-only: the oracle is injected as mesher/simplifier (checker role);
the GPU engine path for the same stages is covered in
tests/test_gpu_parity.py.
"""
import gzip
import json

import numpy as np
import pytest

from igneous_amd import (
    MeshTask, PrecomputedVolume, create_meshing_tasks,
    create_sharded_multires_mesh_tasks, create_unsharded_multires_mesh_tasks,
)
from igneous_amd.formats import draco as draco_fmt
from igneous_amd.formats import sharding
from igneous_amd.formats.mapbuffer import MapBuffer
from igneous_amd.formats.multilod import MultiLevelPrecomputedMeshManifest
from igneous_amd.meshes import Mesh
from igneous_amd.spatial_index import SpatialIndex
from igneous_amd.storage import CloudFiles
from igneous_amd.tasks.multires import MultiResShardedMeshMergeTask


def _make_two_chunk_layer(path):
    """128x64x64 volume, two 64^3 task chunks; a box label spanning the
    chunk seam plus a second label in chunk 0."""
    data = np.zeros((128, 64, 64), dtype=np.uint64)
    data[20:100, 8:56, 8:56] = 77          # spans both chunks
    data[2:12, 2:12, 2:12] = 5             # chunk 0 only
    PrecomputedVolume.from_numpy(
        data, path, resolution=(4, 4, 40), chunk_size=(64, 64, 64),
        mesh_dir="mesh")
    return data


def _run_sharded_mesh_tasks(layer):
    for t in create_meshing_tasks(
            layer, mip=0, shape=(64, 64, 64), sharded=True,
            spatial_index=True, simplification=False):
        t.execute()


def test_sharded_meshtask_writes_frags_and_spatial(tmp_layer_path,
                                                   oracle_mesher):
    _make_two_chunk_layer(tmp_layer_path)
    _run_sharded_mesh_tasks(tmp_layer_path)
    cf = CloudFiles(tmp_layer_path)
    names = list(cf.list("mesh/"))
    frags = [n for n in names if n.endswith(".frags")]
    spatial = [n for n in names if n.endswith(".spatial")]
    assert len(frags) == 2
    assert len(spatial) == 2
    # no unsharded fragment files
    assert not [n for n in names if ":0:" in n]
    # each .frags is a readable MapBuffer of precomputed meshes
    mb = MapBuffer(cf.get("mesh/0-64_0-64_0-64.frags"),
                   frombytesfn=Mesh.from_precomputed)
    assert mb.validate()
    assert set(mb.keys()) == {5, 77}
    m = mb[77]
    assert len(m.vertices) > 0 and len(m.faces) > 0
    # spatial index sees both labels
    idx = SpatialIndex(tmp_layer_path, "mesh")
    assert idx.query() == [5, 77]
    locs = idx.file_locations_per_label([77])
    assert len(locs[77]) == 2  # label 77 spans both chunks


def test_sharded_multires_merge_end_to_end(tmp_layer_path, oracle_mesher,
                                           oracle_simplifier):
    _make_two_chunk_layer(tmp_layer_path)
    _run_sharded_mesh_tasks(tmp_layer_path)

    tasks = create_sharded_multires_mesh_tasks(
        tmp_layer_path, num_lod=1, vertex_quantization_bits=16,
        min_chunk_size=(16, 16, 16))
    for t in tasks:
        t()

    cf = CloudFiles(tmp_layer_path)
    mesh_info = cf.get_json("mesh/info")
    assert mesh_info["@type"] == "neuroglancer_multilod_draco"
    assert mesh_info["vertex_quantization_bits"] == 16
    assert mesh_info["transform"][0] == 4  # resolution on the diagonal
    spec = sharding.ShardingSpecification.from_dict(mesh_info["sharding"])

    shard_files = [n for n in cf.list("mesh/") if n.endswith(".shard")]
    assert shard_files, "no shard files written"

    def fetch(name):
        return cf.get(f"mesh/{name}")

    reader = sharding.ShardReader(spec, fetch)
    for label in (5, 77):
        manifest_bytes = reader.get(label)
        assert manifest_bytes is not None, f"label {label} missing"
        man = MultiLevelPrecomputedMeshManifest.from_binary(
            manifest_bytes, segment_id=label)
        assert man.num_lods >= 1
        assert sum(man.num_fragments_per_lod) == len(man.fragment_offsets)
        # fragments precede the manifest in the shard; decode each and
        # check the dequantized geometry lands inside the label's bbox
        blob, start, size = reader.byte_range(label)
        frag_total = sum(man.fragment_offsets)
        frag_bytes = blob[start - frag_total:start]
        pos = 0
        k = 0
        seen_tris = 0
        for lod in range(man.num_lods):
            for fi in range(man.num_fragments_per_lod[lod]):
                ln = man.fragment_offsets[k]
                k += 1
                if ln == 0:
                    continue
                verts_q, faces = draco_fmt.decode(
                    frag_bytes[pos:pos + ln])
                pos += ln
                seen_tris += len(faces)
                # dequantize into model (voxel) space
                fpos = np.array(man.fragment_positions[lod][fi])
                chunk = np.asarray(man.chunk_shape)
                scale = man.lod_scales[lod]
                model = (np.asarray(man.grid_origin)
                         + chunk * scale
                         * (fpos + verts_q / (2 ** 16 - 1)))
                assert model.min() >= -1.0
                assert model.max() <= 130.0
        assert seen_tris > 0


def test_unsharded_multires_merge(tmp_layer_path, oracle_mesher,
                                  oracle_simplifier):
    """MultiResUnshardedMeshMergeTask path: unsharded fragments ->
    {label}.index + {label} multilod files."""
    _make_two_chunk_layer(tmp_layer_path)
    for t in create_meshing_tasks(
            tmp_layer_path, mip=0, shape=(64, 64, 64), sharded=False,
            spatial_index=True, simplification=False):
        t.execute()
    tasks = create_unsharded_multires_mesh_tasks(
        tmp_layer_path, num_lod=0, min_chunk_size=(16, 16, 16))
    for t in tasks:
        t()
    cf = CloudFiles(tmp_layer_path)
    for label in (5, 77):
        idx = cf.get(f"mesh/{label}.index")
        assert idx is not None
        man = MultiLevelPrecomputedMeshManifest.from_binary(
            idx, segment_id=label)
        data = cf.get(f"mesh/{label}")
        assert data is not None
        assert len(data) == sum(man.fragment_offsets)


def test_labels_file_written_per_shard(tmp_layer_path, oracle_mesher):
    _make_two_chunk_layer(tmp_layer_path)
    _run_sharded_mesh_tasks(tmp_layer_path)
    tasks = create_sharded_multires_mesh_tasks(
        tmp_layer_path, num_lod=0, min_chunk_size=(16, 16, 16))
    cf = CloudFiles(tmp_layer_path)
    label_files = [n for n in cf.list("mesh/") if n.endswith(".labels")]
    assert label_files
    all_labels = []
    for n in label_files:
        all_labels.extend(cf.get_json(n))
    assert sorted(all_labels) == [5, 77]
    # provenance recorded once
    prov = json.loads(CloudFiles(tmp_layer_path).get("provenance"))
    assert any(p["method"]["task"] == "MultiResShardedMeshMergeTask"
               for p in prov["processing"])


def test_draco_roundtrip_integer_and_float():
    rng = np.random.default_rng(5)
    v = rng.integers(0, 2 ** 16 - 1, size=(300, 3)).astype(np.uint32)
    f = rng.integers(0, 300, size=(500, 3)).astype(np.uint32)
    f = f[(f[:, 0] != f[:, 1]) & (f[:, 1] != f[:, 2]) & (f[:, 0] != f[:, 2])]
    blob = draco_fmt.encode(v, f)
    v2, f2 = draco_fmt.decode(blob)
    assert np.array_equal(v, v2)
    assert np.array_equal(f, f2)
    # float path (the unsharded MeshTask draco encoding): quantized
    vf = rng.uniform(0, 1000, size=(100, 3)).astype(np.float32)
    ff = np.arange(99, dtype=np.uint32)
    ff = np.stack([ff, ff + 1, np.roll(ff, 1)], axis=1).astype(np.uint32)
    blob = draco_fmt.encode(vf, ff, quantization_bits=14,
                            quantization_range=1000.0,
                            quantization_origin=np.zeros(3))
    v3, f3 = draco_fmt.decode(blob)
    assert np.array_equal(ff, f3)
    assert np.abs(v3 - vf).max() < 1000.0 / (2 ** 14 - 1) * 0.51 + 1e-3


def test_meshtask_draco_encoding(tmp_layer_path, oracle_mesher):
    """encoding='draco' fragment files decode via the draco restatement
    and land near the precomputed-encoding geometry."""
    data = np.zeros((64, 64, 64), dtype=np.uint32)
    data[1:-1, 1:-1, 1:-1] = 1
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(4, 4, 40),
        chunk_size=(64, 64, 64), mesh_dir="mesh")
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
             layer_path=tmp_layer_path, encoding='draco',
             simplification_factor=0).execute()
    cf = CloudFiles(tmp_layer_path)
    blob = cf.get('mesh/1:0:0-64_0-64_0-64')
    verts, faces = draco_fmt.decode(blob)
    assert verts.dtype == np.float32  # quantization transform applied
    # geometry in nm: box [1..63] voxels * resolution
    assert verts[:, 0].min() >= 4 * 1 - 4.1
    assert verts[:, 0].max() <= 4 * 63 + 4.1
    assert verts[:, 2].max() <= 40 * 63 + 40.1
    assert len(faces) > 0


def test_spatial_index_task(tmp_layer_path):
    """Standalone SpatialIndexTask (reference tasks/spatial_index.py):
    rebuild the .spatial files without MeshTask, precision-formatted."""
    from igneous_amd.tasks import SpatialIndexTask
    data = np.zeros((64, 64, 64), dtype=np.uint32)
    data[2:20, 2:20, 2:20] = 9
    data[30:60, 30:60, 30:60] = 12
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(4, 4, 40),
        chunk_size=(64, 64, 64), mesh_dir="mesh")
    SpatialIndexTask(tmp_layer_path, shape=(64, 64, 64), offset=(0, 0, 0),
                     subdir="mesh", precision=0, mip=0)
    cf = CloudFiles(tmp_layer_path)
    names = [n for n in cf.list("mesh/") if n.endswith(".spatial")]
    assert len(names) == 1
    content = cf.get_json(names[0])
    assert set(content.keys()) == {"9", "12"}
    # bbox of label 9 in nm: voxels [2,20) * resolution
    assert content["9"] == [8, 8, 80, 80, 80, 800]


def test_sharded_from_unsharded_merge(tmp_layer_path, oracle_mesher,
                                      oracle_simplifier, tmp_path):
    """MultiResShardedFromUnshardedMeshMergeTask: legacy unsharded
    fragments in src -> multires shard files in dest."""
    from igneous_amd.formats import sharding as _sh
    from igneous_amd.tasks import MultiResShardedFromUnshardedMeshMergeTask
    from igneous_amd.task_creation import configure_multires_info
    _make_two_chunk_layer(tmp_layer_path)
    for t in create_meshing_tasks(
            tmp_layer_path, mip=0, shape=(64, 64, 64), sharded=False,
            spatial_index=False, simplification=False):
        t.execute()

    dest = f"file://{tmp_path}/dest"
    data = np.zeros((4, 4, 4), dtype=np.uint64)
    PrecomputedVolume.from_numpy(
        data, dest, resolution=(4, 4, 40), chunk_size=(4, 4, 4),
        mesh_dir="mesh")
    mesh_info = configure_multires_info(dest, 16, "mesh")
    spec = _sh.ShardingSpecification(
        preshift_bits=0, minishard_bits=2, shard_bits=1,
        hash="murmurhash3_x86_128", minishard_index_encoding="gzip",
        data_encoding="raw")
    mesh_info["sharding"] = spec.to_dict()
    cf_dest = CloudFiles(dest)
    cf_dest.put_json("mesh/info", mesh_info)
    shard_labels = _sh.assign_labels_to_shards(
        np.array([5, 77], dtype=np.uint64), 0, 1, 2)
    for shardno, labels in shard_labels.items():
        cf_dest.put_json(f"mesh/{shardno}.labels", labels)
        MultiResShardedFromUnshardedMeshMergeTask(
            tmp_layer_path, dest, shardno, num_lod=0,
            min_chunk_size=(16, 16, 16))
    shard_files = [n for n in cf_dest.list("mesh/")
                   if n.endswith(".shard")]
    assert shard_files
    reader = _sh.ShardReader(spec,
                             lambda n: cf_dest.get(f"mesh/{n}"))
    for label in (5, 77):
        man = reader.get(label)
        assert man is not None and len(man) > 0


def test_create_spatial_index_mesh_tasks(tmp_layer_path):
    from igneous_amd.task_creation import create_spatial_index_mesh_tasks
    data = np.zeros((64, 64, 64), dtype=np.uint32)
    data[5:40, 5:40, 5:40] = 3
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(4, 4, 40),
        chunk_size=(64, 64, 64), mesh_dir="mesh")
    tasks = list(create_spatial_index_mesh_tasks(
        tmp_layer_path, shape=(64, 64, 64)))
    assert len(tasks) == 1
    for t in tasks:
        t()
    cf = CloudFiles(tmp_layer_path)
    names = [n for n in cf.list("mesh/") if n.endswith(".spatial")]
    assert len(names) == 1
    assert cf.get_json("mesh/info")["spatial_index"]["resolution"] == [4, 4, 40]
    assert set(cf.get_json(names[0]).keys()) == {"3"}


def test_mesh_deletion_and_xfer_tasks(tmp_layer_path, oracle_mesher,
                                      tmp_path):
    from igneous_amd.task_creation import (create_mesh_deletion_tasks,
                                           create_xfer_meshes_tasks)
    _make_two_chunk_layer(tmp_layer_path)
    for t in create_meshing_tasks(
            tmp_layer_path, mip=0, shape=(64, 64, 64), sharded=False,
            spatial_index=False, simplification=False):
        t.execute()
    cf = CloudFiles(tmp_layer_path)
    assert [n for n in cf.list("mesh/") if ":0:" in n]

    # transfer to a second layer
    dest = f"file://{tmp_path}/xfer"
    data = np.zeros((4, 4, 4), dtype=np.uint64)
    PrecomputedVolume.from_numpy(
        data, dest, resolution=(4, 4, 40), chunk_size=(4, 4, 4),
        mesh_dir="mesh")
    for t in create_xfer_meshes_tasks(tmp_layer_path, dest,
                                      mesh_dir="mesh"):
        t()
    cf_dest = CloudFiles(dest)
    assert sorted(n for n in cf_dest.list("mesh/") if ":0:" in n) == \
        sorted(n for n in cf.list("mesh/") if ":0:" in n)

    # delete the source meshes
    for t in create_mesh_deletion_tasks(tmp_layer_path):
        t()
    assert not [n for n in cf.list("mesh/") if ":0:" in n]


def test_create_sharded_from_unsharded_tasks(tmp_layer_path, oracle_mesher,
                                             oracle_simplifier, tmp_path):
    from igneous_amd.formats import sharding as _sh
    from igneous_amd.formats.multilod import (
        MultiLevelPrecomputedMeshManifest)
    from igneous_amd.task_creation import (
        create_sharded_multires_mesh_from_unsharded_tasks)
    _make_two_chunk_layer(tmp_layer_path)
    for t in create_meshing_tasks(
            tmp_layer_path, mip=0, shape=(64, 64, 64), sharded=False,
            spatial_index=False, simplification=False):
        t.execute()
    dest = f"file://{tmp_path}/dest2"
    data = np.zeros((4, 4, 4), dtype=np.uint64)
    PrecomputedVolume.from_numpy(
        data, dest, resolution=(4, 4, 40), chunk_size=(4, 4, 4),
        mesh_dir="mesh")
    tasks = create_sharded_multires_mesh_from_unsharded_tasks(
        tmp_layer_path, dest, num_lod=0)
    for t in tasks:
        t()
    cf_dest = CloudFiles(dest)
    info = cf_dest.get_json("mesh/info")
    spec = _sh.ShardingSpecification.from_dict(info["sharding"])
    reader = _sh.ShardReader(spec, lambda n: cf_dest.get(f"mesh/{n}"))
    for label in (5, 77):
        man_bytes = reader.get(label)
        assert man_bytes, f"label {label} missing in dest shards"
        man = MultiLevelPrecomputedMeshManifest.from_binary(
            man_bytes, segment_id=label)
        assert sum(man.num_fragments_per_lod) >= 1
