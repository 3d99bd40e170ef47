"""MeshTask host-logic tests, mirroring the reference's own mesh tests
(test/test_tasks.py:407-462: output naming + listing) plus the host steps
the reference suite leaves uncovered (padding, boundary closure, dust,
remap semantics, manifests, spatial index). Runs on CPU with the oracle
injected as the mesher (checker role); the real engine path is covered by
tests/test_gpu_parity.py."""
import json

import numpy as np
import pytest

from igneous_amd import (
    Mesh, MeshTask, PrecomputedVolume, LocalTaskQueue, create_meshing_tasks,
)
from igneous_amd.lib import Bbox
from igneous_amd.storage import CloudFiles
from igneous_amd.tasks import MeshManifestFilesystemTask


def _make_box_layer(path, dtype=np.uint32):
    """The reference's test_mesh fixture: 64^3, ones at [1:-1]^3
    (test_tasks.py:410-414)."""
    data = np.zeros((64, 64, 64), dtype=dtype)
    data[1:-1, 1:-1, 1:-1] = 1
    PrecomputedVolume.from_numpy(
        data, path, resolution=(1, 1, 1), chunk_size=(64, 64, 64),
        mesh_dir="mesh")
    return data


@pytest.mark.parametrize("compress", ("gzip", None))
def test_mesh(tmp_layer_path, oracle_mesher, compress):
    """Mirror of reference test_mesh (test_tasks.py:407-431)."""
    _make_box_layer(tmp_layer_path)
    cf = CloudFiles(tmp_layer_path)
    t = MeshTask(
        shape=(64, 64, 64),
        offset=(0, 0, 0),
        layer_path=tmp_layer_path,
        mip=0,
        remap_table={"1": "10"},
        low_padding=0,
        high_padding=1,
        compress=compress,
        simplification_factor=0,
    )
    t.execute()
    assert cf.get('mesh/10:0:0-64_0-64_0-64') is not None
    assert list(cf.list('mesh/')) == ['mesh/10:0:0-64_0-64_0-64']


def test_mesh_object_ids(tmp_layer_path, oracle_mesher):
    """Mirror of reference test_mesh_object_ids (test_tasks.py:433-462)."""
    data = np.zeros((64, 64, 64), dtype=np.uint32)
    data[1:-1, 1:-1, 1:-1] = 1
    data[1:-1, 1:-1, 32:63] = 2
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(64, 64, 64),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)

    MeshTask(
        shape=(64, 64, 64), offset=(0, 0, 0), layer_path=tmp_layer_path,
        mip=0, exclude_object_ids=[2], simplification_factor=0,
    ).execute()
    assert cf.get('mesh/1:0:0-64_0-64_0-64') is not None
    assert list(cf.list('mesh/')) == ['mesh/1:0:0-64_0-64_0-64']

    MeshTask(
        shape=(64, 64, 64), offset=(0, 0, 0), layer_path=tmp_layer_path,
        mip=0, object_ids=[2], simplification_factor=0,
    ).execute()
    assert cf.get('mesh/2:0:0-64_0-64_0-64') is not None


def test_mesh_fragment_decodes(tmp_layer_path, oracle_mesher):
    """The uploaded fragment is valid precomputed bytes whose geometry
    matches the oracle's mesh shifted into global nm coordinates."""
    _make_box_layer(tmp_layer_path)
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0,
             simplification_factor=0).execute()
    binary = cf.get('mesh/1:0:0-64_0-64_0-64')
    m = Mesh.from_precomputed(binary)
    assert m.vertices.shape[0] == 6 * 62 * 62
    assert m.faces.shape[0] == 2 * m.vertices.shape[0] - 4
    # offset 0, resolution 1 -> chunk-local == global
    assert np.allclose(m.vertices.min(axis=0), [0.5, 0.5, 0.5])


def test_global_offset_applied(tmp_layer_path, oracle_mesher):
    """Vertices shift by (bounds.minpt - low_padding)*resolution
    (mesh.py:434-435)."""
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[4:10, 4:10, 4:10] = 5
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(16, 16, 40),
        voxel_offset=(128, 64, 32), chunk_size=(32, 32, 32), mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(32, 32, 32), offset=(128, 64, 32),
             layer_path=tmp_layer_path, mip=0,
             simplification_factor=0).execute()
    name = 'mesh/5:0:128-160_64-96_32-64'
    binary = cf.get(name)
    assert binary is not None
    m = Mesh.from_precomputed(binary)
    assert np.allclose(m.vertices.min(axis=0),
                       [(128 + 3.5) * 16, (64 + 3.5) * 16, (32 + 3.5) * 40])


def test_closed_dataset_edges(tmp_layer_path, oracle_mesher):
    """A label touching the dataset boundary is closed by a zero border
    (mesh.py:267-303): its surface is still watertight and extends half a
    voxel past the boundary."""
    data = np.full((16, 16, 16), 9, dtype=np.uint32)
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(16, 16, 16),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(16, 16, 16), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0,
             simplification_factor=0).execute()
    m = Mesh.from_precomputed(cf.get('mesh/9:0:0-16_0-16_0-16'))
    assert np.allclose(m.vertices.min(axis=0), [-0.5, -0.5, -0.5])
    assert np.allclose(m.vertices.max(axis=0), [15.5, 15.5, 15.5])
    # closed box of 16^3: V = 6*16^2 + ... watertight genus-0
    assert m.faces.shape[0] == 2 * m.vertices.shape[0] - 4


def test_no_closed_dataset_edges(tmp_layer_path, oracle_mesher):
    data = np.full((16, 16, 16), 9, dtype=np.uint32)
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(16, 16, 16),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(16, 16, 16), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, closed_dataset_edges=False,
             simplification_factor=0).execute()
    m = Mesh.from_precomputed(cf.get('mesh/9:0:0-16_0-16_0-16'))
    # open at the low boundary: no vertices below 0
    assert m.vertices.min() >= 0


def test_dust_threshold(tmp_layer_path, oracle_mesher):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[2:20, 2:20, 2:20] = 1   # big
    data[24:26, 24:26, 24:26] = 2  # 8 voxels of dust
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(32, 32, 32),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(32, 32, 32), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, dust_threshold=100,
             simplification_factor=0).execute()
    assert list(cf.list('mesh/')) == ['mesh/1:0:0-32_0-32_0-32']


def test_generate_manifests(tmp_layer_path, oracle_mesher):
    _make_box_layer(tmp_layer_path)
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, generate_manifests=True,
             simplification_factor=0).execute()
    manifest = cf.get_json('mesh/1:0')
    assert manifest == {"fragments": ["1:0:0-64_0-64_0-64"]}


def test_manifest_filesystem_task(tmp_layer_path, oracle_mesher):
    """Mirror of reference test_mesh_manifests_filesystem
    (test_tasks.py:507-549): fabricate fragment files, assert manifest
    content exactly."""
    PrecomputedVolume.from_numpy(
        np.zeros((8, 8, 8), np.uint32), tmp_layer_path, mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    for segid in (1, 50, 300):
        for lod in (0, 1):
            for frag in range(3):
                cf.put(f"mesh/{segid}:{lod}:{frag}", b"")
    MeshManifestFilesystemTask(layer_path=tmp_layer_path, lod=0)
    manifest = cf.get_json("mesh/50:0")
    assert manifest == {"fragments": ["50:0:0", "50:0:1", "50:0:2"]}
    assert cf.get_json("mesh/300:0") == {
        "fragments": ["300:0:0", "300:0:1", "300:0:2"]}


def test_spatial_index(tmp_layer_path, oracle_mesher):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[4:10, 4:10, 4:10] = 5
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(2, 2, 2), chunk_size=(32, 32, 32),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(32, 32, 32), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, spatial_index=True,
             simplification_factor=0).execute()
    idx = cf.get_json('mesh/0-64_0-64_0-64.spatial')
    assert set(idx.keys()) == {"5"}
    # voxels 4..9, voxel-centered surface 3.5..9.5, x2nm -> [7,19]
    assert np.allclose(idx["5"], [7, 7, 7, 19, 19, 19])


def test_spatial_index_empty_volume(tmp_layer_path, oracle_mesher):
    data = np.zeros((16, 16, 16), dtype=np.uint32)
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(16, 16, 16),
        mesh_dir="mesh")
    cf = CloudFiles(tmp_layer_path)
    MeshTask(shape=(16, 16, 16), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, spatial_index=True,
             simplification_factor=0).execute()
    assert cf.get_json('mesh/0-16_0-16_0-16.spatial') == {}
    assert list(cf.list('mesh/')) == ['mesh/0-16_0-16_0-16.spatial']


def test_unsupported_paths_raise(tmp_layer_path, oracle_mesher):
    _make_box_layer(tmp_layer_path)
    with pytest.raises(ValueError):
        MeshTask(shape=(64,) * 3, offset=(0,) * 3,
                 layer_path=tmp_layer_path, encoding='obj')
    # draco + sharded are implemented now (formats/); only the
    # fastmorph-dependent fill_holes path stays a scoped refusal
    for kw in ({"fill_holes": 1}, {"dust_threshold": 10,
                                   "dust_global": True}):
        t = MeshTask(shape=(64,) * 3, offset=(0,) * 3,
                     layer_path=tmp_layer_path, **kw)
        with pytest.raises(NotImplementedError):
            t.execute()


def test_task_payload_roundtrip(tmp_layer_path):
    t = MeshTask(shape=(64, 64, 64), offset=(0, 0, 64),
                 layer_path=tmp_layer_path, mip=0,
                 remap_table={"1": "10"}, simplification_factor=17)
    from igneous_amd.queue import RegisteredTask
    t2 = RegisteredTask.deserialize(t.payload())
    assert isinstance(t2, MeshTask)
    assert list(t2.shape) == [64, 64, 64]
    assert list(t2.offset) == [0, 0, 64]
    assert t2.options['simplification_factor'] == 17
    assert t2.options['remap_table'] == {"1": "10"}


def test_create_meshing_tasks_end_to_end(tmp_layer_path, oracle_mesher):
    """Fan-out + LocalTaskQueue over a 2x2x1-task volume; every task's
    fragments land with the right names; mesh info + provenance written
    (task_creation/mesh.py:197-208,237-265)."""
    data = np.zeros((100, 100, 50), dtype=np.uint32)
    data[10:90, 10:90, 10:40] = 77
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(64, 64, 64))
    tasks = create_meshing_tasks(
        tmp_layer_path, mip=0, shape=(64, 64, 64), simplification=False,
        spatial_index=True)
    assert len(tasks) == 4
    with LocalTaskQueue(parallel=1) as tq:
        tq.insert(tasks)
    cf = CloudFiles(tmp_layer_path)
    vol = PrecomputedVolume(tmp_layer_path)
    mesh_dir = vol.info['mesh']
    assert mesh_dir == 'mesh_mip_0_err_40.0'
    names = [n for n in cf.list(f'{mesh_dir}/') if ':0:' in n]
    assert sorted(names) == sorted([
        f'{mesh_dir}/77:0:0-64_0-64_0-50',
        f'{mesh_dir}/77:0:64-100_0-64_0-50',
        f'{mesh_dir}/77:0:0-64_64-100_0-50',
        f'{mesh_dir}/77:0:64-100_64-100_0-50',
    ])
    info = cf.get_json(f'{mesh_dir}/info')
    assert info['@type'] == 'neuroglancer_legacy_mesh'
    assert info['chunk_size'] == [64, 64, 64]
    assert 'spatial_index' in info
    prov = cf.get_json('provenance')
    assert prov['processing'][0]['method']['task'] == 'MeshTask'


def test_chunk_seams_stitch(tmp_layer_path, oracle_mesher):
    """Adjacent tasks' meshes share identical vertices on the seam plane
    (the 1vx overlap contract, mesh.py:155-160)."""
    data = np.zeros((20, 10, 10), dtype=np.uint32)
    data[2:18, 2:8, 2:8] = 4
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(10, 10, 10))
    tasks = create_meshing_tasks(
        tmp_layer_path, mip=0, shape=(10, 10, 10), simplification=False,
        spatial_index=False)
    with LocalTaskQueue() as tq:
        tq.insert(tasks)
    cf = CloudFiles(tmp_layer_path)
    mesh_dir = PrecomputedVolume(tmp_layer_path).info['mesh']
    m0 = Mesh.from_precomputed(cf.get(f'{mesh_dir}/4:0:0-10_0-10_0-10'))
    m1 = Mesh.from_precomputed(cf.get(f'{mesh_dir}/4:0:10-20_0-10_0-10'))
    seam0 = {tuple(v) for v in m0.vertices[np.isclose(m0.vertices[:, 0], 10.0)]}
    seam1 = {tuple(v) for v in m1.vertices[np.isclose(m1.vertices[:, 0], 10.0)]}
    assert seam0 and seam0 == seam1


def test_transfer_and_delete_mesh_files(tmp_path, oracle_mesher):
    """Mirror of reference TransferMeshFilesTask/DeleteMeshFilesTask
    (mesh.py:726-749)."""
    from igneous_amd.tasks import TransferMeshFilesTask, DeleteMeshFilesTask
    src = f"file://{tmp_path}/src"
    dst = f"file://{tmp_path}/dst"
    _make_box_layer(src)
    PrecomputedVolume.from_numpy(
        np.zeros((8, 8, 8), np.uint32), dst, mesh_dir="mesh")
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0), layer_path=src,
             mip=0, simplification_factor=0).execute()
    TransferMeshFilesTask(src=src, dest=dst, prefix="1:")
    cf_dst = CloudFiles(dst)
    assert cf_dst.get("mesh/1:0:0-64_0-64_0-64") is not None
    DeleteMeshFilesTask(cloudpath=dst, prefix="1:")
    assert list(cf_dst.list("mesh/")) == []


def test_frag_path_unsharded_matches_reference(tmp_layer_path,
                                                oracle_mesher, tmp_path):
    """In the reference, frag_path is consumed ONLY by the sharded
    MapBuffer uploader (mesh.py:385-387); the unsharded path
    (_upload_individuals, mesh.py:399-417) always writes fragments to
    layer_path. Mirror that: frag_path with sharded=False is inert."""
    _make_box_layer(tmp_layer_path)
    frag = f"file://{tmp_path}/frags"
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, frag_path=frag,
             simplification_factor=0).execute()
    assert CloudFiles(tmp_layer_path).get('mesh/1:0:0-64_0-64_0-64') \
        is not None
    assert CloudFiles(frag).get('mesh/1:0:0-64_0-64_0-64') is None


def test_dry_run_writes_nothing(tmp_layer_path, oracle_mesher):
    """dry_run computes but uploads nothing (the reference's dry_run path
    at mesh.py:249-252 hits a latent NameError; ours returns the data —
    DESIGN.md §7)."""
    _make_box_layer(tmp_layer_path)
    result = MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
                      layer_path=tmp_layer_path, mip=0, dry_run=True,
                      simplification_factor=0).execute()
    assert result is not None
    meshes, bboxes = result
    assert 1 in meshes and 1 in bboxes
    assert list(CloudFiles(tmp_layer_path).list('mesh/')) == []


def test_fill_missing_through_task(tmp_layer_path, oracle_mesher):
    """fill_missing zero-fills absent chunks instead of raising
    (mesh.py:177-182 download kwargs)."""
    data = np.zeros((64, 64, 32), dtype=np.uint32)
    data[1:-1, 1:-1, 1:-1] = 1
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1),
        chunk_size=(32, 32, 32), mesh_dir="mesh")
    # delete one stored chunk file out from under the task
    cf = CloudFiles(tmp_layer_path)
    gone = "1_1_1/32-64_32-64_0-32"
    assert cf.get(gone) is not None
    cf.delete(gone)
    with pytest.raises(FileNotFoundError):
        MeshTask(shape=(64, 64, 32), offset=(0, 0, 0),
                 layer_path=tmp_layer_path, mip=0,
                 simplification_factor=0).execute()
    MeshTask(shape=(64, 64, 32), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0, fill_missing=True,
             simplification_factor=0).execute()
    assert cf.get('mesh/1:0:0-64_0-64_0-32') is not None


def test_manifest_prefix_task(tmp_layer_path, oracle_mesher):
    """MeshManifestPrefixTask (reference mesh.py:672-724): only segids
    under the given prefix get manifests."""
    from igneous_amd import MeshManifestPrefixTask
    data = np.zeros((64, 64, 64), dtype=np.uint32)
    data[1:30, 1:30, 1:30] = 1
    data[32:63, 32:63, 32:63] = 21
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(64, 64, 64),
        mesh_dir="mesh")
    MeshTask(shape=(64, 64, 64), offset=(0, 0, 0),
             layer_path=tmp_layer_path, mip=0,
             simplification_factor=0).execute()
    cf = CloudFiles(tmp_layer_path)
    MeshManifestPrefixTask(layer_path=tmp_layer_path, prefix="2")
    assert cf.get_json('mesh/21:0') == {
        "fragments": ["21:0:0-64_0-64_0-64"]}
    assert cf.get_json('mesh/1:0') is None   # prefix "2" excludes segid 1
    MeshManifestPrefixTask(layer_path=tmp_layer_path, prefix="1")
    assert cf.get_json('mesh/1:0') == {
        "fragments": ["1:0:0-64_0-64_0-64"]}
