"""create_meshing_tasks — drop-in for the reference generator at
/root/reference/igneous/task_creation/mesh.py:158-267: same signature and
defaults, same mesh info JSON (:197-208), same provenance record
(:237-265), same FinelyDividedTaskIterator fan-out into MeshTask objects.
The fan-out's consumers are GPU dispatch workers (igneous_amd.dispatch)
instead of an SQS worker fleet.
"""
from __future__ import annotations

from time import strftime
from typing import Optional

from ..lib import Vec
from ..storage import CloudFiles
from ..tasks.mesh import MeshTask
from ..volume import PrecomputedVolume
from .common import FinelyDividedTaskIterator, operator_contact

__all__ = ["create_meshing_tasks", "create_mesh_manifest_tasks",
           "create_sharded_multires_mesh_tasks",
           "create_unsharded_multires_mesh_tasks",
           "configure_multires_info",
           "create_spatial_index_mesh_tasks",
           "create_mesh_deletion_tasks", "create_xfer_meshes_tasks",
           "create_sharded_multires_mesh_from_unsharded_tasks"]


def create_meshing_tasks(
    layer_path: str,
    mip: int,
    shape=(448, 448, 448),
    simplification: bool = True,
    max_simplification_error: float = 40.0,
    mesh_dir: Optional[str] = None,
    cdn_cache: bool = False,
    dust_threshold: Optional[int] = None,
    object_ids=None,
    progress: bool = False,
    fill_missing: bool = False,
    encoding: str = 'precomputed',
    spatial_index: bool = True,
    frag_path: Optional[str] = None,
    sharded: bool = False,
    compress: Optional[str] = 'gzip',
    closed_dataset_edges: bool = True,
    dust_global: bool = False,
    fill_holes: int = 0,
    dry_run: bool = False,
    exclude_object_ids=[],
):
    shape = Vec(*shape)

    assert 0 <= fill_holes <= 103, "fill_holes must be between 0 to 103 inclusive."

    vol = PrecomputedVolume(layer_path, mip)

    if mesh_dir is None:
        if 'mesh' in vol.info:
            mesh_dir = vol.info['mesh']
        else:
            mesh_dir = 'mesh_mip_{}_err_{}'.format(mip, max_simplification_error)

    if 'mesh' not in vol.info:
        vol.info['mesh'] = mesh_dir
        vol.commit_info()

    cf = CloudFiles(layer_path)
    info_filename = '{}/info'.format(mesh_dir)
    mesh_info = cf.get_json(info_filename) or {}
    mesh_info['@type'] = 'neuroglancer_legacy_mesh'
    mesh_info['mip'] = int(vol.mip)
    mesh_info['chunk_size'] = shape.tolist()
    if spatial_index:
        mesh_info['spatial_index'] = {
            'resolution': vol.resolution.tolist(),
            'chunk_size': (shape * vol.resolution).tolist(),
        }
    cf.put_json(info_filename, mesh_info)

    class MeshTaskIterator(FinelyDividedTaskIterator):
        def task(self, shape, offset):
            return MeshTask(
                shape=shape.clone(),
                offset=offset.clone(),
                layer_path=layer_path,
                mip=vol.mip,
                simplification_factor=(0 if not simplification else 100),
                max_simplification_error=max_simplification_error,
                mesh_dir=mesh_dir,
                cache_control=('' if cdn_cache else 'no-cache'),
                dust_threshold=dust_threshold,
                dust_global=bool(dust_global),
                progress=progress,
                object_ids=object_ids,
                exclude_object_ids=exclude_object_ids,
                fill_missing=fill_missing,
                encoding=encoding,
                spatial_index=spatial_index,
                frag_path=frag_path,
                sharded=sharded,
                compress=compress,
                closed_dataset_edges=closed_dataset_edges,
                fill_holes=fill_holes,
                dry_run=dry_run,
            )

        def on_finish(self):
            vol.provenance.processing.append({
                'method': {
                    'task': 'MeshTask',
                    'layer_path': layer_path,
                    'mip': vol.mip,
                    'shape': shape.tolist(),
                    'simplification': simplification,
                    'max_simplification_error': max_simplification_error,
                    'mesh_dir': mesh_dir,
                    'fill_missing': fill_missing,
                    'cdn_cache': cdn_cache,
                    'dust_threshold': dust_threshold,
                    'encoding': encoding,
                    'object_ids': object_ids,
                    'exclude_object_ids': exclude_object_ids,
                    'spatial_index': spatial_index,
                    'frag_path': frag_path,
                    'sharded': sharded,
                    'compress': compress,
                    'closed_dataset_edges': closed_dataset_edges,
                    'dust_global': bool(dust_global),
                    'fill_holes': int(fill_holes),
                    'dry_run': bool(dry_run),
                },
                'by': operator_contact(),
                'date': strftime('%Y-%m-%d %H:%M %Z'),
            })
            vol.commit_provenance()

    return MeshTaskIterator(vol.mip_bounds(mip), shape)


def create_mesh_manifest_tasks(layer_path: str, magnitude: int = 3,
                               mesh_dir: Optional[str] = None):
    """Mirror of the reference's manifest fan-out
    (task_creation/mesh.py:54-89). file:// protocol -> one filesystem task."""
    from functools import partial
    from ..tasks.mesh import MeshManifestFilesystemTask
    assert int(magnitude) == magnitude and magnitude > 0
    return [partial(MeshManifestFilesystemTask,
                    layer_path=layer_path, mesh_dir=mesh_dir)]


# ---------------------------------------------------------------------------
# Multires / sharded task creation (reference task_creation/mesh.py:
# configure_multires_info :437-479, create_unsharded_multires_mesh_tasks
# :481-530, create_sharded_multires_mesh_tasks :706-813).

import copy as _copy

import numpy as _np

from ..formats import sharding as _sharding


def configure_multires_info(cloudpath: str,
                            vertex_quantization_bits: int,
                            mesh_dir: Optional[str]):
    """Write the neuroglancer_multilod_draco mesh info
    (task_creation/mesh.py:437-479)."""
    assert vertex_quantization_bits in (10, 16), vertex_quantization_bits
    vol = PrecomputedVolume(cloudpath)
    mesh_dir = mesh_dir or vol.info.get("mesh", None)
    if "mesh" not in vol.info:
        vol.info["mesh"] = mesh_dir
        vol.commit_info()

    cf = CloudFiles(cloudpath)
    info_filename = f"{mesh_dir}/info"
    mesh_info = cf.get_json(info_filename) or {}
    mip = int(mesh_info.get("mip", 0))
    res = PrecomputedVolume(cloudpath, mip=mip).resolution
    new_mesh_info = _copy.deepcopy(mesh_info)
    new_mesh_info['@type'] = "neuroglancer_multilod_draco"
    new_mesh_info['vertex_quantization_bits'] = vertex_quantization_bits
    new_mesh_info['transform'] = [
        int(res[0]), 0, 0, 0,
        0, int(res[1]), 0, 0,
        0, 0, int(res[2]), 0,
    ]
    new_mesh_info['lod_scale_multiplier'] = 1.0
    if new_mesh_info != mesh_info:
        cf.put_json(info_filename, new_mesh_info,
                    cache_control="no-cache")
    return new_mesh_info


def create_unsharded_multires_mesh_tasks(
        cloudpath: str, num_lod: int = 0,
        magnitude: int = 3, mesh_dir: Optional[str] = None,
        vertex_quantization_bits: int = 16,
        min_chunk_size=(256, 256, 256)):
    """One prefix-task list over existing unsharded fragments
    (task_creation/mesh.py:481-530; file:// -> single full-prefix task)."""
    from functools import partial
    from ..tasks.multires import MultiResUnshardedMeshMergeTask

    configure_multires_info(cloudpath, vertex_quantization_bits, mesh_dir)
    return [partial(MultiResUnshardedMeshMergeTask,
                    cloudpath, prefix="",
                    mesh_dir=mesh_dir, num_lod=num_lod,
                    min_chunk_size=min_chunk_size)]


def create_sharded_multires_mesh_tasks(
        cloudpath: str,
        shard_index_bytes: int = 2 ** 13,
        minishard_index_bytes: int = 2 ** 15,
        min_shards: int = 1,
        num_lod: int = 0,
        draco_compression_level: int = 7,
        vertex_quantization_bits: int = 16,
        minishard_index_encoding: str = "gzip",
        mesh_dir: Optional[str] = None,
        spatial_index_db: Optional[str] = None,
        frag_path: Optional[str] = None,
        cache: Optional[bool] = False,
        min_chunk_size=(256, 256, 256),
        max_labels_per_shard: Optional[int] = None,
        progress: bool = True):
    """Mirror of task_creation/mesh.py:706-813: write the multilod info
    + sharding spec, assign labels to shards from the spatial index,
    persist {shard}.labels, return one MultiResShardedMeshMergeTask per
    shard."""
    from functools import partial
    from ..spatial_index import SpatialIndex
    from ..tasks.multires import MultiResShardedMeshMergeTask

    mesh_info = configure_multires_info(
        cloudpath, vertex_quantization_bits, mesh_dir)
    mdir = mesh_dir or PrecomputedVolume(cloudpath).info.get("mesh")

    all_labels = SpatialIndex(cloudpath, mdir).query()

    if max_labels_per_shard is not None:
        assert max_labels_per_shard >= 1
        min_shards = max(
            int(_np.ceil(len(all_labels) / max_labels_per_shard)),
            min_shards)

    (shard_bits, minishard_bits, preshift_bits) = \
        _sharding.compute_shard_params_for_hashed(
            num_labels=len(all_labels),
            shard_index_bytes=int(shard_index_bytes),
            minishard_index_bytes=int(minishard_index_bytes),
            min_shards=min_shards)

    spec = _sharding.ShardingSpecification(
        type='neuroglancer_uint64_sharded_v1',
        preshift_bits=preshift_bits,
        hash='murmurhash3_x86_128',
        minishard_bits=minishard_bits,
        shard_bits=shard_bits,
        minishard_index_encoding=minishard_index_encoding,
        data_encoding="raw",  # draco encoded meshes
    )

    cf = CloudFiles(cloudpath)
    mesh_info['sharding'] = spec.to_dict()
    cf.put_json(f"{mdir}/info", mesh_info, cache_control="no-cache")

    shard_labels = _sharding.assign_labels_to_shards(
        _np.asarray(all_labels, dtype=_np.uint64),
        preshift_bits, shard_bits, minishard_bits)
    cf_mesh = CloudFiles(f"{cloudpath.rstrip('/')}/{mdir}")
    for shardno, labels in shard_labels.items():
        cf_mesh.put_json(str(shardno) + '.labels', labels,
                         compress="gzip", cache_control="no-cache")

    vol = PrecomputedVolume(cloudpath)
    vol.provenance.processing.append({
        'method': {
            'task': 'MultiResShardedMeshMergeTask',
            'cloudpath': cloudpath,
            'mip': int(mesh_info.get('mip', 0)),
            'num_lod': num_lod,
            'vertex_quantization_bits': vertex_quantization_bits,
            'preshift_bits': preshift_bits,
            'minishard_bits': minishard_bits,
            'shard_bits': shard_bits,
            'mesh_dir': mdir,
            'frag_path': frag_path,
            'draco_compression_level': draco_compression_level,
            'min_chunk_size': list(min_chunk_size),
        },
        'by': operator_contact(),
        'date': strftime('%Y-%m-%d %H:%M %Z'),
    })
    vol.commit_provenance()

    return [
        partial(MultiResShardedMeshMergeTask,
                cloudpath, shard_no,
                num_lod=num_lod,
                mesh_dir=mdir,
                frag_path=frag_path,
                cache=cache,
                spatial_index_db=spatial_index_db,
                draco_compression_level=draco_compression_level,
                min_chunk_size=min_chunk_size)
        for shard_no in shard_labels.keys()
    ]


def create_spatial_index_mesh_tasks(
        cloudpath: str,
        shape=(448, 448, 448),
        mip: int = 0,
        fill_missing: bool = False,
        compress='gzip',
        mesh_dir: Optional[str] = None):
    """Mirror of task_creation/mesh.py:363-435: (re)build the mesh
    spatial index over a grid of SpatialIndexTasks."""
    import copy as _c
    from functools import partial
    from ..tasks.spatial_index import SpatialIndexTask

    shape = Vec(*shape)
    vol = PrecomputedVolume(cloudpath, mip=mip)
    if mesh_dir is None and not vol.info.get("mesh", None):
        mesh_dir = f"mesh_mip_{mip}_err_40"
    elif mesh_dir is None:
        mesh_dir = vol.info["mesh"]
    if "mesh" not in vol.info:
        vol.info["mesh"] = mesh_dir
        vol.commit_info()

    cf = CloudFiles(cloudpath)
    info_filename = f"{mesh_dir}/info"
    mesh_info = cf.get_json(info_filename) or {}
    new_mesh_info = _c.deepcopy(mesh_info)
    new_mesh_info['@type'] = new_mesh_info.get(
        '@type', 'neuroglancer_legacy_mesh')
    new_mesh_info['mip'] = new_mesh_info.get("mip", int(mip))
    new_mesh_info['chunk_size'] = shape.tolist()
    new_mesh_info['spatial_index'] = {
        'resolution': vol.resolution.tolist(),
        'chunk_size': (shape * vol.resolution).tolist(),
    }
    if new_mesh_info != mesh_info:
        cf.put_json(info_filename, new_mesh_info)

    precision = (new_mesh_info['spatial_index'].get('precision')
                 if isinstance(new_mesh_info.get('spatial_index'), dict)
                 else None)

    class SpatialIndexMeshTaskIterator(FinelyDividedTaskIterator):
        def task(self, tshape, offset):
            return partial(SpatialIndexTask,
                           cloudpath=cloudpath,
                           shape=tuple(int(x) for x in tshape),
                           offset=tuple(int(x) for x in offset),
                           subdir=mesh_dir,
                           precision=precision,
                           mip=int(mip),
                           fill_missing=bool(fill_missing),
                           compress=compress)

    return SpatialIndexMeshTaskIterator(vol.mip_bounds(mip), shape)


def create_mesh_deletion_tasks(layer_path: str, magnitude: int = 3,
                               mesh_dir: Optional[str] = None):
    """Mirror of task_creation/mesh.py:91-156: delete the mesh info and
    fan prefix-parallel DeleteMeshFilesTask over the fragment names.
    (file:// stores list cheaply, so one full-prefix task suffices —
    same final state as the reference's 10^magnitude prefix split.)"""
    from functools import partial
    from ..tasks.mesh import DeleteMeshFilesTask
    assert int(magnitude) == magnitude and magnitude >= 0
    vol = PrecomputedVolume(layer_path)
    mdir = mesh_dir or vol.info.get('mesh', 'mesh')
    cf = CloudFiles(f"{layer_path.rstrip('/')}/{mdir}")
    cf.delete(['info'])
    try:
        return [partial(DeleteMeshFilesTask, cloudpath=layer_path,
                        prefix="", mesh_dir=mesh_dir)]
    finally:
        vol.provenance.processing.append({
            'method': {
                'task': 'DeleteMeshFilesTask',
                'layer_path': layer_path,
                'mesh_dir': mesh_dir,
            },
            'by': operator_contact(),
            'date': strftime('%Y-%m-%d %H:%M %Z'),
        })
        vol.commit_provenance()


def create_xfer_meshes_tasks(src: str, dest: str,
                             mesh_dir: Optional[str] = None,
                             magnitude: int = 2):
    """Mirror of task_creation/mesh.py:548-588: copy a mesh directory
    between layers (one full-prefix task on file:// stores)."""
    from functools import partial
    from ..tasks.mesh import TransferMeshFilesTask
    cf_dest = CloudFiles(dest)
    if not mesh_dir:
        info = cf_dest.get_json("info") or {}
        if info.get("mesh", None):
            mesh_dir = info.get("mesh")
    src_vol = PrecomputedVolume(src)
    smdir = mesh_dir or src_vol.info.get('mesh', 'mesh')
    src_mesh_info = CloudFiles(src).get_json(f"{smdir}/info")
    if src_mesh_info is not None:
        cf_dest.put_json(f"{mesh_dir or smdir}/info", src_mesh_info)
    return [partial(TransferMeshFilesTask, src=src, dest=dest,
                    prefix="", mesh_dir=mesh_dir)]


def create_sharded_multires_mesh_from_unsharded_tasks(
        src: str, dest: str,
        shard_index_bytes: int = 2 ** 13,
        minishard_index_bytes: int = 2 ** 15,
        min_shards: int = 1,
        num_lod: int = 0,
        draco_compression_level: int = 7,
        vertex_quantization_bits: int = 16,
        minishard_index_encoding: str = "gzip",
        mesh_dir: Optional[str] = None):
    """Mirror of task_creation/mesh.py:590-705: convert an UNSHARDED
    legacy mesh layer into sharded multires — enumerate labels from the
    source fragment files, compute shard params, write the multilod
    info + {shard}.labels into dest, return one
    MultiResShardedFromUnshardedMeshMergeTask per shard."""
    from functools import partial
    from ..tasks.multires import (MultiResShardedFromUnshardedMeshMergeTask,
                                  get_mesh_filenames_subset)

    src_vol = PrecomputedVolume(src)
    smdir = mesh_dir or src_vol.info.get('mesh', 'mesh')
    all_labels = sorted(get_mesh_filenames_subset(src, smdir, "").keys())

    (shard_bits, minishard_bits, preshift_bits) = \
        _sharding.compute_shard_params_for_hashed(
            num_labels=len(all_labels),
            shard_index_bytes=int(shard_index_bytes),
            minishard_index_bytes=int(minishard_index_bytes),
            min_shards=min_shards)
    spec = _sharding.ShardingSpecification(
        type='neuroglancer_uint64_sharded_v1',
        preshift_bits=preshift_bits,
        hash='murmurhash3_x86_128',
        minishard_bits=minishard_bits,
        shard_bits=shard_bits,
        minishard_index_encoding=minishard_index_encoding,
        data_encoding="raw")

    mesh_info = configure_multires_info(
        dest, vertex_quantization_bits, mesh_dir or smdir)
    mesh_info['sharding'] = spec.to_dict()
    dmdir = mesh_dir or PrecomputedVolume(dest).info.get('mesh', smdir)
    cf_dest = CloudFiles(dest)
    cf_dest.put_json(f"{dmdir}/info", mesh_info,
                     cache_control="no-cache")

    shard_labels = _sharding.assign_labels_to_shards(
        _np.asarray(all_labels, dtype=_np.uint64),
        preshift_bits, shard_bits, minishard_bits)
    cf_mesh = CloudFiles(f"{dest.rstrip('/')}/{dmdir}")
    for shardno, labels in shard_labels.items():
        cf_mesh.put_json(str(shardno) + '.labels', labels,
                         compress="gzip", cache_control="no-cache")

    dvol = PrecomputedVolume(dest)
    dvol.provenance.processing.append({
        'method': {
            'task': 'MultiResShardedFromUnshardedMeshMergeTask',
            'src': src, 'dest': dest,
            'num_lod': num_lod,
            'vertex_quantization_bits': vertex_quantization_bits,
            'preshift_bits': preshift_bits,
            'minishard_bits': minishard_bits,
            'shard_bits': shard_bits,
            'mesh_dir': dmdir,
            'draco_compression_level': draco_compression_level,
        },
        'by': operator_contact(),
        'date': strftime('%Y-%m-%d %H:%M %Z'),
    })
    dvol.commit_provenance()

    return [
        partial(MultiResShardedFromUnshardedMeshMergeTask,
                src=src, dest=dest, shard_no=shard_no,
                num_lod=num_lod, mesh_dir=dmdir,
                draco_compression_level=draco_compression_level)
        for shard_no in shard_labels.keys()
    ]
