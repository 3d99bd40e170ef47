"""Multi-resolution sharded mesh merge — SURVEY §8f row 3.

Mirrors /root/reference/igneous/tasks/mesh/multires.py with the same
task names and argument meanings:
  MultiResShardedMeshMergeTask        (multires.py:207-260)
  MultiResUnshardedMeshMergeTask      (multires.py:44-81)
  process_mesh                        (multires.py:83-178)
  generate_lods                       (multires.py:308-359)
  create_octree_level_from_mesh       (multires.py:556-585)
  retriangulate_mesh                  (multires.py:546-553)
  locations_for_labels / labels_for_shard / collect_mesh_fragments
                                      (multires.py:404-508)

External-package replacements (all in-repo, from their published
specs/algorithms — DESIGN.md §7):
  zmesh.chunk_mesh / merge_close_vertices -> igneous_amd.meshops
  zmesh.simplify_fqmr                     -> the HIP quadric simplifier
        via the `set_simplifier` seam (default: engine.simplify_mesh on
        the GPU; the pyfqmr aggressiveness/K error schedule is NOT
        replicated — our engine's reduction_factor/max_error contract
        drives each LOD halving instead, a documented deviation)
  cloudvolume MultiLevelPrecomputedMeshManifest / to_stored_model_space
                                          -> igneous_amd.formats.multilod
  cloudvolume synthesize_shard_files      -> igneous_amd.formats.sharding
  DracoPy.encode                          -> igneous_amd.formats.draco
  mapbuffer.MapBuffer                     -> igneous_amd.formats.mapbuffer
"""
from __future__ import annotations

import functools
import itertools
import re
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..formats import draco as draco_fmt
from ..formats.mapbuffer import MapBuffer
from ..formats.multilod import (MultiLevelPrecomputedMeshManifest,
                                to_stored_model_space)
from ..formats import sharding
from ..lib import Bbox
from ..meshes import Mesh
from .. import meshops
from ..spatial_index import SpatialIndex
from ..storage import CloudFiles
from ..volume import PrecomputedVolume

__all__ = [
    "MultiResShardedMeshMergeTask",
    "MultiResUnshardedMeshMergeTask",
    "MultiResShardedFromUnshardedMeshMergeTask",
    "process_mesh",
    "generate_lods",
    "set_simplifier",
]

# ---------------------------------------------------------------------------
# LOD simplifier seam. The product path is the HIP engine
# (engine.simplify_mesh, no CPU fallback); tests may inject the oracle's
# simplifier as the CHECKER. fn(mesh: Mesh, target_count: int) -> Mesh.

_simplifier_fn = None


def set_simplifier(fn) -> None:
    """Test seam. Pass None to restore the HIP engine simplifier."""
    global _simplifier_fn
    _simplifier_fn = fn


def _get_simplifier():
    if _simplifier_fn is not None:
        return _simplifier_fn
    from .. import engine

    def gpu_simplify(mesh: Mesh, target_count: int) -> Mesh:
        return engine.simplify_mesh(mesh, target_count)
    return gpu_simplify


# ---------------------------------------------------------------------------

def _mesh_meta(layer_path: str, mesh_dir: Optional[str]):
    """(volume-at-mesh-mip, mesh_dir, mesh_info)."""
    vol0 = PrecomputedVolume(layer_path)
    if mesh_dir is None and 'mesh' in vol0.info:
        mesh_dir = vol0.info['mesh']
    cf = CloudFiles(layer_path)
    mesh_info = cf.get_json(f"{mesh_dir}/info") or {}
    mip = int(mesh_info.get("mip", 0))
    vol = PrecomputedVolume(layer_path, mip=mip) if mip else vol0
    return vol, mesh_dir, mesh_info


def generate_lods(label: int, mesh: Mesh,
                  mesh_shape: np.ndarray,
                  min_chunk_size,
                  num_lods: int,
                  decimation_factor: int = 2,
                  max_triangle_target_per_chunk: int = int(2e6)
                  ) -> List[Mesh]:
    """LOD chain: lods[0] = input; each further LOD halves the triangle
    budget through the quadric simplifier (multires.py:308-359; the
    zmesh.simplify_fqmr schedule is replaced by our simplifier contract)."""
    assert num_lods >= 0, num_lods
    simplify = _get_simplifier()
    lods = [mesh]
    for lod in range(1, num_lods + 1):
        target_count = int(len(mesh.faces) / (decimation_factor ** lod))
        target_count = max(target_count, 4)
        num_chunks = np.asarray(mesh_shape, dtype=np.float64) \
            / (np.asarray(min_chunk_size) * (2 ** lod))
        max_triangles = int(np.prod(num_chunks)
                            * max_triangle_target_per_chunk)
        target_count = min(target_count, max_triangles)
        simplified = simplify(lods[-1], target_count)
        simplified.id = label
        lods.append(simplified)
        mesh = lods[-1]
    return lods


def determine_mesh_shape_from_lods(lods: List[Mesh]):
    mesh_starts = [np.min(lod.vertices, axis=0) for lod in lods]
    mesh_ends = [np.max(lod.vertices, axis=0) for lod in lods]
    grid_origin = np.floor(np.min(mesh_starts, axis=0))
    grid_end = np.ceil(np.max(mesh_ends, axis=0))
    mesh_shape = (grid_end - grid_origin).astype(int)
    return grid_origin, mesh_shape


def retriangulate_mesh(mesh: Mesh, offset, scale) -> Mesh:
    """Cut triangles at the UPPER octree level's cell boundaries so no
    triangle leaks across a coarser node (multires.py:546-553)."""
    chunks = meshops.chunk_mesh(mesh, scale, offset)
    if not chunks:
        return mesh
    new_mesh = Mesh.concatenate(*chunks.values(), id=mesh.id)
    return meshops.merge_close_vertices(new_mesh, radius=1e-5)


def cmp_zorder(lhs, rhs) -> int:
    def less_msb(x: int, y: int) -> bool:
        return x < y and x < (x ^ y)
    msd = 2
    for dim in [1, 0]:
        if less_msb(lhs[msd] ^ rhs[msd], lhs[dim] ^ rhs[dim]):
            msd = dim
    return lhs[msd] - rhs[msd]


def create_octree_level_from_mesh(mesh: Mesh, chunk_shape, lod: int,
                                  num_lods: int, offset, grid_length):
    """(submeshes, node positions) for one LOD (multires.py:556-585)."""
    scale = np.asarray(chunk_shape, dtype=np.float64) * (2 ** lod)
    if lod > 0:
        upper = np.asarray(chunk_shape, dtype=np.float64) * (2 ** (lod - 1))
        mesh = retriangulate_mesh(mesh, offset, upper)
    if lod == num_lods - 1:
        return ([Mesh(mesh.vertices, mesh.faces, id=mesh.id)],
                ((0, 0, 0),))
    grid = meshops.chunk_mesh(mesh, scale, offset)
    if not grid:
        return ([], tuple())
    nodes, submeshes = zip(*sorted(
        grid.items(),
        key=functools.cmp_to_key(lambda x, y: cmp_zorder(x[0], y[0]))))
    return (list(submeshes), nodes)


def process_mesh(vol: PrecomputedVolume,
                 mesh_info: dict,
                 label: int,
                 mesh: Mesh,
                 num_lod: int,
                 min_chunk_size=(512, 512, 512),
                 draco_compression_level: int = 7
                 ) -> Tuple[Optional[MultiLevelPrecomputedMeshManifest],
                            Optional[bytes]]:
    """Multires fragment builder (multires.py:83-178): LOD chain, octree
    chunking, per-fragment quantization + draco encode, manifest."""
    mesh = Mesh(mesh.vertices.copy(), mesh.faces, id=label)
    mesh.vertices /= np.asarray(vol.resolution, dtype=np.float32)

    grid_origin = np.floor(np.min(mesh.vertices, axis=0))
    mesh_shape = (np.max(mesh.vertices, axis=0) - grid_origin).astype(int)

    if np.any(mesh_shape == 0):
        return (None, None)

    min_chunk_size = np.array(min_chunk_size, dtype=int)
    max_lod = int(max(np.min(np.log2(
        np.maximum(mesh_shape / min_chunk_size, 1e-9))), 0))
    max_lod = min(max_lod, num_lod)

    lods = generate_lods(label, mesh, mesh_shape, min_chunk_size, max_lod)
    grid_origin, mesh_shape = determine_mesh_shape_from_lods(lods)
    if np.any(mesh_shape < 0):
        return (None, None)

    chunk_shape = np.ceil(mesh_shape / (2 ** (len(lods) - 1)))
    if np.any(chunk_shape == 0):
        return (None, None)

    lods = [
        create_octree_level_from_mesh(
            lods[lod], chunk_shape, lod, len(lods), grid_origin,
            mesh_shape)
        for lod in range(len(lods))
    ]
    fragment_positions = [nodes for submeshes, nodes in lods]
    lods = [submeshes for submeshes, nodes in lods]

    manifest = MultiLevelPrecomputedMeshManifest(
        segment_id=label,
        chunk_shape=chunk_shape,
        grid_origin=grid_origin,
        num_lods=len(lods),
        lod_scales=[2 ** i for i in range(len(lods))],
        vertex_offsets=[[0, 0, 0]] * len(lods),
        num_fragments_per_lod=[len(lods[lod]) for lod in range(len(lods))],
        fragment_positions=fragment_positions,
        fragment_offsets=[],
    )

    vqb = int(mesh_info["vertex_quantization_bits"])

    mesh_binaries = []
    for lod, submeshes in enumerate(lods):
        for frag_no, submesh in enumerate(submeshes):
            if len(submesh.faces) == 0:
                manifest.fragment_offsets.append(0)
                mesh_binaries.append(b"")
                continue
            stored = to_stored_model_space(
                submesh.vertices, manifest, lod=lod,
                vertex_quantization_bits=vqb, frag=frag_no)
            try:
                binary = draco_fmt.encode(
                    stored, submesh.faces,
                    quantization_bits=vqb,
                    compression_level=draco_compression_level,
                )
            except draco_fmt.EncodingFailedException:
                binary = b""
            manifest.fragment_offsets.append(len(binary))
            mesh_binaries.append(binary)

    return (manifest, b"".join(mesh_binaries))


# ---------------------------------------------------------------------------

def get_mesh_filenames_subset(layer_path: str, mesh_dir: str, prefix: str):
    """(multires.py:181-203)"""
    prefix = f'{mesh_dir}/{prefix}'
    segids = defaultdict(list)
    cf = CloudFiles(layer_path)
    meshexpr = re.compile(r'(\d+):(\d+):')
    for filename in cf.list(prefix=prefix):
        filename = filename.split("/")[-1]
        matches = re.search(meshexpr, filename)
        if not matches:
            continue
        segid, lod = matches.groups()
        segid, lod = int(segid), int(lod)
        if lod != 0:
            continue
        segids[segid].append(filename)
    return segids


def MultiResUnshardedMeshMergeTask(
        cloudpath: str, prefix: str,
        cache_control: bool = False,
        draco_compression_level: int = 1,
        mesh_dir: Optional[str] = None,
        num_lod: int = 1,
        min_chunk_size=(512, 512, 512),
        progress: bool = False):
    """(multires.py:44-81): gather each label's unsharded fragment
    files, build the multilod draco representation, write
    {label}.index + {label}."""
    vol, mesh_dir, mesh_info = _mesh_meta(cloudpath, mesh_dir)
    files_per_label = get_mesh_filenames_subset(cloudpath, mesh_dir, prefix)
    cf = CloudFiles(f"{cloudpath.rstrip('/')}/{mesh_dir}")
    for label, filenames in files_per_label.items():
        files = [cf.get(fn) for fn in sorted(filenames)]
        meshes = [Mesh.from_precomputed(f) for f in files if f]
        mesh = meshops.consolidate(
            Mesh.concatenate(*meshes, id=label))
        manifest, binary = process_mesh(
            vol, mesh_info, label, mesh, num_lod, min_chunk_size,
            draco_compression_level)
        if manifest is None:
            continue
        cf.put(f"{label}.index", manifest.to_binary(),
               cache_control="no-cache")
        cf.put(f"{label}", binary, cache_control="no-cache")


def locations_for_labels(layer_path: str, mesh_dir: str,
                         resolution,
                         labels: List[int]) -> Dict[int, List[str]]:
    """(multires.py:471-482): spatial-index filenames -> ".frags" names
    (chunk bbox in voxels)."""
    index = SpatialIndex(layer_path, mesh_dir)
    index_filenames = index.file_locations_per_label(labels)
    resolution = np.asarray(resolution, dtype=np.float64)
    out = {}
    for label, locations in index_filenames.items():
        frags = []
        for location in locations:
            stem = re.sub(r'\.spatial$', '', location)
            coords = [float(x) for p in stem.split('_')
                      for x in p.split('-')]
            mins = np.array(coords[0::2]) / resolution
            maxs = np.array(coords[1::2]) / resolution
            bbx = Bbox(np.round(mins).astype(int),
                       np.round(maxs).astype(int))
            frags.append(bbx.to_filename() + '.frags')
        out[label] = frags
    return out


def labels_for_shard(layer_path: str, mesh_dir: str,
                     spec: sharding.ShardingSpecification,
                     shard_no: str) -> List[int]:
    """(multires.py:484-508): precomputed {shard}.labels if present,
    else recompute from the spatial index."""
    cf = CloudFiles(f"{layer_path.rstrip('/')}/{mesh_dir}")
    labels = cf.get_json(shard_no + '.labels')
    if labels is not None:
        return labels
    all_labels = SpatialIndex(layer_path, mesh_dir).query()
    shard_labels = sharding.assign_labels_to_shards(
        np.asarray(all_labels, dtype=np.uint64),
        spec.preshift_bits, spec.shard_bits, spec.minishard_bits,
        hash=spec.hash)
    return shard_labels.get(shard_no, [])


def collect_mesh_fragments(layer_path: str, mesh_dir: str,
                           frag_path: Optional[str],
                           labels, filenames,
                           ) -> Dict[int, List[Mesh]]:
    """(multires.py:404-460): read each ".frags" MapBuffer once, pull
    every wanted label; fragments sorted by filename for determinism."""
    frag_prefix = frag_path or layer_path
    cf = CloudFiles(frag_prefix)
    all_meshes = defaultdict(list)
    for filename in sorted(filenames):
        content = cf.get(f"{mesh_dir}/{filename}")
        if content is None:
            continue
        fragment = MapBuffer(content, frombytesfn=Mesh.from_precomputed)
        for label in labels:
            try:
                mesh = fragment[label]
            except KeyError:
                continue
            mesh.id = label
            all_meshes[label].append((filename, mesh))
    for label in all_meshes:
        all_meshes[label].sort(key=lambda pair: pair[0])
        all_meshes[label] = [pair[1] for pair in all_meshes[label]]
    return all_meshes


def create_mesh_shard(vol, mesh_info: dict, meshes: Dict[int, Mesh],
                      num_lod: int, draco_compression_level: int,
                      shard_no: str, min_chunk_size):
    """(multires.py:362-401)"""
    spec = sharding.ShardingSpecification.from_dict(mesh_info["sharding"])
    processed = {
        label: process_mesh(
            vol, mesh_info, label, mesh, num_lod, min_chunk_size,
            draco_compression_level)
        for label, mesh in meshes.items()
    }
    data_offset = {
        label: len(manifest)
        for label, (manifest, binary) in processed.items()
        if manifest is not None and len(binary) > 0
    }
    blobs = {
        label: binary + manifest.to_binary()
        for label, (manifest, binary) in processed.items()
        if manifest is not None and len(binary) > 0
    }
    if len(blobs) == 0:
        return None, None
    shard_files = sharding.synthesize_shard_files(spec, blobs, data_offset)
    if len(shard_files) != 1:
        raise ValueError(
            "Only one shard file should be generated per task. "
            "Expected: {} Got: {} ".format(
                str(shard_no), ", ".join(shard_files.keys())))
    filename = next(iter(shard_files.keys()))
    return filename, shard_files[filename]


def MultiResShardedMeshMergeTask(
        cloudpath: str,
        shard_no: str,
        draco_compression_level: int = 1,
        mesh_dir: Optional[str] = None,
        frag_path: Optional[str] = None,
        cache: Optional[bool] = False,
        num_lod: int = 1,
        spatial_index_db: Optional[str] = None,
        min_chunk_size=(128, 128, 128),
        progress: bool = False):
    """(multires.py:207-260): gather this shard's labels' fragments from
    the ".frags" MapBuffer files located via the spatial index,
    LOD-simplify + octree-chunk + draco-encode each label, synthesize
    ONE neuroglancer shard file."""
    vol, mesh_dir, mesh_info = _mesh_meta(cloudpath, mesh_dir)
    spec = sharding.ShardingSpecification.from_dict(mesh_info["sharding"])

    labels = labels_for_shard(cloudpath, mesh_dir, spec, shard_no)
    locations = locations_for_labels(cloudpath, mesh_dir, vol.resolution,
                                     labels)
    filenames = set(itertools.chain(*locations.values()))
    labels = set(locations.keys())
    del locations
    meshes = collect_mesh_fragments(
        cloudpath, mesh_dir, frag_path, labels, filenames)
    del filenames

    for label in labels:
        mesh = Mesh.concatenate(*meshes[label], id=label)
        meshes[label] = meshops.consolidate(mesh)
    del labels

    fname, shard = create_mesh_shard(
        vol, mesh_info, meshes, num_lod, draco_compression_level,
        shard_no, min_chunk_size)
    del meshes

    if shard is None:
        return

    cf = CloudFiles(f"{cloudpath.rstrip('/')}/{mesh_dir}")
    cf.put(
        fname, shard,
        compress=False,
        content_type='application/octet-stream',
        cache_control='no-cache',
    )


def MultiResShardedFromUnshardedMeshMergeTask(
        src: str,
        dest: str,
        shard_no: str,
        cache_control: bool = False,
        draco_compression_level: int = 1,
        mesh_dir: Optional[str] = None,
        num_lod: int = 1,
        progress: bool = False,
        min_chunk_size=(512, 512, 512)):
    """(multires.py:262-306): convert an existing UNSHARDED legacy mesh
    layer into sharded multires — gather each shard label's legacy
    fragment files from src, shard-merge into dest."""
    svol, smesh_dir, _ = _mesh_meta(src, mesh_dir)
    dvol, dmesh_dir, dmesh_info = _mesh_meta(dest, mesh_dir or smesh_dir)
    spec = sharding.ShardingSpecification.from_dict(dmesh_info["sharding"])

    cf_dest = CloudFiles(f"{dest.rstrip('/')}/{dmesh_dir}")
    labels = cf_dest.get_json(shard_no + ".labels") or []

    cf_src = CloudFiles(f"{src.rstrip('/')}/{smesh_dir}")
    files_per_label = get_mesh_filenames_subset(src, smesh_dir, "")
    meshes = {}
    for label in labels:
        fnames = sorted(files_per_label.get(int(label), []))
        frags = [cf_src.get(fn) for fn in fnames]
        frags = [Mesh.from_precomputed(f) for f in frags if f]
        if not frags:
            continue
        meshes[int(label)] = meshops.consolidate(
            Mesh.concatenate(*frags, id=int(label)))

    fname, shard = create_mesh_shard(
        dvol, dmesh_info, meshes, num_lod, draco_compression_level,
        shard_no, min_chunk_size)
    if shard is None:
        return
    cf_dest.put(fname, shard, compress=False,
                content_type='application/octet-stream',
                cache_control='no-cache')
