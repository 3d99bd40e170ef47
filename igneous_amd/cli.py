"""Thin CLI mirroring the reference's meshing commands
(/root/reference/igneous_cli/cli.py:966-1071 `mesh forge`, :888-965
`execute`). The SQS/FileQueue worker fleet is replaced by the in-process
GPU dispatcher: `mesh forge` creates the tasks and, unless
--no-execute, runs them immediately on this node's GPU(s); under
`torch.distributed.run` each rank executes its shard.

    python -m igneous_amd mesh forge file:///data/seg --mip 0 --shape 448,448,448
"""
from __future__ import annotations

import click

from .dispatch import execute_tasks, rank_world
from .task_creation import create_meshing_tasks


def _vec3(_ctx, _param, value):
    if value is None:
        return None
    parts = [int(x) for x in str(value).replace("x", ",").split(",")]
    if len(parts) == 1:
        parts = parts * 3
    if len(parts) != 3:
        raise click.BadParameter("expected x,y,z")
    return tuple(parts)


@click.group()
def main():
    """igneous_amd — MI355X-native meshing with igneous's API."""


@main.group()
def mesh():
    """Create 3D meshes from a segmentation. (reference cli.py:966-1034)"""


@mesh.command()
@click.argument("path")
@click.option("--queue", default=None, help="ignored: tasks run in-process "
              "on this node's GPUs (the reference's SQS/FileQueue fleet is "
              "out of scope)")
@click.option("--mip", default=0, help="mesh this mip level of the pyramid")
@click.option("--shape", default="448,448,448", callback=_vec3,
              help="task shape (reference default 448^3)")
@click.option("--simplify/--skip-simplify", is_flag=True, default=True,
              help="quadric edge-collapse simplification (factor 100)")
@click.option("--max-error", default=40.0,
              help="max simplification error in physical units")
@click.option("--dust-threshold", default=None, type=int,
              help="skip labels smaller than this voxel count")
@click.option("--dir", "mesh_dir", default=None,
              help="mesh subdirectory (overrides info)")
@click.option("--compress", default="gzip", help="gzip or none")
@click.option("--spatial-index/--no-spatial-index", default=True)
@click.option("--sharded", is_flag=True, default=False,
              help="write MapBuffer .frags fragment files for the "
                   "sharded multires merge instead of individual meshes")
@click.option("--fill-missing", is_flag=True, default=False)
@click.option("--closed-edge/--open-edge", default=True,
              help="close meshes at dataset boundaries")
@click.option("--no-execute", is_flag=True, default=False,
              help="only create tasks + mesh info; do not run them")
def forge(path, queue, mip, shape, simplify, max_error, dust_threshold,
          mesh_dir, compress, spatial_index, sharded, fill_missing,
          closed_edge, no_execute):
    """(Re)Generate meshes for the segmentation at PATH
    (reference mesh_forge, cli.py:1035-1071)."""
    tasks = create_meshing_tasks(
        path, mip, shape,
        simplification=simplify,
        max_simplification_error=max_error,
        mesh_dir=mesh_dir,
        dust_threshold=dust_threshold,
        spatial_index=spatial_index,
        sharded=sharded,
        fill_missing=fill_missing,
        compress=(None if compress in ("none", "False", "") else compress),
        closed_dataset_edges=closed_edge,
    )
    click.echo(f"{len(tasks)} MeshTasks over {path}")
    if no_execute:
        # consume nothing: the iterator stays lazy; info/provenance written
        return
    rank, world = rank_world()
    n = execute_tasks(tasks)
    click.echo(f"rank {rank}/{world}: executed {n} tasks")


@mesh.command()
@click.argument("path")
@click.option("--dir", "mesh_dir", default=None,
              help="mesh subdirectory (overrides info)")
@click.option("--magnitude", default=3)
def merge(path, mesh_dir, magnitude):
    """Stage 2: merge mesh fragment manifests (reference mesh merge,
    cli.py:1082-1103)."""
    from .task_creation import create_mesh_manifest_tasks
    tasks = create_mesh_manifest_tasks(path, magnitude=magnitude,
                                       mesh_dir=mesh_dir)
    n = execute_tasks(tasks)
    click.echo(f"executed {n} manifest tasks")


@mesh.command(name="merge-sharded")
@click.argument("path")
@click.option("--nlod", "num_lod", default=0,
              help="number of additional levels of detail")
@click.option("--vqb", "vertex_quantization_bits", default=16,
              type=click.Choice(["10", "16"]))
@click.option("--shard-index-bytes", default=2 ** 13)
@click.option("--minishard-index-bytes", default=2 ** 15)
@click.option("--min-shards", default=1)
@click.option("--min-chunk-size", default="256,256,256", callback=_vec3)
@click.option("--dir", "mesh_dir", default=None)
@click.option("--frag-path", default=None)
@click.option("--draco-compression-level", default=7)
def merge_sharded(path, num_lod, vertex_quantization_bits,
                  shard_index_bytes, minishard_index_bytes, min_shards,
                  min_chunk_size, mesh_dir, frag_path,
                  draco_compression_level):
    """Stage 2 (sharded): merge .frags fragments into multires
    neuroglancer shard files (reference mesh merge-sharded,
    cli.py:1105-1161)."""
    from .task_creation import create_sharded_multires_mesh_tasks
    tasks = create_sharded_multires_mesh_tasks(
        path,
        shard_index_bytes=shard_index_bytes,
        minishard_index_bytes=minishard_index_bytes,
        min_shards=min_shards,
        num_lod=num_lod,
        draco_compression_level=draco_compression_level,
        vertex_quantization_bits=int(vertex_quantization_bits),
        mesh_dir=mesh_dir,
        frag_path=frag_path,
        min_chunk_size=min_chunk_size,
    )
    n = execute_tasks(tasks)
    click.echo(f"executed {n} sharded multires merge tasks")


@main.command()
@click.argument("queue", required=False)
def execute(queue):
    """The reference's worker poll loop (cli.py:888-965). Not applicable:
    meshgine runs tasks in-process at forge time; use `mesh forge`
    (optionally under torch.distributed.run for multi-GPU)."""
    raise click.ClickException(
        "meshgine has no external queue: `mesh forge` executes tasks "
        "in-process on this node's GPUs (see INTEGRATION.md)")


if __name__ == "__main__":
    main()
