"""igneous_amd — MI355X-native engine for igneous's MeshTask hot path.

Drop-in surface mirroring the reference package root
(/root/reference/igneous/__init__.py): Mesher (GPU-backed zmesh.Mesher
equivalent), LocalTaskQueue/RegisteredTask (in-process queue), the mesh
tasks, and create_meshing_tasks. Compute runs as hand-written HIP/CDNA4
kernels behind the C ABI in include/meshgine.h.
"""
from .mesher import Mesher, simplify_fqmr, chunk_mesh
from .meshes import Mesh
from .queue import LocalTaskQueue, RegisteredTask
from .tasks import (
    MeshTask, MeshManifestPrefixTask, MeshManifestFilesystemTask,
    TransferMeshFilesTask, DeleteMeshFilesTask,
    MultiResShardedMeshMergeTask, MultiResUnshardedMeshMergeTask,
)
from .task_creation import (
    create_meshing_tasks, create_sharded_multires_mesh_tasks,
    create_unsharded_multires_mesh_tasks,
)
from .volume import PrecomputedVolume
from .lib import Bbox, Vec

__version__ = "0.1.0"
