from .mesh import (
    MeshTask,
    MeshManifestPrefixTask,
    MeshManifestFilesystemTask,
    TransferMeshFilesTask,
    DeleteMeshFilesTask,
    set_mesher,
)
from .multires import (
    MultiResShardedMeshMergeTask,
    MultiResUnshardedMeshMergeTask,
    MultiResShardedFromUnshardedMeshMergeTask,
)
from .spatial_index import SpatialIndexTask
