from .mesh import (
    MeshTask,
    MeshManifestPrefixTask,
    MeshManifestFilesystemTask,
    set_mesher,
)
