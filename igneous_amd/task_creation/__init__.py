from .mesh import (
    create_meshing_tasks,
    create_mesh_manifest_tasks,
    create_sharded_multires_mesh_tasks,
    create_unsharded_multires_mesh_tasks,
    configure_multires_info,
    create_spatial_index_mesh_tasks,
    create_mesh_deletion_tasks,
    create_xfer_meshes_tasks,
    create_sharded_multires_mesh_from_unsharded_tasks,
)
from .common import FinelyDividedTaskIterator, num_tasks
