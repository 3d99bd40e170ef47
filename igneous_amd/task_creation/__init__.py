from .mesh import create_meshing_tasks, create_mesh_manifest_tasks
from .common import FinelyDividedTaskIterator, num_tasks
