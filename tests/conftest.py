import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run with -m gpu on a GPU host)")


def pytest_collection_modifyitems(config, items):
    if config.getoption("-m", default=""):
        return
    # default runs skip gpu tests unless explicitly selected
    skip = pytest.mark.skip(reason="gpu test: run with -m gpu on a GPU host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tmp_layer_path(tmp_path):
    return f"file://{tmp_path}/layer"


@pytest.fixture
def oracle_mesher():
    """Inject the CPU oracle as MeshTask's mesher for host-logic tests
    (checker role only — the product path stays HIP-only)."""
    import oracle as _oracle
    from igneous_amd.tasks import mesh as mesh_mod

    def fn(data, resolution=(1, 1, 1), reduction_factor=0, max_error=40.0,
           voxel_centered=True, **kw):
        return _oracle.mesh_chunk(
            data, resolution=resolution, reduction_factor=reduction_factor,
            max_error=max_error, voxel_centered=voxel_centered)

    mesh_mod.set_mesher(fn)
    yield fn
    mesh_mod.set_mesher(None)


@pytest.fixture
def oracle_simplifier():
    """Inject the CPU oracle's quadric simplifier as the multires LOD
    simplifier (checker role; the product path is mg_simplify_mesh)."""
    import oracle as _oracle
    from igneous_amd.meshes import Mesh
    from igneous_amd.tasks import multires as multires_mod

    def fn(mesh, target_count):
        nt = int(mesh.faces.shape[0])
        target = max(int(target_count), 1)
        if nt <= target:
            return Mesh(mesh.vertices.copy(), mesh.faces.copy(),
                        id=mesh.id)
        rf = max(nt // target, 2)
        v, f = _oracle.simplify_mesh(mesh.vertices, mesh.faces, rf, 1e30)
        return Mesh(v, f, id=mesh.id)

    multires_mod.set_simplifier(fn)
    yield fn
    multires_mod.set_simplifier(None)
