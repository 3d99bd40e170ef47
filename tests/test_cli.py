"""CLI surface tests (mirror of the reference's `igneous mesh forge`)."""
import numpy as np
import pytest
from click.testing import CliRunner

from igneous_amd import PrecomputedVolume
from igneous_amd.cli import main
from igneous_amd.storage import CloudFiles


def test_forge_no_execute_writes_info(tmp_layer_path):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[2:30, 2:30, 2:30] = 4
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(32, 32, 32))
    runner = CliRunner()
    res = runner.invoke(main, [
        "mesh", "forge", tmp_layer_path, "--mip", "0",
        "--shape", "32,32,32", "--no-execute"])
    assert res.exit_code == 0, res.output
    assert "1 MeshTasks" in res.output
    cf = CloudFiles(tmp_layer_path)
    vol = PrecomputedVolume(tmp_layer_path)
    info = cf.get_json(f"{vol.info['mesh']}/info")
    assert info["@type"] == "neuroglancer_legacy_mesh"


def test_forge_executes_with_oracle(tmp_layer_path, oracle_mesher):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[2:30, 2:30, 2:30] = 4
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(32, 32, 32),
        mesh_dir="mesh")
    runner = CliRunner()
    res = runner.invoke(main, [
        "mesh", "forge", tmp_layer_path, "--mip", "0",
        "--shape", "32,32,32", "--skip-simplify", "--no-spatial-index"])
    assert res.exit_code == 0, res.output
    cf = CloudFiles(tmp_layer_path)
    assert cf.get("mesh/4:0:0-32_0-32_0-32") is not None


def test_execute_points_to_forge():
    res = CliRunner().invoke(main, ["execute", "queue://x"])
    assert res.exit_code != 0
    assert "in-process" in res.output


def test_merge_writes_manifests(tmp_layer_path, oracle_mesher):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[2:30, 2:30, 2:30] = 4
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(1, 1, 1), chunk_size=(32, 32, 32),
        mesh_dir="mesh")
    runner = CliRunner()
    res = runner.invoke(main, [
        "mesh", "forge", tmp_layer_path, "--mip", "0",
        "--shape", "32,32,32", "--skip-simplify", "--no-spatial-index"])
    assert res.exit_code == 0, res.output
    res = runner.invoke(main, ["mesh", "merge", tmp_layer_path])
    assert res.exit_code == 0, res.output
    cf = CloudFiles(tmp_layer_path)
    manifest = cf.get_json("mesh/4:0")
    assert manifest == {"fragments": ["4:0:0-32_0-32_0-32"]}


def test_merge_sharded_end_to_end(tmp_layer_path, oracle_mesher,
                                  oracle_simplifier):
    data = np.zeros((32, 32, 32), dtype=np.uint32)
    data[2:30, 2:30, 2:30] = 4
    PrecomputedVolume.from_numpy(
        data, tmp_layer_path, resolution=(4, 4, 40),
        chunk_size=(32, 32, 32), mesh_dir="mesh")
    runner = CliRunner()
    res = runner.invoke(main, [
        "mesh", "forge", tmp_layer_path, "--mip", "0",
        "--shape", "32,32,32", "--skip-simplify", "--sharded",
        "--spatial-index"])
    assert res.exit_code == 0, res.output
    res = runner.invoke(main, [
        "mesh", "merge-sharded", tmp_layer_path, "--nlod", "0",
        "--min-chunk-size", "16,16,16"])
    assert res.exit_code == 0, res.output
    cf = CloudFiles(tmp_layer_path)
    info = cf.get_json("mesh/info")
    assert info["@type"] == "neuroglancer_multilod_draco"
    shards = [n for n in cf.list("mesh/") if n.endswith(".shard")]
    assert shards
