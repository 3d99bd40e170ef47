"""LocalTaskQueue parallel path: payload-serialized tasks execute in
worker processes (the reference's --parallel model, cli.py:915-933)."""
import os

from igneous_amd.queue import LocalTaskQueue, RegisteredTask


class _TouchTask(RegisteredTask):
    def __init__(self, path, name):
        super().__init__(path, name)
        self.path = path
        self.name = name

    def execute(self):
        with open(os.path.join(self.path, self.name), "w") as f:
            f.write(self.name)


def test_parallel_queue_executes_all(tmp_path):
    tasks = [_TouchTask(str(tmp_path), f"t{i}") for i in range(8)]
    with LocalTaskQueue(parallel=2) as tq:
        n = tq.insert(tasks)
    assert n == 8
    assert sorted(os.listdir(tmp_path)) == sorted(f"t{i}" for i in range(8))


def test_payload_roundtrip_executes(tmp_path):
    t = _TouchTask(str(tmp_path), "x")
    t2 = RegisteredTask.deserialize(t.payload())
    t2.execute()
    assert (tmp_path / "x").exists()
