"""In-process task queue replacing the reference's external `task-queue`
dependency for the MeshTask path (RegisteredTask JSON serialization +
LocalTaskQueue, /root/reference/igneous/tasks/mesh/mesh.py:19,39 and
igneous_cli/cli.py:888-965). SQS/FileQueue are out of scope;
the fan-out target is the GPUs of one node, not a worker fleet.
"""
from __future__ import annotations

import inspect
import json
import multiprocessing as mp
from typing import Iterable


class RegisteredTask:
    """Task serializable as JSON by its constructor arguments (mirrors
    taskqueue.RegisteredTask semantics used by MeshTask: ctor kwargs are
    recorded; .payload() round-trips through JSON)."""

    _registry: dict = {}

    def __init_subclass__(cls, **kw):
        super().__init_subclass__(**kw)
        RegisteredTask._registry[cls.__name__] = cls

    def __init__(self, *args, **kwargs):
        params = list(inspect.signature(
            type(self).__init__).parameters.values())[1:]  # drop self
        record = {}
        for i, a in enumerate(args):
            record[params[i].name] = a
        record.update(kwargs)
        self._args = _jsonable(record)

    def payload(self) -> str:
        return json.dumps({
            "class": type(self).__name__,
            "module": type(self).__module__,
            "args": self._args,
        })

    @staticmethod
    def deserialize(payload: str) -> "RegisteredTask":
        d = json.loads(payload)
        cls = RegisteredTask._registry.get(d["class"])
        if cls is None and d.get("module"):
            # worker processes (spawn) have an empty registry: import the
            # defining module, which registers the class on import
            import importlib
            mod = importlib.import_module(d["module"])
            cls = getattr(mod, d["class"], None) or                 RegisteredTask._registry.get(d["class"])
        if cls is None:
            raise KeyError(f"unknown task class {d['class']}")
        return cls(**d["args"])

    def execute(self):
        raise NotImplementedError


def _jsonable(obj):
    import numpy as np
    if isinstance(obj, dict):
        return {k: _jsonable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_jsonable(v) for v in obj]
    if isinstance(obj, np.ndarray):
        return obj.tolist()
    if isinstance(obj, np.generic):
        return obj.item()
    return obj


def _run_payload(payload: str) -> None:
    RegisteredTask.deserialize(payload).execute()


class LocalTaskQueue:
    """Executes tasks in-process (parallel=1) or via a process pool,
    matching the reference's LocalTaskQueue usage (README.md:63-77)."""

    def __init__(self, parallel: int = 1, progress: bool = False):
        self.parallel = max(1, int(parallel))

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False

    def insert(self, tasks: Iterable, **kw) -> int:
        n = 0
        if self.parallel == 1:
            for t in tasks:
                _execute_one(t)
                n += 1
        else:
            payloads = []
            for t in tasks:
                if isinstance(t, RegisteredTask):
                    payloads.append(t.payload())
                else:  # partial / plain callable: run serially
                    t()
                    n += 1
            with mp.get_context("spawn").Pool(self.parallel) as pool:
                pool.map(_run_payload, payloads)
            n += len(payloads)
        return n

    insert_all = insert


def _execute_one(task) -> None:
    if isinstance(task, RegisteredTask):
        task.execute()
    else:  # functools.partial of a @queueable function, or plain callable
        task()
