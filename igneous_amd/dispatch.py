"""Multi-GPU task dispatch — the reference's queue-worker fan-out
(igneous execute, /root/reference/igneous_cli/cli.py:888-965) repointed at
the GPUs of one node: one process per GPU (torch.distributed launch, rank
= GPU), tasks sharded round-robin. No collective on the data path —
chunks are independent by construction (1vx overlap duplication instead
of halo exchange, mesh.py:155-160).
"""
from __future__ import annotations

import os
from typing import Iterable, Optional


def rank_world() -> tuple:
    return (int(os.environ.get("RANK", "0")),
            int(os.environ.get("WORLD_SIZE", "1")))


def shard_tasks(tasks: Iterable, rank: Optional[int] = None,
                world: Optional[int] = None):
    """Round-robin assignment of a task iterator across ranks — the same
    linear-index split FinelyDividedTaskIterator's __getitem__ supports in
    the reference (common.py:77-81), without materializing the list."""
    r, w = rank_world()
    rank = r if rank is None else rank
    world = w if world is None else world
    for i, task in enumerate(tasks):
        if i % world == rank:
            yield task


def _run_one(task) -> int:
    if callable(getattr(task, "execute", None)):
        task.execute()
    else:
        task()
    return 1


def execute_tasks(tasks: Iterable, progress: bool = False,
                  barrier: bool = True, streams: int = 1) -> int:
    """Execute this rank's shard of `tasks` on this rank's GPU
    (MESHGINE_DEVICE defaults to LOCAL_RANK inside the engine). Returns
    the number of tasks this rank executed. With torch.distributed
    initialized (or WORLD_SIZE>1), synchronizes all ranks at the end.

    streams>1: a thread pool with one HIP stream/context per thread —
    in-flight chunks overlap H2D, kernels and D2H on one GPU (ctypes
    releases the GIL during engine calls)."""
    n = 0
    if streams > 1:
        from concurrent.futures import ThreadPoolExecutor
        from . import engine
        engine.PER_THREAD_CTX = True
        with ThreadPoolExecutor(max_workers=streams) as pool:
            n = sum(pool.map(_run_one, shard_tasks(tasks)))
    else:
        for task in shard_tasks(tasks):
            n += _run_one(task)
    if barrier:
        _, world = rank_world()
        if world > 1:
            import torch.distributed as dist
            if not dist.is_initialized():
                import torch
                backend = "nccl" if torch.cuda.is_available() else "gloo"
                dist.init_process_group(backend=backend)
            dist.barrier()
    return n
