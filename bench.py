#!/usr/bin/env python3
"""Benchmark of the meshgine hot path (BASELINE.json metric): Mvoxels/s
meshed on the 512^3 uint64 ~50k-label synthetic segmentation chunk
(configs[2] — the configuration the metric is quoted on), one process per
GPU, weak scaling (each rank meshes its own chunk per step).

    python bench.py --gpus N --steps K --warmup W

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU; ranks synchronize with a barrier + torch.cuda.synchronize around
the timed region and report the MAX elapsed over ranks. Rank 0 prints ONE
JSON line.

A "step" = one mg_mesh_chunk over the full chunk: marching-cubes count +
emit, label partition, vertex weld — outputs complete in HBM
(MG_FLAG_DEVICE_ONLY; the PCIe-inclusive host-extract rate is reported in
DESIGN.md, never as `value`). Inputs are resident in HBM when the timed
region starts (staged in warmup; MG_FLAG_SKIP_H2D).

cpu_baseline: this repo's CPU oracle (oracle/, kind "port" — zmesh is not
installable offline) timed on the host's cores over a bounded
sample of the same workload, igneous's own process-per-chunk parallelism
model (cli.py:915-933).
"""
import argparse
import json
import multiprocessing as mp
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

WORKLOAD = "512^3 uint64 Voronoi chunk, ~50k labels, seed=303 (BASELINE configs[2])"
SHAPE = (512, 512, 512)
K_SEEDS = 50000
SEED = 303
RESOLUTION = (16.0, 16.0, 40.0)
PEAK_HBM = 8.0e12  # B/s, MI355X spec (MI355X_MICROARCH.md)


_BASE_DATA = None   # fork-shared (copy-on-write) chunk for baseline workers
_BASE_RED = 0


def _oracle_baseline_worker(idx):
    """Mesh one 64^3 (+1 overlap) subchunk of the fork-shared chunk —
    the reference's process-per-chunk --parallel worker model."""
    import oracle  # oracle/ is on sys.path (checker/baseline leg only)
    n = SHAPE[0] // 64
    z, rem = divmod(idx, n * n)
    y, x = divmod(rem, n)
    sub = np.asfortranarray(
        _BASE_DATA[x * 64:x * 64 + 65,
                   y * 64:y * 64 + 65,
                   z * 64:z * 64 + 65])
    oracle.mesh_chunk(sub, resolution=RESOLUTION,
                      reduction_factor=_BASE_RED, max_error=40.0)
    return sub.size


def cpu_baseline(data: np.ndarray, reduction: int = 0,
                 budget_s: float = 25.0) -> dict:
    """Time the CPU oracle on a bounded sample of the same chunk:
    process-per-subchunk over all host cores, ~budget_s of wall time.
    Data is fork-shared so workers pay no serialization."""
    global _BASE_DATA, _BASE_RED
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import oracle
    oracle.build()
    _BASE_DATA = data
    _BASE_RED = reduction
    cores = os.cpu_count() or 1
    n = SHAPE[0] // 64
    total_chunks = n ** 3  # 512 tasks of 65^3
    # calibrate with one subchunk, single process
    t0 = time.perf_counter()
    _oracle_baseline_worker(0)
    per_chunk = time.perf_counter() - t0
    target = max(cores, min(total_chunks,
                            int(budget_s / per_chunk * cores)))
    chosen = list(range(target))
    t0 = time.perf_counter()
    with mp.get_context("fork").Pool(cores) as pool:
        vox = sum(pool.map(_oracle_baseline_worker, chosen, chunksize=1))
    elapsed = time.perf_counter() - t0
    return {
        "value": round(vox / elapsed / 1e6, 2),
        "unit": "Mvox/s",
        "cores": cores,
        "kind": "port",
        "sample": (f"{len(chosen)} x 65^3 subchunks of the same 512^3 "
                   f"chunk (reduction_factor={reduction}), "
                   f"{cores}-process pool, {elapsed:.1f}s"),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--simplify", action="store_true",
                    help="BASELINE config 5: simplification_factor=100, "
                         "max_error=40 through the quadric-collapse kernels")
    ap.add_argument("--mode", choices=["chunk512", "chunks256"],
                    default="chunk512",
                    help="chunk512: the headline single-chunk device rate; "
                         "chunks256: BASELINE configs[3] shape — a fan-out "
                         "of 256^3 chunks through the FULL path (H2D + "
                         "kernels + host extract) on --streams HIP "
                         "streams/contexts per GPU")
    ap.add_argument("--streams", type=int, default=3)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(args.gpus, world)

    import torch
    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        # nccl (RCCL) wants one DISTINCT device per rank; fall back to
        # gloo when ranks outnumber devices (single-GPU smoke of the
        # distributed path)
        ndev = torch.cuda.device_count() if torch.cuda.is_available() else 0
        backend = "nccl" if 0 < world <= ndev else "gloo"
        dist.init_process_group(backend=backend)
        if ndev:
            torch.cuda.set_device(local_rank % ndev)

    from igneous_amd.engine import Engine
    from igneous_amd.synth import voronoi_labels

    # weak scaling: every rank meshes its own copy of the same synthetic
    # chunk (chunks are independent; identical per-rank work). Rank 0
    # generates (or loads) the cached chunk; other ranks wait, then read
    # the cache — avoids N concurrent 30s generations.
    if dist is not None and rank != 0:
        dist.barrier()
    data = voronoi_labels(SHAPE, K_SEEDS, SEED, dtype=np.uint64)
    if dist is not None and rank == 0:
        dist.barrier()
    from igneous_amd.engine import load_library
    ndev = max(1, load_library().mg_device_count())
    eng = Engine.get(local_rank % ndev)

    red = 100 if args.simplify else 0

    if args.mode == "chunks256":
        # configs[3] shape: independent 256^3 chunks, full path incl.
        # H2D and host extract, overlapped on per-thread HIP contexts
        from concurrent.futures import ThreadPoolExecutor
        from igneous_amd import engine as engine_mod
        engine_mod.PER_THREAD_CTX = True
        nchunks = 8
        chunks = [voronoi_labels((256, 256, 256), 6250, 1000 + i,
                                 dtype=np.uint64) for i in range(nchunks)]
        pool = ThreadPoolExecutor(max_workers=args.streams)
        thread_engines = []

        def mesh_one(chunk):
            e = engine_mod.Engine.get(local_rank % ndev)
            if e not in thread_engines:
                thread_engines.append(e)
            # results discarded before the thread's next call: zero-copy
            return e.mesh_chunk(chunk, resolution=RESOLUTION,
                                reduction_factor=red, max_error=40.0,
                                copy=False)

        def step(skip_h2d=True):
            list(pool.map(mesh_one, chunks))

        step_vox = nchunks * 256 ** 3
    else:
        def step(skip_h2d=True):
            eng.mesh_chunk(data, resolution=RESOLUTION, reduction_factor=red,
                           max_error=40.0, device_only=True,
                           skip_h2d=skip_h2d)

        step_vox = int(np.prod(SHAPE))

    # warmup (first call stages the labels into HBM)
    step(skip_h2d=False)
    for _ in range(max(0, args.warmup - 1)):
        step()

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if args.mode == "chunks256" and thread_engines:
        stats = thread_engines[0].stats()  # per-thread ctxs did the work
    else:
        stats = eng.stats()
    total_vox = step_vox * args.steps * world
    mvox_s = total_vox / elapsed / 1e6

    if rank != 0:
        if dist is not None:
            dist.destroy_process_group()
        return

    # roofline for the dominant kernel (k_emit: the marching-cubes scan +
    # triangle emit over the whole volume). Algorithmic bytes per launch =
    # one read of each label voxel.
    algo_bytes = float(stats["bytes_read_algorithmic"])
    ms_emit = stats["ms_emit"]
    achieved = algo_bytes / (ms_emit * 1e-3) if ms_emit > 0 else 0.0
    traffic = None
    tfile = os.path.join(REPO, "profiles", "roofline_traffic.json")
    if os.path.exists(tfile):
        try:
            tj = json.load(open(tfile))
            traffic = tj.get("k_emit_bytes_per_launch")
        except Exception:
            traffic = None
    roofline = {
        "bound": "hbm",
        "achieved": round(achieved / 1e9, 1),
        "peak": round(PEAK_HBM / 1e9, 1),
        "unit": "GB/s",
        "frac": round(achieved / PEAK_HBM, 4),
        "traffic": traffic,
    }

    cpu = None
    if not args.no_cpu_baseline and rank == 0 and world == 1:
        cpu = cpu_baseline(data, reduction=red)

    line = {
        "metric": "Mvoxels/s meshed (512^3 uint64 seg chunk)",
        "value": round(mvox_s, 1),
        "unit": "Mvox/s",
        "n_gpus": n_gpus if world == 1 else world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1e3, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "u64",
        "data": "synthetic",
        "config": {
            "workload": (WORKLOAD if args.mode == "chunk512" else
                         f"{8} x 256^3 u64 chunks (configs[3] shape), full "
                         f"H2D+extract path, {args.streams} streams/GPU")
                        + (" + simplification_factor=100" if args.simplify else ""),
            "chunk": list(SHAPE),
            "labels": K_SEEDS,
            "resolution_nm": list(RESOLUTION),
            "simplification": red,
            "n_labels_meshed": int(stats["n_labels"]),
            "total_tris": int(stats["total_tris"]),
            "kernel_ms": {k: round(stats[k], 3) for k in (
                "ms_count", "ms_scan", "ms_emit", "ms_partition",
                "ms_weld", "ms_simplify", "ms_total")},
        },
        "roofline": roofline,
        "cpu_baseline": cpu,
    }
    print(json.dumps(line), flush=True)
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
