#!/usr/bin/env python3
"""Benchmark of the meshgine hot path (BASELINE.json metric): Mvoxels/s
meshed on the 512^3 uint64 ~50k-label synthetic segmentation chunk
(configs[2] — the configuration the metric is quoted on), one process per
GPU, weak scaling (each rank meshes its own chunk per step).

    python bench.py --gpus N --steps K --warmup W

With no mode flags, FOUR benches run back-to-back and all appear in the
single JSON line's "configs" array (the top-level value/ms_per_step stay
the headline configs[2] device rate):
  - mc             BASELINE configs[2]: 512^3 u64 device-resident MC-only
  - simplify       BASELINE configs[4] ("config 5"): same chunk with
                   simplification_factor=100, max_error=40
  - chunks256      BASELINE configs[3] shape: 8 x 256^3 chunks through
                   the FULL path (H2D + kernels + host extract) on
                   --streams HIP streams/contexts per GPU
  - chunks256_prod the same fan-out with the reference's production
                   default simplification_factor=100
--mode/--simplify restrict the run to one bench (targeted profiling).

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU; ranks synchronize with a barrier + torch.cuda.synchronize around
the timed region and report the MAX elapsed over ranks. Rank 0 prints ONE
JSON line.

A "step" = one mg_mesh_chunk over the chunk (or the 8-chunk fan-out).
For the device-resident benches inputs are in HBM when the timed region
starts (staged in warmup; MG_FLAG_SKIP_H2D); the chunks256 bench times the
PCIe-inclusive production path and says so in its config.

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


def kernel_ms(stats: dict) -> dict:
    return {k: round(stats[k], 3) for k in (
        "ms_count", "ms_scan", "ms_emit", "ms_partition",
        "ms_weld", "ms_simplify", "ms_total")}


def run_chunk512(eng, data, steps, warmup, simplify, barrier_sync, max_rank):
    """Device-resident single-512^3-chunk bench (BASELINE configs[2] at
    simplify=0, configs[4]/'config 5' at simplify=1)."""
    red = 100 if simplify else 0

    def step(skip_h2d=True):
        eng.mesh_chunk(data, resolution=RESOLUTION, reduction_factor=red,
                       max_error=40.0, device_only=True, skip_h2d=skip_h2d)

    step(skip_h2d=False)
    for _ in range(max(0, warmup - 1)):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    barrier_sync()
    elapsed = max_rank(time.perf_counter() - t0)
    stats = eng.stats()
    return {
        "name": "simplify" if simplify else "mc",
        "value": None,  # filled by caller (needs world size)
        "elapsed": elapsed,
        "step_vox": int(np.prod(SHAPE)),
        "config": {
            "workload": WORKLOAD + (
                " + simplification_factor=100, max_error=40 "
                "(BASELINE configs[4])" if simplify else ""),
            "chunk": list(SHAPE),
            "labels": K_SEEDS,
            "resolution_nm": list(RESOLUTION),
            "simplification": red,
            "input_residency": "device (MG_FLAG_SKIP_H2D, MG_FLAG_DEVICE_ONLY)",
            "n_labels_meshed": int(stats["n_labels"]),
            "total_tris": int(stats["total_tris"]),
            "kernel_ms": kernel_ms(stats),
        },
        "stats": stats,
    }


def run_chunks256(local_rank, ndev, steps, warmup, streams,
                  barrier_sync, max_rank, red=0, name="chunks256"):
    """BASELINE configs[3] shape: independent 256^3 chunks, FULL path
    (H2D + kernels + host extract), overlapped on per-thread HIP
    contexts. PCIe-inclusive by design — reported as its own configs[]
    entry, never as the headline value."""
    from concurrent.futures import ThreadPoolExecutor
    from igneous_amd import engine as engine_mod
    from igneous_amd.synth import voronoi_labels
    engine_mod.PER_THREAD_CTX = True
    nchunks = 8
    seeds_per_chunk = 6250
    chunks = [voronoi_labels((256, 256, 256), seeds_per_chunk, 1000 + i,
                             dtype=np.uint64) for i in range(nchunks)]
    pool = ThreadPoolExecutor(max_workers=streams)

    def mesh_one(chunk):
        e = engine_mod.Engine.get(local_rank % ndev)
        # results discarded before the thread's next call: zero-copy
        e.mesh_chunk(chunk, resolution=RESOLUTION,
                     reduction_factor=red, max_error=40.0, copy=False)
        return e

    def step():
        return list(pool.map(mesh_one, chunks))

    step()
    for _ in range(max(0, warmup - 1)):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        engines = step()
    barrier_sync()
    elapsed = max_rank(time.perf_counter() - t0)
    # per-chunk stats: each engine ctx holds its LAST chunk's stats; one
    # extra untimed pass gives an aggregate across all 8 chunks
    agg = {"n_labels": 0, "total_tris": 0}
    for chunk in chunks:
        e = engine_mod.Engine.get(local_rank % ndev)
        e.mesh_chunk(chunk, resolution=RESOLUTION, reduction_factor=red,
                     max_error=40.0, copy=False)
        st = e.stats()
        agg["n_labels"] += int(st["n_labels"])
        agg["total_tris"] += int(st["total_tris"])
    pool.shutdown()
    return {
        "name": name,
        "value": None,
        "elapsed": elapsed,
        "step_vox": nchunks * 256 ** 3,
        "config": {
            "workload": (f"{nchunks} x 256^3 u64 Voronoi chunks "
                         f"({seeds_per_chunk} labels each, BASELINE "
                         f"configs[3] shape), FULL path: H2D + kernels + "
                         f"host extract, {streams} streams/GPU"
                         + (f", simplification_factor={red} (the "
                            f"reference's production default)"
                            if red else "")),
            "chunk": [256, 256, 256],
            "n_chunks": nchunks,
            "labels": seeds_per_chunk * nchunks,
            "resolution_nm": list(RESOLUTION),
            "simplification": red,
            "input_residency": "host (PCIe-inclusive production path)",
            "n_labels_meshed": agg["n_labels"],
            "total_tris": agg["total_tris"],
        },
        "stats": None,
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--simplify", action="store_true",
                    help="run ONLY the config-5 bench (simplification_"
                         "factor=100, max_error=40)")
    ap.add_argument("--mode", choices=["all", "chunk512", "chunks256"],
                    default="all",
                    help="all: mc + simplify + chunks256 benches (default); "
                         "chunk512: only the single-chunk device bench; "
                         "chunks256: only the full-path fan-out bench")
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

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def max_rank(elapsed):
        if dist is None:
            return elapsed
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())

    from igneous_amd.engine import Engine, load_library
    from igneous_amd.synth import voronoi_labels

    # weak scaling: every rank meshes its own copy of the same synthetic
    # chunk (chunks are independent; identical per-rank work). Rank 0
    # generates first (synth caches to disk); other ranks then hit the
    # cache instead of N concurrent generations.
    if dist is not None and rank != 0:
        dist.barrier()
    data = voronoi_labels(SHAPE, K_SEEDS, SEED, dtype=np.uint64)
    if dist is not None and rank == 0:
        dist.barrier()
    ndev = max(1, load_library().mg_device_count())
    eng = Engine.get(local_rank % ndev)

    which = []
    if args.simplify:
        which = ["simplify"]
    elif args.mode == "chunk512":
        which = ["mc"]
    elif args.mode == "chunks256":
        which = ["chunks256"]
    else:
        which = ["mc", "simplify", "chunks256", "chunks256_prod"]

    results = []
    for name in which:
        if name == "mc":
            results.append(run_chunk512(eng, data, args.steps, args.warmup,
                                        False, barrier_sync, max_rank))
        elif name == "simplify":
            results.append(run_chunk512(eng, data, args.steps, args.warmup,
                                        True, barrier_sync, max_rank))
        elif name == "chunks256":
            # chunks256 benches run LAST: they flip the engine module to
            # per-thread contexts
            results.append(run_chunks256(local_rank, ndev, args.steps,
                                         args.warmup, args.streams,
                                         barrier_sync, max_rank))
        else:
            # the reference's production DEFAULT: fan-out WITH
            # simplification_factor=100 (task_creation/mesh.py:216)
            results.append(run_chunks256(local_rank, ndev, args.steps,
                                         args.warmup, args.streams,
                                         barrier_sync, max_rank,
                                         red=100, name="chunks256_prod"))

    for r in results:
        total_vox = r["step_vox"] * args.steps * world
        r["value"] = round(total_vox / r["elapsed"] / 1e6, 1)
        r["ms_per_step"] = round(r["elapsed"] / args.steps * 1e3, 2)

    if rank != 0:
        if dist is not None:
            dist.destroy_process_group()
        return

    head = results[0]  # headline = first bench run (mc in the default set)

    # roofline for the dominant kernel (k_emit: the marching-cubes scan +
    # triangle emit over the whole volume). Algorithmic bytes per launch =
    # one read of each label voxel; launch duration from HIP
    # events around the kernel, recorded live by the engine.
    roofline = None
    if head["stats"] is not None:
        stats = head["stats"]
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
        cpu = cpu_baseline(data, reduction=100 if args.simplify else 0)

    line = {
        "metric": "Mvoxels/s meshed (512^3 uint64 seg chunk)",
        "value": head["value"],
        "unit": "Mvox/s",
        "n_gpus": n_gpus if world == 1 else world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": head["ms_per_step"],
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "u64",
        "data": "synthetic",
        "config": head["config"],
        "configs": [
            {"name": r["name"], "value": r["value"], "unit": "Mvox/s",
             "ms_per_step": r["ms_per_step"], "config": r["config"]}
            for r in results
        ],
        "roofline": roofline,
        "cpu_baseline": cpu,
    }
    print(json.dumps(line), flush=True)
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
