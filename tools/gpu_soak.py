#!/usr/bin/env python3
"""Randomized GPU-vs-oracle parity soak (run on an MI355X host):

    python tools/gpu_soak.py [--n 20] [--seed 1000]

Meshes N random configurations (dims, label counts, dtype, resolution,
reduction_factor/max_error, voxel_centered) through BOTH the HIP engine
and the CPU oracle and asserts bit-exactness — label sets, uint32 face
arrays, float32 vertex arrays. Complements the fixed-seed tests in
tests/test_gpu_parity.py by sweeping the parameter space; sizes are kept
small enough that the oracle finishes each case in seconds.
"""
import argparse
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "oracle"))

import numpy as np  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=20)
    ap.add_argument("--seed", type=int, default=1000)
    ap.add_argument("--maxdim", type=int, default=72)
    args = ap.parse_args()

    import oracle
    from igneous_amd.engine import Engine
    from igneous_amd.synth import voronoi_labels

    eng = Engine.get(0)
    rng = np.random.default_rng(args.seed)
    for case in range(args.n):
        dims = tuple(int(d) for d in rng.integers(3, args.maxdim, size=3))
        dtype = np.uint64 if rng.integers(2) else np.uint32
        style = int(rng.integers(3))
        seed = int(rng.integers(2**31))
        if style == 0:  # random labels (many tiny disconnected pieces)
            r2 = np.random.default_rng(seed)
            data = np.asfortranarray(
                r2.integers(0, int(rng.integers(2, 9)), size=dims).astype(dtype))
        elif style == 1:  # voronoi cells
            npts = max(2, int(np.prod(dims) // int(rng.integers(500, 5000))))
            data = voronoi_labels(dims, npts, seed=seed, dtype=dtype)
        else:  # sparse boxes in empty space
            data = np.zeros(dims, dtype=dtype, order="F")
            for b in range(int(rng.integers(1, 4))):
                lo = [int(rng.integers(0, max(1, d - 2))) for d in dims]
                hi = [int(rng.integers(lo[i] + 1, dims[i] + 1)) for i in range(3)]
                data[lo[0]:hi[0], lo[1]:hi[1], lo[2]:hi[2]] = b + 1
        res = tuple(float(x) for x in rng.choice([1.0, 4.0, 16.0, 40.0], 3))
        rf = int(rng.choice([0, 0, 2, 10, 100]))
        err = float(rng.choice([0.0, 40.0, 1e9]))
        vc = bool(rng.integers(2))
        got = eng.mesh_chunk(data, resolution=res, reduction_factor=rf,
                             max_error=err, voxel_centered=vc)
        want = oracle.mesh_chunk(data, resolution=res, reduction_factor=rf,
                                 max_error=err, voxel_centered=vc)
        assert sorted(got) == sorted(want), \
            f"case {case}: label sets differ ({dims} {dtype} rf={rf})"
        for lab in want:
            gv, gf = got[lab]
            wv, wf = want[lab]
            assert np.array_equal(gf, wf), \
                f"case {case} label {lab}: faces differ ({dims} rf={rf} e={err})"
            assert np.array_equal(gv, wv), \
                f"case {case} label {lab}: verts differ ({dims} rf={rf} e={err})"
        print(f"case {case:3d}: dims={dims} {np.dtype(dtype).name} "
              f"style={style} rf={rf} err={err:g} vc={int(vc)} "
              f"labels={len(want)} OK")
    print(f"SOAK PASS: {args.n} random configs bit-exact")


if __name__ == "__main__":
    main()
