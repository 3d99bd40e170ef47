"""Timing probe: device dust passes at the bench config."""
import sys, time
sys.path.insert(0, ".")
import numpy as np
from igneous_amd.engine import Engine
from igneous_amd.synth import voronoi_labels

data = voronoi_labels((512, 512, 512), 50000, 303, dtype=np.uint64)
eng = Engine.get(0)
for thr in (0, 1000):
    eng.mesh_chunk(data, resolution=(16., 16., 40.), dust_threshold=thr,
                   device_only=True)
    t0 = time.perf_counter()
    for _ in range(5):
        eng.mesh_chunk(data, resolution=(16., 16., 40.),
                       dust_threshold=thr, device_only=True, skip_h2d=True)
    dt = (time.perf_counter() - t0) / 5
    st = eng.stats()
    print(f"dust_threshold={thr}: step {dt*1e3:.2f} ms, "
          f"labels {st['n_labels']}, tris {st['total_tris']}")
