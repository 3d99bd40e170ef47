import sys, time
sys.path.insert(0, ".")
import numpy as np
from igneous_amd.engine import Engine
from igneous_amd.synth import voronoi_labels
data = voronoi_labels((512, 512, 512), 50000, 303, dtype=np.uint64)
eng = Engine.get(0)
for me in (40.0, 1e9):
    eng.mesh_chunk(data, resolution=(16., 16., 40.), reduction_factor=100,
                   max_error=me, device_only=True)
    t0 = time.perf_counter()
    for _ in range(3):
        eng.mesh_chunk(data, resolution=(16., 16., 40.),
                       reduction_factor=100, max_error=me,
                       device_only=True, skip_h2d=True)
    dt = (time.perf_counter() - t0) / 3
    st = eng.stats()
    ms = st["ms_simplify"]
    print(f"max_error={me}: step {dt*1e3:.1f} ms, ms_simplify {ms:.1f}")
