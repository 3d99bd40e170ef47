import sys, os, time
sys.path.insert(0, ".")
import numpy as np
from igneous_amd.engine import Engine
from igneous_amd.synth import voronoi_labels

eng = Engine.get(0)
data = voronoi_labels((512, 512, 512), 200, seed=11, dtype=np.uint64)
for subs in ("1", "6"):
    os.environ["MG_SIMP_SUBS"] = subs
    eng.mesh_chunk(data, resolution=(16., 16., 40.), reduction_factor=100,
                   max_error=40.0, device_only=True)
    for _ in range(2):
        eng.mesh_chunk(data, resolution=(16., 16., 40.), reduction_factor=100,
                       max_error=40.0, device_only=True)
        st = eng.stats()
        ms = st["ms_simplify"] if isinstance(st, dict) else st.ms_simplify
        tot = st["ms_total"] if isinstance(st, dict) else st.ms_total
        print("SUBS=%s: simplify %.1f ms total %.1f ms" % (subs, ms, tot))
