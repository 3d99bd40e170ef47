import sys, os
sys.path.insert(0, '/root/repo'); sys.path.insert(0, '/root/repo/oracle')
import numpy as np
import oracle
from igneous_amd.engine import Engine
eng = Engine.get(0)
data = np.zeros((65, 65, 65), dtype=np.uint32, order="F")
data[1:63, 1:63, 1:63] = 1
res = (16.0, 16.0, 40.0)
for factor, err in ((100, 40.0), (10, 1e9), (4, 0.0), (0, 40.0)):
    got = eng.mesh_chunk(data, resolution=res, reduction_factor=factor, max_error=err)
    want = oracle.mesh_chunk(data, resolution=res, reduction_factor=factor, max_error=err)
    gv, gf = got[1]; wv, wf = want[1]
    same = gv.shape == wv.shape and gf.shape == wf.shape and np.array_equal(gv, wv) and np.array_equal(gf, wf)
    print(f"f={factor} e={err}: gpu V={gv.shape[0]} F={gf.shape[0]} | orc V={wv.shape[0]} F={wf.shape[0]} exact={same}")
    if gv.shape == wv.shape and not np.array_equal(gv, wv):
        d = np.abs(gv - wv); print("  max vert diff:", d.max(), "ndiff:", (d.sum(1) > 0).sum())
    if gf.shape == wf.shape and not np.array_equal(gf, wf):
        print("  face mismatch count:", (gf != wf).any(1).sum(), "of", gf.shape[0])
