import sys
sys.path.insert(0, '/root/repo'); sys.path.insert(0, '/root/repo/oracle')
import numpy as np
import oracle
from igneous_amd.engine import Engine
from igneous_amd.synth import voronoi_labels
eng = Engine.get(0)
data = voronoi_labels((64, 64, 64), 30, 17, dtype=np.uint64)
res = (16.0, 16.0, 40.0)
for factor, err in ((100, 40.0), (10, 1e9), (4, 0.0)):
    got = eng.mesh_chunk(data, resolution=res, reduction_factor=factor, max_error=err)
    want = oracle.mesh_chunk(data, resolution=res, reduction_factor=factor, max_error=err)
    bad = 0
    for lab in want:
        gv, gf = got[lab]; wv, wf = want[lab]
        ok = gv.shape == wv.shape and gf.shape == wf.shape and np.array_equal(gv, wv) and np.array_equal(gf, wf)
        if not ok:
            bad += 1
            if bad <= 3:
                print(f"  f={factor} lab={lab}: gpu V{gv.shape[0]} F{gf.shape[0]} vs orc V{wv.shape[0]} F{wf.shape[0]}", 
                      "shape-same" if gv.shape==wv.shape and gf.shape==wf.shape else "")
                if gv.shape == wv.shape and gf.shape == wf.shape:
                    dv = np.flatnonzero((gv != wv).any(1))
                    df = np.flatnonzero((gf != wf).any(1))
                    print(f"    vdiff {len(dv)} first {dv[:3]}, fdiff {len(df)} first {df[:3]}")
    print(f"f={factor} e={err}: {bad}/{len(want)} labels differ")
