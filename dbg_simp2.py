import sys
sys.path.insert(0, '/root/repo'); sys.path.insert(0, '/root/repo/oracle')
import numpy as np
import oracle
from igneous_amd.engine import Engine
eng = Engine.get(0)
data = np.zeros((17,17,17), dtype=np.uint32, order='F')
data[1:15,1:15,1:15] = 1
res = (16.0, 16.0, 40.0)
got = eng.mesh_chunk(data, resolution=res, reduction_factor=10, max_error=1e9)
print("gpu final:", got[1][0].shape, got[1][1].shape)

# replica round-1 on the oracle's unsimplified mesh
r = oracle.mesh_chunk(data, resolution=res)
V, F = r[1]
V = V.astype(np.float32).copy(); F = F.astype(np.int64).copy()
print("replica faces0-3:", F[:4].ravel().tolist())
p0, p1, p2 = V[F[:,0]], V[F[:,1]], V[F[:,2]]
n = np.cross(p1-p0, p2-p0).astype(np.float32)
ln = np.sqrt((n*n).sum(1), dtype=np.float32)
valid = ln > 0
nn = np.where(valid[:,None], n/np.maximum(ln,1e-30)[:,None], 0).astype(np.float32)
d = -(nn*p0).sum(1).astype(np.float32)
planes = np.concatenate([nn, d[:,None]], 1).astype(np.float32)
idx = [(0,0),(0,1),(0,2),(0,3),(1,1),(1,2),(1,3),(2,2),(2,3),(3,3)]
fq = np.stack([planes[:,a]*planes[:,b] for a,b in idx],1).astype(np.float32)
fq[~valid] = 0
Q = np.zeros((len(V),10), np.float32)
for t in range(len(F)):
    if not valid[t]: continue
    for v in F[t]:
        Q[v] += fq[t]
print("replica Q0:", " ".join(float(x).hex() for x in Q[0]))
print("replica Q1:", " ".join(float(x).hex() for x in Q[1]))
