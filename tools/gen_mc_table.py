#!/usr/bin/env python3
"""Generate the canonical marching-cubes case table used by BOTH the CPU
oracle (oracle/mc_oracle.c) and the HIP kernels (igneous_amd/csrc/).

This restates the multi-label marching-cubes contract of the reference's
zmesh dependency (called at /root/reference/igneous/tasks/mesh/mesh.py:245,
pin zmesh>=1.13.1,<2.0 per requirements.txt:26; zmesh's own sources are not
vendored in the reference, so the triangulation below is this repo's
canonical restatement — see DESIGN.md "parity" section).

Semantics of the generated table
--------------------------------
A cell is a 2x2x2 block of voxels. Corner index c = x + 2*y + 4*z with
x,y,z in {0,1}. For a given label L, corner bit c of `mask` is set iff the
corner voxel's label == L.  The table maps mask -> triangles, each triangle
a triple of cell-edge ids.  The 12 cell edges, in canonical order:

    x-edges: 0:(0,1) 1:(2,3) 2:(4,5) 3:(6,7)
    y-edges: 4:(0,2) 5:(1,3) 6:(4,6) 7:(5,7)
    z-edges: 8:(0,4) 9:(1,5) 10:(2,6) 11:(3,7)

Every surface vertex sits at the MIDPOINT of a crossing edge (binary field),
i.e. at doubled-integer coordinates: edge e of cell (cx,cy,cz) sits at
(2*cx,2*cy,2*cz) + EDGE_DOUBLED_OFFSET[e], exactly one component odd.

Face-ambiguity rule (fixed, orientation-consistent): on every cut face the
maximal arcs of label corners are each cut off by one chord running from the
arc's LEAVING crossing (1->0 in CCW-from-outside face order) to its ENTERING
crossing (0->1).  The rule depends only on the face's four corner values, so
the two cells sharing a face always cut identical chords -> crack-free
surfaces.  Diagonal-ambiguous faces resolve to "separate the label corners".

Triangle orientation: loops are traced so the label region lies on the left
viewed from outside; fan-triangulated from the loop's canonical start (the
smallest edge id), giving outward normals (away from the labeled region,
right-hand rule).  Verified below for every single-corner mask.

Output: igneous_amd/csrc/mc_table.h (C header) and oracle reuses the same
header.  Run:  python tools/gen_mc_table.py
"""
import os

CORNER_POS = [(c & 1, (c >> 1) & 1, (c >> 2) & 1) for c in range(8)]

EDGES = [
    (0, 1), (2, 3), (4, 5), (6, 7),   # x-edges
    (0, 2), (1, 3), (4, 6), (5, 7),   # y-edges
    (0, 4), (1, 5), (2, 6), (3, 7),   # z-edges
]
EDGE_ID = {}
for i, (a, b) in enumerate(EDGES):
    EDGE_ID[(a, b)] = i
    EDGE_ID[(b, a)] = i

# doubled-coordinate offset of each edge midpoint from the cell origin
EDGE_DOFF = []
for (a, b) in EDGES:
    pa, pb = CORNER_POS[a], CORNER_POS[b]
    EDGE_DOFF.append(tuple(pa[i] + pb[i] for i in range(3)))

# 6 faces, corners listed CCW when viewed from OUTSIDE the cube.
FACES = [
    (0, 2, 6, 4),  # -x
    (1, 5, 7, 3),  # +x
    (0, 4, 5, 1),  # -y
    (2, 3, 7, 6),  # +y
    (0, 1, 3, 2),  # -z
    (4, 6, 7, 5),  # +z
]


def face_segments(mask, face):
    """Directed chords (src_edge -> dst_edge) on one face for `mask`."""
    vals = [(mask >> c) & 1 for c in face]
    segs = []
    # crossings between consecutive corners (cyclic)
    # identify maximal arcs of 1s; chord: leave-crossing -> enter-crossing
    if all(vals) or not any(vals):
        return segs
    n = 4
    for i in range(n):
        # start of a 1-arc: vals[i]==1 and vals[i-1]==0 -> entering crossing
        if vals[i] == 1 and vals[(i - 1) % n] == 0:
            # walk to end of arc
            j = i
            while vals[(j + 1) % n] == 1:
                j = (j + 1) % n
            enter = EDGE_ID[(face[(i - 1) % n], face[i])]
            leave = EDGE_ID[(face[j], face[(j + 1) % n])]
            segs.append((leave, enter))
    return segs


def loops_for_mask(mask):
    nxt = {}
    for face in FACES:
        for (src, dst) in face_segments(mask, face):
            assert src not in nxt, f"mask {mask}: double out-edge at {src}"
            nxt[src] = dst
    loops = []
    seen = set()
    for start in sorted(nxt):
        if start in seen:
            continue
        loop = [start]
        seen.add(start)
        cur = nxt[start]
        while cur != start:
            loop.append(cur)
            seen.add(cur)
            cur = nxt[cur]
        # canonical rotation: start at smallest edge id (preserves direction)
        k = loop.index(min(loop))
        loop = loop[k:] + loop[:k]
        loops.append(loop)
    loops.sort(key=lambda l: l[0])
    return loops


def triangulate(loops):
    tris = []
    for loop in loops:
        for i in range(1, len(loop) - 1):
            tris.append((loop[0], loop[i], loop[i + 1]))
    return tris


def check_orientation():
    """Every single-corner mask must produce one triangle whose normal
    points AWAY from the labeled corner (outward)."""
    import itertools
    for c in range(8):
        mask = 1 << c
        tris = triangulate(loops_for_mask(mask))
        assert len(tris) == 1, (c, tris)
        p = [EDGE_DOFF[e] for e in tris[0]]
        ux = [p[1][i] - p[0][i] for i in range(3)]
        vx = [p[2][i] - p[0][i] for i in range(3)]
        nrm = (ux[1] * vx[2] - ux[2] * vx[1],
               ux[2] * vx[0] - ux[0] * vx[2],
               ux[0] * vx[1] - ux[1] * vx[0])
        ctr = [sum(q[i] for q in p) / 3.0 for i in range(3)]
        toward = [2 * CORNER_POS[c][i] - ctr[i] for i in range(3)]
        d = sum(nrm[i] * toward[i] for i in range(3))
        assert d < 0, f"corner {c}: normal not outward (dot={d})"


def main():
    check_orientation()
    table = []
    maxt = 0
    for mask in range(256):
        tris = triangulate(loops_for_mask(mask))
        table.append(tris)
        maxt = max(maxt, len(tris))

    here = os.path.dirname(os.path.abspath(__file__))
    out = os.path.join(here, "..", "igneous_amd", "csrc", "mc_table.h")
    with open(out, "w") as f:
        f.write("// AUTO-GENERATED by tools/gen_mc_table.py — do not edit.\n")
        f.write("// Canonical multi-label marching-cubes table; see the\n")
        f.write("// generator's docstring for corner/edge numbering and the\n")
        f.write("// face-ambiguity rule. Shared by oracle/ and HIP kernels.\n")
        f.write("#ifndef MC_TABLE_H\n#define MC_TABLE_H\n\n")
        f.write("// HIP device code defines MC_TABLE_QUAL as __device__ static\n")
        f.write("// const before including; C host code gets plain static const.\n")
        f.write("#ifndef MC_TABLE_QUAL\n#define MC_TABLE_QUAL static const\n#endif\n\n")
        f.write(f"#define MC_MAX_TRIS {maxt}\n\n")
        f.write("// number of triangles for each 8-bit corner mask\n")
        f.write("MC_TABLE_QUAL unsigned char MC_TRI_COUNT[256] = {\n")
        for row in range(0, 256, 16):
            f.write("  " + ", ".join(str(len(table[m])) for m in range(row, row + 16)) + ",\n")
        f.write("};\n\n")
        f.write("// edge ids, 3 per triangle, MC_MAX_TRIS*3 slots per mask, -1 padded\n")
        f.write(f"MC_TABLE_QUAL signed char MC_TRI_TABLE[256][{maxt * 3}] = {{\n")
        for m in range(256):
            flat = [e for t in table[m] for e in t]
            flat += [-1] * (maxt * 3 - len(flat))
            f.write("  {" + ", ".join(f"{v}" for v in flat) + "},\n")
        f.write("};\n\n")
        f.write("// doubled-coordinate offset (x,y,z) of each edge midpoint\n")
        f.write("MC_TABLE_QUAL unsigned char MC_EDGE_DOFF[12][3] = {\n")
        for e in range(12):
            f.write("  {%d, %d, %d},\n" % EDGE_DOFF[e])
        f.write("};\n\n")
        f.write("// packed per-(mask,tri) corner descriptors for the HIP emit\n")
        f.write("// kernel: 5 bits per corner = edge id (4) | side (1), where\n")
        f.write("// side = mask bit of the edge's UPPER corner (is the label\n")
        f.write("// the edge's upper endpoint?). 3 corners -> 15 bits, u16.\n")
        f.write(f"MC_TABLE_QUAL unsigned short MC_TRI_PACK[256][{maxt}] = {{\n")
        for m in range(256):
            packs = []
            for t in table[m]:
                p = 0
                for v, e in enumerate(t):
                    side = (m >> EDGES[e][1]) & 1
                    p |= (e | (side << 4)) << (5 * v)
                packs.append(p)
            packs += [0] * (maxt - len(packs))
            f.write("  {" + ", ".join(f"0x{v:04x}" for v in packs) + "},\n")
        f.write("};\n\n")
        f.write("// corner ids (a,b) of each edge, a < b (a = lower voxel)\n")
        f.write("MC_TABLE_QUAL unsigned char MC_EDGE_CORNERS[12][2] = {\n")
        for (a, b) in EDGES:
            f.write(f"  {{{a}, {b}}},\n")
        f.write("};\n\n")
        f.write("// packed 12-bit-per-axis key offset of each edge midpoint\n")
        f.write("// relative to the cell's packed doubled origin:\n")
        f.write("// key = ((2cz)<<24 | (2cy)<<12 | 2cx) + MC_EDGE_KEYOFF[e]\n")
        f.write("// (no cross-field carries: doubled coords stay < 4096)\n")
        f.write("MC_TABLE_QUAL unsigned long long MC_EDGE_KEYOFF[12] = {\n")
        for e in range(12):
            ox, oy, oz = EDGE_DOFF[e]
            f.write(f"  0x{(oz << 24) | (oy << 12) | ox:x}ULL,\n")
        f.write("};\n\n#endif // MC_TABLE_H\n")
    print(f"wrote {out}: max {maxt} tris/cell")

    # stats
    total = sum(len(t) for t in table)
    print(f"total tris across 256 masks: {total}")


if __name__ == "__main__":
    main()
