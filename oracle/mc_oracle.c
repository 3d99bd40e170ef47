/* oracle/mc_oracle.c — CPU oracle: clean-room restatement of the reference's
 * per-chunk meshing hot path (multi-label marching cubes + per-label vertex
 * welding), i.e. the work zmesh's C++ `Mesher.mesh()` / `Mesher.get()` do for
 * igneous's MeshTask:
 *   /root/reference/igneous/tasks/mesh/mesh.py:245  (Mesher.mesh: one pass,
 *       all labels, label-vs-rest binary surfaces, F-order chunk with 1vx
 *       high padding)
 *   /root/reference/igneous/tasks/mesh/mesh.py:374-381 (Mesher.get: per-label
 *       extract, voxel_centered=True, reduction_factor, max_error in nm)
 *   /root/reference/igneous/tasks/mesh/mesh.py:151   (Mesher(resolution):
 *       anisotropic nm scaling of vertex coordinates)
 *
 * PARITY STATUS: **parity vs zmesh is UNPINNED.** The third-party module at
 * the arithmetic boundary is `zmesh` (pinned >=1.13.1,<2.0 by the
 * reference's requirements.txt:26). Its sources are not vendored under
 * /root/reference, no wheel is installable offline, and the reference's own
 * test suite pins NO geometry at this boundary (test/test_tasks.py:407-462
 * asserts output-file existence only). This oracle therefore restates the
 * published algorithm (marching cubes over binary label-vs-rest fields,
 * midpoint vertices, welded per-label meshes, quadric edge-collapse
 * simplification) with a fixed canonical contract documented in DESIGN.md;
 * the HIP engine is pinned BIT-EXACT against THIS oracle, and both are
 * pinned against the reference's call-site semantics and its 64^3 box test
 * fixture (exact V/F counts derivable in closed form).
 *
 * TEST INFRASTRUCTURE ONLY: this file may be imported/linked/executed only
 * by tests/, __graft_entry__.smoke() (as the checker) and bench.py's
 * cpu_baseline leg. It is never the product path: igneous_amd's MeshTask
 * fails loudly if the HIP engine is missing on a GPU host.
 *
 * Canonical contract (shared, bit-exact, with igneous_amd/csrc kernels):
 *  - cells scanned in global F-order (x fastest, then y, then z);
 *  - per cell, distinct non-zero labels in first-seen corner order (corner
 *    index c = x + 2y + 4z);
 *  - per (cell,label): 8-bit mask (bit c set iff corner label == L), triangle
 *    list from mc_table.h (generated; see tools/gen_mc_table.py for the
 *    face-ambiguity rule), triangles in table order, corners in table order;
 *  - per-label vertex numbering: order of first appearance in that label's
 *    triangle-corner stream ("first-seen welding");
 *  - vertex position (float32): p = (0.5f*k + (voxel_centered?0.0f:0.5f)) * r
 *    per axis, k = doubled integer edge-midpoint coordinate, r = resolution
 *    in nm. Chunk-local: the Python host adds the global offset afterwards,
 *    exactly like the reference (mesh.py:434-435).
 *
 * Build: gcc -O2 -shared -fPIC oracle/mc_oracle.c -o oracle/liboracle.so -lm
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>
#include <math.h>

#include "mc_table.h"

typedef struct {
  uint64_t label;
  uint32_t nverts, ntris;
  uint32_t vcap, tcap;
  uint64_t *vkeys;    /* packed doubled coords, in first-seen order */
  uint32_t *faces;    /* 3*ntris vertex indices */
  /* weld hash: key -> vertex index */
  uint64_t *hkeys;
  uint32_t *hvals;
  uint32_t hsize;     /* power of two */
  float    *verts;    /* 3*nverts, filled at finalize */
} omc_builder;

typedef struct {
  uint64_t label;
  uint32_t nverts, ntris;
  float    *verts;
  uint32_t *faces;
} omc_mesh;

typedef struct {
  uint32_t nmeshes;
  omc_mesh *meshes;
} omc_meshset;

/* ------------------------------------------------------------------ */

static uint64_t mix64(uint64_t x) { /* splitmix64 finalizer */
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

static void builder_init(omc_builder *b, uint64_t label) {
  memset(b, 0, sizeof(*b));
  b->label = label;
  b->vcap = 64; b->tcap = 64;
  b->vkeys = (uint64_t*)malloc(b->vcap * sizeof(uint64_t));
  b->faces = (uint32_t*)malloc(b->tcap * 3 * sizeof(uint32_t));
  b->hsize = 256;
  b->hkeys = (uint64_t*)calloc(b->hsize, sizeof(uint64_t));
  b->hvals = (uint32_t*)malloc(b->hsize * sizeof(uint32_t));
}

static void builder_rehash(omc_builder *b) {
  uint32_t ns = b->hsize * 2;
  uint64_t *nk = (uint64_t*)calloc(ns, sizeof(uint64_t));
  uint32_t *nv = (uint32_t*)malloc(ns * sizeof(uint32_t));
  for (uint32_t i = 0; i < b->hsize; i++) {
    uint64_t k = b->hkeys[i];
    if (!k) continue;
    uint64_t h = mix64(k) & (ns - 1);
    while (nk[h]) h = (h + 1) & (ns - 1);
    nk[h] = k; nv[h] = b->hvals[i];
  }
  free(b->hkeys); free(b->hvals);
  b->hkeys = nk; b->hvals = nv; b->hsize = ns;
}

/* weld: return vertex index for edge-midpoint key (first-seen numbering) */
static uint32_t builder_vertex(omc_builder *b, uint64_t key) {
  if (b->nverts * 2 >= b->hsize) builder_rehash(b);
  uint64_t h = mix64(key) & (b->hsize - 1);
  while (b->hkeys[h]) {
    if (b->hkeys[h] == key) return b->hvals[h];
    h = (h + 1) & (b->hsize - 1);
  }
  if (b->nverts == b->vcap) {
    b->vcap *= 2;
    b->vkeys = (uint64_t*)realloc(b->vkeys, b->vcap * sizeof(uint64_t));
  }
  uint32_t idx = b->nverts++;
  b->vkeys[idx] = key;
  b->hkeys[h] = key; b->hvals[h] = idx;
  return idx;
}

static void builder_tri(omc_builder *b, uint64_t k0, uint64_t k1, uint64_t k2) {
  if (b->ntris == b->tcap) {
    b->tcap *= 2;
    b->faces = (uint32_t*)realloc(b->faces, b->tcap * 3 * sizeof(uint32_t));
  }
  uint32_t *f = b->faces + 3 * b->ntris;
  f[0] = builder_vertex(b, k0);
  f[1] = builder_vertex(b, k1);
  f[2] = builder_vertex(b, k2);
  b->ntris++;
}

/* ------------------------------------------------------------------ */
/* label -> builder map (open addressing on label value) */

typedef struct {
  uint64_t *keys;
  omc_builder **vals;
  uint32_t size;   /* power of two */
  uint32_t count;
} label_map;

static void lmap_init(label_map *m) {
  m->size = 1024; m->count = 0;
  m->keys = (uint64_t*)calloc(m->size, sizeof(uint64_t));
  m->vals = (omc_builder**)calloc(m->size, sizeof(omc_builder*));
}

static void lmap_rehash(label_map *m) {
  uint32_t ns = m->size * 2;
  uint64_t *nk = (uint64_t*)calloc(ns, sizeof(uint64_t));
  omc_builder **nv = (omc_builder**)calloc(ns, sizeof(omc_builder*));
  for (uint32_t i = 0; i < m->size; i++) {
    if (!m->keys[i]) continue;
    uint64_t h = mix64(m->keys[i]) & (ns - 1);
    while (nk[h]) h = (h + 1) & (ns - 1);
    nk[h] = m->keys[i]; nv[h] = m->vals[i];
  }
  free(m->keys); free(m->vals);
  m->keys = nk; m->vals = nv; m->size = ns;
}

static omc_builder *lmap_get(label_map *m, uint64_t label) {
  if (m->count * 2 >= m->size) lmap_rehash(m);
  uint64_t h = mix64(label) & (m->size - 1);
  while (m->keys[h]) {
    if (m->keys[h] == label) return m->vals[h];
    h = (h + 1) & (m->size - 1);
  }
  omc_builder *b = (omc_builder*)malloc(sizeof(omc_builder));
  builder_init(b, label);
  m->keys[h] = label; m->vals[h] = b;
  m->count++;
  return b;
}

/* ------------------------------------------------------------------ */

#define PACK_KEY(dx, dy, dz) \
  (((uint64_t)(dz) << 24) | ((uint64_t)(dy) << 12) | (uint64_t)(dx))

static int cmp_label(const void *a, const void *b) {
  uint64_t la = ((const omc_mesh*)a)->label, lb = ((const omc_mesh*)b)->label;
  return (la < lb) ? -1 : (la > lb) ? 1 : 0;
}

/* core: templated over label width via macro */
#define DEFINE_MESH_FN(NAME, LTYPE)                                          \
static void NAME(const LTYPE *lab, int sx, int sy, int sz, label_map *map) { \
  const int64_t stx = 1, sty = sx, stz = (int64_t)sx * sy;                   \
  for (int cz = 0; cz < sz - 1; cz++)                                        \
  for (int cy = 0; cy < sy - 1; cy++) {                                      \
    const LTYPE *base = lab + (int64_t)cy * sty + (int64_t)cz * stz;         \
    for (int cx = 0; cx < sx - 1; cx++) {                                    \
      const LTYPE *p = base + cx;                                            \
      LTYPE c[8];                                                            \
      c[0] = p[0];             c[1] = p[stx];                                \
      c[2] = p[sty];           c[3] = p[sty + stx];                          \
      c[4] = p[stz];           c[5] = p[stz + stx];                          \
      c[6] = p[stz + sty];     c[7] = p[stz + sty + stx];                    \
      /* uniform-cell early out */                                           \
      if (c[0] == c[1] && c[0] == c[2] && c[0] == c[3] && c[0] == c[4] &&    \
          c[0] == c[5] && c[0] == c[6] && c[0] == c[7]) continue;            \
      for (int i = 0; i < 8; i++) {                                          \
        LTYPE L = c[i];                                                      \
        if (L == 0) continue;                                                \
        int seen = 0;                                                        \
        for (int j = 0; j < i; j++) if (c[j] == L) { seen = 1; break; }      \
        if (seen) continue;                                                  \
        unsigned mask = 0;                                                   \
        for (int j = 0; j < 8; j++) if (c[j] == L) mask |= 1u << j;          \
        int nt = MC_TRI_COUNT[mask];                                         \
        if (!nt) continue;                                                   \
        omc_builder *b = lmap_get(map, (uint64_t)L);                         \
        const signed char *tt = MC_TRI_TABLE[mask];                          \
        for (int t = 0; t < nt; t++) {                                       \
          uint64_t k[3];                                                     \
          for (int v = 0; v < 3; v++) {                                      \
            int e = tt[3 * t + v];                                           \
            int dx = 2 * cx + MC_EDGE_DOFF[e][0];                            \
            int dy = 2 * cy + MC_EDGE_DOFF[e][1];                            \
            int dz = 2 * cz + MC_EDGE_DOFF[e][2];                            \
            k[v] = PACK_KEY(dx, dy, dz);                                     \
          }                                                                  \
          builder_tri(b, k[0], k[1], k[2]);                                  \
        }                                                                    \
      }                                                                      \
    }                                                                        \
  }                                                                          \
}

DEFINE_MESH_FN(mesh_u32, uint32_t)
DEFINE_MESH_FN(mesh_u64, uint64_t)

/* defined in simplify.c (linked into liboracle.so) */
void omc_simplify_mesh(float *verts, uint32_t *nverts_io,
                       uint32_t *faces, uint32_t *ntris_io,
                       uint32_t reduction_factor, float max_error);

int omc_mesh_chunk(const void *labels, int sx, int sy, int sz, int dtype,
                   float rx, float ry, float rz,
                   uint32_t reduction_factor, float max_error,
                   int voxel_centered, omc_meshset **out) {
  if (!labels || !out || sx < 1 || sy < 1 || sz < 1) return 1;
  if (sx > 2047 || sy > 2047 || sz > 2047) return 2; /* 12-bit packed axes */
  label_map map;
  lmap_init(&map);
  if (dtype == 0)      mesh_u32((const uint32_t*)labels, sx, sy, sz, &map);
  else if (dtype == 1) mesh_u64((const uint64_t*)labels, sx, sy, sz, &map);
  else return 3;

  /* finalize: float vertices, sorted-by-label meshset */
  omc_meshset *ms = (omc_meshset*)malloc(sizeof(omc_meshset));
  ms->nmeshes = map.count;
  ms->meshes = (omc_mesh*)malloc((map.count ? map.count : 1) * sizeof(omc_mesh));
  const float shift = voxel_centered ? 0.0f : 0.5f;
  uint32_t mi = 0;
  for (uint32_t i = 0; i < map.size; i++) {
    if (!map.keys[i]) continue;
    omc_builder *b = map.vals[i];
    float *verts = (float*)malloc((size_t)b->nverts * 3 * sizeof(float));
    for (uint32_t v = 0; v < b->nverts; v++) {
      uint64_t k = b->vkeys[v];
      float dx = (float)(uint32_t)(k & 0xFFF);
      float dy = (float)(uint32_t)((k >> 12) & 0xFFF);
      float dz = (float)(uint32_t)((k >> 24) & 0xFFF);
      verts[3 * v + 0] = (0.5f * dx + shift) * rx;
      verts[3 * v + 1] = (0.5f * dy + shift) * ry;
      verts[3 * v + 2] = (0.5f * dz + shift) * rz;
    }
    uint32_t nv = b->nverts, nt = b->ntris;
    uint32_t *faces = b->faces;   /* builder's array adopted by the mesh */
    if (reduction_factor > 1 && nt > 0) {
      omc_simplify_mesh(verts, &nv, faces, &nt, reduction_factor, max_error);
    }
    ms->meshes[mi].label  = b->label;
    ms->meshes[mi].nverts = nv;
    ms->meshes[mi].ntris  = nt;
    ms->meshes[mi].verts  = verts;
    ms->meshes[mi].faces  = faces;
    mi++;
    free(b->vkeys); free(b->hkeys); free(b->hvals); free(b);
  }
  free(map.keys); free(map.vals);
  qsort(ms->meshes, ms->nmeshes, sizeof(omc_mesh), cmp_label);
  *out = ms;
  return 0;
}

void omc_meshset_free(omc_meshset *ms) {
  if (!ms) return;
  for (uint32_t i = 0; i < ms->nmeshes; i++) {
    free(ms->meshes[i].verts);
    free(ms->meshes[i].faces);
  }
  free(ms->meshes);
  free(ms);
}
