/* oracle/simplify.c — CPU restatement of the per-label quadric edge-collapse
 * simplifier the reference runs through zmesh's `Mesher.get(id,
 * reduction_factor, max_error, voxel_centered=True)`
 * (/root/reference/igneous/tasks/mesh/mesh.py:376-381). zmesh sources are
 * not vendored (parity unpinned — see mc_oracle.c header); this restates the
 * published algorithm (Garland-Heckbert error quadrics, edge collapse,
 * reduction target = ntris/reduction_factor, max_error = max allowed
 * vertex displacement-error in physical nm) with a DETERMINISTIC
 * matched-pair independent-set schedule that the HIP kernel mirrors
 * exactly (see DESIGN.md "simplifier contract").
 *
 * Contract (canonical; shared bit-exact with igneous_amd/csrc):
 *  - vertex quadric Q_v = sum over incident faces (in face-index order) of
 *    the face's UNIT plane quadric (w=1; f32 arithmetic, plain summation in
 *    ascending face index order) — unit weight keeps the cost in nm^2 so it
 *    compares directly against max_error^2;
 *  - candidate per edge (u,v), u<v: placement = the Garland-Heckbert
 *    OPTIMAL point (Cramer solve of the summed quadric's normal
 *    equations, quad_place below), falling back to the midpoint when
 *    the 3x3 system is near-singular; cost = (Qu+Qv)(placement) in nm^2;
 *  - a round: every vertex picks its cheapest incident edge (ties -> the
 *    smaller peer index); an edge collapses iff both endpoints picked it
 *    AND cost <= max_error^2;
 *  - collapse moves both endpoints to the placement, remaps v->u (u<v),
 *    ADDS the loser's quadric into the winner (Q_u += Q_w, f32), and
 *    drops degenerate faces;
 *  - rounds are grouped: one full quadric recompute (step 1) is followed
 *    by up to OMC_SUBS pick/collapse/compact sub-rounds that reuse the
 *    merged quadrics (standard GH quadric merging; a full recompute
 *    resets the drift every group). Labels that START above 65536 faces
 *    use 1 sub-round per group (they run the engine's global-rounds
 *    path, where the sub passes dwarf the recompute they would skip);
 *  - iteration ends when ntris <= target or a group makes no progress
 *    (the first sub-round of a group collapses nothing).
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdlib.h>
#include <math.h>
#include <stdio.h>

typedef struct { float q[10]; } quad10; /* symmetric 4x4: a2,ab,ac,ad,b2,bc,bd,c2,cd,d2 */

static void quad_add_plane(quad10 *Q, float a, float b, float c, float d, float w) {
  Q->q[0] += w * a * a; Q->q[1] += w * a * b; Q->q[2] += w * a * c; Q->q[3] += w * a * d;
  Q->q[4] += w * b * b; Q->q[5] += w * b * c; Q->q[6] += w * b * d;
  Q->q[7] += w * c * c; Q->q[8] += w * c * d;
  Q->q[9] += w * d * d;
}

static float quad_eval(const quad10 *Q, float x, float y, float z) {
  return Q->q[0]*x*x + 2.0f*Q->q[1]*x*y + 2.0f*Q->q[2]*x*z + 2.0f*Q->q[3]*x
       + Q->q[4]*y*y + 2.0f*Q->q[5]*y*z + 2.0f*Q->q[6]*y
       + Q->q[7]*z*z + 2.0f*Q->q[8]*z
       + Q->q[9];
}

/* Canonical collapse placement (contract; the HIP kernels mirror this
 * arithmetic VERBATIM — f32, fixed expression order, -ffp-contract=off):
 * the Garland-Heckbert optimal point solving A x = -b by Cramer's rule
 * on the summed quadric S (A = upper-left 3x3, b = (q3,q6,q8)), falling
 * back to the edge midpoint when A is near-singular (flat regions,
 * where the midpoint is already cost-0). Writes the placement, returns
 * its cost. */
static float quad_place(const quad10 *S, float mx, float my, float mz,
                        float *px, float *py, float *pz) {
  float a00 = S->q[0], a01 = S->q[1], a02 = S->q[2], b0 = S->q[3];
  float a11 = S->q[4], a12 = S->q[5], b1 = S->q[6];
  float a22 = S->q[7], b2 = S->q[8];
  float m00 = a11*a22 - a12*a12;
  float m01 = a02*a12 - a01*a22;
  float m02 = a01*a12 - a02*a11;
  float m11 = a00*a22 - a02*a02;
  float m12 = a01*a02 - a00*a12;
  float m22 = a00*a11 - a01*a01;
  float det = a00*m00 + a01*m01 + a02*m02;
  float tr = a00 + a11 + a22;
  float x = mx, y = my, z = mz;
  if (fabsf(det) > 1e-6f * tr * tr * tr) {
    float inv = 1.0f / det;
    x = -(m00*b0 + m01*b1 + m02*b2) * inv;
    y = -(m01*b0 + m11*b1 + m12*b2) * inv;
    z = -(m02*b0 + m12*b1 + m22*b2) * inv;
  }
  *px = x; *py = y; *pz = z;
  float cost = quad_eval(S, x, y, z);
  if (cost < 0.0f) cost = 0.0f;
  return cost;
}

void omc_simplify_mesh(float *verts, uint32_t *nverts_io,
                       uint32_t *faces, uint32_t *ntris_io,
                       uint32_t reduction_factor, float max_error) {
  uint32_t nv = *nverts_io, nt = *ntris_io;
  if (reduction_factor <= 1 || nt == 0) return;
  uint32_t target = nt / reduction_factor;
  if (target < 1) target = 1;
  const float max_cost = max_error * max_error;

  /* sub-rounds per quadric recompute; MG_SIMP_SUBS is a dev knob shared
   * with the HIP engine (both default 6 -- part of the contract) */
  uint32_t SUBS = 6;
  {
    const char *e = getenv("MG_SIMP_SUBS");
    if (e && e[0]) SUBS = (uint32_t)atoi(e);
    if (SUBS < 1) SUBS = 1;
  }
  if (nt > 65536) SUBS = 1; /* global-rounds path: 1 sub-round per group */
  uint32_t *remap = (uint32_t*)malloc(nv * sizeof(uint32_t));
  quad10 *Q = (quad10*)malloc(nv * sizeof(quad10));
  /* pick[v]: encoded best edge for vertex v */
  uint64_t *pick = (uint64_t*)malloc(nv * sizeof(uint64_t));

  int progress = 1;
  while (nt > target && progress) {
    progress = 0;
    /* 1. vertex quadrics, ascending face order */
    {
    memset(Q, 0, nv * sizeof(quad10));
    for (uint32_t t = 0; t < nt; t++) {
      uint32_t i0 = faces[3*t], i1 = faces[3*t+1], i2 = faces[3*t+2];
      float *p0 = verts + 3*i0, *p1 = verts + 3*i1, *p2 = verts + 3*i2;
      float ux = p1[0]-p0[0], uy = p1[1]-p0[1], uz = p1[2]-p0[2];
      float vx = p2[0]-p0[0], vy = p2[1]-p0[1], vz = p2[2]-p0[2];
      float nx = uy*vz - uz*vy, ny = uz*vx - ux*vz, nz = ux*vy - uy*vx;
      float len = sqrtf(nx*nx + ny*ny + nz*nz);
      if (len <= 0.0f) continue;
      float inv = 1.0f / len;
      nx *= inv; ny *= inv; nz *= inv;
      float d = -(nx*p0[0] + ny*p0[1] + nz*p0[2]);
      quad_add_plane(&Q[i0], nx, ny, nz, d, 1.0f);
      quad_add_plane(&Q[i1], nx, ny, nz, d, 1.0f);
      quad_add_plane(&Q[i2], nx, ny, nz, d, 1.0f);
    }
    }
    for (uint32_t sub = 0; sub < SUBS && nt > target; sub++) {
    /* 2. per-vertex best incident edge: encode (costbits<<32 | peer) and
     * take min. Cost as raw f32 bits (all costs >= 0 so bit order == value
     * order); tie-break by smaller peer index. Deterministic: min over
     * edges is order-independent. */
    for (uint32_t v = 0; v < nv; v++) pick[v] = UINT64_MAX;
    for (uint32_t t = 0; t < nt; t++) {
      for (int e = 0; e < 3; e++) {
        uint32_t a = faces[3*t + e], b = faces[3*t + (e+1)%3];
        if (a == b) continue;
        uint32_t u = a < b ? a : b, w = a < b ? b : a;
        float mx = 0.5f*(verts[3*u]+verts[3*w]);
        float my = 0.5f*(verts[3*u+1]+verts[3*w+1]);
        float mz = 0.5f*(verts[3*u+2]+verts[3*w+2]);
        quad10 S;
        for (int k = 0; k < 10; k++) S.q[k] = Q[u].q[k] + Q[w].q[k];
        float px, py, pz;
        float cost = quad_place(&S, mx, my, mz, &px, &py, &pz);
        if (cost > max_cost) continue;
        uint32_t cb; memcpy(&cb, &cost, 4);
        /* deterministic per-edge jitter on the 3 low cost bits: breaks
         * the equal-cost pick chains of flat regions (which would give
         * O(1/sqrt(n)) matches per round) into random-preference
         * matchings (~1/degree of vertices collapse per round). Part of
         * the canonical contract; the HIP kernel mirrors it exactly. */
        uint32_t hsh = u ^ (w * 2654435761u);
        hsh ^= hsh >> 16; hsh *= 2246822519u; hsh ^= hsh >> 13;
        cb ^= (hsh & 7u);
        uint64_t enc_u = ((uint64_t)cb << 32) | w;
        uint64_t enc_w = ((uint64_t)cb << 32) | u;
        if (enc_u < pick[u]) pick[u] = enc_u;
        if (enc_w < pick[w]) pick[w] = enc_w;
      }
    }
    /* 3. matched pairs collapse (u<v keeps u) */
    for (uint32_t v = 0; v < nv; v++) remap[v] = v;
    uint32_t collapses = 0;
    for (uint32_t u = 0; u < nv; u++) {
      if (pick[u] == UINT64_MAX) continue;
      uint32_t w = (uint32_t)pick[u];
      if (w <= u) continue;               /* handle each pair from its min end */
      if (pick[w] == UINT64_MAX || (uint32_t)pick[w] != u) continue; /* not matched */
      if (remap[u] != u || remap[w] != w) continue; /* already touched */
      {
        float mx = 0.5f*(verts[3*u]+verts[3*w]);
        float my = 0.5f*(verts[3*u+1]+verts[3*w+1]);
        float mz = 0.5f*(verts[3*u+2]+verts[3*w+2]);
        quad10 S;
        for (int k = 0; k < 10; k++) S.q[k] = Q[u].q[k] + Q[w].q[k];
        float px, py, pz;
        (void)quad_place(&S, mx, my, mz, &px, &py, &pz);
        verts[3*u] = px; verts[3*u+1] = py; verts[3*u+2] = pz;
      }
      for (int k = 0; k < 10; k++) Q[u].q[k] += Q[w].q[k];
      remap[w] = u;
      /* blocked marking: matched vertices leave the pick graph. A live
       * pick entry therefore means "unmatched with an eligible edge";
       * note a proposal target ALWAYS has a live pick if unmatched,
       * because the edge (u,w) was offered to both endpoints. */
      pick[u] = UINT64_MAX; pick[w] = UINT64_MAX;
      collapses++;
    }
    /* 3b. PROPOSAL-ACCEPTANCE second wave (contract, r02): mutual picks
     * converge on local cost minima ("stars": many vertices pick one),
     * so the mutual wave alone matches ~10-20%/round. Unmatched
     * vertices whose pick points UP (peer id > self) PROPOSE to that
     * peer; an unmatched NON-proposing peer accepts its minimum
     * (costbits, proposer+1) proposal. Deterministic: acceptance is a
     * min-reduction; proposers never accept; each proposer targets one
     * vertex, so the extra pairs are disjoint among themselves and
     * with the mutual wave. Placement/cost identical to wave 1.
     * MG_SIMP_PROPOSE=0 disables (engine reads the same knob). */
    {
      static int propose_on = -1;
      if (propose_on < 0) {
        const char *e = getenv("MG_SIMP_PROPOSE");
        propose_on = !(e && e[0] == '0');
      }
      if (propose_on) {
        /* id-only key: the acceptor takes its MINIMUM-ID proposer
         * (every proposal already satisfies the cost bound; a 32-bit
         * key lets the engine keep the accept table in otherwise-dead
         * LDS bytes) */
        uint32_t *accept = (uint32_t*)malloc(nv * sizeof(uint32_t));
        for (uint32_t v = 0; v < nv; v++) accept[v] = UINT32_MAX;
        for (uint32_t u = 0; u < nv; u++) {
          if (pick[u] == UINT64_MAX) continue;   /* matched or pickless */
          uint32_t w = (uint32_t)pick[u];
          if (w <= u) continue;          /* propose-up only */
          if (pick[w] == UINT64_MAX) continue;   /* blocked target */
          if (u + 1 < accept[w]) accept[w] = u + 1;
        }
        for (uint32_t w = 0; w < nv; w++) {
          if (accept[w] == UINT32_MAX) continue;
          /* acceptors: unmatched (live pick) non-proposers */
          if (pick[w] == UINT64_MAX) continue;
          if ((uint32_t)pick[w] > w) continue;
          uint32_t u = accept[w] - 1;
          {
            float mx = 0.5f*(verts[3*u]+verts[3*w]);
            float my = 0.5f*(verts[3*u+1]+verts[3*w+1]);
            float mz = 0.5f*(verts[3*u+2]+verts[3*w+2]);
            quad10 S;
            for (int k = 0; k < 10; k++) S.q[k] = Q[u].q[k] + Q[w].q[k];
            float px, py, pz;
            (void)quad_place(&S, mx, my, mz, &px, &py, &pz);
            verts[3*u] = px; verts[3*u+1] = py; verts[3*u+2] = pz;
          }
          for (int k = 0; k < 10; k++) Q[u].q[k] += Q[w].q[k];
          remap[w] = u;
          collapses++;
        }
        free(accept);
      }
    }
    if (!collapses) break;
    /* 4. rewrite faces, drop degenerates */
    uint32_t out = 0;
    for (uint32_t t = 0; t < nt; t++) {
      uint32_t i0 = remap[faces[3*t]], i1 = remap[faces[3*t+1]], i2 = remap[faces[3*t+2]];
      if (i0 == i1 || i1 == i2 || i0 == i2) continue;
      faces[3*out] = i0; faces[3*out+1] = i1; faces[3*out+2] = i2;
      out++;
    }
    if (out < nt) progress = 1;
    if (getenv("OMC_DEBUG"))
      fprintf(stderr, "[omc] round sub=%u: collapses=%u nt %u -> %u\n",
              sub, collapses, nt, out);
    nt = out;
    }
  }

  /* 5. compact vertices to those referenced, preserving index order */
  uint32_t *newidx = (uint32_t*)malloc(nv * sizeof(uint32_t));
  memset(newidx, 0xFF, nv * sizeof(uint32_t));
  uint32_t nnv = 0;
  for (uint32_t t = 0; t < nt; t++)
    for (int e = 0; e < 3; e++) {
      uint32_t v = faces[3*t + e];
      if (newidx[v] == 0xFFFFFFFFu) newidx[v] = 1;
    }
  for (uint32_t v = 0; v < nv; v++) {
    if (newidx[v] == 1) {
      newidx[v] = nnv;
      verts[3*nnv] = verts[3*v]; verts[3*nnv+1] = verts[3*v+1]; verts[3*nnv+2] = verts[3*v+2];
      nnv++;
    } else newidx[v] = 0xFFFFFFFFu;
  }
  for (uint32_t t = 0; t < nt; t++)
    for (int e = 0; e < 3; e++) faces[3*t + e] = newidx[faces[3*t + e]];

  free(newidx); free(pick); free(Q); free(remap);
  *nverts_io = nnv;
  *ntris_io = nt;
}
