// simplify.hip — per-label quadric edge-collapse simplifier on the GPU.
//
// Restates zmesh's Mesher.get(id, reduction_factor, max_error,
// voxel_centered) simplification step
// (/root/reference/igneous/tasks/mesh/mesh.py:376-381) with the SAME
// deterministic matched-pair independent-set schedule as the CPU oracle
// (oracle/simplify.c) — bit-exact: identical f32 expressions (plane from
// cross product + correctly-rounded sqrtf/div, unit-weight quadrics summed
// in ascending face order, midpoint placement, cost = (Qu+Qw)(m) <=
// max_error^2, per-vertex cheapest edge with deterministic per-edge tie
// jitter on the 3 low cost bits + smaller-peer tie-break, matched pairs
// collapse, rounds until a label hits its triangle target or stops
// shrinking).
//
// All labels simplify SIMULTANEOUSLY: faces stay label-partitioned, the
// vertex space is label-disjoint (vbase slices), so per-label rounds are
// independent; a finished label freezes while others continue.
//
// Included by meshgine.hip (single TU).

struct SimpPlane {
  float nx, ny, nz, d;
};

// quadric helpers — textually identical arithmetic to oracle/simplify.c
__device__ __forceinline__ void sq_add_plane(float *q, float a, float b,
                                             float c_, float d, float w) {
  q[0] += w * a * a; q[1] += w * a * b; q[2] += w * a * c_; q[3] += w * a * d;
  q[4] += w * b * b; q[5] += w * b * c_; q[6] += w * b * d;
  q[7] += w * c_ * c_; q[8] += w * c_ * d;
  q[9] += w * d * d;
}

__device__ __forceinline__ float sq_eval(const float *q, float x, float y,
                                         float z) {
  return q[0]*x*x + 2.0f*q[1]*x*y + 2.0f*q[2]*x*z + 2.0f*q[3]*x
       + q[4]*y*y + 2.0f*q[5]*y*z + 2.0f*q[6]*y
       + q[7]*z*z + 2.0f*q[8]*z
       + q[9];
}

// Canonical collapse placement — textually identical arithmetic to
// oracle/simplify.c quad_place (the contract expression): GH optimal
// point via Cramer on the summed quadric, midpoint fallback when the
// 3x3 system is near-singular. Writes the placement, returns its cost.
__device__ __forceinline__ float sq_place(const float *S, float mx,
                                          float my, float mz,
                                          float *px, float *py,
                                          float *pz) {
  float a00 = S[0], a01 = S[1], a02 = S[2], b0 = S[3];
  float a11 = S[4], a12 = S[5], b1 = S[6];
  float a22 = S[7], b2 = S[8];
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
  float cost = sq_eval(S, x, y, z);
  if (cost < 0.0f) cost = 0.0f;
  return cost;
}

// cost-only form for the pick phase (identical arithmetic; the
// placement registers don't outlive the call — keeps the per-label
// kernel at <=128 VGPRs / 4 waves per SIMD)
__device__ __forceinline__ float sq_place_cost(const float *S, float mx,
                                               float my, float mz) {
  float a00 = S[0], a01 = S[1], a02 = S[2], b0 = S[3];
  float a11 = S[4], a12 = S[5], b1 = S[6];
  float a22 = S[7], b2 = S[8];
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
  float cost = sq_eval(S, x, y, z);
  if (cost < 0.0f) cost = 0.0f;
  return cost;
}

// [S1] per-face plane (recomputed each round; verts move)
__global__ void k_face_planes(const uint32_t *__restrict__ faces_g,
                              const float *__restrict__ verts,
                              const uint8_t *__restrict__ active_lab,
                              const uint32_t *__restrict__ flab,
                              SimpPlane *__restrict__ fq,
                              uint8_t *__restrict__ fvalid,
                              uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  if (!active_lab[flab[t]]) { fvalid[t] = 0; return; }
  uint32_t i0 = faces_g[3*t], i1 = faces_g[3*t+1], i2 = faces_g[3*t+2];
  const float *p0 = verts + 3*i0, *p1 = verts + 3*i1, *p2 = verts + 3*i2;
  float ux = p1[0]-p0[0], uy = p1[1]-p0[1], uz = p1[2]-p0[2];
  float vx = p2[0]-p0[0], vy = p2[1]-p0[1], vz = p2[2]-p0[2];
  float nx = uy*vz - uz*vy, ny = uz*vx - ux*vz, nz = ux*vy - uy*vx;
  float len = sqrtf(nx*nx + ny*ny + nz*nz);
  if (len <= 0.0f) {
    // zero plane: accumulating it adds exact +0 products, bitwise
    // identical to the oracle's skip (quadric sums are never -0)
    fq[t] = SimpPlane{0.0f, 0.0f, 0.0f, 0.0f};
    fvalid[t] = 0;
    return;
  }
  float inv = 1.0f / len;
  nx *= inv; ny *= inv; nz *= inv;
  float d = -(nx*p0[0] + ny*p0[1] + nz*p0[2]);
  fq[t] = SimpPlane{nx, ny, nz, d};
  fvalid[t] = 1;
}

// [S2] emit (vertex, face) pairs for quadric accumulation; inactive
// labels get the sentinel key (sorted to the end, skipped)
__global__ void k_emit_vf_pairs(const uint32_t *__restrict__ faces_g,
                                const uint8_t *__restrict__ active_lab,
                                const uint32_t *__restrict__ flab,
                                uint32_t *__restrict__ pk,
                                uint32_t *__restrict__ pv,
                                uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  bool act = active_lab[flab[t]];
  #pragma unroll
  for (int v = 0; v < 3; ++v) {
    pk[3*t + v] = act ? faces_g[3*t + v] : 0xFFFFFFFFu;
    pv[3*t + v] = (uint32_t)t;
  }
}

// [S3] per-vertex quadric: walk the vertex's pair segment in ascending
// face order (stable sort preserves it) — the oracle's summation order
__global__ void k_accum_quadrics(const uint32_t *__restrict__ pk,
                                 const uint32_t *__restrict__ pv,
                                 const SimpPlane *__restrict__ fq,
                                 const uint8_t *__restrict__ fvalid,
                                 float *__restrict__ Q,
                                 uint64_t npairs) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= npairs) return;
  uint32_t v = pk[i];
  if (v == 0xFFFFFFFFu) return;
  if (i > 0 && pk[i-1] == v) return;  // not a segment head
  float q[10];
  #pragma unroll
  for (int k = 0; k < 10; ++k) q[k] = 0.0f;
  for (uint64_t j = i; j < npairs && pk[j] == v; ++j) {
    uint32_t f = pv[j];
    if (!fvalid[f]) continue;
    SimpPlane p = fq[f];
    sq_add_plane(q, p.nx, p.ny, p.nz, p.d, 1.0f);
  }
  float4 *qo = (float4 *)(Q + 12ull*v);
  qo[0] = make_float4(q[0], q[1], q[2], q[3]);
  qo[1] = make_float4(q[4], q[5], q[6], q[7]);
  qo[2] = make_float4(q[8], q[9], 0.0f, 0.0f);
}

// [S4] per-vertex cheapest incident edge (cost-bits<<32 | peer, min)
__global__ void k_edge_pick(const uint32_t *__restrict__ faces_g,
                            const uint8_t *__restrict__ active_lab,
                            const uint32_t *__restrict__ flab,
                            const uint32_t *__restrict__ vbase,
                            const float *__restrict__ verts,
                            const float *__restrict__ Q,
                            unsigned long long *__restrict__ pick,
                            float max_cost, uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t lab = flab[t];
  if (!active_lab[lab]) return;
  const uint32_t vb = vbase[lab];
  // stage the 3 corner quadrics + positions in registers once per face
  // (the a<b swap below defeats CSE of the Q loads otherwise)
  uint32_t fc[3] = {faces_g[3*t], faces_g[3*t+1], faces_g[3*t+2]};
  float cq[3][10], cp[3][3];
  #pragma unroll
  for (int ci = 0; ci < 3; ++ci) {
    const float4 *qp = (const float4 *)(Q + 12ull*fc[ci]);
    float4 a = qp[0], b = qp[1], cc = qp[2];
    cq[ci][0] = a.x; cq[ci][1] = a.y; cq[ci][2] = a.z; cq[ci][3] = a.w;
    cq[ci][4] = b.x; cq[ci][5] = b.y; cq[ci][6] = b.z; cq[ci][7] = b.w;
    cq[ci][8] = cc.x; cq[ci][9] = cc.y;
    #pragma unroll
    for (int k = 0; k < 3; ++k) cp[ci][k] = verts[3ull*fc[ci] + k];
  }
  #pragma unroll
  for (int e = 0; e < 3; ++e) {
    const int ea = e, eb = (e + 1) % 3;   // compile-time after unroll
    uint32_t a = fc[ea], b = fc[eb];
    if (a == b) continue;
    // canonical u < w; addition is commutative-safe here only because
    // both operand orders are evaluated identically (x+y) — keep the
    // oracle's Q[u]+Q[w] order via selects on compile-time indices
    bool fwd = a < b;
    uint32_t u = fwd ? a : b, w = fwd ? b : a;
    float mx = 0.5f*((fwd ? cp[ea][0] : cp[eb][0]) + (fwd ? cp[eb][0] : cp[ea][0]));
    float my = 0.5f*((fwd ? cp[ea][1] : cp[eb][1]) + (fwd ? cp[eb][1] : cp[ea][1]));
    float mz = 0.5f*((fwd ? cp[ea][2] : cp[eb][2]) + (fwd ? cp[eb][2] : cp[ea][2]));
    float S[10];
    #pragma unroll
    for (int k = 0; k < 10; ++k)
      S[k] = (fwd ? cq[ea][k] : cq[eb][k]) + (fwd ? cq[eb][k] : cq[ea][k]);
    float cost = sq_place_cost(S, mx, my, mz);
    if (cost > max_cost) continue;
    uint32_t cb = __float_as_uint(cost);
    // per-edge tie jitter — identical to oracle/simplify.c, which works
    // in label-LOCAL vertex ids: subtract the label's vertex base
    uint32_t ul = u - vb, wl = w - vb;
    uint32_t hsh = ul ^ (wl * 2654435761u);
    hsh ^= hsh >> 16; hsh *= 2246822519u; hsh ^= hsh >> 13;
    cb ^= (hsh & 7u);
    atomicMin(&pick[u], ((unsigned long long)cb << 32) | w);
    atomicMin(&pick[w], ((unsigned long long)cb << 32) | u);
  }
}

// [S5] matched pairs collapse; u (smaller id) survives. Matched
// vertices leave the pick graph (blocked marking for the proposal
// wave; the plain-store races are decision-invariant).
__global__ void k_collapse(unsigned long long *__restrict__ pick,
                           float *__restrict__ verts,
                           uint32_t *__restrict__ remap,
                           float *__restrict__ Q,
                           uint64_t nverts) {
  uint64_t u = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= nverts) return;
  unsigned long long pu = pick[u];
  if (pu == ~0ull) return;
  uint32_t w = (uint32_t)pu;
  if (w <= u) return;
  unsigned long long pw = pick[w];
  if (pw == ~0ull || (uint32_t)pw != u) return;
  {
    float mx = 0.5f*(verts[3*u]+verts[3*w]);
    float my = 0.5f*(verts[3*u+1]+verts[3*w+1]);
    float mz = 0.5f*(verts[3*u+2]+verts[3*w+2]);
    float S[10];
    #pragma unroll
    for (int k = 0; k < 10; ++k)
      S[k] = Q[12*u + k] + Q[12*(uint64_t)w + k];
    float px, py, pz;
    (void)sq_place(S, mx, my, mz, &px, &py, &pz);
    verts[3*u] = px; verts[3*u+1] = py; verts[3*u+2] = pz;
  }
  #pragma unroll
  for (int k = 0; k < 10; ++k) Q[12*u + k] += Q[12*(uint64_t)w + k];
  remap[w] = (uint32_t)u;
  pick[u] = ~0ull;
  pick[w] = ~0ull;
}

// [S5b] proposal wave (oracle step 3b), global-rounds path. Vertex ids
// in pick are global; the proposer id in the acceptance key preserves
// the oracle's local tie-break order (ids within one label are
// monotone in the global numbering).
__global__ void k_propose(const unsigned long long *__restrict__ pick,
                          uint32_t *__restrict__ accept,
                          uint64_t nverts) {
  uint64_t v = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (v >= nverts) return;
  unsigned long long pu = pick[v];
  if (pu == ~0ull) return;
  uint32_t w = (uint32_t)pu;
  if (w <= v) return;                  // propose-up only
  if (pick[w] == ~0ull) return;        // blocked target
  atomicMin(&accept[w], (uint32_t)(v + 1));
}

__global__ void k_accept(const unsigned long long *__restrict__ pick,
                         const uint32_t *__restrict__ accept,
                         float *__restrict__ verts,
                         uint32_t *__restrict__ remap,
                         float *__restrict__ Q,
                         uint64_t nverts) {
  uint64_t w = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (w >= nverts) return;
  uint32_t aw = accept[w];
  if (aw == 0xFFFFFFFFu) return;
  unsigned long long pw = pick[w];
  if (pw == ~0ull) return;             // matched or pickless
  if ((uint32_t)pw > w) return;        // proposers never accept
  uint64_t u = (uint64_t)aw - 1;
  {
    float mx = 0.5f*(verts[3*u]+verts[3*w]);
    float my = 0.5f*(verts[3*u+1]+verts[3*w+1]);
    float mz = 0.5f*(verts[3*u+2]+verts[3*w+2]);
    float S[10];
    #pragma unroll
    for (int k = 0; k < 10; ++k)
      S[k] = Q[12*u + k] + Q[12*(uint64_t)w + k];
    float px, py, pz;
    (void)sq_place(S, mx, my, mz, &px, &py, &pz);
    verts[3*u] = px; verts[3*u+1] = py; verts[3*u+2] = pz;
  }
  #pragma unroll
  for (int k = 0; k < 10; ++k) Q[12*u + k] += Q[12*(uint64_t)w + k];
  remap[w] = (uint32_t)u;
}

// [S6] remap face corners in place; keep flag for non-degenerates
__global__ void k_remap_faces(uint32_t *__restrict__ faces_g,
                              const uint32_t *__restrict__ remap,
                              uint32_t *__restrict__ keep,
                              uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t i0 = remap[faces_g[3*t]], i1 = remap[faces_g[3*t+1]],
           i2 = remap[faces_g[3*t+2]];
  faces_g[3*t] = i0; faces_g[3*t+1] = i1; faces_g[3*t+2] = i2;
  keep[t] = (i0 != i1 && i1 != i2 && i0 != i2) ? 1u : 0u;
}

// [S7] stable compaction of kept faces (+ labels); per-label new counts
__global__ void k_compact_faces(const uint32_t *__restrict__ faces_g,
                                const uint32_t *__restrict__ flab,
                                const uint32_t *__restrict__ keep,
                                const uint32_t *__restrict__ keep_scan,
                                uint32_t *__restrict__ faces_out,
                                uint32_t *__restrict__ flab_out,
                                uint32_t *__restrict__ nt_new,
                                uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  if (!keep[t]) return;
  uint64_t o = keep_scan[t];
  faces_out[3*o] = faces_g[3*t];
  faces_out[3*o+1] = faces_g[3*t+1];
  faces_out[3*o+2] = faces_g[3*t+2];
  uint32_t lab = flab[t];
  flab_out[o] = lab;
  atomicAdd(&nt_new[lab], 1u);
}

// [S8] per-label round bookkeeping
__global__ void k_update_active(const uint32_t *__restrict__ nt_new,
                                uint32_t *__restrict__ nt_cur,
                                const uint32_t *__restrict__ target,
                                uint8_t *__restrict__ active_lab,
                                uint32_t *__restrict__ any_active,
                                uint32_t nlabels,
                                uint32_t enforce_progress) {
  // enforce_progress=1 on a group's FIRST sub-round: a label whose
  // fresh-quadric sub collapses nothing is done (oracle termination).
  // Later subs leave no-progress labels active — they recompute next
  // group (their remaining subs this group are deterministic no-ops).
  uint32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l >= nlabels) return;
  if (!active_lab[l]) return;
  uint32_t nn = nt_new[l];
  bool progress = nn != nt_cur[l];
  nt_cur[l] = nn;
  bool act = nn > target[l] && (progress || !enforce_progress);
  active_lab[l] = act ? 1 : 0;
  if (act) atomicExch(any_active, 1u);
}

__global__ void k_init_simplify(const uint32_t *__restrict__ tri_off,
                                uint32_t *__restrict__ nt_cur,
                                uint32_t *__restrict__ target,
                                uint8_t *__restrict__ active_lab,
                                uint32_t reduction_factor,
                                uint32_t nlabels) {
  uint32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l >= nlabels) return;
  uint32_t nt = tri_off[l+1] - tri_off[l];
  uint32_t tg = nt / reduction_factor;
  if (tg < 1) tg = 1;
  nt_cur[l] = nt;
  target[l] = tg;
  active_lab[l] = (nt > tg) ? 1 : 0;
}

// faces local -> global vertex ids (vbase offset per label)
__global__ void k_globalize_faces(uint32_t *__restrict__ faces,
                                  const uint32_t *__restrict__ lab_sorted,
                                  const uint32_t *__restrict__ vbase,
                                  uint32_t *__restrict__ flab,
                                  uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t lab = lab_sorted[t];
  flab[t] = lab;
  uint32_t base = vbase[lab];
  faces[3*t] += base; faces[3*t+1] += base; faces[3*t+2] += base;
}

// ---- final referenced-vertex compaction ------------------------------

__global__ void k_mark_ref(const uint32_t *__restrict__ faces_g,
                           uint32_t *__restrict__ ref, uint64_t ncorners) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= ncorners) return;
  ref[faces_g[i]] = 1u;  // idempotent store, order-free
}

__global__ void k_scatter_verts(const float *__restrict__ verts,
                                const uint32_t *__restrict__ ref,
                                const uint32_t *__restrict__ newid,
                                float *__restrict__ verts_out,
                                uint64_t nverts) {
  uint64_t v = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (v >= nverts) return;
  if (!ref[v]) return;
  uint64_t o = newid[v];
  verts_out[3*o] = verts[3*v];
  verts_out[3*o+1] = verts[3*v+1];
  verts_out[3*o+2] = verts[3*v+2];
}

__global__ void k_new_vbase(const uint32_t *__restrict__ vbase_old,
                            const uint32_t *__restrict__ newid,
                            uint32_t *__restrict__ vbase_new,
                            uint32_t nlabels, uint32_t total_new) {
  uint32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l > nlabels) return;
  vbase_new[l] = (l == nlabels) ? total_new : newid[vbase_old[l]];
}

__global__ void k_localize_faces(const uint32_t *__restrict__ faces_g,
                                 const uint32_t *__restrict__ newid,
                                 const uint32_t *__restrict__ flab,
                                 const uint32_t *__restrict__ vbase_new,
                                 uint32_t *__restrict__ faces_out,
                                 uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t base = vbase_new[flab[t]];
  faces_out[3*t] = newid[faces_g[3*t]] - base;
  faces_out[3*t+1] = newid[faces_g[3*t+1]] - base;
  faces_out[3*t+2] = newid[faces_g[3*t+2]] - base;
}

// ---- working-set parking: a label that goes inactive leaves the round
// loop entirely; its faces are stored verbatim (in face order) at the
// label's ORIGINAL tri_off offset in the park store, and the working
// array keeps only active labels' faces. Late rounds then touch only the
// shrinking active tail instead of all 155M faces.

__global__ void k_flag_active_faces(const uint32_t *__restrict__ flab,
                                    const uint8_t *__restrict__ active_lab,
                                    uint32_t *__restrict__ aflag,
                                    uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  aflag[t] = active_lab[flab[t]] ? 1u : 0u;
}

__global__ void k_first_label_idx(const uint32_t *__restrict__ flab,
                                  uint32_t *__restrict__ first,
                                  uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  if (t == 0 || flab[t] != flab[t - 1]) first[flab[t]] = (uint32_t)t;
}

__global__ void k_park_scatter(const uint32_t *__restrict__ faces_g,
                               const uint32_t *__restrict__ flab,
                               const uint32_t *__restrict__ aflag,
                               const uint32_t *__restrict__ apos,
                               const uint32_t *__restrict__ first,
                               const uint32_t *__restrict__ orig_tri_off,
                               uint32_t *__restrict__ work_faces,
                               uint32_t *__restrict__ work_flab,
                               uint32_t *__restrict__ park_faces,
                               uint32_t skip_cap,  // labels <= this size
                               // were parked by k_simplify_label: drop
                               uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t lab = flab[t];
  if (!aflag[t] &&
      orig_tri_off[lab + 1] - orig_tri_off[lab] <= skip_cap)
    return;  // small label: its park slice is already final
  if (aflag[t]) {
    uint64_t o = apos[t];
    work_faces[3*o] = faces_g[3*t];
    work_faces[3*o+1] = faces_g[3*t+1];
    work_faces[3*o+2] = faces_g[3*t+2];
    work_flab[o] = lab;
  } else {
    uint64_t o = (uint64_t)orig_tri_off[lab] + ((uint32_t)t - first[lab]);
    park_faces[3*o] = faces_g[3*t];
    park_faces[3*o+1] = faces_g[3*t+1];
    park_faces[3*o+2] = faces_g[3*t+2];
  }
}

// assemble the final compact face array from the park store
__global__ void k_gather_final(const uint32_t *__restrict__ park_faces,
                               const uint32_t *__restrict__ orig_tri_off,
                               const uint32_t *__restrict__ final_tri_off,
                               uint32_t *__restrict__ faces_out,
                               uint32_t *__restrict__ flab_out,
                               uint32_t nlabels, uint64_t final_total) {
  uint64_t f = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (f >= final_total) return;
  // binary search: label whose [final_tri_off[l], final_tri_off[l+1]) holds f
  uint32_t lo = 0, hi = nlabels;
  while (hi - lo > 1) {
    uint32_t mid = (lo + hi) >> 1;
    if ((uint64_t)final_tri_off[mid] <= f) lo = mid; else hi = mid;
  }
  uint32_t l = lo;
  uint64_t src = (uint64_t)orig_tri_off[l] + (f - final_tri_off[l]);
  faces_out[3*f] = park_faces[3*src];
  faces_out[3*f+1] = park_faces[3*src+1];
  faces_out[3*f+2] = park_faces[3*src+2];
  flab_out[f] = l;
}

// ---------------------------------------------------------------------------
// Per-label simplification: one workgroup runs a label's ENTIRE round
// loop over its own slices of the global arrays (faces/verts/Q/pick/
// remap + a per-label CSR). A label's working set (~100-300 KB) stays
// cache-resident and there are no global sorts, no host round trips and
// no device-wide synchronization — all 50k labels simplify concurrently.
// Labels larger than `big_cap` faces are left to the global-rounds path.
//
// Arithmetic and schedule are IDENTICAL to oracle/simplify.c per label
// (same plane/quadric/cost expressions, ascending-face quadric order via
// the sorted CSR, jittered pick encoding on label-local ids, mutual-pick
// matched collapse, stable compaction, same termination conditions).

// wave-shuffle prefix machinery shared by the block-wide helpers below:
// each thread sums its contiguous chunk, a 64-lane shuffle scan orders
// the per-thread partials within the wave (no LDS), ONE barrier shares
// the NW wave totals, and every thread derives its global offset from
// the (tiny) wave-total array. 2 barriers per helper call instead of the
// 16+ of the old Hillis-Steele block scan — the round loop's barrier
// count was the per-label kernel's dominant WAIT source.
template <int BS>
__device__ __forceinline__ uint32_t blk_prefix(uint32_t sum,
                                               uint32_t *s_wsum /*NW*/,
                                               uint32_t *p_total) {
  constexpr int NW = BS / 64;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  uint32_t incl = sum;
  #pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    uint32_t y = __shfl_up(incl, d, 64);
    if (lane >= d) incl += y;
  }
  if (lane == 63) s_wsum[wid] = incl;
  __syncthreads();
  uint32_t wbase = 0, total = 0;
  #pragma unroll
  for (int w = 0; w < NW; ++w) {
    uint32_t x = s_wsum[w];
    if (w < wid) wbase += x;
    total += x;
  }
  *p_total = total;
  return wbase + incl - sum;  // exclusive prefix of this thread's sum
}

// block-wide exclusive scan of src[0..n) into dst (+base), returns total
template <int BS>
__device__ uint32_t blk_exscan(const uint32_t *src, uint32_t *dst,
                               uint32_t n, uint32_t base,
                               uint32_t *s_sums /*>= BS/64*/,
                               uint32_t *dst2 = nullptr) {
  const uint32_t tid = threadIdx.x;
  const uint32_t chunk = (n + BS - 1) / BS;
  const uint32_t lo = tid * chunk;
  const uint32_t hi = lo + chunk < n ? lo + chunk : n;
  uint32_t sum = 0;
  for (uint32_t i = lo; i < hi; ++i) sum += src[i];
  uint32_t total;
  uint32_t excl = blk_prefix<BS>(sum, s_sums, &total);
  uint32_t run = base + excl;
  for (uint32_t i = lo; i < hi; ++i) {
    uint32_t t = src[i];
    dst[i] = run;
    if (dst2) dst2[i] = run;  // optional cursor copy (src may alias
    run += t;                 // dst2: t was read first)
  }
  __syncthreads();  // s_sums reusable after this
  return total;
}

// fused face rewrite + stable compaction: remap corners through the
// label-local rm[] (LDS), keep-flag in a register bitmask (no valid[]
// array traffic), scan via blk_prefix, scatter kept remapped faces into
// fb. fa is never written: the second pass re-reads fa and re-applies
// the (cheap, LDS) rm gathers instead of paying a 12 B/face global
// write + re-read. Identical output order to the old rewrite + compact
// pair.
template <int BS, int MAXN, typename RMT>
__device__ uint32_t blk_rewrite_compact(const uint32_t *__restrict__ fa,
                                        uint32_t *__restrict__ fb,
                                        const RMT *__restrict__ rm,
                                        uint32_t v0,
                                        uint32_t n, uint32_t *s_sums) {
  const uint32_t tid = threadIdx.x;
  const uint32_t chunk = (n + BS - 1) / BS;
  const uint32_t lo = tid * chunk;
  const uint32_t hi = lo + chunk < n ? lo + chunk : n;
  // bitmask register array sized for the band's face-count cap MAXN
  constexpr uint32_t MAXW = ((uint32_t)MAXN / BS + 63) / 64 + 1;
  unsigned long long bm[MAXW] = {};
  uint32_t sum = 0;
  for (uint32_t i = lo; i < hi; ++i) {
    uint32_t i0 = (uint32_t)rm[fa[3*i] - v0];
    uint32_t i1 = (uint32_t)rm[fa[3*i+1] - v0];
    uint32_t i2 = (uint32_t)rm[fa[3*i+2] - v0];
    if (i0 != i1 && i1 != i2 && i0 != i2) {
      bm[(i - lo) >> 6] |= 1ull << ((i - lo) & 63);
      ++sum;
    }
  }
  uint32_t total;
  uint32_t run = blk_prefix<BS>(sum, s_sums, &total);
  for (uint32_t i = lo; i < hi; ++i)
    if (bm[(i - lo) >> 6] & (1ull << ((i - lo) & 63))) {
      fb[3*run] = v0 + (uint32_t)rm[fa[3*i] - v0];
      fb[3*run+1] = v0 + (uint32_t)rm[fa[3*i+1] - v0];
      fb[3*run+2] = v0 + (uint32_t)rm[fa[3*i+2] - v0];
      ++run;
    }
  __syncthreads();  // s_sums reusable after this
  return total;
}

// (an amdgpu_waves_per_eu(4) floor was measured here: it fits 127 VGPRs
// with 52 B/lane scratch, and the spill traffic LOSES 7% net — the
// natural 148-VGPR / 3-waves-per-SIMD allocation stands)
// The 16-wide bitonic sort tier is the kernel's VGPR peak: with it the
// allocation is 148 VGPRs (3 waves/SIMD), without it 116 (4 waves/SIMD)
// — degree>8 vertices are rare enough that the insertion-sort fallback
// plus the extra wave slot wins (A/B'd on the 512^3/50k config).
#ifndef SIMP_D16_TIER
#define SIMP_D16_TIER 1
#endif
// WAVEMODE (BS=64, one wave per label): the whole round loop runs
// wave-synchronously — every __syncthreads() in a 64-thread block
// lowers to a waitcnt, the pick table ALIASES the (phase-disjoint)
// degree/cursor array, and remap is u16 — 20 KB LDS per label instead
// of 33, so ~7 labels are resident per CU instead of 3. Owns the
// (nt_lo, nt_hi] x (nv_lo, nv_hi] band its launch names.
template <bool CLLDS, int BS, int CAPVT, bool WAVEMODE = false>
__global__ __launch_bounds__(BS)
__attribute__((amdgpu_waves_per_eu(4))) void k_simplify_label(
    uint32_t *__restrict__ faces_g,       // slices at 3*tri_off[b]
    uint32_t *__restrict__ faces_tmp,     // same slicing (scratch)
    const uint32_t *__restrict__ tri_off, // L+1 (original offsets)
    const uint32_t *__restrict__ vbase,   // L+1
    float *__restrict__ verts,
    float *__restrict__ Q,                // 10 per vertex
    unsigned long long *__restrict__ pick,
    uint32_t *__restrict__ remap,
    uint32_t *__restrict__ deg,           // per vertex (doubles as cursor)
    uint32_t *__restrict__ adj_off,       // per vertex
    uint32_t *__restrict__ cols,          // 3 per face slot (CSR payload)
    SimpPlane *__restrict__ fq,           // per face (round scratch)
    uint8_t *__restrict__ fvalid,         // per face (plane valid / keep)
    uint32_t *__restrict__ nt_cur,
    const uint32_t *__restrict__ target,
    uint8_t *__restrict__ active,
    uint32_t *__restrict__ park_faces,
    unsigned long long *__restrict__ prof,  // 6 phase counters or null
    uint32_t *__restrict__ roundhist,  // 160 u32 or null (see host)
    float max_cost, uint32_t nlabels, uint32_t big_cap,
    uint32_t subs, uint32_t nv_lo, uint32_t nv_hi,
    uint32_t nt_lo, uint32_t nt_hi,
    uint32_t *__restrict__ accept_g,  // per-vertex (wave 2)
    uint32_t propose,
    const uint32_t *__restrict__ sched /*block->label, largest first*/) {
  if (blockIdx.x >= nlabels) return;
  // biggest-label-first dispatch: per-label serial time scales with nt0,
  // and a 65k-face label dispatched late extends the whole launch by its
  // full runtime — schedule stragglers first so they overlap the swarm
  const uint32_t b = sched ? sched[blockIdx.x] : blockIdx.x;
  unsigned long long t_start =
      roundhist ? __builtin_amdgcn_s_memtime() : 0;
  const uint32_t f0 = tri_off[b];
  const uint32_t nt0 = tri_off[b + 1] - f0;
  if (nt0 > big_cap) return;  // global-rounds path handles big labels
  const uint32_t v0 = vbase[b];
  const uint32_t nv = vbase[b + 1] - v0;
  // size-class dispatch: each launch variant owns an (nv_lo, nv_hi] x
  // (nt_lo, nt_hi] band
  if (nv <= nv_lo || nv > nv_hi) return;
  if (nt0 <= nt_lo || nt0 > nt_hi) return;
  if (!active[b]) {
    // already at/below target: final faces = original faces; park them
    for (uint32_t i = threadIdx.x; i < 3 * nt0; i += BS)
      park_faces[3ull * f0 + i] = faces_g[3ull * f0 + i];
    return;
  }
  const uint32_t tgt = target[b];
  const uint32_t tid = threadIdx.x;
  uint32_t *faces = faces_g + 3ull * f0;
  uint32_t *ftmp = faces_tmp + 3ull * f0;
  SimpPlane *pl = fq + f0;
  uint32_t *aoff = adj_off + v0;
  uint32_t *cl = cols + 3ull * f0;

  __shared__ uint32_t s_sums[BS / 64];  // wave totals for blk_prefix
  __shared__ uint32_t s_nt, s_collapses;
  // the two ATOMIC-hot per-vertex arrays live in LDS when the label fits:
  // neighboring faces' vertices share cache lines, so the global
  // atomicAdd/atomicMin streams serialize on hot L2 lines — LDS atomics
  // are bank-parallel. (~25 KB -> ~6 blocks/CU.)
  constexpr uint32_t CAPV = CAPVT;
  __shared__ uint32_t s_deg[WAVEMODE ? 1 : CAPV];
  __shared__ unsigned long long s_pick[CAPV];
  // optional LDS-resident CSR payload (face ids, u16) for rounds whose
  // face count fits; costs 24 KB LDS -> fewer blocks/CU, so env-gated
  // for A/B (MG_SIMP_CLLDS=0 disables; measured ~5% faster on, default on).
  constexpr uint32_t CAPF = CLLDS ? 4096u : 1u;
  __shared__ uint16_t s_cl[3 * CAPF];
  // remap holds label-LOCAL canonical ids; LDS-resident when the label
  // fits (rewrite's 3 gathers/face are the hot readers). WAVEMODE packs
  // it to u16 (label-local ids < CAPV).
  using RMT = typename std::conditional<WAVEMODE, uint16_t,
                                        uint32_t>::type;
  __shared__ RMT s_remap[CAPV];
  const bool lds_mode = (nv <= CAPV);
  // WAVEMODE: the degree/cursor array lives in the PICK bytes — deg is
  // dead before the first pick write of each group (the reset pass
  // below runs after the last cursor read, behind a barrier)
  uint32_t *dg = WAVEMODE ? (uint32_t *)s_pick
                          : (lds_mode ? s_deg : (deg + v0));
  unsigned long long *pick_l =
      lds_mode ? s_pick : (pick + v0);
  RMT *rm = lds_mode ? s_remap : (RMT *)(remap + v0);
  // accept table (proposal wave): u32 id-only keys — lives in the
  // degree array's bytes (dead during the collapse phases) when the
  // label is LDS-resident; global fallback otherwise
  uint32_t *accept_l = (!WAVEMODE && lds_mode)
                           ? s_deg : (accept_g + v0);
  // ping-pong face buffers: rewrite reads fa, compaction scatters into
  // fb, then the buffers swap — no copy-back pass. Parking at the end
  // reads whichever buffer is current.
  uint32_t *fa = faces, *fb = ftmp;
  if (tid == 0) s_nt = nt0;
  __syncthreads();

  // env-gated phase profiling (thread 0 wall cycles per phase)
  unsigned long long ph[6] = {0, 0, 0, 0, 0, 0};
  unsigned long long t_last = prof ? __builtin_amdgcn_s_memtime() : 0;
#define PHASE_MARK(k)                                          \
  if (prof && tid == 0) {                                      \
    unsigned long long now = __builtin_amdgcn_s_memtime();     \
    ph[k] += now - t_last;                                     \
    t_last = now;                                              \
  }

  uint32_t n_groups = 0, n_subs = 0;
  for (int round = 0; round < 65536; ++round) {
    uint32_t nt = s_nt;
    if (nt <= tgt) break;
    ++n_groups;
    const uint32_t nt_group = nt;  // group-progress watermark
    const bool clmode = CLLDS && (nt <= CAPF);
    PHASE_MARK(0)  // loop head

    // [1+2] face planes fused with the CSR degree count (one face pass)
    for (uint32_t v = tid; v < nv; v += BS) dg[v] = 0;
    __syncthreads();
    for (uint32_t f = tid; f < nt; f += BS) {
      uint32_t i0 = fa[3*f], i1 = fa[3*f+1], i2 = fa[3*f+2];
      atomicAdd(&dg[i0 - v0], 1u);
      atomicAdd(&dg[i1 - v0], 1u);
      atomicAdd(&dg[i2 - v0], 1u);
      const float *p0 = verts + 3ull*i0, *p1 = verts + 3ull*i1,
                  *p2 = verts + 3ull*i2;
      float ux = p1[0]-p0[0], uy = p1[1]-p0[1], uz = p1[2]-p0[2];
      float vx = p2[0]-p0[0], vy = p2[1]-p0[1], vz = p2[2]-p0[2];
      float nx = uy*vz - uz*vy, ny = uz*vx - ux*vz, nz = ux*vy - uy*vx;
      float len = sqrtf(nx*nx + ny*ny + nz*nz);
      if (len <= 0.0f) {
        pl[f] = SimpPlane{0.0f, 0.0f, 0.0f, 0.0f};  // +0 products = skip
        continue;
      }
      float inv = 1.0f / len;
      nx *= inv; ny *= inv; nz *= inv;
      float d = -(nx*p0[0] + ny*p0[1] + nz*p0[2]);
      pl[f] = SimpPlane{nx, ny, nz, d};
    }
    __syncthreads();
    PHASE_MARK(1)  // planes + degree count
    // [3] offsets; the scan also writes the fill cursors (dst2 = dg)
    blk_exscan<BS>(dg, aoff, nv, 0, s_sums, dg);
    if (clmode) {
      for (uint32_t f = tid; f < nt; f += BS) {
        s_cl[atomicAdd(&dg[fa[3*f] - v0], 1u)] = (uint16_t)f;
        s_cl[atomicAdd(&dg[fa[3*f+1] - v0], 1u)] = (uint16_t)f;
        s_cl[atomicAdd(&dg[fa[3*f+2] - v0], 1u)] = (uint16_t)f;
      }
    } else {
      for (uint32_t f = tid; f < nt; f += BS) {
        cl[atomicAdd(&dg[fa[3*f] - v0], 1u)] = f;
        cl[atomicAdd(&dg[fa[3*f+1] - v0], 1u)] = f;
        cl[atomicAdd(&dg[fa[3*f+2] - v0], 1u)] = f;
      }
    }
    __syncthreads();
    PHASE_MARK(2)  // offsets scan + CSR fill
    // [5] per-vertex: sort incident faces ascending (insertion sort),
    // accumulate quadrics in that order (oracle step 1)
    for (uint32_t v = tid; v < nv; v += BS) {
      uint32_t lo = aoff[v];
      uint32_t hi = dg[v];  // cursor ended at one-past-last
      uint32_t d = hi - lo;
      float q[10];
      #pragma unroll
      for (int k = 0; k < 10; ++k) q[k] = 0.0f;
      if (d <= 8) {
        // dominant tier: interior vertices have degree ~6 — an 8-wide
        // bitonic network + 8 plane gathers costs ~1/3 of the 16-wide
        // path it replaces. Sorted order is unique (face ids distinct),
        // so this matches the oracle's insertion sort exactly.
        uint32_t fl[8];
        #pragma unroll
        for (int k = 0; k < 8; ++k)
          fl[k] = (k < (int)d)
                      ? (clmode ? (uint32_t)s_cl[lo + k] : cl[lo + k])
                      : 0xFFFFFFFFu;
        #pragma unroll
        for (int ksz = 2; ksz <= 8; ksz <<= 1) {
          #pragma unroll
          for (int j = ksz >> 1; j > 0; j >>= 1) {
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
              int l = i ^ j;
              if (l > i) {
                bool up = ((i & ksz) == 0);
                uint32_t a = fl[i], b2 = fl[l];
                bool sw = up ? (a > b2) : (a < b2);
                fl[i] = sw ? b2 : a;
                fl[l] = sw ? a : b2;
              }
            }
          }
        }
        SimpPlane ps[8];
        #pragma unroll
        for (int k = 0; k < 8; ++k)
          ps[k] = pl[k < (int)d ? fl[k] : 0u];
        #pragma unroll
        for (int k = 0; k < 8; ++k)
          if (k < (int)d)
            sq_add_plane(q, ps[k].nx, ps[k].ny, ps[k].nz, ps[k].d, 1.0f);
      } else if (SIMP_D16_TIER && d <= 16) {
        // register path: one batched load of the face list, bitonic
        // sort network (compile-time indices, no global RMW chains).
        uint32_t fl[16];
        #pragma unroll
        for (int k = 0; k < 16; ++k)
          fl[k] = (k < (int)d)
                      ? (clmode ? (uint32_t)s_cl[lo + k] : cl[lo + k])
                      : 0xFFFFFFFFu;
        #pragma unroll
        for (int ksz = 2; ksz <= 16; ksz <<= 1) {
          #pragma unroll
          for (int j = ksz >> 1; j > 0; j >>= 1) {
            #pragma unroll
            for (int i = 0; i < 16; ++i) {
              int l = i ^ j;
              if (l > i) {
                bool up = ((i & ksz) == 0);
                uint32_t a = fl[i], b2 = fl[l];
                bool sw = up ? (a > b2) : (a < b2);
                fl[i] = sw ? b2 : a;
                fl[l] = sw ? a : b2;
              }
            }
          }
        }
        // gather planes in two 8-batches (8 independent loads in flight
        // each) — staging all 16 at once cost 64 VGPRs and pushed the
        // kernel past the 128-VGPR / 4-waves-per-SIMD occupancy step.
        // Accumulation order (ascending fl) is unchanged.
        #pragma unroll
        for (int h = 0; h < 2; ++h) {
          SimpPlane ps[8];
          #pragma unroll
          for (int k = 0; k < 8; ++k)
            // pad with face 0 (nt >= 1 inside the loop); fl[k] is the
            // 0xFFFFFFFF sort sentinel beyond d, must not be indexed
            ps[k] = pl[8*h + k < (int)d ? fl[8*h + k] : 0u];
          #pragma unroll
          for (int k = 0; k < 8; ++k)
            if (8*h + k < (int)d)
              sq_add_plane(q, ps[k].nx, ps[k].ny, ps[k].nz, ps[k].d, 1.0f);
        }
      } else if (clmode) {
        for (uint32_t i = lo + 1; i < hi; ++i) {
          uint16_t x = s_cl[i];
          uint32_t j = i;
          while (j > lo && s_cl[j-1] > x) { s_cl[j] = s_cl[j-1]; --j; }
          s_cl[j] = x;
        }
        for (uint32_t i = lo; i < hi; ++i) {
          SimpPlane p = pl[s_cl[i]];
          sq_add_plane(q, p.nx, p.ny, p.nz, p.d, 1.0f);
        }
      } else {
        for (uint32_t i = lo + 1; i < hi; ++i) {
          uint32_t x = cl[i];
          uint32_t j = i;
          while (j > lo && cl[j-1] > x) { cl[j] = cl[j-1]; --j; }
          cl[j] = x;
        }
        for (uint32_t i = lo; i < hi; ++i) {
          SimpPlane p = pl[cl[i]];
          sq_add_plane(q, p.nx, p.ny, p.nz, p.d, 1.0f);
        }
      }
      {  // 12-float row (2 pad zeros), 3 dwordx4 stores
        float4 *qo = (float4 *)(Q + 12ull*(v0+v));
        qo[0] = make_float4(q[0], q[1], q[2], q[3]);
        qo[1] = make_float4(q[4], q[5], q[6], q[7]);
        qo[2] = make_float4(q[8], q[9], 0.0f, 0.0f);
      }
      if (!WAVEMODE) {
        pick_l[v] = ~0ull;  // fused pick reset (same-thread slot)
        rm[v] = (RMT)v;     // fused remap identity (consumed in collapse)
        if (propose) accept_l[v] = 0xFFFFFFFFu;
      }
    }
    __syncthreads();
    if (WAVEMODE) {
      // dg (aliasing the pick bytes) had its last read above; now the
      // pick table takes the space back
      for (uint32_t v = tid; v < nv; v += BS) {
        pick_l[v] = ~0ull;
        rm[v] = (RMT)v;
        if (propose) accept_l[v] = 0xFFFFFFFFu;
      }
      __syncthreads();
    }
    PHASE_MARK(3)  // sort + quadric accumulate
    // sub-rounds: reuse merged quadrics (Q[u] += Q[w] on collapse) for
    // up to `subs` pick/collapse/compact passes per recompute (oracle
    // group structure; a full recompute resets the drift)
    for (uint32_t sub = 0; sub < subs; ++sub) {
    nt = s_nt;
    if (nt <= tgt) break;
    ++n_subs;
    if (sub > 0) {
      for (uint32_t v = tid; v < nv; v += BS) {
        pick_l[v] = ~0ull;
        rm[v] = (RMT)v;
        if (propose) accept_l[v] = 0xFFFFFFFFu;
      }
      __syncthreads();
    }
    // [6] picks (oracle step 2). Q rows are 12 floats (48 B, 16-B
    // aligned): stage each corner's quadric ONCE per face (3 dwordx4
    // loads) and build every edge's sum from registers — the edge loop
    // re-loading both quadrics per edge doubled the phase's traffic.
    // f32 order preserved via selects on compile-time indices, exactly
    // like the global k_edge_pick.
    for (uint32_t f = tid; f < nt; f += BS) {
      uint32_t fc[3] = {fa[3*f], fa[3*f+1], fa[3*f+2]};
      float cq[3][10], cp[3][3];
      #pragma unroll
      for (int ci = 0; ci < 3; ++ci) {
        const float4 *qp = (const float4 *)(Q + 12ull*fc[ci]);
        float4 a4 = qp[0], b4 = qp[1], c4 = qp[2];
        cq[ci][0] = a4.x; cq[ci][1] = a4.y; cq[ci][2] = a4.z;
        cq[ci][3] = a4.w; cq[ci][4] = b4.x; cq[ci][5] = b4.y;
        cq[ci][6] = b4.z; cq[ci][7] = b4.w; cq[ci][8] = c4.x;
        cq[ci][9] = c4.y;
        #pragma unroll
        for (int k = 0; k < 3; ++k) cp[ci][k] = verts[3ull*fc[ci] + k];
      }
      #pragma unroll
      for (int e = 0; e < 3; ++e) {
        const int ea = e, eb = (e + 1) % 3;  // compile-time after unroll
        uint32_t a = fc[ea], bb = fc[eb];
        if (a == bb) continue;
        bool fwd = a < bb;
        uint32_t u = fwd ? a : bb, w = fwd ? bb : a;
        float mx = 0.5f*((fwd ? cp[ea][0] : cp[eb][0]) + (fwd ? cp[eb][0] : cp[ea][0]));
        float my = 0.5f*((fwd ? cp[ea][1] : cp[eb][1]) + (fwd ? cp[eb][1] : cp[ea][1]));
        float mz = 0.5f*((fwd ? cp[ea][2] : cp[eb][2]) + (fwd ? cp[eb][2] : cp[ea][2]));
        float S[10];
        #pragma unroll
        for (int k = 0; k < 10; ++k)
          S[k] = (fwd ? cq[ea][k] : cq[eb][k]) + (fwd ? cq[eb][k] : cq[ea][k]);
        float cost = sq_place_cost(S, mx, my, mz);
        if (cost > max_cost) continue;
        uint32_t cb = __float_as_uint(cost);
        uint32_t ul = u - v0, wl = w - v0;  // label-local ids (oracle)
        uint32_t hsh = ul ^ (wl * 2654435761u);
        hsh ^= hsh >> 16; hsh *= 2246822519u; hsh ^= hsh >> 13;
        cb ^= (hsh & 7u);
        atomicMin(&pick_l[ul], ((unsigned long long)cb << 32) | w);
        atomicMin(&pick_l[wl], ((unsigned long long)cb << 32) | u);
      }
    }
    PHASE_MARK(4)  // edge picks
    // [7] matched-pair collapse (oracle step 3); rm was pre-set to the
    // identity in the same pass that reset the pick table
    if (tid == 0) s_collapses = 0;
    __syncthreads();
    for (uint32_t v = tid; v < nv; v += BS) {
      uint32_t u = v0 + v;
      unsigned long long pu = pick_l[v];
      if (pu == ~0ull) continue;
      uint32_t w = (uint32_t)pu;
      if (w <= u) continue;
      unsigned long long pw = pick_l[w - v0];
      if (pw == ~0ull || (uint32_t)pw != u) continue;
      {
        float mx = 0.5f*(verts[3ull*u]+verts[3ull*w]);
        float my = 0.5f*(verts[3ull*u+1]+verts[3ull*w+1]);
        float mz = 0.5f*(verts[3ull*u+2]+verts[3ull*w+2]);
        float S[10];
        #pragma unroll
        for (int k = 0; k < 10; ++k)
          S[k] = Q[12ull*u + k] + Q[12ull*w + k];
        float px, py, pz;
        (void)sq_place(S, mx, my, mz, &px, &py, &pz);
        verts[3ull*u] = px; verts[3ull*u+1] = py; verts[3ull*u+2] = pz;
      }
      #pragma unroll
      for (int k = 0; k < 10; ++k) Q[12ull*u + k] += Q[12ull*w + k];
      rm[w - v0] = u - v0;
      // blocked marking: matched vertices leave the pick graph (the
      // races on these plain stores are benign: any reader's decision
      // is identical for the old and the cleared value — DESIGN §2)
      pick_l[v] = ~0ull;
      pick_l[w - v0] = ~0ull;
      atomicAdd(&s_collapses, 1u);
    }
    __syncthreads();
    // [7b] proposal-acceptance second wave (oracle step 3b): mutual
    // picks converge on cost-minima stars, so unmatched vertices whose
    // pick points UP propose to that peer; an unmatched non-proposing
    // peer accepts its minimum (costbits, local proposer+1) proposal.
    if (propose) {
      for (uint32_t v = tid; v < nv; v += BS) {
        unsigned long long pu = pick_l[v];
        if (pu == ~0ull) continue;
        uint32_t w = (uint32_t)pu;
        if (w <= v0 + v) continue;            // propose-up only
        if (pick_l[w - v0] == ~0ull) continue;  // blocked target
        atomicMin(&accept_l[w - v0], v + 1);
      }
      __syncthreads();
      for (uint32_t v = tid; v < nv; v += BS) {
        uint32_t aw = accept_l[v];
        if (aw == 0xFFFFFFFFu) continue;
        unsigned long long pw = pick_l[v];
        if (pw == ~0ull) continue;            // matched or pickless
        uint32_t w = v0 + v;
        if ((uint32_t)pw > w) continue;       // proposers never accept
        uint32_t ul = aw - 1;
        uint32_t u = v0 + ul;
        {
          float mx = 0.5f*(verts[3ull*u]+verts[3ull*w]);
          float my = 0.5f*(verts[3ull*u+1]+verts[3ull*w+1]);
          float mz = 0.5f*(verts[3ull*u+2]+verts[3ull*w+2]);
          float S[10];
          #pragma unroll
          for (int k = 0; k < 10; ++k)
            S[k] = Q[12ull*u + k] + Q[12ull*w + k];
          float px, py, pz;
          (void)sq_place(S, mx, my, mz, &px, &py, &pz);
          verts[3ull*u] = px; verts[3ull*u+1] = py; verts[3ull*u+2] = pz;
        }
        #pragma unroll
        for (int k = 0; k < 10; ++k) Q[12ull*u + k] += Q[12ull*w + k];
        rm[v] = (RMT)ul;
        atomicAdd(&s_collapses, 1u);
      }
      __syncthreads();
    }
    if (s_collapses == 0) break;
    // [8] fused rewrite + stable compact (oracle step 4)
    uint32_t kept = blk_rewrite_compact<BS, WAVEMODE ? 8192 : 65536, RMT>(
        fa, fb, rm, v0, nt, s_sums);
    { uint32_t *t = fa; fa = fb; fb = t; }  // compacted faces now in fa
    if (tid == 0) s_nt = kept;
    __syncthreads();
    PHASE_MARK(5)  // collapse + rewrite + compact
    if (kept == nt) break;  // no face dropped this sub
    }  // sub loop
    if (s_nt == nt_group) break;  // group made no progress: terminate
  }
#undef PHASE_MARK
  if (prof && tid == 0) {
    #pragma unroll
    for (int k = 0; k < 6; ++k)
      if (ph[k]) atomicAdd(&prof[k], ph[k]);
  }
  // park the final faces at the label's original offset
  for (uint32_t i = tid; i < 3 * s_nt; i += BS)
    park_faces[3ull * f0 + i] = fa[i];
  if (tid == 0) {
    nt_cur[b] = s_nt;
    active[b] = 0;  // done: global rounds skip this label
    if (roundhist) {
      // layout: [0..63] group-count hist, [64..127] sub-count hist
      // (capped), [128] sum groups, [129] sum subs, [130] labels seen,
      // [131] sum nt0 (faces entering), [132] max label cycles,
      // [133] nt0 of (a) max-cycle label, [134..159] log2 cycle hist
      unsigned long long cyc = __builtin_amdgcn_s_memtime() - t_start;
      uint32_t ck = (uint32_t)(cyc > 0xFFFFFFFFull ? 0xFFFFFFFFull : cyc);
      uint32_t prev = atomicMax(&roundhist[132], ck);
      if (ck > prev) roundhist[133] = nt0;  // racy but indicative
      int lg = 32 - __clz(ck | 1);          // 1..32
      atomicAdd(&roundhist[134 + (lg < 26 ? lg : 25)], 1u);
      // cycle totals split by LDS vs global-array mode ([160..163] u64x2,
      // [164] lds count, [165] global count — buffer is 176 u32)
      atomicAdd((unsigned long long *)&roundhist[lds_mode ? 160 : 162],
                cyc);
      atomicAdd(&roundhist[164 + (lds_mode ? 0 : 1)], 1u);
      atomicAdd(&roundhist[n_groups < 64 ? n_groups : 63], 1u);
      atomicAdd(&roundhist[64 + (n_subs < 64 ? n_subs : 63)], 1u);
      atomicAdd(&roundhist[128], n_groups);
      atomicAdd(&roundhist[129], n_subs);
      atomicAdd(&roundhist[130], 1u);
      atomicAdd(&roundhist[131], nt0);
    }
  }
}
