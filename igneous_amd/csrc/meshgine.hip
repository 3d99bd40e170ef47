// meshgine.hip — MI355X (gfx950/CDNA4) per-chunk meshing engine.
//
// Implements the C ABI of include/meshgine.h: the compute the reference
// delegates to zmesh's CPU C++ (multi-label marching cubes at
// /root/reference/igneous/tasks/mesh/mesh.py:245, per-label welded extract
// + quadric simplify at mesh.py:374-381) as hand-written HIP kernels.
//
// Canonical contract (bit-exact with oracle/mc_oracle.c — see DESIGN.md):
//   cells in global F-order; per cell distinct non-zero labels in
//   first-seen corner order; triangles from mc_table.h in table order;
//   per-label first-seen vertex welding; vertex position
//   (0.5f*k + shift)*res in f32.
//
// Device pipeline (one mg_mesh_chunk call):
//   H2D labels
//   [1] k_count        wave-per-64-cell-segment triangle counts
//                      (+ label hash build)                    -> segcnt
//   [2] scan           rocPRIM exclusive scan                  -> segoff, T
//   [3] k_emit         recompute counts, wave prefix, write one 16-B
//                      record (3 weld slots + label id) per triangle in
//                      canonical global order (case tables in LDS)
//   [4] partition      rocPRIM stable radix-sort by label id + 16-B gather
//   [5] weld           direct-addressed (edge,side) table: atomicMax(~pos)
//                      first-seen positions (LDS pre-dedup), bit-packed
//                      first-occurrence flags + word-granular scan ->
//                      vertex ids, vertex emit, per-label face indices
//   [6] (optional) per-label quadric simplification (one workgroup per
//                      label; global matched-pair rounds for huge labels)
//   [7] D2H slices + meshset assembly (ctx-owned pinned staging)
//
// Determinism: every kernel's output is a pure function of its inputs —
// atomics are only used for first-position minima (order-free), hash slot
// claims (value-keyed) and the label counter (affects internal ids only;
// triangle partition order is by internal id but per-label content and the
// final label-sorted meshset are invariant).

#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <mutex>
#include <string>
#include <vector>
#include <algorithm>

#include <hip/hip_runtime.h>
#include <rocprim/rocprim.hpp>

#define MC_TABLE_QUAL __device__ static const
#include "mc_table.h"

#include "../../include/meshgine.h"

// ---------------------------------------------------------------------------
// small utilities

#define WAVE 64

static inline uint64_t next_pow2_u64(uint64_t x) {
  if (x < 2) return 2;
  return 1ull << (64 - __builtin_clzll(x - 1));
}

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// wave-wide inclusive prefix sum (64 lanes)
__device__ __forceinline__ uint32_t wave_incl_scan(uint32_t x, int lane) {
  for (int d = 1; d < WAVE; d <<= 1) {
    uint32_t y = __shfl_up(x, d, WAVE);
    if (lane >= d) x += y;
  }
  return x;
}

// ---------------------------------------------------------------------------
// label hash: raw label value -> dense internal id (build in count pass)

struct LabelHash {
  uint64_t *keys;   // 0 = empty (label 0 never inserted)
  uint32_t *vals;
  uint32_t *counter;   // next id
  uint32_t *overflow;  // error flag
  uint64_t nslots;     // power of two
};

__device__ __forceinline__ void label_insert(LabelHash h, uint64_t label) {
  uint64_t slot = mix64(label) & (h.nslots - 1);
  for (uint64_t probe = 0; probe < h.nslots; ++probe) {
    uint64_t cur = h.keys[slot];
    if (cur == label) return;
    if (cur == 0) {
      uint64_t prev = atomicCAS((unsigned long long *)&h.keys[slot], 0ull,
                                (unsigned long long)label);
      if (prev == 0) {
        h.vals[slot] = atomicAdd(h.counter, 1u);
        return;
      }
      if (prev == label) return;
    }
    slot = (slot + 1) & (h.nslots - 1);
  }
  atomicExch(h.overflow, 1u);
}

__device__ __forceinline__ uint32_t label_lookup(LabelHash h, uint64_t label) {
  uint64_t slot = mix64(label) & (h.nslots - 1);
  for (uint64_t probe = 0; probe < h.nslots; ++probe) {
    // ids were published by k_count, which completed before any lookup
    if (h.keys[slot] == label) return h.vals[slot];
    slot = (slot + 1) & (h.nslots - 1);
  }
  atomicExch(h.overflow, 1u);  // bug surfaces as error, not a GPU hang
  return 0;
}

// ---------------------------------------------------------------------------
// device dust removal (MeshTask dust_threshold, reference mesh.py:313-323
// via fastremap.unique + mask): three volume passes — build the label
// hash, histogram voxels per label, zero voxels of labels below the
// threshold. Replaces a multi-second host numpy unique/mask at 512^3.

template <typename T>
__global__ void k_dust_build(const T *__restrict__ labels, uint64_t nvox,
                             LabelHash lh) {
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvox; i += (uint64_t)gridDim.x * blockDim.x) {
    T L = labels[i];
    if (L != 0) label_insert(lh, (uint64_t)L);
  }
}

template <typename T>
__global__ void k_dust_count(const T *__restrict__ labels, uint64_t nvox,
                             LabelHash lh, uint32_t *__restrict__ counts) {
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvox; i += (uint64_t)gridDim.x * blockDim.x) {
    T L = labels[i];
    if (L != 0) atomicAdd(&counts[label_lookup(lh, (uint64_t)L)], 1u);
  }
}

template <typename T>
__global__ void k_dust_zero(T *__restrict__ labels, uint64_t nvox,
                            LabelHash lh,
                            const uint32_t *__restrict__ counts,
                            uint64_t thr) {
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvox; i += (uint64_t)gridDim.x * blockDim.x) {
    T L = labels[i];
    if (L != 0 && (uint64_t)counts[label_lookup(lh, (uint64_t)L)] < thr)
      labels[i] = 0;
  }
}

// invert: label id -> label value (fill after count)
__global__ void k_label_values(LabelHash h, uint64_t *values, uint64_t nslots) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nslots) return;
  uint64_t k = h.keys[i];
  if (k != 0) values[h.vals[i]] = k;
}

// ---------------------------------------------------------------------------
// per-cell triangle enumeration, shared by count and emit
//
// Segment = up to 64 consecutive cells of one x-row (aligned to 64 cells).
// One wave per segment; lane = cell index within the segment.

template <typename T>
__device__ __forceinline__ void load_corners(const T *__restrict__ labels,
                                             int64_t sx, int64_t sxy,
                                             int64_t cx, int64_t cy, int64_t cz,
                                             T c[8]) {
  const T *p = labels + cx + cy * sx + cz * sxy;
  c[0] = p[0];            c[1] = p[1];
  c[2] = p[sx];           c[3] = p[sx + 1];
  c[4] = p[sxy];          c[5] = p[sxy + 1];
  c[6] = p[sxy + sx];     c[7] = p[sxy + sx + 1];
}

template <typename T>
__device__ __forceinline__ uint32_t cell_tri_count(const T c[8],
                                                   const uint8_t *cnt_tab) {
  if (c[0] == c[1] && c[0] == c[2] && c[0] == c[3] && c[0] == c[4] &&
      c[0] == c[5] && c[0] == c[6] && c[0] == c[7])
    return 0;
  uint32_t total = 0;
  #pragma unroll
  for (int i = 0; i < 8; ++i) {
    T L = c[i];
    if (L == 0) continue;
    bool seen = false;
    for (int j = 0; j < i; ++j) seen |= (c[j] == L);
    if (seen) continue;
    unsigned mask = 0;
    #pragma unroll
    for (int j = 0; j < 8; ++j) mask |= (c[j] == L) ? (1u << j) : 0u;
    total += cnt_tab[mask];
  }
  return total;
}

struct GridDims {
  int64_t sx, sy, sz;     // voxel dims
  int64_t ncx, ncy, ncz;  // cell dims
  int64_t nsegx;          // segments per row
  int64_t nseg;           // total segments
};

// XCD-aware segment schedule: block b runs on XCD b%8 (observed CDNA4
// dispatch), so give each XCD one CONTIGUOUS 1/8 slab of the segment
// range — adjacent rows (which share a voxel row) then stream through
// the SAME XCD's L2 instead of being split round-robin across all 8
// (measured 2.1x row re-read from HBM with the naive linear stride).
struct SegSched {
  int64_t beg, end, step;
};

__device__ __forceinline__ SegSched xcd_seg_sched(int64_t nseg,
                                                  int waves_per_blk,
                                                  int wave_in_blk) {
  const int64_t xcd = blockIdx.x & 7;
  const int64_t j = blockIdx.x >> 3;
  const int64_t nbx = (gridDim.x + 7) >> 3;  // blocks per XCD class
  const int64_t slab = (nseg + 7) >> 3;
  SegSched ss;
  ss.beg = xcd * slab + j * waves_per_blk + wave_in_blk;
  ss.end = std::min<int64_t>((xcd + 1) * slab, nseg);
  ss.step = nbx * waves_per_blk;
  return ss;
}

// [1] count: one wave per segment; build label hash; write per-segment count
template <typename T>
__global__ void k_count(const T *__restrict__ labels, GridDims g,
                        uint32_t *__restrict__ segcnt, LabelHash lh) {
  __shared__ uint8_t s_cnt[256];
  for (int k = threadIdx.x; k < 256; k += blockDim.x)
    s_cnt[k] = MC_TRI_COUNT[k];
  __syncthreads();
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave_in_blk = threadIdx.x / WAVE;
  const int waves_per_blk = blockDim.x / WAVE;
  const int64_t sxy = g.sx * g.sy;
  const SegSched ss = xcd_seg_sched(g.nseg, waves_per_blk, wave_in_blk);
  for (int64_t seg = ss.beg; seg < ss.end; seg += ss.step) {
    const int64_t row = seg / g.nsegx;
    const int64_t segx = seg - row * g.nsegx;
    const int64_t cy = row % g.ncy;
    const int64_t cz = row / g.ncy;
    const int64_t cx = segx * WAVE + lane;
    uint32_t cnt = 0;
    if (cx < g.ncx) {
      T c[8];
      load_corners(labels, g.sx, sxy, cx, cy, cz, c);
      if (!(c[0] == c[1] && c[0] == c[2] && c[0] == c[3] && c[0] == c[4] &&
            c[0] == c[5] && c[0] == c[6] && c[0] == c[7])) {
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
          T L = c[i];
          if (L == 0) continue;
          bool seen = false;
          for (int j = 0; j < i; ++j) seen |= (c[j] == L);
          if (seen) continue;
          unsigned mask = 0;
          #pragma unroll
          for (int j = 0; j < 8; ++j) mask |= (c[j] == L) ? (1u << j) : 0u;
          uint32_t nt = s_cnt[mask];
          if (nt) {
            cnt += nt;
            label_insert(lh, (uint64_t)L);
          }
        }
      }
    }
    // wave reduce
    uint32_t incl = wave_incl_scan(cnt, lane);
    if (lane == WAVE - 1) segcnt[seg] = incl;
  }
}

// Compact 8-B triangle record: {cell_lin, mask | (t<<8)}. The three weld
// slots (edge axis, voxel, side) and the doubled vertex coordinates are
// all recomputable from it via the case tables — consumers decode instead
// of re-reading a fat 16-B record (the record stream dominated the MC
// pipeline's HBM traffic at 16 B/tri).
//
// slot = ((cell_lin*3 + comb[e]) << 1) | side, where comb[e] folds the
// edge's voxel offset and axis; side = is the label the edge's UPPER
// endpoint, baked into MC_TRI_PACK at table-gen time. Collision-free: a
// midpoint is a vertex only for its two endpoint labels; 6*nvox < 2^32
// enforced on the host.

// stage the decode tables into LDS (s_pack: 256*MC_MAX_TRIS u16,
// s_comb: 12 u32). Call before __syncthreads().
__device__ __forceinline__ void stage_decode_tables(
    uint16_t *s_pack, uint32_t *s_comb, int64_t sx, int64_t sxy) {
  for (int k = threadIdx.x; k < 256; k += blockDim.x) {
    #pragma unroll
    for (int t = 0; t < MC_MAX_TRIS; ++t)
      s_pack[k * MC_MAX_TRIS + t] = MC_TRI_PACK[k][t];
  }
  if (threadIdx.x < 12) {
    int e = threadIdx.x;
    uint32_t eoff = (uint32_t)(MC_EDGE_DOFF[e][0] >> 1) +
                    (uint32_t)(MC_EDGE_DOFF[e][1] >> 1) * (uint32_t)sx +
                    (uint32_t)(MC_EDGE_DOFF[e][2] >> 1) * (uint32_t)sxy;
    uint32_t axis = (MC_EDGE_DOFF[e][0] & 1)
                        ? 0u
                        : ((MC_EDGE_DOFF[e][1] & 1) ? 1u : 2u);
    s_comb[e] = eoff * 3u + axis;
  }
}

// decode one corner's weld slot (v in 0..2)
__device__ __forceinline__ uint32_t decode_rec_slot1(
    uint2 rec, const uint16_t *s_pack, const uint32_t *s_comb, uint32_t v) {
  const uint32_t pk =
      s_pack[(rec.y & 255u) * MC_MAX_TRIS + (rec.y >> 8)];
  const uint32_t nib = (pk >> (5u * v)) & 31u;
  return ((rec.x * 3u + s_comb[nib & 15u]) << 1) | (nib >> 4);
}

__device__ __forceinline__ void decode_rec_slots(
    uint2 rec, const uint16_t *s_pack, const uint32_t *s_comb,
    uint32_t s[3]) {
  const uint32_t cl3 = rec.x * 3u;
  const uint32_t pk =
      s_pack[(rec.y & 255u) * MC_MAX_TRIS + (rec.y >> 8)];
  s[0] = ((cl3 + s_comb[pk & 15]) << 1) | ((pk >> 4) & 1);
  s[1] = ((cl3 + s_comb[(pk >> 5) & 15]) << 1) | ((pk >> 9) & 1);
  s[2] = ((cl3 + s_comb[(pk >> 10) & 15]) << 1) | ((pk >> 14) & 1);
}

// [3] emit: recompute counts, wave prefix, write one 8-B record + the
// 4-B label id per triangle in canonical global order.
template <typename T>
__global__ void k_emit(const T *__restrict__ labels, GridDims g,
                       const uint32_t *__restrict__ segoff, LabelHash lh,
                       uint32_t *__restrict__ tri_label,
                       uint2 *__restrict__ tri_recs) {
  // count table staged in LDS: global-memory byte gathers on the small
  // case tables were the emit kernel's dominant cost (L1 line replays)
  __shared__ uint8_t s_cnt[256];
  for (int k = threadIdx.x; k < 256; k += blockDim.x)
    s_cnt[k] = MC_TRI_COUNT[k];
  __syncthreads();
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave_in_blk = threadIdx.x / WAVE;
  const int waves_per_blk = blockDim.x / WAVE;
  const int64_t sxy = g.sx * g.sy;
  const SegSched ss = xcd_seg_sched(g.nseg, waves_per_blk, wave_in_blk);
  for (int64_t seg = ss.beg; seg < ss.end; seg += ss.step) {
    const int64_t row = seg / g.nsegx;
    const int64_t segx = seg - row * g.nsegx;
    const int64_t cy = row % g.ncy;
    const int64_t cz = row / g.ncy;
    const int64_t cx = segx * WAVE + lane;
    T c[8];
    uint32_t cnt = 0;
    bool active = false;
    if (cx < g.ncx) {
      load_corners(labels, g.sx, sxy, cx, cy, cz, c);
      cnt = cell_tri_count(c, s_cnt);
      active = cnt > 0;
    }
    uint32_t incl = wave_incl_scan(cnt, lane);
    uint32_t base = segoff[seg] + incl - cnt;
    if (active) {
      uint32_t pos = base;
      const uint32_t cell_lin = (uint32_t)((cz * g.sy + cy) * g.sx + cx);
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        T L = c[i];
        if (L == 0) continue;
        bool seen = false;
        for (int j = 0; j < i; ++j) seen |= (c[j] == L);
        if (seen) continue;
        unsigned mask = 0;
        #pragma unroll
        for (int j = 0; j < 8; ++j) mask |= (c[j] == L) ? (1u << j) : 0u;
        uint32_t nt = s_cnt[mask];
        if (!nt) continue;
        uint32_t lid = label_lookup(lh, (uint64_t)L);
        for (uint32_t t = 0; t < nt; ++t) {
          tri_label[pos] = lid;
          tri_recs[pos] = make_uint2(cell_lin, mask | (t << 8));
          ++pos;
        }
      }
    }
  }
}

// [4c] per-label triangle ranges (labels sorted, every id present)
__global__ void k_label_ranges(const uint32_t *__restrict__ lab_sorted,
                               uint32_t *__restrict__ tri_off,
                               uint64_t ntris, uint32_t nlabels) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= ntris) return;
  if (i == 0) {
    tri_off[lab_sorted[0]] = 0;  // first label's range starts at 0
    tri_off[nlabels] = (uint32_t)ntris;
  } else if (lab_sorted[i] != lab_sorted[i - 1]) {
    tri_off[lab_sorted[i]] = (uint32_t)i;
  }
}

// ---------------------------------------------------------------------------
// welding

// Direct-addressed weld table: an edge midpoint is a vertex only for the
// labels of the edge's two endpoint voxels, so (edge, side) is a
// collision-free address — no hash, no probing, and label-partitioned
// corner streams touch CONTIGUOUS table lines (the reference's spatial
// locality survives). minp_enc stores ~(first position) via atomicMax so
// the table zero-memsets (0 = "no position yet").
//
// slot = axis*nvox + linear_voxel(vx,vy,vz), doubled; axis = the key's
// odd coordinate. Capacity bound: 6*nvox must fit in u32 (checked on the
// host; matches the reference's own 32-bit mesher task bound,
// igneous_cli/cli.py:1049-1052).

// [5a] record first (minimum) stream position per (edge, side).
// One thread per TRIANGLE (reads its 16-B record once). A per-workgroup
// LDS table pre-merges duplicate slots within the block's 768-corner
// window (the label-partitioned stream is spatially coherent, so ~half
// the corners repeat a slot seen moments earlier) — cutting the global
// atomic count roughly in half.
template <int BLK, int TPT>
__global__ __launch_bounds__(BLK) void k_weld_insert(
    const uint2 *__restrict__ tri_recs,   // emit order (8-B records)
    const uint32_t *__restrict__ order,   // label partition permutation
    uint32_t *__restrict__ slots_sorted,  // decoded 3 slots/tri (12 B) —
                                          // downstream weld passes read
                                          // plain slots, no case tables
    uint32_t *__restrict__ wminp,
    int64_t sx, int64_t sxy,
    uint64_t ntris) {
  // LDS table sized for ~BLK*TPT*3 corners at ~0.4 load; wider windows
  // dedup more of a vertex's ~6 corner occurrences before the global
  // atomics (the label-sorted stream is spatially coherent)
  constexpr int WI_LDS_SLOTS = BLK * TPT * 4;
  constexpr int LG = __builtin_ctz(WI_LDS_SLOTS);
  __shared__ uint32_t lkey[WI_LDS_SLOTS];
  __shared__ uint32_t lval[WI_LDS_SLOTS];
  __shared__ uint16_t s_pack[256 * MC_MAX_TRIS];
  __shared__ uint32_t s_comb[12];
  stage_decode_tables(s_pack, s_comb, sx, sxy);
  for (int k = threadIdx.x; k < WI_LDS_SLOTS; k += BLK) {
    lkey[k] = 0;
    lval[k] = 0;
  }
  __syncthreads();
  const uint64_t span0 = (uint64_t)blockIdx.x * BLK * TPT;
  #pragma unroll
  for (int rep = 0; rep < TPT; ++rep) {
    uint64_t t = span0 + (uint64_t)rep * BLK + threadIdx.x;
    if (t >= ntris) break;
    uint2 rec = order ? tri_recs[order[t]] : tri_recs[t];
    uint32_t s[3];
    decode_rec_slots(rec, s_pack, s_comb, s);
    slots_sorted[3 * t] = s[0];
    slots_sorted[3 * t + 1] = s[1];
    slots_sorted[3 * t + 2] = s[2];
    #pragma unroll
    for (int v = 0; v < 3; ++v) {
      uint32_t slot = s[v];
      uint32_t enc = ~(uint32_t)(3 * t + v);
      uint32_t key = slot + 1;
      uint32_t h = (slot * 2654435761u) >> (32 - LG);
      bool placed = false;
      for (int probe = 0; probe < 32; ++probe) {
        uint32_t cur = lkey[h];
        if (cur == 0) cur = atomicCAS(&lkey[h], 0u, key);
        if (cur == 0 || cur == key) {
          atomicMax(&lval[h], enc);
          placed = true;
          break;
        }
        h = (h + 1) & (WI_LDS_SLOTS - 1);
      }
      if (!placed) atomicMax(&wminp[slot], enc);  // rare spill
    }
  }
  __syncthreads();
  for (int k = threadIdx.x; k < WI_LDS_SLOTS; k += BLK) {
    uint32_t key = lkey[k];
    if (key) atomicMax(&wminp[key - 1], lval[k]);
  }
}

// [5b] first-occurrence flags as a bit array: each wave handles 64
// consecutive corners and ballots the flags into one u64 word. Vertex
// ids then come from a word-granular scan + popcount instead of a full
// per-element scan array.
__global__ void k_weld_flag_bits(const uint32_t *__restrict__ slots_sorted,
                                 const uint32_t *__restrict__ wminp,
                                 unsigned long long *__restrict__ bits,
                                 uint64_t ncorners, uint64_t nwords) {
  uint64_t gid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t word = gid / 64;
  if (word >= nwords) return;
  uint32_t lane = (uint32_t)(gid & 63);
  uint64_t i = word * 64 + lane;
  bool flag = false;
  if (i < ncorners)
    flag = (wminp[slots_sorted[i]] == ~(uint32_t)i);
  unsigned long long m = __ballot(flag);
  if (lane == 0) bits[word] = m;
}

struct PopcWord {
  __device__ uint32_t operator()(unsigned long long w) const {
    return (uint32_t)__popcll(w);
  }
};

// vertex id of corner i (must be below its word's scan base + rank)
__device__ __forceinline__ uint32_t vtx_id_of(
    const uint32_t *__restrict__ wscan,
    const unsigned long long *__restrict__ bits, uint64_t i) {
  uint64_t w = i >> 6;
  unsigned long long mask = (1ull << (i & 63)) - 1;
  return wscan[w] + (uint32_t)__popcll(bits[w] & mask);
}

__global__ void k_total_verts(const uint32_t *__restrict__ wscan,
                              const unsigned long long *__restrict__ bits,
                              uint64_t nwords, uint32_t *out) {
  *out = wscan[nwords - 1] + (uint32_t)__popcll(bits[nwords - 1]);
}

// [5d] first occurrences: record vertex id in the table, write the vertex
// (doubled coordinates = 2*cell + edge offset, decoded from the record)
__global__ void k_weld_verts(const uint32_t *__restrict__ slots_sorted,
                             const uint32_t *__restrict__ wscan,
                             const unsigned long long *__restrict__ bits,
                             const uint32_t *__restrict__ wminp,
                             uint32_t *__restrict__ wvtx,
                             float *__restrict__ verts,
                             uint32_t usx, uint32_t usxy,
                             float rx, float ry, float rz, float shift,
                             uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  const uint32_t s[3] = {slots_sorted[3 * t], slots_sorted[3 * t + 1],
                         slots_sorted[3 * t + 2]};
  #pragma unroll
  for (int v = 0; v < 3; ++v) {
    uint32_t i = (uint32_t)(3 * t + v);
    uint32_t slot = s[v];
    if (wminp[slot] != ~i) continue;
    uint32_t vid = vtx_id_of(wscan, bits, i);
    wvtx[slot] = vid;
    // doubled coordinates decoded from the slot (first occurrences
    // only, ~1/6 of corners): slot = (lin*3 + axis)*2 | side
    uint32_t eslot = slot >> 1;
    uint32_t axis = eslot % 3u;
    uint32_t lin = eslot / 3u;
    uint32_t vz = lin / usxy;
    uint32_t rem = lin - vz * usxy;
    uint32_t vy = rem / usx;
    uint32_t vx = rem - vy * usx;
    float dx = (float)(2 * vx + (axis == 0));
    float dy = (float)(2 * vy + (axis == 1));
    float dz = (float)(2 * vz + (axis == 2));
    verts[3ull * vid + 0] = (0.5f * dx + shift) * rx;
    verts[3ull * vid + 1] = (0.5f * dy + shift) * ry;
    verts[3ull * vid + 2] = (0.5f * dz + shift) * rz;
  }
}

// [5e] per-label vertex bases: vbase[l] = vtx_scan at the label's first corner
__global__ void k_vbase(const uint32_t *__restrict__ tri_off,
                        const uint32_t *__restrict__ wscan,
                        const unsigned long long *__restrict__ bits,
                        uint32_t *__restrict__ vbase,
                        uint32_t nlabels, uint64_t total_verts) {
  uint32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l > nlabels) return;
  if (l == nlabels) vbase[l] = (uint32_t)total_verts;
  else vbase[l] = vtx_id_of(wscan, bits, 3ull * tri_off[l]);
}

// [5f] faces: per-label local vertex indices (label id from the sorted
// label array — the partition sort's key output)
__global__ void k_faces(const uint32_t *__restrict__ slots_sorted,
                        const uint32_t *__restrict__ lab_sorted,
                        const uint32_t *__restrict__ wvtx,
                        const uint32_t *__restrict__ vbase,
                        uint32_t *__restrict__ faces,
                        uint64_t ntris) {
  uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ntris) return;
  uint32_t base = vbase[lab_sorted[t]];
  faces[3 * t + 0] = wvtx[slots_sorted[3 * t]] - base;
  faces[3 * t + 1] = wvtx[slots_sorted[3 * t + 1]] - base;
  faces[3 * t + 2] = wvtx[slots_sorted[3 * t + 2]] - base;
}

__global__ void k_iota(uint32_t *p, uint64_t n) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = (uint32_t)i;
}

// out = exclusive_scan[n-1] + input[n-1]  (total of a scanned array)
__global__ void k_last_sum(const uint32_t *scan, const uint32_t *input,
                           uint64_t n, uint32_t *out) {
  *out = scan[n - 1] + input[n - 1];
}

__global__ void k_any_active(const uint8_t *active, uint32_t nlabels,
                             uint32_t *any) {
  uint32_t l = blockIdx.x * blockDim.x + threadIdx.x;
  if (l < nlabels && active[l]) atomicExch(any, 1u);
}

#include "simplify.hip"

// ---------------------------------------------------------------------------
// context / host side

struct DevBuf {
  void *ptr = nullptr;
  size_t cap = 0;
};

struct HostBuf {
  void *ptr = nullptr;
  size_t cap = 0;
};

struct mg_ctx {
  int device = 0;
  hipStream_t stream = nullptr;
  hipStream_t stream2 = nullptr;  // concurrent simplify band
  std::mutex lock;
  std::string err;
  mg_stats stats = {};
  // device buffers (grow-only cache)
  DevBuf labels, segcnt, segoff, scan_tmp,
      lh_keys, lh_vals, lh_misc,
      tri_label, tri_label_alt, order, order_alt, tri_keys, keys_sorted,
      tri_off, sort_tmp,
      wh_keys, wh_vtx, vtx_scan, verts, faces, vbase,
      label_values,
      simp_fq, simp_valid, simp_pk, simp_pk_alt, simp_pv, simp_pv_alt,
      simp_Q, simp_pick, simp_remap, simp_flab, simp_flab_alt,
      simp_faces_alt, simp_verts_alt, simp_vbase_alt, simp_meta, simp_ref,
      simp_keep, simp_keep_scan, simp_park, simp_first, simp_troff_final,
      simp_deg, simp_adj, simp_rh, simp_sched, simp_accept;
  uint64_t lh_slots = 1ull << 20;
  HostBuf h_verts, h_faces;  // pinned output staging, reused across calls
  hipEvent_t ev[16] = {};
};

static thread_local std::string g_err;  // for ctx==NULL failures

#define SET_ERR(ctx, ...)                                     \
  do {                                                        \
    char _buf[512];                                           \
    snprintf(_buf, sizeof(_buf), __VA_ARGS__);                \
    if (ctx) (ctx)->err = _buf; else g_err = _buf;            \
  } while (0)

#define HIP_TRY(ctx, call, retcode)                           \
  do {                                                        \
    hipError_t _e = (call);                                   \
    if (_e != hipSuccess) {                                   \
      SET_ERR(ctx, "%s failed: %s (%s:%d)", #call,            \
              hipGetErrorString(_e), __FILE__, __LINE__);     \
      return retcode;                                         \
    }                                                         \
  } while (0)

static int ensure_host(mg_ctx *c, HostBuf &b, size_t bytes) {
  if (b.cap >= bytes) return 0;
  if (b.ptr) (void)hipHostFree(b.ptr);
  b.ptr = nullptr;
  b.cap = 0;
  size_t want = bytes + bytes / 4;
  if (hipHostMalloc(&b.ptr, want) != hipSuccess) {
    if (hipHostMalloc(&b.ptr, bytes) != hipSuccess) {
      SET_ERR(c, "hipHostMalloc(%zu) failed", bytes);
      return 1;
    }
    b.cap = bytes;
    return 0;
  }
  b.cap = want;
  return 0;
}

static int ensure(mg_ctx *c, DevBuf &b, size_t bytes) {
  if (b.cap >= bytes) return 0;
  if (b.ptr) (void)hipFree(b.ptr);
  b.ptr = nullptr;
  b.cap = 0;
  size_t want = bytes + bytes / 4;  // 25% headroom to damp re-allocs
  hipError_t e = hipMalloc(&b.ptr, want);
  if (e != hipSuccess) {
    e = hipMalloc(&b.ptr, bytes);  // retry exact
    if (e != hipSuccess) {
      SET_ERR(c, "hipMalloc(%zu) failed: %s", bytes, hipGetErrorString(e));
      return 1;
    }
    b.cap = bytes;
    return 0;
  }
  b.cap = want;
  return 0;
}

extern "C" {

const char *mg_version(void) { return "meshgine 0.1.0 (gfx950)"; }

int mg_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

mg_ctx *mg_init(int device_id) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || device_id >= n || n == 0) {
    return nullptr;
  }
  if (hipSetDevice(device_id) != hipSuccess) return nullptr;
  mg_ctx *c = new mg_ctx();
  c->device = device_id;
  if (hipStreamCreateWithFlags(&c->stream, hipStreamNonBlocking) !=
      hipSuccess) {
    delete c;
    return nullptr;
  }
  if (hipStreamCreateWithFlags(&c->stream2, hipStreamNonBlocking) !=
      hipSuccess) {
    delete c;
    return nullptr;
  }
  for (auto &e : c->ev)
    if (hipEventCreate(&e) != hipSuccess) { delete c; return nullptr; }
  return c;
}

void mg_destroy(mg_ctx *c) {
  if (!c) return;
  (void)hipSetDevice(c->device);
  for (auto *b : {&c->labels, &c->segcnt, &c->segoff, &c->scan_tmp,
                  &c->lh_keys, &c->lh_vals, &c->lh_misc, &c->tri_label,
                  &c->tri_label_alt, &c->order, &c->order_alt, &c->tri_keys,
                  &c->keys_sorted, &c->tri_off, &c->sort_tmp, &c->wh_keys,
                  &c->wh_vtx, &c->vtx_scan,
                  &c->verts, &c->faces, &c->vbase, &c->label_values,
                  &c->simp_fq, &c->simp_valid, &c->simp_pk, &c->simp_pk_alt,
                  &c->simp_pv, &c->simp_pv_alt, &c->simp_Q, &c->simp_pick,
                  &c->simp_remap, &c->simp_flab, &c->simp_flab_alt,
                  &c->simp_faces_alt, &c->simp_verts_alt, &c->simp_vbase_alt,
                  &c->simp_meta, &c->simp_ref, &c->simp_keep,
                  &c->simp_keep_scan, &c->simp_park, &c->simp_first,
                  &c->simp_troff_final, &c->simp_deg, &c->simp_adj,
                  &c->simp_rh, &c->simp_sched, &c->simp_accept}) {
    if (b->ptr) (void)hipFree(b->ptr);
  }
  if (c->h_verts.ptr) (void)hipHostFree(c->h_verts.ptr);
  if (c->h_faces.ptr) (void)hipHostFree(c->h_faces.ptr);
  for (auto &e : c->ev) if (e) (void)hipEventDestroy(e);
  if (c->stream2) (void)hipStreamDestroy(c->stream2);
  if (c->stream) (void)hipStreamDestroy(c->stream);
  delete c;
}

const char *mg_last_error(mg_ctx *c) {
  return c ? c->err.c_str() : g_err.c_str();
}

int mg_get_stats(mg_ctx *c, mg_stats *out) {
  if (!c || !out) return 1;
  *out = c->stats;
  return 0;
}

void mg_meshset_free(mg_meshset *ms) {
  // the flat vertex/face storage is ctx-owned (pinned, reused); the
  // meshset is just the descriptor + mesh array
  free(ms);
}

// ---------------------------------------------------------------------------

static int mesh_chunk_impl(mg_ctx *c, const void *labels_host,
                           int sx, int sy, int sz, int dtype,
                           float rx, float ry, float rz,
                           uint32_t reduction_factor, float max_error,
                           int voxel_centered, uint64_t dust_threshold,
                           uint32_t flags_,
                           mg_meshset **out);

int mg_mesh_chunk(mg_ctx *c, const void *labels, int sx, int sy, int sz,
                  int dtype, float rx, float ry, float rz,
                  uint32_t reduction_factor, float max_error,
                  int voxel_centered, uint64_t dust_threshold,
                  uint32_t flags, mg_meshset **out) {
  if (!c) { SET_ERR(c, "null ctx"); return 1; }
  std::lock_guard<std::mutex> g(c->lock);
  c->err.clear();
  if (!labels || !out) { SET_ERR(c, "null argument"); return 1; }
  if (sx < 1 || sy < 1 || sz < 1 || sx > 2047 || sy > 2047 || sz > 2047) {
    SET_ERR(c, "dims out of range (1..2047): %d %d %d", sx, sy, sz);
    return 2;
  }
  if (dtype != MG_U32 && dtype != MG_U64) {
    SET_ERR(c, "bad dtype %d", dtype);
    return 3;
  }
  HIP_TRY(c, hipSetDevice(c->device), 4);
  return mesh_chunk_impl(c, labels, sx, sy, sz, dtype, rx, ry, rz,
                         reduction_factor, max_error, voxel_centered,
                         dust_threshold, flags, out);
}

}  // extern "C"

template <typename T>
static int run_count_emit(mg_ctx *c, const T *d_labels, const GridDims &g,
                          LabelHash lh, uint32_t *d_segcnt,
                          uint32_t *d_segoff, uint64_t *p_total,
                          uint32_t *p_nlabels);

static double ev_ms(mg_ctx *c, int a, int b) {
  float ms = 0.f;
  if (hipEventElapsedTime(&ms, c->ev[a], c->ev[b]) != hipSuccess) return 0.0;
  return (double)ms;
}


// ---------------------------------------------------------------------------
// GPU quadric simplification driver (kernels in simplify.hip); mirrors
// oracle/simplify.c round for round. Updates faces/verts/vbase/tri_off
// in the ctx; p_T/p_V become the post-simplification totals.
static int run_simplify(mg_ctx *c, uint32_t nlabels,
                        const uint32_t *lab_sorted,
                        uint32_t reduction_factor, float max_error,
                        uint64_t *p_T, uint64_t *p_V) {
  hipStream_t s = c->stream;
  uint64_t T = *p_T;
  const uint64_t V = *p_V;
  const float max_cost = max_error * max_error;
  const int blk = 256;
  if (T == 0 || V == 0) return 0;

  // meta layout: [0,L) nt_cur | [L,2L) target | [2L,3L) nt_new |
  //              3L..: active u8[L] | any u32 (aligned)
  const uint64_t L = nlabels;
  const uint64_t meta_bytes = L * 12 + ((L + 3) & ~3ull) + 4;
  if (ensure(c, c->simp_meta, meta_bytes)) return 40;
  uint32_t *nt_cur = (uint32_t *)c->simp_meta.ptr;
  uint32_t *target = nt_cur + L;
  uint32_t *nt_new = target + L;
  uint8_t *active = (uint8_t *)(nt_new + L);
  uint32_t *d_any = (uint32_t *)((char *)c->simp_meta.ptr +
                                 L * 12 + ((L + 3) & ~3ull));

  if (ensure(c, c->simp_flab, T * 4)) return 40;
  if (ensure(c, c->simp_flab_alt, T * 4)) return 40;
  if (ensure(c, c->simp_faces_alt, T * 12)) return 40;
  if (ensure(c, c->simp_fq, T * sizeof(SimpPlane))) return 40;
  if (ensure(c, c->simp_valid, T)) return 40;
  if (ensure(c, c->simp_pk, 3 * T * 4)) return 40;
  if (ensure(c, c->simp_pk_alt, 3 * T * 4)) return 40;
  if (ensure(c, c->simp_pv, 3 * T * 4)) return 40;
  if (ensure(c, c->simp_pv_alt, 3 * T * 4)) return 40;
  if (ensure(c, c->simp_Q, V * 48)) return 40;  // 12-float rows
  if (ensure(c, c->simp_pick, V * 8)) return 40;
  if (ensure(c, c->simp_remap, V * 4)) return 40;
  if (ensure(c, c->simp_keep, T * 4)) return 40;
  if (ensure(c, c->simp_keep_scan, T * 4)) return 40;
  if (ensure(c, c->simp_park, 3 * T * 4)) return 40;
  if (ensure(c, c->simp_first, (L + 1) * 4)) return 40;
  if (ensure(c, c->simp_troff_final, (L + 1) * 4)) return 40;
  if (ensure(c, c->simp_deg, V * 4)) return 40;
  if (ensure(c, c->simp_adj, V * 4)) return 40;
  if (ensure(c, c->simp_accept, V * 4)) return 40;
  // proposal-acceptance second matching wave (contract knob shared
  // with the oracle; MG_SIMP_PROPOSE=0 disables on both sides)
  uint32_t propose = 1;
  {
    const char *e = getenv("MG_SIMP_PROPOSE");
    if (e && e[0] == '0') propose = 0;
  }

  uint32_t *faces_g = (uint32_t *)c->faces.ptr;
  float *verts = (float *)c->verts.ptr;

  // faces -> global vertex ids, label per face
  {
    uint64_t nbt = (T + blk - 1) / blk;
    hipLaunchKernelGGL(k_globalize_faces, dim3((uint32_t)nbt), dim3(blk), 0,
                       s, faces_g, lab_sorted,
                       (const uint32_t *)c->vbase.ptr,
                       (uint32_t *)c->simp_flab.ptr, T);
    uint32_t nbl = (uint32_t)((L + 255) / 256);
    hipLaunchKernelGGL(k_init_simplify, dim3(nbl), dim3(256), 0, s,
                       (const uint32_t *)c->tri_off.ptr, nt_cur, target,
                       active, reduction_factor, (uint32_t)L);
  }
  HIP_TRY(c, hipGetLastError(), 40);

  // labels up to SIMP_BIG_CAP faces simplify entirely inside one
  // workgroup each (cache-resident round loop, no global sorts); only
  // bigger labels go through the global-rounds machinery below.
  const uint32_t SIMP_BIG_CAP = 65536;
  // sub-rounds per quadric recompute (contract constant; the oracle
  // reads the same env knob, default 6 on both sides)
  uint32_t simp_subs = 6;
  {
    const char *e = getenv("MG_SIMP_SUBS");
    if (e && e[0]) { int v = atoi(e); simp_subs = v < 1 ? 1u : (uint32_t)v; }
  }
  if (getenv("MG_SIMP_PROF"))
    HIP_TRY(c, hipMemsetAsync((unsigned long long *)c->lh_misc.ptr + 8, 0,
                              48, s), 40);
  // round-count histogram (MG_SIMP_ROUNDHIST=1): groups/subs per label
  uint32_t *d_rh = nullptr;
  if (getenv("MG_SIMP_ROUNDHIST")) {
    if (ensure(c, c->simp_rh, 176 * 4)) return 40;
    d_rh = (uint32_t *)c->simp_rh.ptr;
    HIP_TRY(c, hipMemsetAsync(d_rh, 0, 176 * 4, s), 40);
  }
  // biggest-label-first dispatch order (MG_SIMP_SCHED=1 enables;
  // measured 8% WORSE on the 512^3/50k config: the front-loaded big
  // labels all run the global-array fallback path and thrash when
  // concentrated, and no straggler tail exists to begin with — per-label
  // cycle histogram shows max 5.7 ms vs 61 ms wall)
  uint32_t *d_sched = nullptr;
  {
    const char *e = getenv("MG_SIMP_SCHED");
    if (e && e[0] == '1') {
      if (ensure(c, c->simp_sched, 3 * L * 4)) return 40;
      uint32_t *iota = (uint32_t *)c->simp_sched.ptr;
      uint32_t *keys_out = iota + L;
      d_sched = keys_out + L;
      uint32_t nbl = (uint32_t)((L + 255) / 256);
      hipLaunchKernelGGL(k_iota, dim3(nbl), dim3(256), 0, s, iota, L);
      size_t tmp = 0;
      hipError_t err = rocprim::radix_sort_pairs_desc(
          nullptr, tmp, nt_cur, keys_out, iota, d_sched, L, 0u, 32u, s);
      if (err != hipSuccess) { SET_ERR(c, "sched sort size"); return 40; }
      if (ensure(c, c->sort_tmp, tmp)) return 40;
      err = rocprim::radix_sort_pairs_desc(
          c->sort_tmp.ptr, tmp, nt_cur, keys_out, iota, d_sched, L, 0u,
          32u, s);
      if (err != hipSuccess) { SET_ERR(c, "sched sort"); return 40; }
    }
  }
  {
    // default OFF since sub-round groups: the +24 KB LDS CSR payload
    // costs blocks/CU and the recompute phases it serves are now ~25%
    const char *clenv = getenv("MG_SIMP_CLLDS");  // "1" enables
    const bool use_cl = (clenv && clenv[0] == '1');
    // block size: 256 default; MG_SIMP_BS in {128,256,512} for A/B
    int bs = 256;
    {
      const char *e = getenv("MG_SIMP_BS");
      if (e && e[0]) { int v = atoi(e); if (v==128||v==256||v==512) bs = v; }
    }
    decltype(&k_simplify_label<false, 256, 2048>) ksl;
    if (bs == 128)
      ksl = use_cl ? k_simplify_label<true, 128, 2048>
                   : k_simplify_label<false, 128, 2048>;
    else if (bs == 512)
      ksl = use_cl ? k_simplify_label<true, 512, 2048>
                   : k_simplify_label<false, 512, 2048>;
    else
      ksl = use_cl ? k_simplify_label<true, 256, 2048>
                   : k_simplify_label<false, 256, 2048>;
    // single-wave small-LDS variant takes labels with nv <= small_cap
    // (~8.3 KB LDS -> ~19 blocks/CU, intra-wave barriers); the default
    // variant takes the rest. Measured ~1% WORSE in-box (the serialized
    // second launch eats the occupancy gain) -> default off;
    // MG_SIMP_SMALL=1 enables for experiments.
    const char *sm = getenv("MG_SIMP_SMALL");
    const uint32_t small_cap = (sm && sm[0] == '1') ? 512u : 0u;
    auto launch_band = [&](auto fn, hipStream_t st, int bsz,
                           uint32_t nv_lo, uint32_t nv_hi,
                           uint32_t nt_lo = 0u,
                           uint32_t nt_hi = 0xFFFFFFFFu) {
      hipLaunchKernelGGL(fn, dim3((uint32_t)L), dim3(bsz), 0, st,
                         faces_g, (uint32_t *)c->simp_faces_alt.ptr,
                         (const uint32_t *)c->tri_off.ptr,
                         (const uint32_t *)c->vbase.ptr,
                         verts, (float *)c->simp_Q.ptr,
                         (unsigned long long *)c->simp_pick.ptr,
                         (uint32_t *)c->simp_remap.ptr,
                         (uint32_t *)c->simp_deg.ptr,
                         (uint32_t *)c->simp_adj.ptr,
                         (uint32_t *)c->simp_pk.ptr,
                         (SimpPlane *)c->simp_fq.ptr,
                         (uint8_t *)c->simp_valid.ptr,
                         nt_cur, target, active,
                         (uint32_t *)c->simp_park.ptr,
                         getenv("MG_SIMP_PROF")
                             ? (unsigned long long *)c->lh_misc.ptr + 8
                             : nullptr,
                         d_rh,
                         max_cost, (uint32_t)L, SIMP_BIG_CAP, simp_subs,
                         nv_lo, nv_hi, nt_lo, nt_hi,
                         (uint32_t *)c->simp_accept.ptr,
                         propose, d_sched);
    };
    if (small_cap)
      launch_band(k_simplify_label<false, 64, 512>, s, 64, 0u, small_cap);
    // nv-banded dispatch (MG_SIMP_BANDS=0 disables): labels with
    // 2048 < nv <= 4096 get their own CAPV=4096 instantiation (64 KB
    // LDS, 2 blocks/CU) on a SECOND stream, concurrent with the main
    // band — previously they fell into the global-array mode whose
    // hot per-vertex atomics serialize on L2 lines
    const char *bandsenv = getenv("MG_SIMP_BANDS");
    const bool bands = !(bandsenv && bandsenv[0] == '0') && !use_cl;
    if (bands) {
      // record BEFORE enqueuing band A: stream2 must wait only for the
      // shared setup, so the other bands run CONCURRENT with A
      // (recording after A serialized them — 51 ms vs 38 ms overlapped)
      HIP_TRY(c, hipEventRecord(c->ev[9], s), 40);
      HIP_TRY(c, hipStreamWaitEvent(c->stream2, c->ev[9], 0), 40);
      // WAVEMODE band (MG_SIMP_WAVE=1 enables; default OFF — measured
      // 19% WORSE on 512^3/50k: 7 labels/CU resident vs 3, but one
      // 64-lane wave pays 4x the per-pass iterations and its in-flight
      // load window can't cover the longer chains): one wave per label
      // for the small-label mass (nt<=8192, nv<=2048), 20 KB LDS/label
      const char *wenv = getenv("MG_SIMP_WAVE");
      const bool wave = (wenv && wenv[0] == '1');
      if (wave) {
        launch_band(k_simplify_label<false, 64, 2048, true>, s, 64,
                    small_cap, 2048u, 0u, 8192u);
        launch_band(ksl, c->stream2, bs, small_cap, 2048u,
                    8192u, 0xFFFFFFFFu);
      } else {
        launch_band(ksl, s, bs, small_cap, 2048u);
      }
      launch_band(k_simplify_label<false, 256, 4096>, c->stream2, 256,
                  2048u, 4096u);
      launch_band(ksl, c->stream2, bs, 4096u, 0xFFFFFFFFu);
      HIP_TRY(c, hipEventRecord(c->ev[10], c->stream2), 40);
      HIP_TRY(c, hipStreamWaitEvent(s, c->ev[10], 0), 40);
    } else {
      launch_band(ksl, s, bs, small_cap, 0xFFFFFFFFu);
    }
    HIP_TRY(c, hipGetLastError(), 40);
    if (getenv("MG_SIMP_PROF")) {
      unsigned long long hp[6];
      HIP_TRY(c, hipMemcpyAsync(hp, (unsigned long long *)c->lh_misc.ptr + 8,
                                48, hipMemcpyDeviceToHost, s), 40);
      HIP_TRY(c, hipStreamSynchronize(s), 40);
      const char *nm[6] = {"loophead", "planes+deg", "scan+fill",
                           "sortaccum", "picks", "collapse+compact"};
      fprintf(stderr, "[mg simp phases, summed tid0 cycles]");
      for (int k = 0; k < 6; ++k)
        fprintf(stderr, " %s=%llu", nm[k], hp[k]);
      fprintf(stderr, "\n");
    }
    if (d_rh) {
      uint32_t h[176];
      HIP_TRY(c, hipMemcpyAsync(h, d_rh, 176 * 4, hipMemcpyDeviceToHost, s),
              40);
      HIP_TRY(c, hipStreamSynchronize(s), 40);
      fprintf(stderr, "[mg simp rounds] labels=%u sum_groups=%u "
              "sum_subs=%u sum_nt0=%u max_cycles=%u (nt0=%u)\n",
              h[130], h[128], h[129], h[131], h[132], h[133]);
      {
        unsigned long long lds_c, glob_c;
        memcpy(&lds_c, &h[160], 8); memcpy(&glob_c, &h[162], 8);
        fprintf(stderr, "[mg simp mode split] lds: n=%u cyc=%llu | "
                "global: n=%u cyc=%llu\n", h[164], lds_c, h[165], glob_c);
      }
      fprintf(stderr, "[mg simp log2-cycle hist]");
      for (int k = 0; k < 26; ++k)
        if (h[134 + k]) fprintf(stderr, " %d:%u", k, h[134 + k]);
      fprintf(stderr, "\n");
      fprintf(stderr, "[mg simp group hist]");
      for (int k = 0; k < 64; ++k)
        if (h[k]) fprintf(stderr, " %d:%u", k, h[k]);
      fprintf(stderr, "\n[mg simp subs hist]");
      for (int k = 0; k < 64; ++k)
        if (h[64 + k]) fprintf(stderr, " %d:%u", k, h[64 + k]);
      fprintf(stderr, "\n");
    }
  }
  // drop the per-label-kernel labels from the working set; park big
  // never-active labels (their final faces are their originals)
  {
    uint64_t nbt2 = (T + blk - 1) / blk;
    uint32_t *flab2 = (uint32_t *)c->simp_flab.ptr;
    hipLaunchKernelGGL(k_flag_active_faces, dim3((uint32_t)nbt2), dim3(blk),
                       0, s, flab2, active, (uint32_t *)c->simp_keep.ptr, T);
    size_t tmp = 0;
    hipError_t e = rocprim::exclusive_scan(
        nullptr, tmp, (uint32_t *)c->simp_keep.ptr,
        (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "prepark scan size"); return 50; }
    if (ensure(c, c->scan_tmp, tmp)) return 50;
    e = rocprim::exclusive_scan(
        c->scan_tmp.ptr, tmp, (uint32_t *)c->simp_keep.ptr,
        (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "prepark scan"); return 50; }
    hipLaunchKernelGGL(k_first_label_idx, dim3((uint32_t)nbt2), dim3(blk),
                       0, s, flab2, (uint32_t *)c->simp_first.ptr, T);
    hipLaunchKernelGGL(k_park_scatter, dim3((uint32_t)nbt2), dim3(blk), 0,
                       s, faces_g, flab2,
                       (const uint32_t *)c->simp_keep.ptr,
                       (const uint32_t *)c->simp_keep_scan.ptr,
                       (const uint32_t *)c->simp_first.ptr,
                       (const uint32_t *)c->tri_off.ptr,
                       (uint32_t *)c->simp_faces_alt.ptr,
                       (uint32_t *)c->simp_flab_alt.ptr,
                       (uint32_t *)c->simp_park.ptr, SIMP_BIG_CAP, T);
    hipLaunchKernelGGL(k_last_sum, dim3(1), dim3(1), 0, s,
                       (const uint32_t *)c->simp_keep_scan.ptr,
                       (const uint32_t *)c->simp_keep.ptr, T,
                       (uint32_t *)c->lh_misc.ptr + 4);
    uint32_t act_total = 0;
    HIP_TRY(c, hipMemcpyAsync(&act_total, (uint32_t *)c->lh_misc.ptr + 4,
                              4, hipMemcpyDeviceToHost, s), 50);
    HIP_TRY(c, hipStreamSynchronize(s), 50);
    std::swap(c->faces, c->simp_faces_alt);
    std::swap(c->simp_flab, c->simp_flab_alt);
    faces_g = (uint32_t *)c->faces.ptr;
    T = act_total;
  }

  bool simp_done = false;
  for (int group = 0; group < 65536 && !simp_done; ++group) {
    // any label still active?
    HIP_TRY(c, hipMemsetAsync(d_any, 0, 4, s), 41);
    {
      uint32_t nbl = (uint32_t)((L + 255) / 256);
      hipLaunchKernelGGL(k_any_active, dim3(nbl), dim3(256), 0, s,
                         active, (uint32_t)L, d_any);
    }
    uint32_t any = 0;
    HIP_TRY(c, hipMemcpyAsync(&any, d_any, 4, hipMemcpyDeviceToHost, s), 41);
    HIP_TRY(c, hipStreamSynchronize(s), 41);
    if (!any) break;

    uint64_t nbt = (T + blk - 1) / blk;
    uint64_t NP = 3 * T;
    uint64_t nbp = (NP + blk - 1) / blk;
    uint32_t *flab = (uint32_t *)c->simp_flab.ptr;

    hipLaunchKernelGGL(k_face_planes, dim3((uint32_t)nbt), dim3(blk), 0, s,
                       faces_g, verts, active, flab,
                       (SimpPlane *)c->simp_fq.ptr,
                       (uint8_t *)c->simp_valid.ptr, T);
    hipLaunchKernelGGL(k_emit_vf_pairs, dim3((uint32_t)nbt), dim3(blk), 0, s,
                       faces_g, active, flab,
                       (uint32_t *)c->simp_pk.ptr,
                       (uint32_t *)c->simp_pv.ptr, T);
    // stable sort pairs by vertex (face order preserved within vertex)
    {
      rocprim::double_buffer<uint32_t> dk((uint32_t *)c->simp_pk.ptr,
                                          (uint32_t *)c->simp_pk_alt.ptr);
      rocprim::double_buffer<uint32_t> dv((uint32_t *)c->simp_pv.ptr,
                                          (uint32_t *)c->simp_pv_alt.ptr);
      size_t tmp = 0;
      hipError_t e = rocprim::radix_sort_pairs(nullptr, tmp, dk, dv, NP,
                                               0u, 32u, s);
      if (e != hipSuccess) { SET_ERR(c, "simplify sort size query"); return 42; }
      if (ensure(c, c->sort_tmp, tmp)) return 42;
      e = rocprim::radix_sort_pairs(c->sort_tmp.ptr, tmp, dk, dv, NP,
                                    0u, 32u, s);
      if (e != hipSuccess) { SET_ERR(c, "simplify sort"); return 42; }
      hipLaunchKernelGGL(k_accum_quadrics, dim3((uint32_t)nbp), dim3(blk), 0,
                         s, dk.current(), dv.current(),
                         (const SimpPlane *)c->simp_fq.ptr,
                         (const uint8_t *)c->simp_valid.ptr,
                         (float *)c->simp_Q.ptr, NP);
    }
    // sub-rounds: reuse the group's quadrics with GH merging (the same
    // group structure as the per-label kernel and the oracle). Measured
    // on 512^3/200 big labels: subs>1 is ~5% SLOWER here (the T- and
    // V-sized sub passes dwarf the recompute they skip, unlike the
    // per-label kernel) -> the contract pins big labels at 1 sub/group.
    const uint32_t subs_global = 1;
    for (uint32_t sub = 0; sub < subs_global; ++sub) {
    uint64_t nbs = (T + blk - 1) / blk;  // T shrinks between subs
    HIP_TRY(c, hipMemsetAsync(c->simp_pick.ptr, 0xFF, V * 8, s), 43);
    if (propose)
      HIP_TRY(c, hipMemsetAsync(c->simp_accept.ptr, 0xFF, V * 4, s), 43);
    hipLaunchKernelGGL(k_edge_pick, dim3((uint32_t)nbs), dim3(blk), 0, s,
                       faces_g, active, flab,
                       (const uint32_t *)c->vbase.ptr, verts,
                       (const float *)c->simp_Q.ptr,
                       (unsigned long long *)c->simp_pick.ptr, max_cost, T);
    if (group == 0 && sub == 0 && getenv("MG_DEBUG_SIMPLIFY")) {
      float hq[40]; unsigned long long hp[8]; uint32_t hf[12];
      (void)hipMemcpyAsync(hq, c->simp_Q.ptr, sizeof(hq), hipMemcpyDeviceToHost, s);
      (void)hipMemcpyAsync(hp, c->simp_pick.ptr, sizeof(hp), hipMemcpyDeviceToHost, s);
      (void)hipMemcpyAsync(hf, faces_g, sizeof(hf), hipMemcpyDeviceToHost, s);
      (void)hipStreamSynchronize(s);
      fprintf(stderr, "[mg] faces0-3:");
      for (int k = 0; k < 12; ++k) fprintf(stderr, " %u", hf[k]);
      fprintf(stderr, "\n[mg] Q0: ");
      for (int k = 0; k < 10; ++k) fprintf(stderr, "%a ", hq[k]);
      fprintf(stderr, "\n[mg] Q1: ");
      for (int k = 10; k < 20; ++k) fprintf(stderr, "%a ", hq[k]);
      fprintf(stderr, "\n[mg] pick0-7:");
      for (int k = 0; k < 8; ++k) fprintf(stderr, " %llx", hp[k]);
      fprintf(stderr, "\n");
    }
    {
      uint64_t nbv = (V + blk - 1) / blk;
      hipLaunchKernelGGL(k_iota, dim3((uint32_t)nbv), dim3(blk), 0, s,
                         (uint32_t *)c->simp_remap.ptr, V);
      hipLaunchKernelGGL(k_collapse, dim3((uint32_t)nbv), dim3(blk), 0, s,
                         (unsigned long long *)c->simp_pick.ptr, verts,
                         (uint32_t *)c->simp_remap.ptr,
                         (float *)c->simp_Q.ptr, V);
      if (propose) {
        hipLaunchKernelGGL(k_propose, dim3((uint32_t)nbv), dim3(blk), 0, s,
                           (const unsigned long long *)c->simp_pick.ptr,
                           (uint32_t *)c->simp_accept.ptr, V);
        hipLaunchKernelGGL(k_accept, dim3((uint32_t)nbv), dim3(blk), 0, s,
                           (const unsigned long long *)c->simp_pick.ptr,
                           (const uint32_t *)c->simp_accept.ptr,
                           verts, (uint32_t *)c->simp_remap.ptr,
                           (float *)c->simp_Q.ptr, V);
      }
    }
    hipLaunchKernelGGL(k_remap_faces, dim3((uint32_t)nbs), dim3(blk), 0, s,
                       faces_g, (const uint32_t *)c->simp_remap.ptr,
                       (uint32_t *)c->simp_keep.ptr, T);
    {
      size_t tmp = 0;
      hipError_t e = rocprim::exclusive_scan(
          nullptr, tmp, (uint32_t *)c->simp_keep.ptr,
          (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
          rocprim::plus<uint32_t>(), s);
      if (e != hipSuccess) { SET_ERR(c, "keep scan size query"); return 44; }
      if (ensure(c, c->scan_tmp, tmp)) return 44;
      e = rocprim::exclusive_scan(
          c->scan_tmp.ptr, tmp, (uint32_t *)c->simp_keep.ptr,
          (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
          rocprim::plus<uint32_t>(), s);
      if (e != hipSuccess) { SET_ERR(c, "keep scan"); return 44; }
    }
    hipLaunchKernelGGL(k_last_sum, dim3(1), dim3(1), 0, s,
                       (const uint32_t *)c->simp_keep_scan.ptr,
                       (const uint32_t *)c->simp_keep.ptr, T,
                       (uint32_t *)c->lh_misc.ptr + 4);
    uint32_t kept = 0;
    HIP_TRY(c, hipMemcpyAsync(&kept, (uint32_t *)c->lh_misc.ptr + 4, 4,
                              hipMemcpyDeviceToHost, s), 44);
    HIP_TRY(c, hipMemsetAsync(nt_new, 0, L * 4, s), 44);
    hipLaunchKernelGGL(k_compact_faces, dim3((uint32_t)nbs), dim3(blk), 0, s,
                       faces_g, flab, (const uint32_t *)c->simp_keep.ptr,
                       (const uint32_t *)c->simp_keep_scan.ptr,
                       (uint32_t *)c->simp_faces_alt.ptr,
                       (uint32_t *)c->simp_flab_alt.ptr, nt_new, T);
    {
      uint32_t nbl = (uint32_t)((L + 255) / 256);
      hipLaunchKernelGGL(k_update_active, dim3(nbl), dim3(256), 0, s,
                         nt_new, nt_cur, target, active, d_any, (uint32_t)L,
                         sub == 0 ? 1u : 0u);
    }
    HIP_TRY(c, hipGetLastError(), 44);
    HIP_TRY(c, hipStreamSynchronize(s), 44);
    std::swap(c->faces, c->simp_faces_alt);
    std::swap(c->simp_flab, c->simp_flab_alt);
    faces_g = (uint32_t *)c->faces.ptr;
    T = kept;
    if (T == 0) { simp_done = true; break; }

    // park faces of labels that just went inactive: working set keeps
    // only active labels (late rounds touch only the active tail)
    {
      uint64_t nbt2 = (T + blk - 1) / blk;
      uint32_t *flab2 = (uint32_t *)c->simp_flab.ptr;
      hipLaunchKernelGGL(k_flag_active_faces, dim3((uint32_t)nbt2),
                         dim3(blk), 0, s, flab2, active,
                         (uint32_t *)c->simp_keep.ptr, T);
      size_t tmp = 0;
      hipError_t e = rocprim::exclusive_scan(
          nullptr, tmp, (uint32_t *)c->simp_keep.ptr,
          (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
          rocprim::plus<uint32_t>(), s);
      if (e != hipSuccess) { SET_ERR(c, "park scan size query"); return 47; }
      if (ensure(c, c->scan_tmp, tmp)) return 47;
      e = rocprim::exclusive_scan(
          c->scan_tmp.ptr, tmp, (uint32_t *)c->simp_keep.ptr,
          (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
          rocprim::plus<uint32_t>(), s);
      if (e != hipSuccess) { SET_ERR(c, "park scan"); return 47; }
      hipLaunchKernelGGL(k_first_label_idx, dim3((uint32_t)nbt2), dim3(blk),
                         0, s, flab2, (uint32_t *)c->simp_first.ptr, T);
      hipLaunchKernelGGL(k_park_scatter, dim3((uint32_t)nbt2), dim3(blk), 0,
                         s, faces_g, flab2,
                         (const uint32_t *)c->simp_keep.ptr,
                         (const uint32_t *)c->simp_keep_scan.ptr,
                         (const uint32_t *)c->simp_first.ptr,
                         (const uint32_t *)c->tri_off.ptr,
                         (uint32_t *)c->simp_faces_alt.ptr,
                         (uint32_t *)c->simp_flab_alt.ptr,
                         (uint32_t *)c->simp_park.ptr, SIMP_BIG_CAP, T);
      hipLaunchKernelGGL(k_last_sum, dim3(1), dim3(1), 0, s,
                         (const uint32_t *)c->simp_keep_scan.ptr,
                         (const uint32_t *)c->simp_keep.ptr, T,
                         (uint32_t *)c->lh_misc.ptr + 4);
      uint32_t act_total = 0;
      HIP_TRY(c, hipMemcpyAsync(&act_total, (uint32_t *)c->lh_misc.ptr + 4,
                                4, hipMemcpyDeviceToHost, s), 47);
      HIP_TRY(c, hipStreamSynchronize(s), 47);
      std::swap(c->faces, c->simp_faces_alt);
      std::swap(c->simp_flab, c->simp_flab_alt);
      faces_g = (uint32_t *)c->faces.ptr;
      T = act_total;
      if (T == 0) break;
    }
    }  // sub loop
  }

  // any faces still in the working set (round-cap exit): force-park them
  if (T > 0) {
    HIP_TRY(c, hipMemsetAsync(active, 0, L, s), 48);
    uint64_t nbt2 = (T + blk - 1) / blk;
    uint32_t *flab2 = (uint32_t *)c->simp_flab.ptr;
    hipLaunchKernelGGL(k_flag_active_faces, dim3((uint32_t)nbt2), dim3(blk),
                       0, s, flab2, active, (uint32_t *)c->simp_keep.ptr, T);
    size_t tmp = 0;
    rocprim::exclusive_scan(nullptr, tmp, (uint32_t *)c->simp_keep.ptr,
                            (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
                            rocprim::plus<uint32_t>(), s);
    if (ensure(c, c->scan_tmp, tmp)) return 48;
    rocprim::exclusive_scan(c->scan_tmp.ptr, tmp,
                            (uint32_t *)c->simp_keep.ptr,
                            (uint32_t *)c->simp_keep_scan.ptr, 0u, T,
                            rocprim::plus<uint32_t>(), s);
    hipLaunchKernelGGL(k_first_label_idx, dim3((uint32_t)nbt2), dim3(blk),
                       0, s, flab2, (uint32_t *)c->simp_first.ptr, T);
    hipLaunchKernelGGL(k_park_scatter, dim3((uint32_t)nbt2), dim3(blk), 0,
                       s, faces_g, flab2,
                       (const uint32_t *)c->simp_keep.ptr,
                       (const uint32_t *)c->simp_keep_scan.ptr,
                       (const uint32_t *)c->simp_first.ptr,
                       (const uint32_t *)c->tri_off.ptr,
                       (uint32_t *)c->simp_faces_alt.ptr,
                       (uint32_t *)c->simp_flab_alt.ptr,
                       (uint32_t *)c->simp_park.ptr, SIMP_BIG_CAP, T);
    HIP_TRY(c, hipGetLastError(), 48);
  }

  // final face array: per-label slices from the park store, lid order
  {
    size_t tmp = 0;
    hipError_t e = rocprim::exclusive_scan(
        nullptr, tmp, nt_cur, (uint32_t *)c->simp_troff_final.ptr, 0u, L,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "final troff scan size"); return 49; }
    if (ensure(c, c->scan_tmp, tmp)) return 49;
    e = rocprim::exclusive_scan(
        c->scan_tmp.ptr, tmp, nt_cur, (uint32_t *)c->simp_troff_final.ptr,
        0u, L, rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "final troff scan"); return 49; }
    hipLaunchKernelGGL(k_last_sum, dim3(1), dim3(1), 0, s,
                       (const uint32_t *)c->simp_troff_final.ptr, nt_cur, L,
                       (uint32_t *)c->lh_misc.ptr + 6);
    HIP_TRY(c, hipMemcpyAsync((uint32_t *)c->simp_troff_final.ptr + L,
                              (uint32_t *)c->lh_misc.ptr + 6, 4,
                              hipMemcpyDeviceToDevice, s), 49);
    uint32_t final_total = 0;
    HIP_TRY(c, hipMemcpyAsync(&final_total, (uint32_t *)c->lh_misc.ptr + 6,
                              4, hipMemcpyDeviceToHost, s), 49);
    HIP_TRY(c, hipStreamSynchronize(s), 49);
    T = final_total;
    if (T > 0) {
      uint64_t nbt2 = (T + blk - 1) / blk;
      hipLaunchKernelGGL(k_gather_final, dim3((uint32_t)nbt2), dim3(blk), 0,
                         s, (const uint32_t *)c->simp_park.ptr,
                         (const uint32_t *)c->tri_off.ptr,
                         (const uint32_t *)c->simp_troff_final.ptr,
                         (uint32_t *)c->faces.ptr,
                         (uint32_t *)c->simp_flab.ptr, (uint32_t)L, T);
    }
    faces_g = (uint32_t *)c->faces.ptr;
    std::swap(c->tri_off, c->simp_troff_final);
  }

  // final: drop unreferenced vertices (stable), re-localize faces
  if (ensure(c, c->simp_ref, (V + 1) * 4)) return 45;
  if (ensure(c, c->simp_verts_alt, V * 12 + 12)) return 45;
  if (ensure(c, c->simp_vbase_alt, (L + 1) * 4)) return 45;
  uint32_t *ref = (uint32_t *)c->simp_ref.ptr;
  uint32_t *newid = (uint32_t *)c->vtx_scan.ptr;  // NC*4 >= V*4, free now
  HIP_TRY(c, hipMemsetAsync(ref, 0, V * 4, s), 45);
  uint64_t NCk = 3 * T;
  if (T > 0) {
    uint64_t nbc = (NCk + blk - 1) / blk;
    hipLaunchKernelGGL(k_mark_ref, dim3((uint32_t)nbc), dim3(blk), 0, s,
                       faces_g, ref, NCk);
  }
  {
    size_t tmp = 0;
    hipError_t e = rocprim::exclusive_scan(
        nullptr, tmp, ref, newid, 0u, V, rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "ref scan size query"); return 45; }
    if (ensure(c, c->scan_tmp, tmp)) return 45;
    e = rocprim::exclusive_scan(
        c->scan_tmp.ptr, tmp, ref, newid, 0u, V,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "ref scan"); return 45; }
  }
  hipLaunchKernelGGL(k_last_sum, dim3(1), dim3(1), 0, s, newid, ref, V,
                     (uint32_t *)c->lh_misc.ptr + 5);
  uint32_t newV = 0;
  HIP_TRY(c, hipMemcpyAsync(&newV, (uint32_t *)c->lh_misc.ptr + 5, 4,
                            hipMemcpyDeviceToHost, s), 45);
  HIP_TRY(c, hipStreamSynchronize(s), 45);
  {
    uint64_t nbv = (V + blk - 1) / blk;
    hipLaunchKernelGGL(k_scatter_verts, dim3((uint32_t)nbv), dim3(blk), 0, s,
                       verts, ref, newid, (float *)c->simp_verts_alt.ptr, V);
    uint32_t nbl = (uint32_t)((L + 2 + 255) / 256);
    hipLaunchKernelGGL(k_new_vbase, dim3(nbl), dim3(256), 0, s,
                       (const uint32_t *)c->vbase.ptr, newid,
                       (uint32_t *)c->simp_vbase_alt.ptr, (uint32_t)L, newV);
    if (T > 0) {
      uint64_t nbt = (T + blk - 1) / blk;
      hipLaunchKernelGGL(k_localize_faces, dim3((uint32_t)nbt), dim3(blk), 0,
                         s, faces_g, newid,
                         (const uint32_t *)c->simp_flab.ptr,
                         (const uint32_t *)c->simp_vbase_alt.ptr,
                         (uint32_t *)c->simp_faces_alt.ptr, T);
    }
  }
  HIP_TRY(c, hipGetLastError(), 46);
  std::swap(c->faces, c->simp_faces_alt);
  std::swap(c->verts, c->simp_verts_alt);
  std::swap(c->vbase, c->simp_vbase_alt);
  *p_T = T;
  *p_V = newV;
  return 0;
}

static int mesh_chunk_impl(mg_ctx *c, const void *labels_host,
                           int sx, int sy, int sz, int dtype,
                           float rx, float ry, float rz,
                           uint32_t reduction_factor, float max_error,
                           int voxel_centered, uint64_t dust_threshold,
                           uint32_t flags_,
                           mg_meshset **out) {
  const size_t esize = (dtype == MG_U64) ? 8 : 4;
  const uint64_t nvox = (uint64_t)sx * sy * sz;
  GridDims g;
  g.sx = sx; g.sy = sy; g.sz = sz;
  g.ncx = sx > 1 ? sx - 1 : 0;
  g.ncy = sy > 1 ? sy - 1 : 0;
  g.ncz = sz > 1 ? sz - 1 : 0;
  g.nsegx = (g.ncx + WAVE - 1) / WAVE;
  g.nseg = g.nsegx * g.ncy * g.ncz;

  memset(&c->stats, 0, sizeof(c->stats));
  c->stats.bytes_read_algorithmic = nvox * esize;

  hipStream_t s = c->stream;

  // trivial empty-cell-grid case
  if (g.nseg == 0) {
    mg_meshset *ms = (mg_meshset *)calloc(
        1, sizeof(mg_meshset) + 2 * sizeof(void *));
    *out = ms;
    return 0;
  }

  HIP_TRY(c, hipEventRecord(c->ev[0], s), 10);

  // H2D (skipped when the caller staged the identical volume already)
  if ((flags_ & MG_FLAG_SKIP_H2D) && c->labels.cap >= nvox * esize) {
    // resident input: nothing to upload
  } else {
    if (ensure(c, c->labels, nvox * esize)) return 11;
    HIP_TRY(c, hipMemcpyAsync(c->labels.ptr, labels_host, nvox * esize,
                              hipMemcpyHostToDevice, s), 11);
  }
  HIP_TRY(c, hipEventRecord(c->ev[1], s), 11);

  // device dust removal (three volume passes; see k_dust_* above)
  if (dust_threshold > 0) {
    const int dblk = 256;
    const uint32_t dnb = 4096;
    for (;;) {
      if (ensure(c, c->lh_keys, c->lh_slots * 8)) return 16;
      if (ensure(c, c->lh_vals, c->lh_slots * 4)) return 16;
      if (ensure(c, c->lh_misc, 256)) return 16;
      HIP_TRY(c, hipMemsetAsync(c->lh_keys.ptr, 0, c->lh_slots * 8, s), 16);
      HIP_TRY(c, hipMemsetAsync(c->lh_misc.ptr, 0, 256, s), 16);
      LabelHash dlh;
      dlh.keys = (uint64_t *)c->lh_keys.ptr;
      dlh.vals = (uint32_t *)c->lh_vals.ptr;
      dlh.counter = (uint32_t *)c->lh_misc.ptr;
      dlh.overflow = (uint32_t *)c->lh_misc.ptr + 1;
      dlh.nslots = c->lh_slots;
      if (dtype == MG_U64)
        hipLaunchKernelGGL(k_dust_build<uint64_t>, dim3(dnb), dim3(dblk),
                           0, s, (const uint64_t *)c->labels.ptr, nvox, dlh);
      else
        hipLaunchKernelGGL(k_dust_build<uint32_t>, dim3(dnb), dim3(dblk),
                           0, s, (const uint32_t *)c->labels.ptr, nvox, dlh);
      uint32_t misc[2] = {0, 0};
      HIP_TRY(c, hipMemcpyAsync(misc, c->lh_misc.ptr, 8,
                                hipMemcpyDeviceToHost, s), 16);
      HIP_TRY(c, hipStreamSynchronize(s), 16);
      if (misc[1]) {  // hash overflow: grow and retry
        if (c->lh_slots >= (1ull << 27)) {
          SET_ERR(c, "label hash overflow (dust) at %llu slots",
                  (unsigned long long)c->lh_slots);
          return 16;
        }
        c->lh_slots <<= 2;
        continue;
      }
      uint32_t ndl = misc[0];
      if (ndl == 0) break;
      if (ensure(c, c->order, (uint64_t)ndl * 4)) return 16;
      HIP_TRY(c, hipMemsetAsync(c->order.ptr, 0, (uint64_t)ndl * 4, s), 16);
      if (dtype == MG_U64) {
        hipLaunchKernelGGL(k_dust_count<uint64_t>, dim3(dnb), dim3(dblk),
                           0, s, (const uint64_t *)c->labels.ptr, nvox,
                           dlh, (uint32_t *)c->order.ptr);
        hipLaunchKernelGGL(k_dust_zero<uint64_t>, dim3(dnb), dim3(dblk),
                           0, s, (uint64_t *)c->labels.ptr, nvox, dlh,
                           (const uint32_t *)c->order.ptr, dust_threshold);
      } else {
        hipLaunchKernelGGL(k_dust_count<uint32_t>, dim3(dnb), dim3(dblk),
                           0, s, (const uint32_t *)c->labels.ptr, nvox,
                           dlh, (uint32_t *)c->order.ptr);
        hipLaunchKernelGGL(k_dust_zero<uint32_t>, dim3(dnb), dim3(dblk),
                           0, s, (uint32_t *)c->labels.ptr, nvox, dlh,
                           (const uint32_t *)c->order.ptr, dust_threshold);
      }
      HIP_TRY(c, hipGetLastError(), 16);
      break;
    }
  }

  // label hash (grow-and-retry on overflow)
  uint64_t total_tris = 0;
  uint32_t nlabels = 0;
  for (;;) {
    if (ensure(c, c->lh_keys, c->lh_slots * 8)) return 12;
    if (ensure(c, c->lh_vals, c->lh_slots * 4)) return 12;
    if (ensure(c, c->lh_misc, 256)) return 12;
    if (ensure(c, c->segcnt, g.nseg * 4)) return 12;
    if (ensure(c, c->segoff, g.nseg * 4)) return 12;
    HIP_TRY(c, hipMemsetAsync(c->lh_keys.ptr, 0, c->lh_slots * 8, s), 12);
    HIP_TRY(c, hipMemsetAsync(c->lh_misc.ptr, 0, 256, s), 12);
    LabelHash lh;
    lh.keys = (uint64_t *)c->lh_keys.ptr;
    lh.vals = (uint32_t *)c->lh_vals.ptr;
    lh.counter = (uint32_t *)c->lh_misc.ptr;
    lh.overflow = (uint32_t *)c->lh_misc.ptr + 1;
    lh.nslots = c->lh_slots;

    int rc;
    if (dtype == MG_U64)
      rc = run_count_emit<uint64_t>(c, (const uint64_t *)c->labels.ptr, g, lh,
                                    (uint32_t *)c->segcnt.ptr,
                                    (uint32_t *)c->segoff.ptr,
                                    &total_tris, &nlabels);
    else
      rc = run_count_emit<uint32_t>(c, (const uint32_t *)c->labels.ptr, g, lh,
                                    (uint32_t *)c->segcnt.ptr,
                                    (uint32_t *)c->segoff.ptr,
                                    &total_tris, &nlabels);
    if (rc == -100) {  // hash overflow: grow and retry
      if (c->lh_slots >= (1ull << 27)) {
        SET_ERR(c, "label hash overflow at %llu slots",
                (unsigned long long)c->lh_slots);
        return 13;
      }
      c->lh_slots <<= 2;
      continue;
    }
    if (rc) return rc;
    break;
  }

  LabelHash lh;
  lh.keys = (uint64_t *)c->lh_keys.ptr;
  lh.vals = (uint32_t *)c->lh_vals.ptr;
  lh.counter = (uint32_t *)c->lh_misc.ptr;
  lh.overflow = (uint32_t *)c->lh_misc.ptr + 1;
  lh.nslots = c->lh_slots;

  c->stats.total_tris = total_tris;
  c->stats.n_labels = nlabels;

  if (total_tris == 0 || nlabels == 0) {
    mg_meshset *ms = (mg_meshset *)calloc(
        1, sizeof(mg_meshset) + 2 * sizeof(void *));
    *out = ms;
    HIP_TRY(c, hipStreamSynchronize(s), 14);
    c->stats.ms_h2d = ev_ms(c, 0, 1);
    return 0;
  }
  if (total_tris > (1ull << 31) / 3 * 2) {  // 3T must fit u32 stream index
    SET_ERR(c, "chunk produces %llu triangles (> corner-index limit); "
            "split the task shape", (unsigned long long)total_tris);
    return 15;
  }
  if (nlabels >= (1u << 27)) {  // label id shares the 64-bit weld key
    SET_ERR(c, "%u labels exceed the 2^27 per-chunk label limit", nlabels);
    return 15;
  }

  const uint64_t T = total_tris;
  const uint64_t NC = 3 * T;  // corners

  // label id -> value
  if (ensure(c, c->label_values, (uint64_t)nlabels * 8)) return 17;
  {
    int blk = 256;
    int64_t nb = (c->lh_slots + blk - 1) / blk;
    hipLaunchKernelGGL(k_label_values, dim3((uint32_t)nb), dim3(blk), 0, s,
                       lh, (uint64_t *)c->label_values.ptr, c->lh_slots);
  }

  // [3] emit
  if (ensure(c, c->tri_label, T * 4)) return 18;
  if (ensure(c, c->tri_keys, T * 8)) return 18;
  {
    int blk = 256;
    int waves_per_blk = blk / WAVE;
    int64_t nb = std::min<int64_t>((g.nseg + waves_per_blk - 1) / waves_per_blk,
                                   8192);
    nb = (nb + 7) & ~7ll;  // multiple of 8: XCD slab schedule coverage
    if (dtype == MG_U64)
      hipLaunchKernelGGL(k_emit<uint64_t>, dim3((uint32_t)nb), dim3(blk), 0, s,
                         (const uint64_t *)c->labels.ptr, g,
                         (const uint32_t *)c->segoff.ptr, lh,
                         (uint32_t *)c->tri_label.ptr,
                         (uint2 *)c->tri_keys.ptr);
    else
      hipLaunchKernelGGL(k_emit<uint32_t>, dim3((uint32_t)nb), dim3(blk), 0, s,
                         (const uint32_t *)c->labels.ptr, g,
                         (const uint32_t *)c->segoff.ptr, lh,
                         (uint32_t *)c->tri_label.ptr,
                         (uint2 *)c->tri_keys.ptr);
  }
  HIP_TRY(c, hipGetLastError(), 18);
  HIP_TRY(c, hipEventRecord(c->ev[4], s), 18);

  // [4] stable partition by label id. Default (MG_SORT_RECS=0 reverts):
  // the 8-B records ride the radix sort as its 64-bit VALUES, so the
  // weld insert reads them back SEQUENTIALLY instead of paying a
  // random 8-B gather through the permutation.
  const char *sre = getenv("MG_SORT_RECS");
  const bool sort_recs = !(sre && sre[0] == '0');
  if (ensure(c, c->keys_sorted, T * 20)) return 19;  // [sort-alt 8B] + [slot triples 12B]
  if (ensure(c, c->tri_label_alt, T * 4)) return 19;
  uint32_t *order_sorted = nullptr;
  uint32_t *lab_sorted = nullptr;  // sort key output: label id per tri
  const uint2 *rec_src = (const uint2 *)c->tri_keys.ptr;
  uint32_t *slots_out = (uint32_t *)c->keys_sorted.ptr + 2 * T;
  {
    int blk = 256;
    uint64_t nb = (T + blk - 1) / blk;
    unsigned begin_bit = 0;
    unsigned end_bit = 1;
    while ((1u << end_bit) < nlabels) ++end_bit;
    if (nlabels == 1) end_bit = 1;
    rocprim::double_buffer<uint32_t> d_keys(
        (uint32_t *)c->tri_label.ptr, (uint32_t *)c->tri_label_alt.ptr);
    if (sort_recs) {
      rocprim::double_buffer<uint64_t> d_vals(
          (uint64_t *)c->tri_keys.ptr, (uint64_t *)c->keys_sorted.ptr);
      size_t tmp_bytes = 0;
      hipError_t e = rocprim::radix_sort_pairs(
          nullptr, tmp_bytes, d_keys, d_vals, T, begin_bit, end_bit, s);
      if (e != hipSuccess) { SET_ERR(c, "radix_sort size query failed"); return 19; }
      if (ensure(c, c->sort_tmp, tmp_bytes)) return 19;
      e = rocprim::radix_sort_pairs(
          c->sort_tmp.ptr, tmp_bytes, d_keys, d_vals, T, begin_bit, end_bit, s);
      if (e != hipSuccess) { SET_ERR(c, "radix_sort failed"); return 19; }
      rec_src = (const uint2 *)d_vals.current();
      order_sorted = nullptr;  // records already label-partitioned
    } else {
      if (ensure(c, c->order, T * 4)) return 19;
      if (ensure(c, c->order_alt, T * 4)) return 19;
      hipLaunchKernelGGL(k_iota, dim3((uint32_t)nb), dim3(blk), 0, s,
                         (uint32_t *)c->order.ptr, T);
      rocprim::double_buffer<uint32_t> d_vals(
          (uint32_t *)c->order.ptr, (uint32_t *)c->order_alt.ptr);
      size_t tmp_bytes = 0;
      hipError_t e = rocprim::radix_sort_pairs(
          nullptr, tmp_bytes, d_keys, d_vals, T, begin_bit, end_bit, s);
      if (e != hipSuccess) { SET_ERR(c, "radix_sort size query failed"); return 19; }
      if (ensure(c, c->sort_tmp, tmp_bytes)) return 19;
      e = rocprim::radix_sort_pairs(
          c->sort_tmp.ptr, tmp_bytes, d_keys, d_vals, T, begin_bit, end_bit, s);
      if (e != hipSuccess) { SET_ERR(c, "radix_sort failed"); return 19; }
      order_sorted = d_vals.current();
    }
    lab_sorted = d_keys.current();
    // label ranges
    if (ensure(c, c->tri_off, ((uint64_t)nlabels + 1) * 4)) return 19;
    hipLaunchKernelGGL(k_label_ranges, dim3((uint32_t)nb), dim3(blk), 0, s,
                       d_keys.current(), (uint32_t *)c->tri_off.ptr, T,
                       nlabels);
  }
  HIP_TRY(c, hipGetLastError(), 19);
  HIP_TRY(c, hipEventRecord(c->ev[5], s), 19);

  // [5] weld — direct-addressed (edge, side) table, collision-free
  const uint64_t wslots = 6 * nvox;  // 3 edges/voxel x 2 sides
  if (wslots >= (1ull << 32)) {
    SET_ERR(c, "chunk too large for the 32-bit weld table (%d x %d x %d); "
            "split the task shape (the reference's own mesher bound is "
            "1023x1023x511, igneous_cli/cli.py:1049-1052)", sx, sy, sz);
    return 20;
  }
  uint64_t total_verts = 0;
  if (ensure(c, c->wh_keys, wslots * 4)) return 20;   // wminp
  if (ensure(c, c->wh_vtx, wslots * 4)) return 20;    // wvtx
  if (ensure(c, c->vtx_scan, NC * 4)) return 20;
  HIP_TRY(c, hipMemsetAsync(c->wh_keys.ptr, 0, wslots * 4, s), 20);
  uint32_t *wminp = (uint32_t *)c->wh_keys.ptr;
  uint32_t *wvtx = (uint32_t *)c->wh_vtx.ptr;
  const uint32_t *slots_sorted = slots_out;
  {
    int blk = 256;
    uint64_t nbt = (T + blk - 1) / blk;
    int wi_cfg = 1;  // 0: 1024x1, 1: 1024x2, 2: 1024x4
    if (const char *e = getenv("MG_WELD_INSERT_CFG")) wi_cfg = atoi(e);
    uint32_t *rs_mut = slots_out;
    const uint2 *tr = rec_src;
    const int64_t sxy64 = g.sx * g.sy;
    if (wi_cfg == 0) {
      uint64_t nb2 = (T + 1023) / 1024;
      hipLaunchKernelGGL((k_weld_insert<1024, 1>), dim3((uint32_t)nb2),
                         dim3(1024), 0, s, tr, order_sorted, rs_mut, wminp,
                         g.sx, sxy64, T);
    } else if (wi_cfg == 1) {
      uint64_t nb2 = (T + 2047) / 2048;
      hipLaunchKernelGGL((k_weld_insert<1024, 2>), dim3((uint32_t)nb2),
                         dim3(1024), 0, s, tr, order_sorted, rs_mut, wminp,
                         g.sx, sxy64, T);
    } else {
      uint64_t nb2 = (T + 4095) / 4096;
      hipLaunchKernelGGL((k_weld_insert<1024, 4>), dim3((uint32_t)nb2),
                         dim3(1024), 0, s, tr, order_sorted, rs_mut, wminp,
                         g.sx, sxy64, T);
    }
    // first-occurrence flags as a bit array + word-granular scan
    const uint64_t nwords = (NC + 63) / 64;
    unsigned long long *bits =
        (unsigned long long *)c->vtx_scan.ptr;          // nwords * 8
    uint32_t *wscan = (uint32_t *)c->vtx_scan.ptr + 2 * nwords;  // nwords * 4
    // (vtx_scan buffer is NC*4 bytes >= nwords*12)
    {
      uint64_t nbw = (nwords * 64 + blk - 1) / blk;
      hipLaunchKernelGGL(k_weld_flag_bits, dim3((uint32_t)nbw), dim3(blk),
                         0, s, slots_sorted, wminp, bits, NC, nwords);
    }
    auto it = rocprim::make_transform_iterator(bits, PopcWord{});
    size_t tmp_bytes = 0;
    hipError_t e = rocprim::exclusive_scan(
        nullptr, tmp_bytes, it, wscan, 0u, nwords,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "weld scan size query failed"); return 20; }
    if (ensure(c, c->scan_tmp, tmp_bytes)) return 20;
    e = rocprim::exclusive_scan(
        c->scan_tmp.ptr, tmp_bytes, it, wscan, 0u, nwords,
        rocprim::plus<uint32_t>(), s);
    if (e != hipSuccess) { SET_ERR(c, "weld scan failed"); return 20; }
    hipLaunchKernelGGL(k_total_verts, dim3(1), dim3(1), 0, s,
                       wscan, bits, nwords,
                       (uint32_t *)c->lh_misc.ptr + 3);
    uint32_t tv = 0;
    HIP_TRY(c, hipMemcpyAsync(&tv, (uint32_t *)c->lh_misc.ptr + 3, 4,
                              hipMemcpyDeviceToHost, s), 21);
    HIP_TRY(c, hipStreamSynchronize(s), 21);
    total_verts = tv;
  }
  c->stats.total_verts = total_verts;

  if (ensure(c, c->verts, total_verts * 12)) return 22;
  if (ensure(c, c->faces, NC * 4)) return 22;
  if (ensure(c, c->vbase, ((uint64_t)nlabels + 1) * 4)) return 22;
  {
    int blk = 256;
    uint64_t nbt = (T + blk - 1) / blk;
    const float shift = voxel_centered ? 0.0f : 0.5f;
    const uint64_t nwords = (NC + 63) / 64;
    const unsigned long long *bits =
        (const unsigned long long *)c->vtx_scan.ptr;
    const uint32_t *wscan = (const uint32_t *)c->vtx_scan.ptr + 2 * nwords;
    hipLaunchKernelGGL(k_weld_verts, dim3((uint32_t)nbt), dim3(blk), 0, s,
                       slots_sorted, wscan, bits, wminp, wvtx,
                       (float *)c->verts.ptr,
                       (uint32_t)g.sx, (uint32_t)(g.sx * g.sy),
                       rx, ry, rz, shift, T);
    uint32_t nbl = (nlabels + 1 + 255) / 256;
    hipLaunchKernelGGL(k_vbase, dim3(nbl), dim3(256), 0, s,
                       (const uint32_t *)c->tri_off.ptr,
                       wscan, bits,
                       (uint32_t *)c->vbase.ptr, nlabels, total_verts);
    hipLaunchKernelGGL(k_faces, dim3((uint32_t)nbt), dim3(blk), 0, s,
                       slots_sorted, lab_sorted, wvtx,
                       (const uint32_t *)c->vbase.ptr,
                       (uint32_t *)c->faces.ptr, T);
  }
  HIP_TRY(c, hipGetLastError(), 22);
  HIP_TRY(c, hipEventRecord(c->ev[6], s), 22);

  // [6] per-label quadric simplification (mesh.py:376-381 semantics)
  uint64_t Tcur = T, Vcur = total_verts;
  if (reduction_factor > 1 && T > 0) {
    int rc = run_simplify(c, nlabels, lab_sorted, reduction_factor,
                          max_error, &Tcur, &Vcur);
    if (rc) return rc;
  }
  HIP_TRY(c, hipEventRecord(c->ev[8], s), 22);
  const uint64_t NCX = 3 * Tcur;

  // [7] extract
  mg_meshset *ms = nullptr;
  if (flags_ & MG_FLAG_DEVICE_ONLY) {
    HIP_TRY(c, hipStreamSynchronize(s), 23);
    ms = (mg_meshset *)calloc(1, sizeof(mg_meshset) + 2 * sizeof(void *));
    *out = ms;
  } else {
    if (ensure_host(c, c->h_verts, Vcur * 12 + 12)) return 24;
    if (ensure_host(c, c->h_faces, NCX * 4 + 4)) return 24;
    float *h_verts = (float *)c->h_verts.ptr;
    uint32_t *h_faces = (uint32_t *)c->h_faces.ptr;
    std::vector<uint32_t> h_tri_off(nlabels + 1), h_vbase(nlabels + 1);
    std::vector<uint64_t> h_label_values(nlabels);
    if (Vcur > 0)
      HIP_TRY(c, hipMemcpyAsync(h_verts, c->verts.ptr, Vcur * 12,
                                hipMemcpyDeviceToHost, s), 24);
    if (NCX > 0)
      HIP_TRY(c, hipMemcpyAsync(h_faces, c->faces.ptr, NCX * 4,
                                hipMemcpyDeviceToHost, s), 24);
    HIP_TRY(c, hipMemcpyAsync(h_tri_off.data(), c->tri_off.ptr,
                              (nlabels + 1) * 4, hipMemcpyDeviceToHost, s), 24);
    HIP_TRY(c, hipMemcpyAsync(h_vbase.data(), c->vbase.ptr, (nlabels + 1) * 4,
                              hipMemcpyDeviceToHost, s), 24);
    HIP_TRY(c, hipMemcpyAsync(h_label_values.data(), c->label_values.ptr,
                              nlabels * 8, hipMemcpyDeviceToHost, s), 24);
    HIP_TRY(c, hipStreamSynchronize(s), 24);

    size_t meta_off = sizeof(mg_meshset) + sizeof(mg_mesh) * nlabels;
    ms = (mg_meshset *)calloc(1, meta_off + (size_t)nlabels * 24);
    ms->nmeshes = nlabels;
    ms->meshes = (mg_mesh *)((char *)ms + sizeof(mg_meshset));
    ms->verts_base = h_verts;
    ms->faces_base = h_faces;
    ms->total_verts = Vcur;
    ms->total_tris = Tcur;
    ms->labels_arr = (uint64_t *)((char *)ms + meta_off);
    ms->voff_arr = (uint32_t *)(ms->labels_arr + nlabels);
    ms->nv_arr = ms->voff_arr + nlabels;
    ms->foff_arr = ms->nv_arr + nlabels;
    ms->nf_arr = ms->foff_arr + nlabels;
    // order meshes by ascending label value
    std::vector<uint32_t> idx(nlabels);
    for (uint32_t i = 0; i < nlabels; ++i) idx[i] = i;
    std::sort(idx.begin(), idx.end(), [&](uint32_t a, uint32_t b) {
      return h_label_values[a] < h_label_values[b];
    });
    for (uint32_t m = 0; m < nlabels; ++m) {
      uint32_t l = idx[m];
      mg_mesh &mm = ms->meshes[m];
      mm.label = h_label_values[l];
      mm.nverts = h_vbase[l + 1] - h_vbase[l];
      mm.ntris = h_tri_off[l + 1] - h_tri_off[l];
      mm.verts = h_verts + 3ull * h_vbase[l];
      mm.faces = h_faces + 3ull * h_tri_off[l];
      ms->labels_arr[m] = mm.label;
      ms->voff_arr[m] = h_vbase[l];
      ms->nv_arr[m] = mm.nverts;
      ms->foff_arr[m] = h_tri_off[l];
      ms->nf_arr[m] = mm.ntris;
    }
    *out = ms;
  }
  HIP_TRY(c, hipEventRecord(c->ev[7], s), 25);
  HIP_TRY(c, hipStreamSynchronize(s), 25);

  c->stats.ms_h2d = ev_ms(c, 0, 1);
  c->stats.ms_count = ev_ms(c, 1, 2);
  c->stats.ms_scan = ev_ms(c, 2, 3);
  c->stats.ms_emit = ev_ms(c, 3, 4);
  c->stats.ms_partition = ev_ms(c, 4, 5);
  c->stats.ms_weld = ev_ms(c, 5, 6);
  c->stats.ms_simplify = ev_ms(c, 6, 8);
  c->stats.ms_d2h = ev_ms(c, 8, 7);
  c->stats.ms_total = ev_ms(c, 0, 7);
  return 0;
}

// ---------------------------------------------------------------------------
// Standalone mesh simplification (multires LOD chain; see meshgine.h).
// Reuses the per-label simplify machinery with ONE label spanning the
// whole mesh.

extern "C" int mg_simplify_mesh(mg_ctx *c,
                                const float *verts, uint32_t nverts,
                                const uint32_t *faces, uint32_t ntris,
                                uint32_t reduction_factor, float max_error,
                                const float **out_verts,
                                uint32_t *out_nverts,
                                const uint32_t **out_faces,
                                uint32_t *out_ntris) {
  if (!c) { SET_ERR(c, "null ctx"); return 1; }
  std::lock_guard<std::mutex> g(c->lock);
  c->err.clear();
  if (!verts || !faces || !out_verts || !out_faces) {
    SET_ERR(c, "null argument");
    return 1;
  }
  if (ntris > (1u << 31) / 3) {
    SET_ERR(c, "mesh too large (%u tris)", ntris);
    return 2;
  }
  HIP_TRY(c, hipSetDevice(c->device), 4);
  hipStream_t s = c->stream;
  uint64_t T = ntris, V = nverts;

  if (reduction_factor > 1 && T > 0 && V > 0) {
    if (ensure(c, c->verts, V * 12 + 12)) return 10;
    if (ensure(c, c->faces, 3 * T * 4 + 4)) return 10;
    if (ensure(c, c->tri_off, 8)) return 10;
    if (ensure(c, c->vbase, 8)) return 10;
    if (ensure(c, c->tri_label, T * 4 + 4)) return 10;
    if (ensure(c, c->vtx_scan, (std::max<uint64_t>(V + 1, 3 * T)) * 4))
      return 10;
    HIP_TRY(c, hipMemcpyAsync(c->verts.ptr, verts, V * 12,
                              hipMemcpyHostToDevice, s), 11);
    HIP_TRY(c, hipMemcpyAsync(c->faces.ptr, faces, 3 * T * 4,
                              hipMemcpyHostToDevice, s), 11);
    uint32_t off_h[2] = {0, (uint32_t)T};
    uint32_t vb_h[2] = {0, (uint32_t)V};
    HIP_TRY(c, hipMemcpyAsync(c->tri_off.ptr, off_h, 8,
                              hipMemcpyHostToDevice, s), 11);
    HIP_TRY(c, hipMemcpyAsync(c->vbase.ptr, vb_h, 8,
                              hipMemcpyHostToDevice, s), 11);
    HIP_TRY(c, hipMemsetAsync(c->tri_label.ptr, 0, T * 4, s), 11);
    if (ensure(c, c->lh_misc, 256)) return 11;
    int rc = run_simplify(c, 1, (const uint32_t *)c->tri_label.ptr,
                          reduction_factor, max_error, &T, &V);
    if (rc) return rc;
    if (ensure_host(c, c->h_verts, V * 12 + 12)) return 12;
    if (ensure_host(c, c->h_faces, 3 * T * 4 + 4)) return 12;
    if (V > 0)
      HIP_TRY(c, hipMemcpyAsync(c->h_verts.ptr, c->verts.ptr, V * 12,
                                hipMemcpyDeviceToHost, s), 12);
    if (T > 0)
      HIP_TRY(c, hipMemcpyAsync(c->h_faces.ptr, c->faces.ptr, 3 * T * 4,
                                hipMemcpyDeviceToHost, s), 12);
    HIP_TRY(c, hipStreamSynchronize(s), 12);
  } else {
    // no-op request: hand back a staged copy (uniform ownership)
    if (ensure_host(c, c->h_verts, V * 12 + 12)) return 12;
    if (ensure_host(c, c->h_faces, 3 * T * 4 + 4)) return 12;
    memcpy(c->h_verts.ptr, verts, V * 12);
    memcpy(c->h_faces.ptr, faces, 3 * T * 4);
  }
  *out_verts = (const float *)c->h_verts.ptr;
  *out_nverts = (uint32_t)V;
  *out_faces = (const uint32_t *)c->h_faces.ptr;
  *out_ntris = (uint32_t)T;
  return 0;
}

template <typename T>
static int run_count_emit(mg_ctx *c, const T *d_labels, const GridDims &g,
                          LabelHash lh, uint32_t *d_segcnt,
                          uint32_t *d_segoff, uint64_t *p_total,
                          uint32_t *p_nlabels) {
  hipStream_t s = c->stream;
  HIP_TRY(c, hipEventRecord(c->ev[1], s), 30);
  int blk = 256;
  int waves_per_blk = blk / WAVE;
  int64_t nb = std::min<int64_t>(
      (g.nseg + waves_per_blk - 1) / waves_per_blk, 8192);
  nb = (nb + 7) & ~7ll;  // multiple of 8: XCD slab schedule coverage
  hipLaunchKernelGGL(k_count<T>, dim3((uint32_t)nb), dim3(blk), 0, s,
                     d_labels, g, d_segcnt, lh);
  HIP_TRY(c, hipGetLastError(), 30);
  HIP_TRY(c, hipEventRecord(c->ev[2], s), 30);

  // check overflow + read label count
  uint32_t misc[2] = {0, 0};
  HIP_TRY(c, hipMemcpyAsync(misc, c->lh_misc.ptr, 8, hipMemcpyDeviceToHost, s),
          30);

  // scan segcnt -> segoff
  size_t tmp_bytes = 0;
  hipError_t e = rocprim::exclusive_scan(
      nullptr, tmp_bytes, d_segcnt, d_segoff, 0u, (size_t)g.nseg,
      rocprim::plus<uint32_t>(), s);
  if (e != hipSuccess) { SET_ERR(c, "seg scan size query failed"); return 30; }
  if (ensure(c, c->scan_tmp, tmp_bytes)) return 30;
  e = rocprim::exclusive_scan(
      c->scan_tmp.ptr, tmp_bytes, d_segcnt, d_segoff, 0u, (size_t)g.nseg,
      rocprim::plus<uint32_t>(), s);
  if (e != hipSuccess) { SET_ERR(c, "seg scan failed"); return 30; }

  uint32_t last_off = 0, last_cnt = 0;
  HIP_TRY(c, hipMemcpyAsync(&last_off, d_segoff + (g.nseg - 1), 4,
                            hipMemcpyDeviceToHost, s), 30);
  HIP_TRY(c, hipMemcpyAsync(&last_cnt, d_segcnt + (g.nseg - 1), 4,
                            hipMemcpyDeviceToHost, s), 30);
  HIP_TRY(c, hipEventRecord(c->ev[3], s), 30);
  HIP_TRY(c, hipStreamSynchronize(s), 30);
  if (misc[1]) return -100;  // label hash overflow
  *p_nlabels = misc[0];
  *p_total = (uint64_t)last_off + last_cnt;
  return 0;
}
