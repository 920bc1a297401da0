// MI355X (gfx950, CDNA4) kernels for Louvain local moving + modularity.
//
// Semantic spec: cuvite_amd/local_move.py (transcribed from the reference CPU
// path, louvain.cpp:2185-2431); this is a from-scratch wave64 design, not a
// port of the reference's CUDA kernels: the reference deduplicates neighbor
// communities with an O(deg^2) in-place sweep (distGetMaxIndex,
// louvain_cuda.cu:1190-1346); here every degree class builds an open-addressing
// LDS hash table (community id -> accumulated edge weight) with wave-
// cooperative inserts, then computes the dQ argmax by scanning the table.
//
// Degree-class routing (the reference's 3-bucket idea, count_size_clmap
// louvain_cuda.cu:1426-1592, rebuilt for wave64). Class boundaries are
// sized so the LDS table per block stays at <= 24 KB for the bulk classes
// (6 resident blocks/CU on 160 KB LDS — the round-1 two-class split left
// class (512,2048] on a 48 KB table at 3 blocks/CU, and PMC showed the
// kernels latency-bound at 80% SQ_WAIT_ANY, so residency is the lever):
//   class 0: deg in [1, 16]      16 lanes/vertex,  32-slot LDS table
//   class 1: deg in (16, 64]     one wave/vertex, 128-slot LDS table
//   class 2: deg in (64, 256]    one wave/vertex, 512-slot LDS (24 KB/blk)
//   class 3: deg in (256, 512]   one wave/vertex, 1024-slot LDS (48 KB/blk)
//   class 4: deg in (512, 1024]  256-thread block/vertex, 2048-slot (24 KB)
//   class 5: deg in (1024, 2048] 256-thread block/vertex, 4096-slot (48 KB)
//   class 6: deg in (2048, cut]  256-thread block/vertex, 8192-slot (96 KB;
//                                cut = CUVITE_HUB_CUT, default 6144 after
//                                the s26 A/B in profiles/)
//   hub    : deg > cut (6144)    hub_moves binding (rocPRIM narrow-bit
//                                segmented radix sort + reduce_by_key +
//                                split-block argmax) — 2.9x faster than
//                                the torch global-sort fallback at s26
//                                (A/B in profiles/; CUVITE_HUB_SEGSORT=0
//                                restores the fallback). A global-memory
//                                hash-table hub pipeline existed in round
//                                1 but was DELETED: open-addressing CAS
//                                tables past the 4 MB XCD L2 measured
//                                pathologically slow with an unexplained
//                                hang at exactly 2^21 slots
//                                (profiles/hub_pathology_and_s26.md), and
//                                the sort pipeline beats it anyway.
//
// All gain arithmetic is fp64 regardless of the weight dtype so trajectories
// match the fp64 CPU oracle (tie-break on equal gains -> smaller GLOBAL id).

#include <hip/hip_runtime.h>
#include <rocprim/device/device_segmented_radix_sort.hpp>
#include <rocprim/device/device_radix_sort.hpp>
#include <rocprim/device/device_reduce_by_key.hpp>

#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

namespace cuvite {

static constexpr int32_t EMPTY_KEY = -1;

DEV_INLINE uint32_t hash_u32(uint32_t x) {
  x *= 0x9E3779B1u;
  x ^= x >> 16;
  return x;
}

template <typename W> DEV_INLINE void lds_atomic_add(W* p, W v);
template <> DEV_INLINE void lds_atomic_add<float>(float* p, float v) {
  unsafeAtomicAdd(p, v);
}
template <> DEV_INLINE void lds_atomic_add<double>(double* p, double v) {
  unsafeAtomicAdd(p, v);
}

// Candidate for the dQ argmax. Comparison implements the reference rule
// (louvain.cpp:2225-2233): strictly larger gain wins; equal nonzero gain with
// smaller global id wins. `gain == 0` entries represent "stay" (cc).
struct Best {
  double gain;
  int64_t gid;   // global community id (tie-break)
  int32_t dense; // dense community id (result)
};

DEV_INLINE void best_combine(Best& a, double g, int64_t gid, int32_t dense) {
  if (g > a.gain || (g == a.gain && g != 0.0 && gid < a.gid)) {
    a.gain = g;
    a.gid = gid;
    a.dense = dense;
  }
}

template <int WIDTH> DEV_INLINE void best_reduce(Best& b) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) {
    double g = __shfl_down(b.gain, off, WIDTH);
    int64_t gid = __shfl_down(b.gid, off, WIDTH);
    int32_t dn = __shfl_down(b.dense, off, WIDTH);
    best_combine(b, g, gid, dn);
  }
}

template <int WIDTH> DEV_INLINE double sum_reduce(double v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, WIDTH);
  return v;
}

// Probe (no insert): accumulated weight for key `k`, 0 if absent.
template <typename W>
DEV_INLINE W table_probe(const int32_t* keys, const W* vals, int cap,
                         int32_t k) {
  uint32_t h = hash_u32((uint32_t)k) & (cap - 1);
  while (true) {
    int32_t kk = keys[h];
    if (kk == k) return vals[h];
    if (kk == EMPTY_KEY) return (W)0;
    h = (h + 1) & (cap - 1);
  }
}

template <typename W>
DEV_INLINE void table_insert(int32_t* keys, W* vals, int cap, int32_t k, W w) {
  uint32_t h = hash_u32((uint32_t)k) & (cap - 1);
  while (true) {
    int32_t old = atomicCAS((int*)&keys[h], EMPTY_KEY, k);
    if (old == EMPTY_KEY || old == k) {
      lds_atomic_add(&vals[h], w);
      return;
    }
    h = (h + 1) & (cap - 1);
  }
}

// Edge-insert loop with a 4-deep software pipeline: the per-edge chain
// (load tail -> gather curr_comm[tail] -> LDS CAS insert) is latency-bound
// (PMC: ~80% SQ_WAIT_ANY at 3 blocks/CU); issuing 4 independent tail loads,
// then 4 independent label gathers, before any insert overlaps the two
// global-load latencies instead of serializing them per edge.
template <typename W, int STRIDE>
DEV_INLINE double insert_edges_pipelined(
    int64_t e0, int64_t e1, int lane, int32_t v,
    const int32_t* __restrict__ tails, const W* __restrict__ weights,
    const int32_t* __restrict__ curr_comm, int32_t* keys, W* vals, int cap) {
  constexpr int PIPE = 4;
  double selfloop = 0.0;
  int64_t e = e0 + lane;
  for (; e + (PIPE - 1) * (int64_t)STRIDE < e1; e += PIPE * (int64_t)STRIDE) {
    int32_t t[PIPE];
    W w[PIPE];
#pragma unroll
    for (int k = 0; k < PIPE; k++) {
      t[k] = tails[e + k * STRIDE];
      w[k] = weights[e + k * STRIDE];
    }
    int32_t c[PIPE];
#pragma unroll
    for (int k = 0; k < PIPE; k++) c[k] = curr_comm[t[k]];
#pragma unroll
    for (int k = 0; k < PIPE; k++) {
      if (t[k] == v) selfloop += (double)w[k];
      table_insert(keys, vals, cap, c[k], w[k]);
    }
  }
  for (; e < e1; e += STRIDE) {
    int32_t t = tails[e];
    W w = weights[e];
    if (t == v) selfloop += (double)w;
    table_insert(keys, vals, cap, curr_comm[t], w);
  }
  return selfloop;
}

// Scan the table slots owned by this lane and fold the argmax.
template <typename W>
DEV_INLINE void scan_slots(const int32_t* keys, const W* vals, int cap,
                           int lane, int lanes, int32_t cc, double eix,
                           double ax, double vdeg, double constant,
                           const W* __restrict__ comm_degree,
                           const int64_t* __restrict__ comm_gid, Best& best) {
  for (int s = lane; s < cap; s += lanes) {
    int32_t y = keys[s];
    if (y == EMPTY_KEY || y == cc) continue;
    double eiy = (double)vals[s];
    double ay = (double)comm_degree[y];
    double g = 2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant;
    best_combine(best, g, comm_gid[y], y);
  }
}

// ---------------------------------------------------------------------------
// Sub-block kernel: LANES lanes per vertex, LDS table of CAP slots per group.
// vlist is padded to a multiple of (BLOCK/LANES) with -1.
// ---------------------------------------------------------------------------

// Smallest power of two >= 2*(deg+1), clamped to [lo, CAP]: distinct keys
// <= deg keeps the load factor under 1/2 while init/probe/scan touch only
// `cap` slots. rocprof at s26 showed the class-3 kernel (CAP 4096, typical
// deg ~600) spending most of its time on table init + argmax scan of slots
// that can never be occupied (profiles/round2_kernel_stats).
DEV_INLINE int cap_for_degree(int64_t deg, int CAP, int lo) {
  int cap = CAP;
  while (cap > lo && (cap >> 2) >= (int)(deg + 1)) cap >>= 1;
  return cap;
}

template <typename W, int LANES, int CAP, int BLOCK>
__global__ __launch_bounds__(BLOCK) void lv_move_sub(
    const int32_t* __restrict__ vlist, int nlist,
    const int64_t* __restrict__ rowptr, const int32_t* __restrict__ tails,
    const W* __restrict__ weights, const int32_t* __restrict__ curr_comm,
    const W* __restrict__ v_degree, const int64_t* __restrict__ comm_size,
    const W* __restrict__ comm_degree, const int64_t* __restrict__ comm_gid,
    double constant, int32_t* __restrict__ target,
    W* __restrict__ cluster_weight) {
  constexpr int GROUPS = BLOCK / LANES;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  W* vals = (W*)smem;
  int32_t* keys = (int32_t*)(smem + (size_t)GROUPS * CAP * sizeof(W));

  const int group = threadIdx.x / LANES;
  const int lane = threadIdx.x % LANES;
  const int gidx = blockIdx.x * GROUPS + group;
  const int32_t v = (gidx < nlist) ? vlist[gidx] : -1;

  int32_t* gkeys = keys + (size_t)group * CAP;
  W* gvals = vals + (size_t)group * CAP;

  int64_t e0 = 0, e1 = 0;
  int32_t cc = 0;
  if (v >= 0) {
    e0 = rowptr[v];
    e1 = rowptr[v + 1];
    cc = curr_comm[v];
  }
  const int cap = cap_for_degree(e1 - e0, CAP, LANES);
#pragma unroll 4
  for (int s = lane; s < cap; s += LANES) {
    gkeys[s] = EMPTY_KEY;
    gvals[s] = (W)0;
  }
  __syncthreads();

  double selfloop = insert_edges_pipelined<W, LANES>(
      e0, e1, lane, v, tails, weights, curr_comm, gkeys, gvals, cap);
  __syncthreads();

  selfloop = sum_reduce<LANES>(selfloop);
  Best best{0.0, 0, cc};
  if (v >= 0 && e0 != e1) {
    W wcc = table_probe(gkeys, gvals, cap, cc);  // counter[cc]
    double eix = (double)wcc - __shfl(selfloop, 0, LANES);
    double vdeg = (double)v_degree[v];
    double ax = (double)comm_degree[cc] - vdeg;
    best.gid = comm_gid[cc];
    scan_slots(gkeys, gvals, cap, lane, LANES, cc, eix, ax, vdeg, constant,
               comm_degree, comm_gid, best);
    best_reduce<LANES>(best);
    if (lane == 0) {
      int32_t tgt = best.dense;
      // singleton-swap guard (louvain.cpp:2238-2239)
      if (comm_size[tgt] == 1 && comm_size[cc] == 1 && best.gid > comm_gid[cc])
        tgt = cc;
      target[v] = tgt;
      cluster_weight[v] = wcc;
    }
  }
}

// ---------------------------------------------------------------------------
// Block kernel: one 256-thread block per vertex, CAP-slot LDS table.
// ---------------------------------------------------------------------------

template <typename W, int CAP, int BLOCK>
__global__ __launch_bounds__(BLOCK) void lv_move_block(
    const int32_t* __restrict__ vlist, int nlist,
    const int64_t* __restrict__ rowptr, const int32_t* __restrict__ tails,
    const W* __restrict__ weights, const int32_t* __restrict__ curr_comm,
    const W* __restrict__ v_degree, const int64_t* __restrict__ comm_size,
    const W* __restrict__ comm_degree, const int64_t* __restrict__ comm_gid,
    double constant, int32_t* __restrict__ target,
    W* __restrict__ cluster_weight) {
  constexpr int WAVES = BLOCK / 64;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  W* vals = (W*)smem;
  int32_t* keys = (int32_t*)(smem + (size_t)CAP * sizeof(W));
  __shared__ double red_gain[WAVES];
  __shared__ int64_t red_gid[WAVES];
  __shared__ int32_t red_dense[WAVES];
  __shared__ double red_self[WAVES];

  const int32_t v = vlist[blockIdx.x];
  const int tid = threadIdx.x;
  const int wave = tid / 64, lane = tid % 64;

  const int64_t e0 = rowptr[v], e1 = rowptr[v + 1];
  const int cap = cap_for_degree(e1 - e0, CAP, BLOCK);
  for (int s = tid; s < cap; s += BLOCK) {
    keys[s] = EMPTY_KEY;
    vals[s] = (W)0;
  }
  __syncthreads();

  const int32_t cc = curr_comm[v];
  double selfloop = insert_edges_pipelined<W, BLOCK>(
      e0, e1, tid, v, tails, weights, curr_comm, keys, vals, cap);
  selfloop = sum_reduce<64>(selfloop);
  if (lane == 0) red_self[wave] = selfloop;
  __syncthreads();

  double self_total = 0.0;
#pragma unroll
  for (int i = 0; i < WAVES; i++) self_total += red_self[i];

  W wcc = table_probe(keys, vals, cap, cc);
  double eix = (double)wcc - self_total;
  double vdeg = (double)v_degree[v];
  double ax = (double)comm_degree[cc] - vdeg;
  Best best{0.0, comm_gid[cc], cc};
  scan_slots(keys, vals, cap, tid, BLOCK, cc, eix, ax, vdeg, constant,
             comm_degree, comm_gid, best);
  best_reduce<64>(best);
  if (lane == 0) {
    red_gain[wave] = best.gain;
    red_gid[wave] = best.gid;
    red_dense[wave] = best.dense;
  }
  __syncthreads();
  if (tid == 0) {
#pragma unroll
    for (int i = 1; i < WAVES; i++)
      best_combine(best, red_gain[i], red_gid[i], red_dense[i]);
    int32_t tgt = best.dense;
    if (comm_size[tgt] == 1 && comm_size[cc] == 1 && best.gid > comm_gid[cc])
      tgt = cc;
    target[v] = tgt;
    cluster_weight[v] = wcc;
  }
}

// ---------------------------------------------------------------------------
// Modularity reduction: (sum cluster_weight, sum comm_degree^2) in fp64.
// Reference analog: compute_modularity (modularity.cu:36-100), but fp64
// accumulation and a device-resident 2-double output the RCCL allreduce
// consumes directly (no D2H round trip).
// ---------------------------------------------------------------------------

template <typename W, int BLOCK>
__global__ __launch_bounds__(BLOCK) void modularity_reduce(
    const W* __restrict__ cluster_weight, const W* __restrict__ comm_degree,
    int64_t nv, double* __restrict__ out) {
  double le = 0.0, la2 = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)BLOCK + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * BLOCK) {
    le += (double)cluster_weight[i];
    double d = (double)comm_degree[i];
    la2 += d * d;
  }
  le = sum_reduce<64>(le);
  la2 = sum_reduce<64>(la2);
  if ((threadIdx.x % 64) == 0) {
    unsafeAtomicAdd(&out[0], le);
    unsafeAtomicAdd(&out[1], la2);
  }
}

// ---------------------------------------------------------------------------
// scatter_add: out[idx[i]] += val[i] with NATIVE fp32/fp64 atomics
// (unsafeAtomicAdd -> global_atomic_add_f64 on gfx950). PyTorch-ROCm's
// index_add_ lowers fp64 atomics to a CAS loop, which measured 16.5 s per
// call on an R-MAT s22 sweep (profiles/); this kernel replaces it on the
// hot paths (degree sums, community-aggregate deltas).
// ---------------------------------------------------------------------------

template <typename W>
__global__ void scatter_add_kernel(W* __restrict__ out,
                                   const int64_t* __restrict__ idx,
                                   const W* __restrict__ val, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    unsafeAtomicAdd(&out[idx[i]], val[i]);
}

// ---------------------------------------------------------------------------
// Fused community-aggregate update after a move sweep (ref 4-case update,
// louvain.cpp:2308-2376): one pass over the vertices applies the +-1 size
// and +-v_degree deltas for every moved vertex directly, with native int64/
// fp atomics. Replaces a 6-op torch chain (moved mask -> .any() host sync ->
// boolean compactions -> cat -> index_add_ -> scatter_add) that rocprof
// measured at ~40 ms/sweep at s26 (profiles/round2_kernel_stats). Only
// LOCALLY-owned community labels are applied here; remotely-owned deltas are
// compacted by the caller for the RCCL push (world>1 only).
// ---------------------------------------------------------------------------

template <typename W>
__global__ void apply_deltas_kernel(
    const int64_t* __restrict__ target, const int64_t* __restrict__ curr,
    const W* __restrict__ v_degree, int64_t nv, int64_t base, int64_t bound,
    int64_t* __restrict__ local_size, W* __restrict__ local_degree) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += stride) {
    const int64_t src = curr[i];
    const int64_t dst = target[i];
    if (src == dst) continue;
    const W d = v_degree[i];
    if (src >= base && src < bound) {
      atomicAdd((unsigned long long*)&local_size[src - base],
                (unsigned long long)(-1ll));
      unsafeAtomicAdd(&local_degree[src - base], (W)(-d));
    }
    if (dst >= base && dst < bound) {
      atomicAdd((unsigned long long*)&local_size[dst - base], 1ull);
      unsafeAtomicAdd(&local_degree[dst - base], d);
    }
  }
}

// ---------------------------------------------------------------------------
// Device-side CSR assembly (no sort; replaces torch.argsort which is capped
// at INT_MAX elements): degree histogram + atomic-cursor placement.
// Reference analog: processGraphData (utils.cpp:10-87) done host-side there.
// Row-internal edge order is nondeterministic; all consumers (local-move
// hash tables, degree sums) are order-invariant.
// ---------------------------------------------------------------------------

__global__ void degree_count_kernel(const int64_t* __restrict__ src, int64_t n,
                                    int64_t base, int32_t* __restrict__ cnt) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    atomicAdd(&cnt[src[i] - base], 1);
}

template <typename W>
__global__ void csr_place_kernel(const int64_t* __restrict__ src,
                                 const int64_t* __restrict__ dst,
                                 const W* __restrict__ w, int64_t n,
                                 int64_t base,
                                 const int64_t* __restrict__ rowptr,
                                 int32_t* __restrict__ cursor,
                                 int64_t* __restrict__ tails_out,
                                 W* __restrict__ w_out) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const int64_t s = src[i] - base;
    const int64_t o = rowptr[s] + atomicAdd(&cursor[s], 1);
    tails_out[o] = dst[i];
    w_out[o] = w[i];
  }
}

// ---------------------------------------------------------------------------
// Distance-1 coloring round: per-vertex strict min/max of every hash over
// COMPETING neighbors (uncolored at round start, not self), one wave per
// vertex with all n_hash hashes in a single edge pass and a wave reduction —
// no atomics, no edge-list materialization. Replaces 2*n_hash torch scatter
// passes over the full edge list per round (13.4 s for the whole coloring
// at R-MAT s26). Hash = the reference's 32-bit mix (coloring.cpp:74-85).
// ---------------------------------------------------------------------------

DEV_INLINE uint32_t color_hash(uint32_t a, uint32_t seed) {
  a ^= seed;
  a = (a + 0x7ED55D16u) + (a << 12);
  a = (a ^ 0xC761C23Cu) + (a >> 19);
  a = (a + 0x165667B1u) + (a << 5);
  a = (a ^ 0xD3A2646Cu) + (a << 9);
  a = (a + 0xFD7046C5u) + (a << 3);
  a = (a ^ 0xB55A4F09u) + (a >> 16);
  return a;
}

constexpr int MAX_NHASH = 8;

template <int BLOCK>
__global__ __launch_bounds__(BLOCK) void coloring_minmax_kernel(
    const int64_t* __restrict__ rowptr, const int32_t* __restrict__ tails,
    const int64_t* __restrict__ gid_all, const bool* __restrict__ uncolored,
    const int64_t* __restrict__ colors, int64_t nv, int64_t base,
    const int64_t* __restrict__ seeds, int n_hash,
    int64_t* __restrict__ mn_out, int64_t* __restrict__ mx_out) {
  constexpr int WAVES = BLOCK / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int64_t vstride = (int64_t)gridDim.x * WAVES;
  for (int64_t v = (int64_t)blockIdx.x * WAVES + wave; v < nv; v += vstride) {
    uint32_t mn[MAX_NHASH], mx[MAX_NHASH];
#pragma unroll
    for (int t = 0; t < MAX_NHASH; t++) {
      mn[t] = 0xFFFFFFFFu;
      mx[t] = 0u;
    }
    bool any = false;
    if (colors[v] < 0) {
      const int64_t e0 = rowptr[v], e1 = rowptr[v + 1];
      const int64_t vg = v + base;
      for (int64_t e = e0 + lane; e < e1; e += 64) {
        const int32_t td = tails[e];
        if (!uncolored[td]) continue;
        const int64_t g = gid_all[td];
        if (g == vg) continue;
        any = true;
        for (int t = 0; t < n_hash; t++) {
          const uint32_t h = color_hash((uint32_t)g, (uint32_t)seeds[t]);
          mn[t] = h < mn[t] ? h : mn[t];
          mx[t] = h > mx[t] ? h : mx[t];
        }
      }
    }
    any = __any(any);
    for (int t = 0; t < n_hash; t++) {
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        const uint32_t omn = (uint32_t)__shfl_down((int)mn[t], off, 64);
        const uint32_t omx = (uint32_t)__shfl_down((int)mx[t], off, 64);
        mn[t] = omn < mn[t] ? omn : mn[t];
        mx[t] = omx > mx[t] ? omx : mx[t];
      }
    }
    if (lane == 0) {
      for (int t = 0; t < n_hash; t++) {
        // no competitor: same sentinels as the torch path (1<<33 / -1)
        mn_out[v * n_hash + t] =
            any ? (int64_t)mn[t] : ((int64_t)1 << 33);
        mx_out[v * n_hash + t] = any ? (int64_t)mx[t] : (int64_t)-1;
      }
    }
  }
}

void launch_coloring_minmax(const int64_t* rowptr, const int32_t* tails,
                            const int64_t* gid_all, const bool* uncolored,
                            const int64_t* colors, int64_t nv, int64_t base,
                            const int64_t* seeds, int n_hash, int64_t* mn_out,
                            int64_t* mx_out, hipStream_t stream) {
  if (nv == 0) return;
  constexpr int BLOCK = 256;
  constexpr int WAVES = BLOCK / 64;
  int64_t grid = (nv + WAVES - 1) / WAVES;
  if (grid > 1048576) grid = 1048576;
  hipLaunchKernelGGL((coloring_minmax_kernel<BLOCK>), dim3((uint32_t)grid),
                     dim3(BLOCK), 0, stream, rowptr, tails, gid_all,
                     uncolored, colors, nv, base, seeds, n_hash, mn_out,
                     mx_out);
}

// Per-vertex weighted degree: one wave per vertex, lane-strided row scan
// (ref distSumVertexDegree, louvain.cpp:2126-2151).
template <typename W, int BLOCK>
__global__ __launch_bounds__(BLOCK) void row_sum_kernel(
    const int64_t* __restrict__ rowptr, const W* __restrict__ weights,
    int64_t nv, W* __restrict__ out) {
  constexpr int WAVES = BLOCK / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  // wave-strided over vertices: grid stays under the 2^32-1 workitem cap
  const int64_t vstride = (int64_t)gridDim.x * WAVES;
  for (int64_t v = (int64_t)blockIdx.x * WAVES + wave; v < nv; v += vstride) {
    const int64_t e0 = rowptr[v], e1 = rowptr[v + 1];
    double acc = 0.0;
    for (int64_t e = e0 + lane; e < e1; e += 64) acc += (double)weights[e];
    acc = sum_reduce<64>(acc);
    if (lane == 0) out[v] = (W)acc;
  }
}

// ---------------------------------------------------------------------------
// Hub candidate generation via rocPRIM (CUVITE_HUB_SEGSORT experiment):
// per-hub segmented radix sort of the community keys (only ceil(log2 C) bits)
// + reduce_by_key for the (hub, community) weight sums. The torch path sorts
// full 64-bit packed keys; narrow-bit segmented sort does ~half the radix
// passes.
// ---------------------------------------------------------------------------

__global__ void gather_comm_kernel(const int32_t* __restrict__ tails_flat,
                                   const int32_t* __restrict__ curr_comm,
                                   int64_t n, int32_t* __restrict__ keys) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    keys[i] = curr_comm[tails_flat[i]];
}

__global__ void pack_key64_kernel(const int32_t* __restrict__ keys,
                                  const int32_t* __restrict__ seg_flat,
                                  int64_t n, int64_t C,
                                  int64_t* __restrict__ key64) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    key64[i] = (int64_t)seg_flat[i] * C + keys[i];
}

// dQ argmax over the per-hub (sorted, deduped) candidate ranges produced by
// reduce_by_key. Two facts make this cheap: (a) keys are UNIQUE, so the
// weight to the own community wcc is a single binary search, not a range
// scan; (b) the argmax over a hub's range is split across ARGMAX_SPLITS
// blocks so an R-MAT mega-hub (10^6 candidates) cannot serialize on one
// wave — rocprof at s26 showed the wave-per-hub version at 20-45 ms/sweep
// from exactly that imbalance (profiles/round2_kernel_stats).

constexpr int ARGMAX_SPLITS = 8;

DEV_INLINE int64_t lower_bound64(const int64_t* __restrict__ a, int64_t n,
                                 int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    const int64_t m = (lo + hi) >> 1;
    if (a[m] < key) lo = m + 1; else hi = m;
  }
  return lo;
}

template <typename W, int BLOCK>
__global__ __launch_bounds__(BLOCK) void hub_argmax_part_kernel(
    const int64_t* __restrict__ uniq, const W* __restrict__ sums,
    const int32_t* __restrict__ cnt_ptr, int64_t C,
    const int32_t* __restrict__ hubs, int nhub,
    const double* __restrict__ hub_self,
    const int32_t* __restrict__ curr_comm, const W* __restrict__ v_degree,
    const W* __restrict__ comm_degree, const int64_t* __restrict__ comm_gid,
    double constant, double* __restrict__ p_gain,
    int64_t* __restrict__ p_gid, int32_t* __restrict__ p_dense) {
  constexpr int WAVES = BLOCK / 64;
  __shared__ double red_gain[WAVES];
  __shared__ int64_t red_gid[WAVES];
  __shared__ int32_t red_dense[WAVES];
  const int hidx = blockIdx.x / ARGMAX_SPLITS;
  const int part = blockIdx.x % ARGMAX_SPLITS;
  const int64_t n = (int64_t)cnt_ptr[0];
  const int32_t v = hubs[hidx];
  const int32_t cc = curr_comm[v];
  const int64_t keylo = (int64_t)hidx * C;
  const int64_t r0 = lower_bound64(uniq, n, keylo);
  const int64_t r1 = lower_bound64(uniq, n, keylo + C);
  // unique keys: the own-community weight is one lookup
  const int64_t pcc = lower_bound64(uniq, n, keylo + cc);
  const double wcc =
      (pcc < n && uniq[pcc] == keylo + cc) ? (double)sums[pcc] : 0.0;
  const double eix = wcc - hub_self[hidx];
  const double vdeg = (double)v_degree[v];
  const double ax = (double)comm_degree[cc] - vdeg;
  const int64_t len = r1 - r0;
  const int64_t s0 = r0 + part * len / ARGMAX_SPLITS;
  const int64_t s1 = r0 + (part + 1) * len / ARGMAX_SPLITS;
  Best best{0.0, comm_gid[cc], cc};
  for (int64_t i = s0 + threadIdx.x; i < s1; i += BLOCK) {
    const int32_t y = (int32_t)(uniq[i] - keylo);
    if (y == cc) continue;
    const double eiy = (double)sums[i];
    const double ay = (double)comm_degree[y];
    const double g = 2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant;
    best_combine(best, g, comm_gid[y], y);
  }
  best_reduce<64>(best);
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  if (lane == 0) {
    red_gain[wave] = best.gain;
    red_gid[wave] = best.gid;
    red_dense[wave] = best.dense;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int i = 1; i < WAVES; i++)
      best_combine(best, red_gain[i], red_gid[i], red_dense[i]);
    p_gain[blockIdx.x] = best.gain;
    p_gid[blockIdx.x] = best.gid;
    p_dense[blockIdx.x] = best.dense;
  }
}

template <typename W, int BLOCK>
__global__ __launch_bounds__(BLOCK) void hub_argmax_final_kernel(
    const int64_t* __restrict__ uniq, const W* __restrict__ sums,
    const int32_t* __restrict__ cnt_ptr, int64_t C,
    const int32_t* __restrict__ hubs, int nhub,
    const int32_t* __restrict__ curr_comm,
    const int64_t* __restrict__ comm_size, const int64_t* __restrict__ comm_gid,
    const double* __restrict__ p_gain, const int64_t* __restrict__ p_gid,
    const int32_t* __restrict__ p_dense, int32_t* __restrict__ target_hub,
    W* __restrict__ cw_hub) {
  constexpr int WAVES = BLOCK / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int hidx = blockIdx.x * WAVES + wave;
  if (hidx >= nhub) return;
  const int64_t n = (int64_t)cnt_ptr[0];
  const int32_t v = hubs[hidx];
  const int32_t cc = curr_comm[v];
  Best best{0.0, comm_gid[cc], cc};
  if (lane < ARGMAX_SPLITS) {
    const int p = hidx * ARGMAX_SPLITS + lane;
    best = Best{p_gain[p], p_gid[p], p_dense[p]};
  }
  best_reduce<64>(best);
  if (lane == 0) {
    int32_t tgt = best.dense;
    if (comm_size[tgt] == 1 && comm_size[cc] == 1 && best.gid > comm_gid[cc])
      tgt = cc;
    target_hub[hidx] = tgt;
    const int64_t keycc = (int64_t)hidx * C + cc;
    const int64_t pcc = lower_bound64(uniq, n, keycc);
    cw_hub[hidx] =
        (pcc < n && uniq[pcc] == keycc) ? sums[pcc] : (W)0;
  }
}

// ------------------------------- launchers ---------------------------------

static int grid_for(int64_t n, int block) {
  int64_t g = (n + block - 1) / block;
  // >> 2048 workgroups keeps all 8 XCDs fed; cap to bound tail latency
  return (int)(g < 1 ? 1 : (g > 65535 ? 65535 : g));
}


template <typename W>
struct MoveArgs {
  const int64_t* rowptr;
  const int32_t* tails;
  const W* weights;
  const int32_t* curr_comm;
  const W* v_degree;
  const int64_t* comm_size;
  const W* comm_degree;
  const int64_t* comm_gid;
  double constant;
  int32_t* target;
  W* cluster_weight;
};

template <typename W, int LANES, int CAP>
void launch_sub(const int32_t* vlist, int nlist, const MoveArgs<W>& a,
                hipStream_t stream) {
  constexpr int BLOCK = 256;
  constexpr int GROUPS = BLOCK / LANES;
  const int grid = (nlist + GROUPS - 1) / GROUPS;
  const size_t shmem = (size_t)GROUPS * CAP * (sizeof(W) + sizeof(int32_t));
  auto kern = lv_move_sub<W, LANES, CAP, BLOCK>;
  if (shmem > 65536)
    (void)hipFuncSetAttribute((const void*)kern,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)shmem);
  hipLaunchKernelGGL(kern, dim3(grid), dim3(BLOCK), shmem, stream, vlist,
                     nlist, a.rowptr, a.tails, a.weights, a.curr_comm,
                     a.v_degree, a.comm_size, a.comm_degree, a.comm_gid,
                     a.constant, a.target, a.cluster_weight);
}

template <typename W, int CAP>
void launch_block(const int32_t* vlist, int nlist, const MoveArgs<W>& a,
                  hipStream_t stream) {
  constexpr int BLOCK = 256;
  const size_t shmem = (size_t)CAP * (sizeof(W) + sizeof(int32_t));
  auto kern = lv_move_block<W, CAP, BLOCK>;
  if (shmem > 65536)
    (void)hipFuncSetAttribute((const void*)kern,
                              hipFuncAttributeMaxDynamicSharedMemorySize,
                              (int)shmem);
  hipLaunchKernelGGL(kern, dim3(nlist), dim3(BLOCK), shmem, stream, vlist,
                     nlist, a.rowptr, a.tails, a.weights, a.curr_comm,
                     a.v_degree, a.comm_size, a.comm_degree, a.comm_gid,
                     a.constant, a.target, a.cluster_weight);
}

// explicit instantiations used by bindings.cpp
#define INSTANTIATE(W)                                                        \
  template void launch_sub<W, 16, 32>(const int32_t*, int, const MoveArgs<W>&,\
                                      hipStream_t);                           \
  template void launch_sub<W, 64, 128>(const int32_t*, int,                  \
                                       const MoveArgs<W>&, hipStream_t);     \
  template void launch_sub<W, 64, 512>(const int32_t*, int,                  \
                                       const MoveArgs<W>&, hipStream_t);     \
  template void launch_sub<W, 64, 1024>(const int32_t*, int,                 \
                                        const MoveArgs<W>&, hipStream_t);    \
  template void launch_block<W, 2048>(const int32_t*, int,                  \
                                      const MoveArgs<W>&, hipStream_t);      \
  template void launch_block<W, 4096>(const int32_t*, int,                  \
                                      const MoveArgs<W>&, hipStream_t);      \
  template void launch_block<W, 8192>(const int32_t*, int,                  \
                                      const MoveArgs<W>&, hipStream_t);

INSTANTIATE(float)
INSTANTIATE(double)

template <typename W>
void launch_modularity(const W* cw, const W* cd, int64_t nv, double* out,
                       hipStream_t stream) {
  constexpr int BLOCK = 256;
  int grid = (int)((nv + BLOCK - 1) / BLOCK);
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL((modularity_reduce<W, BLOCK>), dim3(grid), dim3(BLOCK), 0,
                     stream, cw, cd, nv, out);
}

template void launch_modularity<float>(const float*, const float*, int64_t,
                                       double*, hipStream_t);
template void launch_modularity<double>(const double*, const double*, int64_t,
                                        double*, hipStream_t);

template <typename W>
void launch_scatter_add(W* out, const int64_t* idx, const W* val, int64_t n,
                        hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL((scatter_add_kernel<W>), dim3(grid_for(n, 256)),
                     dim3(256), 0, stream, out, idx, val, n);
}
template void launch_scatter_add<float>(float*, const int64_t*, const float*,
                                        int64_t, hipStream_t);
template void launch_scatter_add<double>(double*, const int64_t*,
                                         const double*, int64_t, hipStream_t);

// Fresh recount of the community aggregates from scratch: one pass, two
// atomics per vertex. At world=1 in a heavy-move regime this beats the
// delta update (4 atomics per MOVED vertex; rocprof: 25 ms vs ~12 at s26
// where most vertices move every oscillating sweep). The distributed path
// keeps deltas (owners need only the cross-rank changes).
template <typename W>
__global__ void recount_kernel(const int64_t* __restrict__ labels,
                               const W* __restrict__ v_degree, int64_t nv,
                               int64_t base, int64_t ncomm,
                               int64_t* __restrict__ size,
                               W* __restrict__ degree) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += stride) {
    const int64_t c = labels[i] - base;
    // world-1 labels are local by construction; the bounds check is a
    // memory-safety guard (an OOB label must not corrupt memory)
    if (c < 0 || c >= ncomm) continue;
    atomicAdd((unsigned long long*)&size[c], 1ull);
    unsafeAtomicAdd(&degree[c], v_degree[i]);
  }
}

template <typename W>
void launch_recount(const int64_t* labels, const W* v_degree, int64_t nv,
                    int64_t base, int64_t ncomm, int64_t* size, W* degree,
                    hipStream_t stream) {
  if (nv == 0) return;
  hipLaunchKernelGGL((recount_kernel<W>), dim3(grid_for(nv, 256)), dim3(256),
                     0, stream, labels, v_degree, nv, base, ncomm, size,
                     degree);
}
template void launch_recount<float>(const int64_t*, const float*, int64_t,
                                    int64_t, int64_t, int64_t*, float*,
                                    hipStream_t);
template void launch_recount<double>(const int64_t*, const double*, int64_t,
                                     int64_t, int64_t, int64_t*, double*,
                                     hipStream_t);

template <typename W>
void launch_apply_deltas(const int64_t* target, const int64_t* curr,
                         const W* v_degree, int64_t nv, int64_t base,
                         int64_t bound, int64_t* local_size, W* local_degree,
                         hipStream_t stream) {
  if (nv == 0) return;
  hipLaunchKernelGGL((apply_deltas_kernel<W>), dim3(grid_for(nv, 256)),
                     dim3(256), 0, stream, target, curr, v_degree, nv, base,
                     bound, local_size, local_degree);
}
template void launch_apply_deltas<float>(const int64_t*, const int64_t*,
                                         const float*, int64_t, int64_t,
                                         int64_t, int64_t*, float*,
                                         hipStream_t);
template void launch_apply_deltas<double>(const int64_t*, const int64_t*,
                                          const double*, int64_t, int64_t,
                                          int64_t, int64_t*, double*,
                                          hipStream_t);

void launch_degree_count(const int64_t* src, int64_t n, int64_t base,
                         int32_t* cnt, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(degree_count_kernel, dim3(grid_for(n, 256)), dim3(256), 0,
                     stream, src, n, base, cnt);
}

template <typename W>
void launch_csr_place(const int64_t* src, const int64_t* dst, const W* w,
                      int64_t n, int64_t base, const int64_t* rowptr,
                      int32_t* cursor, int64_t* tails_out, W* w_out,
                      hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL((csr_place_kernel<W>), dim3(grid_for(n, 256)), dim3(256),
                     0, stream, src, dst, w, n, base, rowptr, cursor,
                     tails_out, w_out);
}
template void launch_csr_place<float>(const int64_t*, const int64_t*,
                                      const float*, int64_t, int64_t,
                                      const int64_t*, int32_t*, int64_t*,
                                      float*, hipStream_t);
template void launch_csr_place<double>(const int64_t*, const int64_t*,
                                       const double*, int64_t, int64_t,
                                       const int64_t*, int32_t*, int64_t*,
                                       double*, hipStream_t);

void launch_gather_comm(const int32_t* tails_flat, const int32_t* curr_comm,
                        int64_t n, int32_t* keys, hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(gather_comm_kernel, dim3(grid_for(n, 256)), dim3(256), 0,
                     stream, tails_flat, curr_comm, n, keys);
}

void launch_pack_key64(const int32_t* keys, const int32_t* seg_flat,
                       int64_t n, int64_t C, int64_t* key64,
                       hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(pack_key64_kernel, dim3(grid_for(n, 256)), dim3(256), 0,
                     stream, keys, seg_flat, n, C, key64);
}

template <typename W>
void launch_hub_argmax(const int64_t* uniq, const W* sums,
                       const int32_t* cnt, int64_t C, const int32_t* hubs,
                       int nhub, const double* hub_self,
                       const int32_t* curr_comm, const W* v_degree,
                       const int64_t* comm_size, const W* comm_degree,
                       const int64_t* comm_gid, double constant,
                       double* p_gain, int64_t* p_gid, int32_t* p_dense,
                       int32_t* target_hub, W* cw_hub, hipStream_t stream) {
  if (nhub == 0) return;
  constexpr int BLOCK = 256;
  constexpr int WAVES = BLOCK / 64;
  hipLaunchKernelGGL((hub_argmax_part_kernel<W, BLOCK>),
                     dim3(nhub * ARGMAX_SPLITS), dim3(BLOCK), 0, stream,
                     uniq, sums, cnt, C, hubs, nhub, hub_self, curr_comm,
                     v_degree, comm_degree, comm_gid, constant, p_gain,
                     p_gid, p_dense);
  hipLaunchKernelGGL((hub_argmax_final_kernel<W, BLOCK>),
                     dim3((nhub + WAVES - 1) / WAVES), dim3(BLOCK), 0, stream,
                     uniq, sums, cnt, C, hubs, nhub, curr_comm, comm_size,
                     comm_gid, p_gain, p_gid, p_dense, target_hub, cw_hub);
}
template void launch_hub_argmax<float>(const int64_t*, const float*,
                                       const int32_t*, int64_t,
                                       const int32_t*, int, const double*,
                                       const int32_t*, const float*,
                                       const int64_t*, const float*,
                                       const int64_t*, double, double*,
                                       int64_t*, int32_t*, int32_t*,
                                       float*, hipStream_t);
template void launch_hub_argmax<double>(const int64_t*, const double*,
                                        const int32_t*, int64_t,
                                        const int32_t*, int, const double*,
                                        const int32_t*, const double*,
                                        const int64_t*, const double*,
                                        const int64_t*, double, double*,
                                        int64_t*, int32_t*, int32_t*,
                                        double*, hipStream_t);

int hub_argmax_splits() { return ARGMAX_SPLITS; }

template <typename W>
void launch_row_sum(const int64_t* rowptr, const W* weights, int64_t nv,
                    W* out, hipStream_t stream) {
  if (nv == 0) return;
  constexpr int BLOCK = 256;
  constexpr int WAVES = BLOCK / 64;
  int64_t grid = (nv + WAVES - 1) / WAVES;
  if (grid > 1048576) grid = 1048576;  // stay under 2^32-1 workitems
  hipLaunchKernelGGL((row_sum_kernel<W, BLOCK>), dim3((uint32_t)grid),
                     dim3(BLOCK), 0, stream, rowptr, weights, nv, out);
}
template void launch_row_sum<float>(const int64_t*, const float*, int64_t,
                                    float*, hipStream_t);
template void launch_row_sum<double>(const int64_t*, const double*, int64_t,
                                     double*, hipStream_t);

// rocPRIM wrappers (two-phase: bytes query with temp==nullptr, then run).
template <typename W>
void segsort_pairs(void* temp, size_t* bytes, const int32_t* keys_in,
                   int32_t* keys_out, const W* vals_in, W* vals_out,
                   int64_t n, int nseg, const int64_t* offs, int end_bit,
                   hipStream_t stream) {
  (void)rocprim::segmented_radix_sort_pairs(
      temp, *bytes, keys_in, keys_out, vals_in, vals_out, (size_t)n,
      (unsigned)nseg, offs, offs + 1, 0u, (unsigned)end_bit, stream);
}
template void segsort_pairs<float>(void*, size_t*, const int32_t*, int32_t*,
                                   const float*, float*, int64_t, int,
                                   const int64_t*, int, hipStream_t);
template void segsort_pairs<double>(void*, size_t*, const int32_t*, int32_t*,
                                    const double*, double*, int64_t, int,
                                    const int64_t*, int, hipStream_t);

// Narrow-bit global radix sort of packed (src, dst) coarse-edge keys with
// their weights (coarsening aggregate; end_bit = bits of gnc^2, ~50 at s26
// vs the 64 bits a generic int64 sort pays).
template <typename W>
void sort_pairs64(void* temp, size_t* bytes, const int64_t* keys_in,
                  int64_t* keys_out, const W* vals_in, W* vals_out,
                  int64_t n, int end_bit, hipStream_t stream) {
  (void)rocprim::radix_sort_pairs(temp, *bytes, keys_in, keys_out, vals_in,
                                  vals_out, (size_t)n, 0u,
                                  (unsigned)end_bit, stream);
}
template void sort_pairs64<float>(void*, size_t*, const int64_t*, int64_t*,
                                  const float*, float*, int64_t, int,
                                  hipStream_t);
template void sort_pairs64<double>(void*, size_t*, const int64_t*, int64_t*,
                                   const double*, double*, int64_t, int,
                                   hipStream_t);

template <typename W>
void reduce_by_key64(void* temp, size_t* bytes, const int64_t* keys,
                     const W* vals, int64_t n, int64_t* uniq_out, W* sums_out,
                     unsigned int* count_out, hipStream_t stream) {
  (void)rocprim::reduce_by_key(temp, *bytes, keys, vals, (size_t)n, uniq_out,
                               sums_out, count_out, rocprim::plus<W>(),
                               rocprim::equal_to<int64_t>(), stream);
}
template void reduce_by_key64<float>(void*, size_t*, const int64_t*,
                                     const float*, int64_t, int64_t*, float*,
                                     unsigned int*, hipStream_t);
template void reduce_by_key64<double>(void*, size_t*, const int64_t*,
                                      const double*, int64_t, int64_t*,
                                      double*, unsigned int*, hipStream_t);

}  // namespace cuvite
