// MI355X (gfx950, CDNA4) kernels for the parameter-server framework.
//
// These are the MI355X-native replacements for the reference's JVM hot
// loops:
//   gather             <- SimplePSLogic.onPullRecv answer assembly  (K3)
//   apply_add          <- SimplePSLogic.onPushRecv, paramUpdate='+' (K1)
//   apply_adagrad      <- new capability (configs[3])               (K2)
//   hash_init[_rows]   <- RangedRandomFactorInitializer             (K6)
//   mf_sgd_step        <- SGDUpdater.delta per rating               (K4)
//   pa_step/_mc        <- passive/aggressive/algorithm margin+tau   (K7)
//   lr_step            <- new capability: sparse logistic reg       (K9)
//   sgns_step          <- new capability: word2vec SGNS             (K8)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes, hard-coded; sub-wave lane groups of GP lanes
//    (power of two <= 64) process one sample each, reducing dots with
//    __shfl_xor across the group.
//  * The MF/SGNS hot path works on KEY-SORTED interleaved records
//    rec[j] = int4{key, slot, val_bits, 0}: one 16-B load per sample
//    replaces three 4-B strided loads (and the pack writes one 16-B
//    store instead of three scattered 4-B stores).  Sorting is a
//    counting sort for small key ranges (LDS histogram) or rocPRIM
//    radix (torch::argsort) otherwise.
//  * The per-sample coefficient (MF: e = r - u.v; SGNS: g = lr*(y -
//    sigmoid(c.x))) is computed INSIDE the first segment walk: the
//    keyed row is loaded once per segment into registers, so the
//    former standalone pair kernel and its gathers disappear; the
//    coefficient is written back into rec.val for the second pass.
//  * the walks read their records from LDS: each block cooperatively
//    stages its chunks (block-contiguous samples) with coalesced loads,
//    removing the dependent global record load from the latency chain
//    (measured +8.5% on the MF32 step).
//  * duplicate keys accumulate in registers per (segment x chunk) and
//    flush with ONE atomicAdd per column — zipfian hot keys otherwise
//    serialize device-scope atomics (measured: first-cut one-kernel MF
//    step ran at ~300 GB/s — profiles/mf32_r01_baseline_atomics.md).
//    REPS replica spreading (blockIdx & rep_mask) splits the hottest
//    rows across up to 8 buffer copies the host sums; near-unique keys
//    take the opposite path — a segment provably interior to one chunk
//    flushes with plain vector stores.
//  * epoch-resweep workloads (reuse_tag) cache the sort positions per
//    batch: revisits replace histogram + scan + counting sort with one
//    scatter pass per side (measured +70% on the MF32 step).
//  * numerics: f32 master values end to end; the semantics oracle is
//    ops/reference.py and tests compare against it elementwise.  The
//    MF L2 term applies analytically from per-row counts (race-free;
//    see k_scale_rows_by_count), and mean_users gives per-row averaged
//    mini-batch updates for stability under skewed keys.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <climits>
#include <cstdlib>
#include <map>
#include <tuple>

#include <rocprim/rocprim.hpp>

#define WAVE 64
#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be a device tensor")
#define CHECK_CONT(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static inline int grid_for(int64_t work, int block) {
  int64_t g = (work + block - 1) / block;
  const int64_t cap = 2048;  // 256 CUs x 8 blocks; grid-stride the rest
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

// ---------------------------------------------------------------------
// splitmix64 — bit-identical to ops/reference.py::splitmix64
// ---------------------------------------------------------------------
__device__ __forceinline__ uint64_t splitmix64(uint64_t z) {
  z += 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

__device__ __forceinline__ float hash_uniform(uint64_t key, float lo,
                                              float hi) {
  float u24 = (float)(splitmix64(key) >> 40);       // [0, 2^24)
  return lo + (hi - lo) * (u24 * (1.0f / 16777216.0f));
}

// ---------------------------------------------------------------------
// K6: deterministic per-id init
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_hash_init_rows(
    float* __restrict__ out, int64_t rows, int64_t dim, int64_t first_id,
    int64_t stride, float lo, float hi, int64_t seed) {
  int64_t total = rows * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / dim, c = i - r * dim;
    uint64_t key = (uint64_t)((first_id + r * stride) * dim + c +
                              seed * 1000003);
    out[i] = hash_uniform(key, lo, hi);
  }
}

__global__ __launch_bounds__(256) void k_hash_init_ids(
    float* __restrict__ out, const int64_t* __restrict__ ids, int64_t n,
    int64_t dim, float lo, float hi, int64_t seed) {
  int64_t total = n * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / dim, c = i - r * dim;
    uint64_t key = (uint64_t)(ids[r] * dim + c + seed * 1000003);
    out[i] = hash_uniform(key, lo, hi);
  }
}

// ---------------------------------------------------------------------
// K3: gather — values[j] = table[rows[j]].  One wave-fraction per row,
// float4 path when dim % 4 == 0 (16 B/lane coalesced).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_gather_vec4(
    const float4* __restrict__ table, const int64_t* __restrict__ rows,
    float4* __restrict__ out, int64_t n, int64_t dim4) {
  int64_t total = n * dim4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim4, c = i - j * dim4;
    out[i] = table[rows[j] * dim4 + c];
  }
}

// streaming variant: working set >> L2, so skip cache retention on the
// row traffic (guide: nt on streamed data one CU reads once).  The
// nontemporal builtins want clang vector types, not HIP_vector_type.
typedef float v4f __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(256) void k_gather_vec4_nt(
    const float4* __restrict__ table, const int64_t* __restrict__ rows,
    float4* __restrict__ out, int64_t n, int64_t dim4) {
  int64_t total = n * dim4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim4, c = i - j * dim4;
    v4f v = __builtin_nontemporal_load(
        (const v4f*)&table[rows[j] * dim4 + c]);
    __builtin_nontemporal_store(v, (v4f*)&out[i]);
  }
}

__global__ __launch_bounds__(256) void k_gather(
    const float* __restrict__ table, const int64_t* __restrict__ rows,
    float* __restrict__ out, int64_t n, int64_t dim) {
  int64_t total = n * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim, c = i - j * dim;
    out[i] = table[rows[j] * dim + c];
  }
}

// ---------------------------------------------------------------------
// K1: apply_add — table[rows[j]] += deltas[j]; duplicate rows accumulate
// via device-scope f32 atomics (L2).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_apply_add(
    float* __restrict__ table, const int64_t* __restrict__ rows,
    const float* __restrict__ deltas, int64_t n, int64_t dim) {
  int64_t total = n * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim, c = i - j * dim;
    atomicAdd(&table[rows[j] * dim + c], deltas[i]);
  }
}

// ---------------------------------------------------------------------
// K1 fast path: rows unique within the batch (our model logics always
// push pre-aggregated deltas) -> plain read-add-store, no atomics.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_apply_add_unique_vec4(
    float4* __restrict__ table, const int64_t* __restrict__ rows,
    const float4* __restrict__ deltas, int64_t n, int64_t dim4) {
  int64_t total = n * dim4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim4, c = i - j * dim4;
    int64_t t = rows[j] * dim4 + c;
    float4 a = table[t], d = deltas[i];
    a.x += d.x; a.y += d.y; a.z += d.z; a.w += d.w;
    table[t] = a;
  }
}

__global__ __launch_bounds__(256) void k_apply_add_unique_vec4_nt(
    float4* __restrict__ table, const int64_t* __restrict__ rows,
    const float4* __restrict__ deltas, int64_t n, int64_t dim4) {
  int64_t total = n * dim4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim4, c = i - j * dim4;
    int64_t t = rows[j] * dim4 + c;
    v4f a = __builtin_nontemporal_load((const v4f*)&table[t]);
    v4f d = __builtin_nontemporal_load((const v4f*)&deltas[i]);
    a += d;
    __builtin_nontemporal_store(a, (v4f*)&table[t]);
  }
}

__global__ __launch_bounds__(256) void k_apply_add_unique(
    float* __restrict__ table, const int64_t* __restrict__ rows,
    const float* __restrict__ deltas, int64_t n, int64_t dim) {
  int64_t total = n * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim, c = i - j * dim;
    table[rows[j] * dim + c] += deltas[i];
  }
}

// ---------------------------------------------------------------------
// K2: apply_adagrad on a SORTED batch (host sorts rows, gathers grads).
// Element (j, c) acts only if j is its row's segment leader; the leader
// walks its segment, pre-aggregating duplicate-row gradients, then does
// ONE read-modify-write of (G, W) — the contract of
// ops/reference.py::apply_adagrad.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_apply_adagrad_sorted(
    float* __restrict__ table, float* __restrict__ accum,
    const int64_t* __restrict__ srows, const float* __restrict__ sgrads,
    int64_t n, int64_t dim, float lr, float eps) {
  int64_t total = n * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t j = i / dim, c = i - j * dim;
    int64_t row = srows[j];
    if (j > 0 && srows[j - 1] == row) continue;  // not the leader
    float g = sgrads[i];
    for (int64_t k = j + 1; k < n && srows[k] == row; ++k)
      g += sgrads[k * dim + c];
    float G = accum[row * dim + c] + g * g;
    accum[row * dim + c] = G;
    table[row * dim + c] -= lr * g * rsqrtf(G + eps);
  }
}

// ---------------------------------------------------------------------
// Sub-wave group reduction: sum over a lane group of GP lanes (GP a
// power of two <= 64).
// ---------------------------------------------------------------------
__device__ __forceinline__ float4 ld_f4(const float4* p, bool nt) {
  if (nt) {
    v4f v = __builtin_nontemporal_load((const v4f*)p);
    return make_float4(v.x, v.y, v.z, v.w);
  }
  return *p;
}

template <int GP>
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int off = GP / 2; off > 0; off >>= 1)
    v += __shfl_xor(v, off, GP);
  return v;
}

// =====================================================================
// K10: sorted-record builders.
// rec[j] = int4 { key, slot, val_bits, 0 } grouped by key ascending
// (order within a key unspecified — consumers only need grouping).
// =====================================================================

// histogram of a small-range key array (counting-sort prologue)
template <class TK>
__global__ __launch_bounds__(256) void k_key_hist(
    const TK* __restrict__ keys, int* __restrict__ ghist, int64_t n,
    int m) {
  extern __shared__ int lhist[];
  for (int i = threadIdx.x; i < m; i += blockDim.x) lhist[i] = 0;
  __syncthreads();
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x)
    atomicAdd(&lhist[(int)keys[j]], 1);
  __syncthreads();
  for (int i = threadIdx.x; i < m; i += blockDim.x)
    if (lhist[i]) atomicAdd(&ghist[i], lhist[i]);
}

// both passes' histograms in ONE read of the batch (MF: item keys for
// pass A + user keys for pass B)
template <class TK>
__global__ __launch_bounds__(256) void k_dual_hist(
    const TK* __restrict__ ka, const TK* __restrict__ kb,
    int* __restrict__ ga, int* __restrict__ gb, int64_t n, int ma,
    int mb) {
  extern __shared__ int lds[];
  int* la = lds;
  int* lb = lds + ma;
  for (int i = threadIdx.x; i < ma + mb; i += blockDim.x) lds[i] = 0;
  __syncthreads();
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    atomicAdd(&la[(int)ka[j]], 1);
    atomicAdd(&lb[(int)kb[j]], 1);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < ma; i += blockDim.x)
    if (la[i]) atomicAdd(&ga[i], la[i]);
  for (int i = threadIdx.x; i < mb; i += blockDim.x)
    if (lb[i]) atomicAdd(&gb[i], lb[i]);
}

// exclusive scan of a small histogram in ONE workgroup (the rocPRIM
// lookback scan costs 3 launches + an elementwise subtract for a
// <=8192-element scan; this is one ~2 us launch)
__global__ __launch_bounds__(256) void k_excl_scan_small(
    const int* __restrict__ hist, int* __restrict__ starts, int m) {
  __shared__ int psum[256];
  const int T = 256, t = threadIdx.x;
  const int per = (m + T - 1) / T;
  const int lo = min(m, t * per), hi = min(m, lo + per);
  int s = 0;
  for (int i = lo; i < hi; ++i) s += hist[i];
  psum[t] = s;
  __syncthreads();
  for (int off = 1; off < T; off <<= 1) {
    int v = (t >= off) ? psum[t - off] : 0;
    __syncthreads();
    psum[t] += v;
    __syncthreads();
  }
  int base = t ? psum[t - 1] : 0;
  for (int i = lo; i < hi; ++i) {
    starts[i] = base;
    base += hist[i];
  }
}

// counting-sort pack into interleaved recs: per chunk, an LDS histogram
// + one global cursor reservation per touched key, then ONE int4 store
// per sample (vs 3 scattered 4-B stores of the un-interleaved layout).
template <class TK>
__global__ __launch_bounds__(256) void k_pack_rec(
    const TK* __restrict__ keys, const TK* __restrict__ slots,
    const float* __restrict__ vals, int* __restrict__ gcursor,
    int4* __restrict__ rec, int* __restrict__ pos_out, int64_t n, int m,
    int chunk) {
  extern __shared__ int lds[];
  int* lhist = lds;
  int* lbase = lds + m;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const int64_t start = c * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    for (int i = threadIdx.x; i < m; i += blockDim.x) lhist[i] = 0;
    __syncthreads();
    for (int64_t j = start + threadIdx.x; j < end; j += blockDim.x)
      atomicAdd(&lhist[(int)keys[j]], 1);
    __syncthreads();
    for (int i = threadIdx.x; i < m; i += blockDim.x) {
      int h = lhist[i];
      lbase[i] = h ? atomicAdd(&gcursor[i], h) : 0;
    }
    __syncthreads();
    for (int i = threadIdx.x; i < m; i += blockDim.x) lhist[i] = 0;
    __syncthreads();
    for (int64_t j = start + threadIdx.x; j < end; j += blockDim.x) {
      const int k = (int)keys[j];
      const int lpos = atomicAdd(&lhist[k], 1);
      const int64_t pos = (int64_t)lbase[k] + lpos;
      rec[pos] = make_int4(k, (int)slots[j], __float_as_int(vals[j]), 0);
      if (pos_out) pos_out[j] = (int)pos;
    }
    __syncthreads();
  }
}

// re-key an existing rec array by its slot field (pass B: key<->slot
// swap, val carried through — by now val holds the pass-A coefficient)
__global__ __launch_bounds__(256) void k_pack_rec_swap(
    const int4* __restrict__ src, int* __restrict__ gcursor,
    int4* __restrict__ rec, int* __restrict__ pos_out, int64_t n, int m,
    int chunk) {
  extern __shared__ int lds[];
  int* lhist = lds;
  int* lbase = lds + m;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const int64_t start = c * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    for (int i = threadIdx.x; i < m; i += blockDim.x) lhist[i] = 0;
    __syncthreads();
    for (int64_t j = start + threadIdx.x; j < end; j += blockDim.x)
      atomicAdd(&lhist[src[j].y], 1);
    __syncthreads();
    for (int i = threadIdx.x; i < m; i += blockDim.x) {
      int h = lhist[i];
      lbase[i] = h ? atomicAdd(&gcursor[i], h) : 0;
    }
    __syncthreads();
    for (int i = threadIdx.x; i < m; i += blockDim.x) lhist[i] = 0;
    __syncthreads();
    for (int64_t j = start + threadIdx.x; j < end; j += blockDim.x) {
      const int4 s = src[j];
      const int lpos = atomicAdd(&lhist[s.y], 1);
      const int64_t pos = (int64_t)lbase[s.y] + lpos;
      rec[pos] = make_int4(s.y, s.x, s.z, 0);
      if (pos_out) pos_out[j] = (int)pos;
    }
    __syncthreads();
  }
}

// large key ranges: build UNSORTED recs, then rocPRIM radix-sorts
// (key, rec) pairs directly over just ceil(log2(m)) bits — no argsort
// permutation, no gather epilogue.
template <class TK>
__global__ __launch_bounds__(256) void k_interleave_rec(
    const TK* __restrict__ keys, const TK* __restrict__ slots,
    const float* __restrict__ vals, int* __restrict__ okeys,
    int4* __restrict__ rec, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    const int k = (int)keys[j];
    okeys[j] = k;
    rec[j] = make_int4(k, (int)slots[j], __float_as_int(vals[j]), 0);
  }
}

// swap key<->slot of an existing rec array; emits the new key column
__global__ __launch_bounds__(256) void k_interleave_swap(
    const int4* __restrict__ src, int* __restrict__ okeys,
    int4* __restrict__ rec, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    const int4 s = src[j];
    okeys[j] = s.y;
    rec[j] = make_int4(s.y, s.x, s.z, 0);
  }
}

// cached-position builders: rec[pos[j]] = record j (one scatter pass —
// no histogram, no scan, no cursor atomics).  Valid only while the
// caller guarantees the key/slot tensors' CONTENT is unchanged (the
// epoch-resweep contract; see mf_sgd_step reuse_tag).
template <class TK>
__global__ __launch_bounds__(256) void k_apply_pos(
    const TK* __restrict__ keys, const TK* __restrict__ slots,
    const float* __restrict__ vals, const int* __restrict__ pos,
    int4* __restrict__ rec, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x)
    rec[pos[j]] = make_int4((int)keys[j], (int)slots[j],
                            __float_as_int(vals[j]), 0);
}

__global__ __launch_bounds__(256) void k_apply_pos_swap(
    const int4* __restrict__ src, const int* __restrict__ pos,
    int4* __restrict__ rec, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    const int4 s = src[j];
    rec[pos[j]] = make_int4(s.y, s.x, s.z, 0);
  }
}

// u[r] *= (1 + coef*cnt[r]) — the L2 term of the MF user update,
// applied analytically from per-row occurrence counts BEFORE the
// delta accumulation.  (Folding it into the segment flush read the row
// DURING concurrent chunk flushes — a measured 0.04-scale divergence on
// chunk-spanning segments; this form is race-free and bit-equal to the
// oracle's per-occurrence subtraction.)
// mean-aggregated user update: u = u*(1 - lr*reg*[cnt>0]) + acc/cnt
// (per-key averaged mini-batch — effective per-row step stays lr no
// matter how often the row appears in the batch)
__global__ __launch_bounds__(256) void k_apply_mean_update(
    float* __restrict__ table, const float* __restrict__ acc,
    const int* __restrict__ cnt, int64_t rows, int64_t dim, float decay) {
  int64_t total = rows * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int c = cnt[i / dim];
    if (c) table[i] = table[i] * (1.0f - decay) + acc[i] / c;
  }
}

// touched-rows variant for huge sparse-touched tables: segment leaders
// of the user-sorted recs emit (row, count) pairs, and only those rows
// are decayed — a 10^8-row table must not be scanned per step.
__global__ __launch_bounds__(256) void k_seg_extents(
    const int4* __restrict__ rec, int64_t n, int* __restrict__ rows,
    int* __restrict__ cnts, int* __restrict__ ntouch) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    const int row = rec[j].x;
    if (j > 0 && rec[j - 1].x == row) continue;  // not the leader
    int len = 1;
    while (j + len < n && rec[j + len].x == row) ++len;
    const int idx = atomicAdd(ntouch, 1);
    rows[idx] = row;
    cnts[idx] = len;
  }
}

__global__ __launch_bounds__(256) void k_scale_touched_rows(
    float* __restrict__ table, const int* __restrict__ rows,
    const int* __restrict__ cnts, const int* __restrict__ ntouch,
    int64_t dim, float coef) {
  const int64_t total = (int64_t)(*ntouch) * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t t = i / dim, c = i - t * dim;
    table[(int64_t)rows[t] * dim + c] *= 1.0f + coef * cnts[t];
  }
}

// vec4 variant: one 32-lane group per touched row, float4 nontemporal
// traffic (the 51 GB mf128 user table streams through once — measured
// 29% of the round on the scalar per-element form)
__global__ __launch_bounds__(256) void k_scale_touched_rows_vec4(
    float4* __restrict__ table, const int* __restrict__ rows,
    const int* __restrict__ cnts, const int* __restrict__ ntouch,
    int64_t dim4, float coef) {
  const int GP = 32;
  const int lane = threadIdx.x % GP;
  const int64_t gpb = blockDim.x / GP;
  const int64_t nt = (int64_t)(*ntouch);
  for (int64_t t = (int64_t)blockIdx.x * gpb + threadIdx.x / GP; t < nt;
       t += (int64_t)gridDim.x * gpb) {
    const float f = 1.0f + coef * cnts[t];
    const int64_t base = (int64_t)rows[t] * dim4;
    for (int c = lane; c < dim4; c += GP) {
      v4f v = __builtin_nontemporal_load((const v4f*)&table[base + c]);
      v *= f;
      __builtin_nontemporal_store(v, (v4f*)&table[base + c]);
    }
  }
}

__global__ __launch_bounds__(256) void k_scale_rows_by_count(
    float* __restrict__ table, const int* __restrict__ cnt, int64_t rows,
    int64_t dim, float coef) {
  int64_t total = rows * dim;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / dim;
    const int c = cnt[r];
    if (c) table[i] *= 1.0f + coef * c;
  }
}

// recB[posB[a]].w = a: each pass-B record carries its pass-A position
// in the spare 4th int, so pass B reads its coefficient as
// ecol[rec.w] — a gather from the L2-resident sequential e column (the
// int4 record is already in registers; no extra index load)
__global__ __launch_bounds__(256) void k_pack_wposB(
    int4* __restrict__ recB, const int* __restrict__ posB, int64_t n) {
  for (int64_t a = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; a < n;
       a += (int64_t)gridDim.x * blockDim.x)
    recB[posB[a]].w = (int)a;
}

// pos[order[j]] = j  (inverse of a sort permutation)
__global__ __launch_bounds__(256) void k_invert_order(
    const int64_t* __restrict__ order, int* __restrict__ pos, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x)
    pos[order[j]] = (int)j;
}

static int bits_for(int64_t m) {
  int b = 1;
  while ((1ll << b) < m) ++b;
  return b;
}

// radix-sort recs by the freshly built key column (int32, bit-bounded)
static torch::Tensor radix_sort_recs(torch::Tensor keys,
                                     torch::Tensor rec, int64_t n,
                                     int64_t m) {
  auto opts = keys.options();
  auto keys_out = torch::empty_like(keys);
  auto rec_out = torch::empty_like(rec);
  size_t tmp_bytes = 0;
  hipError_t err = rocprim::radix_sort_pairs(
      nullptr, tmp_bytes, keys.data_ptr<int>(), keys_out.data_ptr<int>(),
      (const int4*)rec.data_ptr<int>(), (int4*)rec_out.data_ptr<int>(),
      n, 0, bits_for(m), cur_stream());
  TORCH_CHECK(err == hipSuccess, "radix_sort_pairs sizing: ",
              hipGetErrorString(err));
  auto tmp = torch::empty({(int64_t)tmp_bytes},
                          opts.dtype(torch::kUInt8));
  err = rocprim::radix_sort_pairs(
      tmp.data_ptr(), tmp_bytes, keys.data_ptr<int>(),
      keys_out.data_ptr<int>(), (const int4*)rec.data_ptr<int>(),
      (int4*)rec_out.data_ptr<int>(), n, 0, bits_for(m), cur_stream());
  TORCH_CHECK(err == hipSuccess, "radix_sort_pairs: ",
              hipGetErrorString(err));
  return rec_out;
}

// =====================================================================
// K4/K8: fused segmented walk over key-sorted recs.
//
// MODE 0 (MF pass A):   g = val - dot(keyed[key], other[slot])
// MODE 1 (SGNS pass C): g = lrc * (val - sigmoid(dot(keyed[key],
//                                                 other[slot])))
// MODE 2 (second pass): g = val (coefficient precomputed by pass A/C)
//
// Each lane group walks one chunk of samples, caching keyed[key] in
// registers for the current segment (so MODE 0/1 compute the per-sample
// dot with NO extra gather), accumulating g * other[slot] in registers,
// and flushing out[key] += scale*acc (+ a*cnt*keyed[key] when SELF —
// the L2-regularization term) with one atomicAdd per column per
// (segment x chunk).  MODE 0/1 write g back into rec.val for the
// second pass.  rep_stride spreads flushes over 8 replica copies of
// `out` (zipfian hot keys).
// =====================================================================
// ``ecol`` non-null (cached-record revisits): MODE 0/1 writes the
// per-sample coefficient SEQUENTIALLY into ecol[j] (coalesced — no
// scattered HBM lines) instead of rec.z, and MODE 2 reads its
// coefficient as ecol[rec.w] (rec.w carries the pass-A position; the
// gather hits the L2-resident e column).  Both record arrays then stay
// fully static across revisits with no scatter rebuilds at all.
template <int GP, int EPL, int MODE, bool SELF, bool ZERO_OUT>
__global__ __launch_bounds__(256) void k_seg_fused(
    float4* __restrict__ out, int4* __restrict__ rec,
    const float4* __restrict__ keyed, const float4* __restrict__ other,
    int64_t n, int64_t dim4, float scale, float a, float lrc, int chunk,
    int64_t rep_stride, int rep_mask, float* __restrict__ ecol,
    bool write_g, bool nt_rows) {
  const int64_t rep_off = (int64_t)(blockIdx.x & rep_mask) * rep_stride;
  const int lane = threadIdx.x % GP;
  const int64_t gpb = blockDim.x / GP;
  const int64_t g0 = (int64_t)blockIdx.x * gpb + threadIdx.x / GP;
  const int64_t gstride = (int64_t)gridDim.x * gpb;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t cidx = g0; cidx < nchunks; cidx += gstride) {
    const int64_t start = cidx * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    float4 acc[EPL], kv[EPL];
    int cnt = 0;
    int cur = rec[start].x;
    // a segment that both BEGINS and ENDS strictly inside this chunk is
    // provably this walk's exclusive property (keys are globally
    // sorted), so its flush is a plain vector store — near-unique key
    // distributions otherwise degenerate into one 4-B-atomic storm per
    // sample (measured: the SGNS context pass was 52% of step time).
    bool left_owned = (start == 0);

    auto load_kv = [&]() {
      if (MODE == 2 && !SELF) return;  // keyed row unused
      const int64_t kr = (int64_t)cur * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        kv[e] = (c < dim4) ? ld_f4(&keyed[kr + c], nt_rows)
                           : make_float4(0, 0, 0, 0);
      }
    };
    auto flush = [&](bool right_owned) {
      const bool excl = left_owned && right_owned;
      const int64_t orow = rep_off + (int64_t)cur * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        if (c < dim4) {
          float4 t = acc[e];
          t.x *= scale; t.y *= scale; t.z *= scale; t.w *= scale;
          if (SELF) {
            const float ac = a * cnt;
            t.x += ac * kv[e].x; t.y += ac * kv[e].y;
            t.z += ac * kv[e].z; t.w += ac * kv[e].w;
          }
          if (excl) {
            if (ZERO_OUT) {
              out[orow + c] = t;
            } else {
              float4 o = out[orow + c];
              o.x += t.x; o.y += t.y; o.z += t.z; o.w += t.w;
              out[orow + c] = o;
            }
          } else {
            float* o = (float*)&out[orow + c];
            atomicAdd(o + 0, t.x); atomicAdd(o + 1, t.y);
            atomicAdd(o + 2, t.z); atomicAdd(o + 3, t.w);
          }
        }
        acc[e] = make_float4(0, 0, 0, 0);
      }
    };

#pragma unroll
    for (int e = 0; e < EPL; ++e) acc[e] = make_float4(0, 0, 0, 0);
    load_kv();
    for (int64_t j = start; j < end; ++j) {
      const int4 rj = rec[j];
      if (rj.x != cur) {
        flush(true);  // key changed inside the chunk: right-bounded
        left_owned = true;
        cur = rj.x;
        cnt = 0;
        load_kv();
      }
      const int64_t sr = (int64_t)rj.y * dim4;
      float4 sv[EPL];
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        sv[e] = (c < dim4) ? ld_f4(&other[sr + c], nt_rows)
                           : make_float4(0, 0, 0, 0);
      }
      float g;
      if (MODE == 2) {
        g = ecol ? ecol[rj.w] : __int_as_float(rj.z);
      } else {
        float dot = 0.f;
#pragma unroll
        for (int e = 0; e < EPL; ++e)
          dot += kv[e].x * sv[e].x + kv[e].y * sv[e].y +
                 kv[e].z * sv[e].z + kv[e].w * sv[e].w;
        dot = group_sum<GP>(dot);
        g = (MODE == 0)
                ? (__int_as_float(rj.z) - dot)
                : lrc * (__int_as_float(rj.z) -
                         1.0f / (1.0f + __expf(-dot)));
        if (!write_g) {
          // pass-B-recomputes-e configuration: g is consumed only by
          // this walk's own accumulation — nothing to persist
        } else if (ecol) {
          // every lane holds g (the group_sum broadcast); rotate the
          // emitting lane so consecutive records' SEQUENTIAL stores
          // coalesce across the group
          if (lane == ((int)j & (GP - 1)))
            ecol[j] = g;
        } else if (lane == 0) {
          rec[j].z = __float_as_int(g);
        }
      }
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        acc[e].x += g * sv[e].x; acc[e].y += g * sv[e].y;
        acc[e].z += g * sv[e].z; acc[e].w += g * sv[e].w;
      }
      ++cnt;
    }
    flush(end == n);  // right-bounded only at the array end
  }
}

// LDS-staged variant: the BLOCK cooperatively copies its groups'
// chunks (block-contiguous samples) into LDS with coalesced loads, and
// the walks read records from LDS — the per-iteration dependent global
// record load disappears from the latency chain (guide: stage tiles
// through LDS).  Coefficient write-back (MODE 0/1) still goes to the
// global rec array for the second pass.
template <int GP, int EPL, int MODE, bool SELF, bool ZERO_OUT>
__global__ __launch_bounds__(256) void k_seg_fused_lds(
    float4* __restrict__ out, int4* __restrict__ rec,
    const float4* __restrict__ keyed, const float4* __restrict__ other,
    int64_t n, int64_t dim4, float scale, float a, float lrc, int chunk,
    int64_t rep_stride, int rep_mask, float* __restrict__ ecol,
    bool write_g, bool nt_rows) {
  extern __shared__ int4 srec[];
  const int64_t rep_off = (int64_t)(blockIdx.x & rep_mask) * rep_stride;
  const int lane = threadIdx.x % GP;
  const int gpb = blockDim.x / GP;
  const int my_g = threadIdx.x / GP;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t base = (int64_t)blockIdx.x * gpb; base < nchunks;
       base += (int64_t)gridDim.x * gpb) {
    const int64_t blk_first = base * chunk;
    const int64_t blk_total = min((int64_t)gpb * chunk, n - blk_first);
    __syncthreads();
    for (int64_t t = threadIdx.x; t < blk_total; t += blockDim.x)
      srec[t] = rec[blk_first + t];
    __syncthreads();
    const int64_t cidx = base + my_g;
    if (cidx >= nchunks) continue;
    const int64_t start = cidx * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    const int4* mine = srec + (int64_t)my_g * chunk;
    const int len = (int)(end - start);

    float4 acc[EPL], kv[EPL];
    int cnt = 0;
    int cur = mine[0].x;
    bool left_owned = (start == 0);

    auto load_kv = [&]() {
      if (MODE == 2 && !SELF) return;
      const int64_t kr = (int64_t)cur * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        kv[e] = (c < dim4) ? ld_f4(&keyed[kr + c], nt_rows)
                           : make_float4(0, 0, 0, 0);
      }
    };
    auto flush = [&](bool right_owned) {
      const bool excl = left_owned && right_owned;
      const int64_t orow = rep_off + (int64_t)cur * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        if (c < dim4) {
          float4 t = acc[e];
          t.x *= scale; t.y *= scale; t.z *= scale; t.w *= scale;
          if (SELF) {
            const float ac = a * cnt;
            t.x += ac * kv[e].x; t.y += ac * kv[e].y;
            t.z += ac * kv[e].z; t.w += ac * kv[e].w;
          }
          if (excl) {
            if (ZERO_OUT) {
              out[orow + c] = t;
            } else {
              float4 o = out[orow + c];
              o.x += t.x; o.y += t.y; o.z += t.z; o.w += t.w;
              out[orow + c] = o;
            }
          } else {
            float* o = (float*)&out[orow + c];
            atomicAdd(o + 0, t.x); atomicAdd(o + 1, t.y);
            atomicAdd(o + 2, t.z); atomicAdd(o + 3, t.w);
          }
        }
        acc[e] = make_float4(0, 0, 0, 0);
      }
    };

#pragma unroll
    for (int e = 0; e < EPL; ++e) acc[e] = make_float4(0, 0, 0, 0);
    load_kv();
    for (int lj = 0; lj < len; ++lj) {
      const int4 rj = mine[lj];
      if (rj.x != cur) {
        flush(true);
        left_owned = true;
        cur = rj.x;
        cnt = 0;
        load_kv();
      }
      const int64_t sr = (int64_t)rj.y * dim4;
      float4 sv[EPL];
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        sv[e] = (c < dim4) ? ld_f4(&other[sr + c], nt_rows)
                           : make_float4(0, 0, 0, 0);
      }
      float g;
      if (MODE == 2) {
        g = ecol ? ecol[rj.w] : __int_as_float(rj.z);
      } else {
        float dot = 0.f;
#pragma unroll
        for (int e = 0; e < EPL; ++e)
          dot += kv[e].x * sv[e].x + kv[e].y * sv[e].y +
                 kv[e].z * sv[e].z + kv[e].w * sv[e].w;
        dot = group_sum<GP>(dot);
        g = (MODE == 0)
                ? (__int_as_float(rj.z) - dot)
                : lrc * (__int_as_float(rj.z) -
                         1.0f / (1.0f + __expf(-dot)));
        if (!write_g) {
          // nothing to persist (see the plain variant)
        } else if (ecol) {
          if (lane == (lj & (GP - 1)))
            ecol[start + lj] = g;
        } else if (lane == 0) {
          ((int*)&rec[start + lj])[2] = __float_as_int(g);
        }
      }
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        acc[e].x += g * sv[e].x; acc[e].y += g * sv[e].y;
        acc[e].z += g * sv[e].z; acc[e].w += g * sv[e].w;
      }
      ++cnt;
    }
    flush(end == n);
  }
}

// ---------------------------------------------------------------------
// K10b: hash-based id coalescing for unbounded id spaces (the 1e9-
// feature LR config and any space too big for the dense-flag
// coalescer).  Replaces torch.unique (device radix over the full
// batch) + torch.searchsorted (a binary search per element — measured
// 54% of the LR round) with an open-addressing hash table:
//   insert  — linear-probe CAS into a pow2 table, record the probe
//             position per element
//   count   — one ballot-popcount pass over the table (64-slot chunks)
//   scan    — exclusive starts per chunk (single-block scan)
//   assign  — occupied slots take chunk-base + atomic rank; emit uids
//   gather  — slots[j] = slotmap[pos[j]]
// uids come out UNIQUE but UNSORTED (hash-position order).  That is
// sufficient everywhere: the engine's unique=True pull skips
// coalescing, shard bucketing handles any order, and the sort-free
// adagrad apply only needs duplicates adjacent — unique ids have none.
// ---------------------------------------------------------------------
#define HASH_EMPTY (-1ll)

__global__ __launch_bounds__(256) void k_hash_insert(
    const int64_t* __restrict__ ids, int64_t n,
    long long* __restrict__ keys, int* __restrict__ pos_out,
    int64_t mask) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       j < n; j += (int64_t)gridDim.x * blockDim.x) {
    const long long id = (long long)ids[j];
    int64_t p = (int64_t)(splitmix64((uint64_t)id) & (uint64_t)mask);
    for (;;) {
      const long long k = keys[p];
      if (k == id) break;
      if (k == HASH_EMPTY) {
        const long long old = atomicCAS(
            (unsigned long long*)&keys[p],
            (unsigned long long)HASH_EMPTY, (unsigned long long)id);
        if (old == HASH_EMPTY || old == id) break;
      }
      p = (p + 1) & mask;
    }
    pos_out[j] = (int)p;
  }
}

__global__ __launch_bounds__(256) void k_hash_count(
    const long long* __restrict__ keys, int64_t cap,
    int* __restrict__ counts) {
  // one 64-slot chunk per wave-iteration: ballot-popcount, one atomic
  const int lane = threadIdx.x % WAVE;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < cap; i += (int64_t)gridDim.x * blockDim.x) {
    const uint64_t b = __ballot(keys[i] != HASH_EMPTY);
    if (lane == 0 && b)
      atomicAdd(&counts[i >> 10], (int)__popcll(b));
  }
}

__global__ __launch_bounds__(256) void k_hash_assign(
    const long long* __restrict__ keys, int64_t cap,
    int* __restrict__ cursor, int* __restrict__ slotmap,
    int64_t* __restrict__ uids) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < cap; i += (int64_t)gridDim.x * blockDim.x) {
    const long long k = keys[i];
    if (k == HASH_EMPTY) continue;
    const int s = atomicAdd(&cursor[i >> 10], 1);
    slotmap[i] = s;
    uids[s] = (int64_t)k;
  }
}

__global__ __launch_bounds__(256) void k_hash_slots(
    const int* __restrict__ pos, const int* __restrict__ slotmap,
    int* __restrict__ slots, int64_t n) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       j < n; j += (int64_t)gridDim.x * blockDim.x)
    slots[j] = slotmap[pos[j]];
}

// resolve ids against an EXISTING table (pull answers arriving in
// shard-bucketed order resolve their buffer rows this way); ids not in
// the table get slot -1 (foreign answers are skipped by the buffer)
__global__ __launch_bounds__(256) void k_hash_lookup(
    const int64_t* __restrict__ ids, int64_t n,
    const long long* __restrict__ keys, const int* __restrict__ slotmap,
    int64_t mask, int* __restrict__ slots_out) {
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       j < n; j += (int64_t)gridDim.x * blockDim.x) {
    const long long id = (long long)ids[j];
    int64_t p = (int64_t)(splitmix64((uint64_t)id) & (uint64_t)mask);
    int out = -1;
    for (;;) {
      const long long k = keys[p];
      if (k == id) { out = slotmap[p]; break; }
      if (k == HASH_EMPTY) break;
      p = (p + 1) & mask;
    }
    slots_out[j] = out;
  }
}

// ---------------------------------------------------------------------
// K5/K8 prologue: fused SGNS pair expansion — one pass builds the
// (1 positive + k hash-RNG negatives) pair rows, context table ids
// (+V offset) and labels that the Python path assembled with ~10
// elementwise kernels (arange/splitmix/abs/mod/views/fills).  Negative
// id sequence is bit-identical to models/sgns.py's reference path:
// key = p*k + (j-1) + base_key;  neg = |splitmix64(key)| % V.
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_sgns_expand(
    const int64_t* __restrict__ centers,
    const int64_t* __restrict__ contexts, int64_t n, int64_t k,
    int64_t V, int64_t base_key, int64_t* __restrict__ out_centers,
    int64_t* __restrict__ out_ctx, float* __restrict__ out_labels) {
  const int64_t total = n * (k + 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t p = i / (k + 1), j = i - p * (k + 1);
    out_centers[i] = centers[p];
    if (j == 0) {
      out_ctx[i] = contexts[p] + V;
      out_labels[i] = 1.0f;
    } else {
      int64_t h = (int64_t)splitmix64(
          (uint64_t)(p * k + (j - 1) + base_key));
      h = h < 0 ? -h : h;
      out_ctx[i] = h % V + V;
      out_labels[i] = 0.0f;
    }
  }
}

// ---------------------------------------------------------------------
// K7 (binary) / K9: PA and LR share the sparse-margin skeleton.
// One wave per instance; lanes stride the instance's nnz.
// mode: 0 = PA, 1 = PA-I, 2 = PA-II, 3 = logistic regression.
// ---------------------------------------------------------------------
// rep_stride > 0 spreads the delta scatter over 8 replica copies of
// `deltas` (hot zipfian features otherwise serialize same-address
// atomics); the host sums the replicas.
template <class TIdx>
__global__ __launch_bounds__(256) void k_sparse_linear_step(
    const int64_t* __restrict__ indptr,
    const TIdx* __restrict__ feat_slots,
    const float* __restrict__ x_vals, const float* __restrict__ labels,
    const float* __restrict__ weights, float* __restrict__ margins,
    float* __restrict__ deltas, int64_t B, int mode, float C,
    int64_t rep_stride, int rep_mask) {
  const int64_t rep_off = (int64_t)(blockIdx.x & rep_mask) * rep_stride;
  const int lane = threadIdx.x % WAVE;
  const int64_t waves_per_block = blockDim.x / WAVE;
  const int64_t w0 = (int64_t)blockIdx.x * waves_per_block +
                     threadIdx.x / WAVE;
  const int64_t wstride = (int64_t)gridDim.x * waves_per_block;
  for (int64_t i = w0; i < B; i += wstride) {
    const int64_t lo = indptr[i], hi = indptr[i + 1];
    float m = 0.f, n2 = 0.f;
    for (int64_t j = lo + lane; j < hi; j += WAVE) {
      float x = x_vals[j];
      m += weights[(int64_t)feat_slots[j]] * x;
      n2 += x * x;
    }
    m = group_sum<WAVE>(m);
    n2 = group_sum<WAVE>(n2);
    if (lane == 0) margins[i] = m;
    const float y = labels[i];
    if (isnan(y)) continue;  // unlabeled: predict only
    float coeff;             // delta_f = coeff * x_f
    if (mode == 3) {         // logistic: push raw gradient (p - y)
      float p = 1.0f / (1.0f + __expf(-m));
      coeff = p - y;
      if (coeff == 0.f) continue;
    } else {
      float loss = 1.0f - y * m;
      if (loss <= 0.f) continue;
      float sn = n2 > 1e-12f ? n2 : 1e-12f;
      float tau;
      if (mode == 0) tau = loss / sn;
      else if (mode == 1) { tau = loss / sn; tau = tau > C ? C : tau; }
      else tau = loss / (sn + 1.0f / (2.0f * C));
      coeff = tau * y;
    }
    for (int64_t j = lo + lane; j < hi; j += WAVE)
      atomicAdd(&deltas[rep_off + (int64_t)feat_slots[j]],
                coeff * x_vals[j]);
  }
}

// ---------------------------------------------------------------------
// K7b/K9b: sparse-linear step over SLOT-SORTED records (the static-
// batch fast path).  The per-element form gathers weights[slot] per
// nnz element (a random 4-B read = a 64-B line each) and scatters
// per-element delta atomics into an >L2 buffer — measured line
// amplification bound.  Sorted by slot (positions cached per static
// batch), the weight is read ONCE per segment into a register and the
// delta flushes ONCE per (segment x chunk) with exclusive-store/atomic
// selection exactly like the MF walk; the per-instance margin is
// assembled with one L2-resident atomicAdd per element.
//   rec = {slot, instance, x_bits, 0} sorted by slot
//   pass 1: margins[inst] += w[slot] * x      (w register-cached)
//   coeff:  per instance from margin, STATIC ||x||^2 and the label
//   pass 2: deltas[slot] += coeff[inst] * x   (segment-accumulated)
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_lin_margin(
    const int4* __restrict__ rec, const float* __restrict__ weights,
    float* __restrict__ margins, int64_t n, int chunk) {
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t cidx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       cidx < nchunks; cidx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t start = cidx * chunk;
    const int64_t end = start + chunk < n ? start + chunk : n;
    int cur = rec[start].x;
    float w = weights[cur];
    for (int64_t j = start; j < end; ++j) {
      const int4 rj = rec[j];
      if (rj.x != cur) {
        cur = rj.x;
        w = weights[cur];
      }
      atomicAdd(&margins[rj.y], w * __int_as_float(rj.z));
    }
  }
}

// coeff from assembled margins (margins then hold the raw margin for
// the caller's prediction output).  mode: 0/1/2 = PA variants (labels
// +-1; NaN = unlabeled), 3 = logistic (labels 0/1; NaN = unlabeled).
__global__ __launch_bounds__(256) void k_lin_coeff(
    const float* __restrict__ margins, const float* __restrict__ n2,
    const float* __restrict__ labels, float* __restrict__ coeff,
    int64_t B, int mode, float C) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < B; i += (int64_t)gridDim.x * blockDim.x) {
    const float y = labels[i];
    const float m = margins[i];
    float out = 0.f;
    if (!isnan(y)) {
      if (mode == 3) {
        out = 1.0f / (1.0f + __expf(-m)) - y;
      } else {
        float loss = 1.0f - y * m;
        if (loss > 0.f) {
          float sn = n2[i] > 1e-12f ? n2[i] : 1e-12f;
          float tau;
          if (mode == 0) tau = loss / sn;
          else if (mode == 1) { tau = loss / sn; tau = tau > C ? C : tau; }
          else tau = loss / (sn + 1.0f / (2.0f * C));
          out = tau * y;
        }
      }
    }
    coeff[i] = out;
  }
}

__global__ __launch_bounds__(256) void k_lin_delta(
    const int4* __restrict__ rec, const float* __restrict__ coeff,
    float* __restrict__ deltas, int64_t n, int chunk, float sign) {
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t cidx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       cidx < nchunks; cidx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t start = cidx * chunk;
    const int64_t end = start + chunk < n ? start + chunk : n;
    int cur = rec[start].x;
    bool left_owned = (start == 0);
    float acc = 0.f;
    auto flush = [&](bool right_owned) {
      if (left_owned && right_owned) deltas[cur] = acc;
      else atomicAdd(&deltas[cur], acc);
      acc = 0.f;
    };
    for (int64_t j = start; j < end; ++j) {
      const int4 rj = rec[j];
      if (rj.x != cur) {
        flush(true);
        left_owned = true;
        cur = rj.x;
      }
      acc += coeff[rj.y] * __int_as_float(rj.z) * sign;
    }
    flush(end == n);
  }
}

// static per-batch precomputation: n2[i] = sum_j x_j^2 per instance
__global__ __launch_bounds__(256) void k_lin_n2(
    const int64_t* __restrict__ indptr, const float* __restrict__ x,
    float* __restrict__ n2, int64_t B) {
  const int lane = threadIdx.x % WAVE;
  const int64_t wpb = blockDim.x / WAVE;
  for (int64_t i = (int64_t)blockIdx.x * wpb + threadIdx.x / WAVE;
       i < B; i += (int64_t)gridDim.x * wpb) {
    const int64_t lo = indptr[i], hi = indptr[i + 1];
    float s = 0.f;
    for (int64_t j = lo + lane; j < hi; j += WAVE) s += x[j] * x[j];
    s = group_sum<WAVE>(s);
    if (lane == 0) n2[i] = s;
  }
}

// instance column for the record build: seg[j] = its instance index
__global__ __launch_bounds__(256) void k_lin_seg(
    const int64_t* __restrict__ indptr, int* __restrict__ seg,
    int64_t B) {
  const int lane = threadIdx.x % WAVE;
  const int64_t wpb = blockDim.x / WAVE;
  for (int64_t i = (int64_t)blockIdx.x * wpb + threadIdx.x / WAVE;
       i < B; i += (int64_t)gridDim.x * wpb) {
    const int64_t lo = indptr[i], hi = indptr[i + 1];
    for (int64_t j = lo + lane; j < hi; j += WAVE) seg[j] = (int)i;
  }
}

// ---------------------------------------------------------------------
// K7 (multiclass): one wave per instance; class columns are walked in
// tiles of 64 (lane c of tile t owns class t*64+c — any n_classes).
// scores[c] = sum_j w[slot_j][c] * x_j — lanes read consecutive class
// columns of each weight row (coalesced).  The argmax tie-break is
// deterministic (smallest class index wins), and rep_stride > 0
// spreads the delta scatter over 8 replica copies (zipf-hot features
// otherwise serialize same-address atomics — same mitigation as the
// binary kernel).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_pa_multiclass_step(
    const int64_t* __restrict__ indptr,
    const int64_t* __restrict__ feat_slots,
    const float* __restrict__ x_vals, const int64_t* __restrict__ labels,
    const float* __restrict__ weights, float* __restrict__ scores,
    float* __restrict__ deltas, int64_t B, int64_t nc, int mode, float C,
    int64_t rep_stride) {
  const int64_t rep_off = (int64_t)(blockIdx.x & 7) * rep_stride;
  const int lane = threadIdx.x % WAVE;
  const int64_t waves_per_block = blockDim.x / WAVE;
  const int64_t w0 = (int64_t)blockIdx.x * waves_per_block +
                     threadIdx.x / WAVE;
  const int64_t wstride = (int64_t)gridDim.x * waves_per_block;
  for (int64_t i = w0; i < B; i += wstride) {
    const int64_t lo = indptr[i], hi = indptr[i + 1];
    float n2 = 0.f;
    for (int64_t j = lo + lane; j < hi; j += WAVE) {
      float x = x_vals[j];
      n2 += x * x;
    }
    n2 = group_sum<WAVE>(n2);
    const int64_t y = labels[i];
    float sy = 0.f, wrong = -INFINITY;
    int ridx = INT_MAX;
    for (int64_t c0 = 0; c0 < nc; c0 += WAVE) {
      const int64_t c = c0 + lane;
      float sc = 0.f;
      if (c < nc) {
        for (int64_t j = lo; j < hi; ++j)
          sc += weights[feat_slots[j] * nc + c] * x_vals[j];
        scores[i * nc + c] = sc;
      }
      if (y < 0) continue;  // unlabeled: scores only
      if (c0 <= y && y < c0 + WAVE) sy = __shfl(sc, (int)(y - c0));
      float w_t = (c < nc && c != y) ? sc : -INFINITY;
      int r_t = (c < nc) ? (int)c : INT_MAX;
#pragma unroll
      for (int off = WAVE / 2; off > 0; off >>= 1) {
        float o = __shfl_xor(w_t, off);
        int oi = __shfl_xor(r_t, off);
        if (o > w_t || (o == w_t && oi < r_t)) { w_t = o; r_t = oi; }
      }
      if (w_t > wrong || (w_t == wrong && r_t < ridx)) {
        wrong = w_t; ridx = r_t;
      }
    }
    if (y < 0) continue;
    float loss = 1.0f - sy + wrong;
    if (loss <= 0.f) continue;
    float sn = 2.0f * n2;
    sn = sn > 1e-12f ? sn : 1e-12f;
    float tau;
    if (mode == 0) tau = loss / sn;
    else if (mode == 1) { tau = loss / sn; tau = tau > C ? C : tau; }
    else tau = loss / (sn + 1.0f / (2.0f * C));
    for (int64_t j = lo + lane; j < hi; j += WAVE) {
      float t = tau * x_vals[j];
      atomicAdd(&deltas[rep_off + feat_slots[j] * nc + y], t);
      atomicAdd(&deltas[rep_off + feat_slots[j] * nc + ridx], -t);
    }
  }
}

// =====================================================================
// Host wrappers
// =====================================================================

torch::Tensor gather(torch::Tensor table, torch::Tensor rows) {
  CHECK_DEV(table); CHECK_DEV(rows); CHECK_CONT(table); CHECK_CONT(rows);
  TORCH_CHECK(table.dtype() == torch::kFloat32 &&
              rows.dtype() == torch::kInt64);
  int64_t n = rows.numel(), dim = table.size(1);
  auto out = torch::empty({n, dim}, table.options());
  if (!n) return out;
  if (dim % 4 == 0) {
    int64_t dim4 = dim / 4;
    // streamed traffic far beyond L2: nontemporal variant
    bool streaming = n * dim * 4 > (32ll << 20);
    auto* kfn = streaming ? k_gather_vec4_nt : k_gather_vec4;
    hipLaunchKernelGGL(kfn, dim3(grid_for(n * dim4, 256)),
                       dim3(256), 0, cur_stream(),
                       (const float4*)table.data_ptr<float>(),
                       rows.data_ptr<int64_t>(),
                       (float4*)out.data_ptr<float>(), n, dim4);
  } else {
    hipLaunchKernelGGL(k_gather, dim3(grid_for(n * dim, 256)), dim3(256),
                       0, cur_stream(), table.data_ptr<float>(),
                       rows.data_ptr<int64_t>(), out.data_ptr<float>(),
                       n, dim);
  }
  return out;
}

void apply_add(torch::Tensor table, torch::Tensor rows,
               torch::Tensor deltas, bool unique) {
  CHECK_DEV(table); CHECK_DEV(rows); CHECK_DEV(deltas);
  CHECK_CONT(table); CHECK_CONT(rows); CHECK_CONT(deltas);
  int64_t n = rows.numel(), dim = table.size(1);
  TORCH_CHECK(deltas.size(0) == n && deltas.size(1) == dim);
  if (!n) return;
  if (unique && dim % 4 == 0) {
    int64_t dim4 = dim / 4;
    bool streaming = n * dim * 4 > (32ll << 20);
    auto* kfn = streaming ? k_apply_add_unique_vec4_nt
                          : k_apply_add_unique_vec4;
    hipLaunchKernelGGL(kfn, dim3(grid_for(n * dim4, 256)), dim3(256), 0,
                       cur_stream(), (float4*)table.data_ptr<float>(),
                       rows.data_ptr<int64_t>(),
                       (const float4*)deltas.data_ptr<float>(), n, dim4);
    return;
  }
  if (unique) {
    hipLaunchKernelGGL(k_apply_add_unique, dim3(grid_for(n * dim, 256)),
                       dim3(256), 0, cur_stream(),
                       table.data_ptr<float>(), rows.data_ptr<int64_t>(),
                       deltas.data_ptr<float>(), n, dim);
    return;
  }
  hipLaunchKernelGGL(k_apply_add, dim3(grid_for(n * dim, 256)), dim3(256),
                     0, cur_stream(), table.data_ptr<float>(),
                     rows.data_ptr<int64_t>(), deltas.data_ptr<float>(),
                     n, dim);
}

void apply_adagrad_sorted(torch::Tensor table, torch::Tensor accum,
                          torch::Tensor srows, torch::Tensor sgrads,
                          double lr, double eps) {
  CHECK_DEV(table); CHECK_DEV(accum); CHECK_DEV(srows); CHECK_DEV(sgrads);
  CHECK_CONT(table); CHECK_CONT(accum); CHECK_CONT(srows); CHECK_CONT(sgrads);
  int64_t n = srows.numel(), dim = table.size(1);
  if (!n) return;
  hipLaunchKernelGGL(k_apply_adagrad_sorted, dim3(grid_for(n * dim, 256)),
                     dim3(256), 0, cur_stream(), table.data_ptr<float>(),
                     accum.data_ptr<float>(), srows.data_ptr<int64_t>(),
                     sgrads.data_ptr<float>(), n, dim, (float)lr,
                     (float)eps);
}

torch::Tensor hash_init(torch::Tensor ids, int64_t dim, double lo,
                        double hi, int64_t seed) {
  CHECK_DEV(ids); CHECK_CONT(ids);
  int64_t n = ids.numel();
  auto out = torch::empty({n, dim}, torch::dtype(torch::kFloat32)
                                        .device(ids.device()));
  if (!n) return out;
  hipLaunchKernelGGL(k_hash_init_ids, dim3(grid_for(n * dim, 256)),
                     dim3(256), 0, cur_stream(), out.data_ptr<float>(),
                     ids.data_ptr<int64_t>(), n, dim, (float)lo, (float)hi,
                     seed);
  return out;
}

void hash_init_rows(torch::Tensor out, int64_t first_id, int64_t stride,
                    double lo, double hi, int64_t seed) {
  CHECK_DEV(out); CHECK_CONT(out);
  int64_t rows = out.size(0), dim = out.size(1);
  if (!rows) return;
  hipLaunchKernelGGL(k_hash_init_rows, dim3(grid_for(rows * dim, 256)),
                     dim3(256), 0, cur_stream(), out.data_ptr<float>(),
                     rows, dim, first_id, stride, (float)lo, (float)hi,
                     seed);
}

// ---------------------------------------------------------------------
// Record-building helpers (K10).  Counting sort for key range m <=
// PACK_MAX_M (2*m*4 B of LDS), rocPRIM radix (torch::argsort) + gather
// otherwise.  Chunk sizes are env-tunable for on-GPU sweeps.
// ---------------------------------------------------------------------
// counting-sort LDS budget: 2*m*4 B (hist + base) <= 64 KB per WG
static constexpr int PACK_MAX_M = 8192;

static int env_int(const char* name, int dflt) {
  const char* e = std::getenv(name);
  return e ? std::atoi(e) : dflt;
}

static int pack_chunk() {
  static int c = env_int("FPS_PACK_CHUNK", 2048);
  return c;
}

static int seg_chunk() {
  static int c = env_int("FPS_SEG_CHUNK", 128);
  return c;
}

static torch::Tensor exclusive_starts(torch::Tensor hist) {
  auto starts = torch::empty_like(hist);
  hipLaunchKernelGGL(k_excl_scan_small, dim3(1), dim3(256), 0,
                     cur_stream(), hist.data_ptr<int>(),
                     starts.data_ptr<int>(), (int)hist.numel());
  return starts;
}

// recs sorted by `keys`; hist may be precomputed (else computed here).
// When `pos_out` is given (cacheable first pass), the final position of
// every input record is also emitted; the radix path then goes through
// argsort + gather so positions exist.
template <class TK>
static torch::Tensor build_rec_t(torch::Tensor keys, torch::Tensor slots,
                                 torch::Tensor vals, int64_t m,
                                 torch::Tensor hist,
                                 torch::Tensor pos_out) {
  int64_t n = keys.numel();
  auto opts_i32 = torch::dtype(torch::kInt32).device(keys.device());
  auto rec = torch::empty({n, 4}, opts_i32);
  int* pos = pos_out.defined() ? pos_out.data_ptr<int>() : nullptr;
  if (m > 0 && m <= PACK_MAX_M) {
    if (!hist.defined()) {
      hist = torch::zeros({m}, opts_i32);
      int hg = (int)std::min<int64_t>((n + 4095) / 4096, 512);
      hipLaunchKernelGGL((k_key_hist<TK>), dim3(std::max(hg, 1)),
                         dim3(256), (size_t)m * sizeof(int), cur_stream(),
                         keys.data_ptr<TK>(), hist.data_ptr<int>(), n,
                         (int)m);
    }
    auto cursor = exclusive_starts(hist).contiguous();
    int chunk = pack_chunk();
    int64_t nchunks = (n + chunk - 1) / chunk;
    int grid = (int)std::min<int64_t>(nchunks, 2048);
    hipLaunchKernelGGL((k_pack_rec<TK>), dim3(grid), dim3(256),
                       2 * (size_t)m * sizeof(int), cur_stream(),
                       keys.data_ptr<TK>(), slots.data_ptr<TK>(),
                       vals.data_ptr<float>(), cursor.data_ptr<int>(),
                       (int4*)rec.data_ptr<int>(), pos, n, (int)m, chunk);
  } else if (pos) {
    auto keys32 = keys.dtype() == torch::kInt32
                      ? keys : keys.to(torch::kInt32);
    auto order = torch::argsort(keys32);
    hipLaunchKernelGGL(k_invert_order, dim3(grid_for(n, 256)), dim3(256),
                       0, cur_stream(), order.data_ptr<int64_t>(), pos, n);
    hipLaunchKernelGGL((k_apply_pos<TK>), dim3(grid_for(n, 256)),
                       dim3(256), 0, cur_stream(), keys.data_ptr<TK>(),
                       slots.data_ptr<TK>(), vals.data_ptr<float>(), pos,
                       (int4*)rec.data_ptr<int>(), n);
  } else {
    auto keys32 = torch::empty({n}, opts_i32);
    hipLaunchKernelGGL((k_interleave_rec<TK>), dim3(grid_for(n, 256)),
                       dim3(256), 0, cur_stream(), keys.data_ptr<TK>(),
                       slots.data_ptr<TK>(), vals.data_ptr<float>(),
                       keys32.data_ptr<int>(), (int4*)rec.data_ptr<int>(),
                       n);
    rec = radix_sort_recs(keys32, rec, n, m);
  }
  return rec;
}

static torch::Tensor build_rec(torch::Tensor keys, torch::Tensor slots,
                               torch::Tensor vals, int64_t m,
                               torch::Tensor hist = {},
                               torch::Tensor pos_out = {}) {
  if (keys.dtype() == torch::kInt32)
    return build_rec_t<int>(keys, slots, vals, m, hist, pos_out);
  return build_rec_t<int64_t>(keys, slots, vals, m, hist, pos_out);
}

// recs re-keyed by the slot field of an existing rec array
static torch::Tensor build_rec_swap(torch::Tensor src, int64_t m,
                                    torch::Tensor hist = {},
                                    torch::Tensor pos_out = {}) {
  int64_t n = src.size(0);
  auto opts_i32 = torch::dtype(torch::kInt32).device(src.device());
  auto rec = torch::empty({n, 4}, opts_i32);
  int* pos = pos_out.defined() ? pos_out.data_ptr<int>() : nullptr;
  if (m > 0 && m <= PACK_MAX_M) {
    TORCH_CHECK(hist.defined(), "build_rec_swap needs a histogram");
    auto cursor = exclusive_starts(hist).contiguous();
    int chunk = pack_chunk();
    int64_t nchunks = (n + chunk - 1) / chunk;
    int grid = (int)std::min<int64_t>(nchunks, 2048);
    hipLaunchKernelGGL(k_pack_rec_swap, dim3(grid), dim3(256),
                       2 * (size_t)m * sizeof(int), cur_stream(),
                       (const int4*)src.data_ptr<int>(),
                       cursor.data_ptr<int>(), (int4*)rec.data_ptr<int>(),
                       pos, n, (int)m, chunk);
  } else {
    auto keys32 = torch::empty({n}, opts_i32);
    hipLaunchKernelGGL(k_interleave_swap, dim3(grid_for(n, 256)),
                       dim3(256), 0, cur_stream(),
                       (const int4*)src.data_ptr<int>(),
                       keys32.data_ptr<int>(), (int4*)rec.data_ptr<int>(),
                       n);
    if (pos) {
      auto order = torch::argsort(keys32);
      hipLaunchKernelGGL(k_invert_order, dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         order.data_ptr<int64_t>(), pos, n);
      auto rec2 = torch::empty({n, 4}, opts_i32);
      hipLaunchKernelGGL(k_apply_pos_swap, dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         (const int4*)src.data_ptr<int>(), pos,
                         (int4*)rec2.data_ptr<int>(), n);
      return rec2;
    }
    rec = radix_sort_recs(keys32, rec, n, m);
  }
  return rec;
}

// (GP, EPL) choice: EPL*GP*4 >= dim with GP as small as practical — a
// wave then holds 64/GP independent chunk walks, which is what hides
// the dependent slot->row load latency (measured: GP=64 on dim=300 left
// one walk in flight per wave — profiles/).
template <class F>
static void by_group4(int64_t dim4, F f) {
  using namespace std;
  // FPS_SMALL_GP: lane-group width for dim<=32.  Default 4 (2 float4/
  // lane): twice the independent chunk walks per wave at one fewer
  // shfl per dot — re-swept in round 2 after the emit-into-recB walk
  // (4: 14.5 G samples/s vs 8: 14.0 on mf32; round 1 had measured no
  // difference pre-emit)
  static int small_gp = env_int("FPS_SMALL_GP", 4);
  if (dim4 <= 8) {
    if (small_gp == 4)
      f(integral_constant<int, 4>{}, integral_constant<int, 2>{});
    else
      f(integral_constant<int, 8>{}, integral_constant<int, 1>{});
  }
  else if (dim4 <= 16) f(integral_constant<int, 8>{}, integral_constant<int, 2>{});
  else if (dim4 <= 32) {
    // FPS_GP_D32=16: half the per-lane elements, twice the lanes per
    // sample (fewer registers, one more shfl) — sweepable on GPU
    static int gp32 = env_int("FPS_GP_D32", 16);
    if (gp32 == 16)
      f(integral_constant<int, 16>{}, integral_constant<int, 2>{});
    else
      f(integral_constant<int, 8>{}, integral_constant<int, 4>{});
  }
  else if (dim4 <= 48) f(integral_constant<int, 16>{}, integral_constant<int, 3>{});
  else if (dim4 <= 80) f(integral_constant<int, 16>{}, integral_constant<int, 5>{});
  else if (dim4 <= 128) f(integral_constant<int, 32>{}, integral_constant<int, 4>{});
  else if (dim4 <= 256) f(integral_constant<int, 64>{}, integral_constant<int, 4>{});
  else { TORCH_CHECK(dim4 <= 512, "internal: dim4 > 512 must route to "
                     "the k_seg_fused_huge envelope path");
         f(integral_constant<int, 64>{}, integral_constant<int, 8>{}); }
}

// ---------------------------------------------------------------------
// Envelope walk for rows the register-tiled walks cannot hold (dim >
// 2048, or dim % 4 != 0): one wave per record, lanes stride the
// columns, per-record atomicAdd scatter (no segment pre-aggregation —
// correct for any dim; hot keys serialize, which is acceptable on this
// envelope path and documented in ops/__init__.py).  SELF adds the
// analytic a*keyed term per occurrence (sums to a*cnt*keyed per
// segment — same math as the tiled walks' flush).
// ---------------------------------------------------------------------
template <int MODE, bool SELF>
__global__ __launch_bounds__(256) void k_seg_fused_huge(
    float* __restrict__ out, int4* __restrict__ rec,
    const float* __restrict__ keyed, const float* __restrict__ other,
    int64_t n, int64_t dim, float scale, float a, float lrc) {
  const int lane = threadIdx.x % WAVE;
  const int64_t wpb = blockDim.x / WAVE;
  for (int64_t j = (int64_t)blockIdx.x * wpb + threadIdx.x / WAVE;
       j < n; j += (int64_t)gridDim.x * wpb) {
    const int4 rj = rec[j];
    const int64_t kr = (int64_t)rj.x * dim, sr = (int64_t)rj.y * dim;
    float g;
    if (MODE == 2) {
      g = __int_as_float(rj.z);
    } else {
      float dot = 0.f;
      for (int64_t c = lane; c < dim; c += WAVE)
        dot += keyed[kr + c] * other[sr + c];
      dot = group_sum<WAVE>(dot);
      g = (MODE == 0)
              ? (__int_as_float(rj.z) - dot)
              : lrc * (__int_as_float(rj.z) -
                       1.0f / (1.0f + __expf(-dot)));
      if (lane == 0) rec[j].z = __float_as_int(g);
    }
    for (int64_t c = lane; c < dim; c += WAVE) {
      float t = scale * g * other[sr + c];
      if (SELF) t += a * keyed[kr + c];
      atomicAdd(&out[kr + c], t);
    }
  }
}

// fused segment-walk launcher.  ZERO_OUT: `out` is zero-initialized and
// exclusively written by this launch, so exclusive-segment flushes may
// store instead of read-add.  ``dim4 == 0`` (caller signal: dim % 4 !=
// 0 or dim > 2048) routes to the per-record atomic envelope walk.
template <int MODE, bool SELF, bool ZERO_OUT>
static void seg_fused(torch::Tensor out, torch::Tensor rec,
                      const torch::Tensor& keyed,
                      const torch::Tensor& other, int64_t dim4,
                      double scale, double a, double lrc,
                      int64_t rep_stride = 0, int rep_mask = 7,
                      float* ecol = nullptr, bool write_g = true) {
  int64_t n = rec.size(0);
  if (!n) return;
  if (dim4 == 0) {
    int64_t dim = keyed.size(1);
    hipLaunchKernelGGL((k_seg_fused_huge<MODE, SELF>),
                       dim3(grid_for(n * WAVE, 256)), dim3(256), 0,
                       cur_stream(), out.data_ptr<float>(),
                       (int4*)rec.data_ptr<int>(),
                       keyed.data_ptr<float>(), other.data_ptr<float>(),
                       n, dim, (float)scale, (float)a, (float)lrc);
    return;
  }
  int chunk = seg_chunk();
  int64_t nchunks = (n + chunk - 1) / chunk;
  static int use_lds = env_int("FPS_SEG_LDS", 1);
  // nontemporal row loads in the walks: swept on the mf128 streaming
  // shape and REJECTED (842 -> 832 M samples/s — the same rows recur
  // across the two passes within a round, so L2 retention still pays);
  // kept env-gated for future shapes
  static int nt_env = env_int("FPS_WALK_NT", 0);
  const bool nt_rows = nt_env != 0;
  by_group4(dim4, [&](auto tag, auto etag) {
    constexpr int GP = decltype(tag)::value;
    constexpr int EPL = decltype(etag)::value;
    if (use_lds) {
      // default cap 64 KB keeps 2 blocks/CU; FPS_LDS_CAP=131072 trades
      // occupancy (1 block/CU at 128 KB) for staging the 4-lane shapes
      static size_t lds_cap = (size_t)env_int("FPS_LDS_CAP", 64 * 1024);
      size_t lds = (size_t)(256 / GP) * chunk * sizeof(int4);
      if (lds <= lds_cap && lds <= 160 * 1024) {
        hipLaunchKernelGGL(
            (k_seg_fused_lds<GP, EPL, MODE, SELF, ZERO_OUT>),
            dim3(grid_for(nchunks * GP, 256)), dim3(256), lds,
            cur_stream(), (float4*)out.data_ptr<float>(),
            (int4*)rec.data_ptr<int>(),
            (const float4*)keyed.data_ptr<float>(),
            (const float4*)other.data_ptr<float>(), n, dim4,
            (float)scale, (float)a, (float)lrc, chunk, rep_stride / 4,
            rep_mask, ecol, write_g, nt_rows);
        return;
      }
    }
    hipLaunchKernelGGL((k_seg_fused<GP, EPL, MODE, SELF, ZERO_OUT>),
                       dim3(grid_for(nchunks * GP, 256)), dim3(256), 0,
                       cur_stream(), (float4*)out.data_ptr<float>(),
                       (int4*)rec.data_ptr<int>(),
                       (const float4*)keyed.data_ptr<float>(),
                       (const float4*)other.data_ptr<float>(), n, dim4,
                       (float)scale, (float)a, (float)lrc, chunk,
                       rep_stride / 4, rep_mask, ecol, write_g, nt_rows);
  });
}

// replica count heuristic for the hot-key delta flush (power of two,
// env-sweepable)
static int64_t delta_reps(int64_t n, int64_t m, int64_t dim) {
  // re-swept at the round-2 final balance (chunk 128, recompute-e):
  // 8/16/32/64/128 -> 18.0/20.3/20.9/21.25/21.28 G samples/s on mf32
  // (the walk's flush atomics were the contention); shapes whose
  // replica buffer would exceed 64 MB fall back to 1 as before
  static int64_t r_env = env_int("FPS_DELTA_REPS", 64);
  static int64_t cap_mb = env_int("FPS_REPS_CAP_MB", 64);
  int64_t r = r_env;
  if (r < 1) r = 1;
  // halve toward the largest replica count the batch density and the
  // buffer cap allow (the old all-or-nothing check silently dropped
  // mid-size shapes to 1)
  while (r > 1 && (n < r * m || r * m * dim * 4 > (cap_mb << 20)))
    r >>= 1;
  return r;
}

// K4: one fused SGD pass over a batch of ratings (semantics =
// ops/reference.py::mf_sgd_step: every rating sees the pre-update user
// row u0 and the pulled item snapshot v):
//   pass A (recs by item): e = r - u0.v cached-keyed dot; item deltas
//     lr*(e*u0 - reg*v) segment-accumulated; e persisted into rec.val
//   pass B (recs by user): user updates lr*(e*v - reg*u0) accumulated
//     into user_table in place
// position cache for epoch-resweep workloads (reuse_tag != 0): the
// caller promises the key/slot tensors' CONTENT is unchanged whenever
// the same (tag, buffers) triple recurs, so the sort positions from the
// first visit replace the histogram+scan+counting-sort machinery with
// one scatter pass per side.  Entries hold references to the key
// tensors, so a pointer can never alias a different live tensor.
struct PosCacheEntry {
  torch::Tensor keys, slots, posA, posB, cntB;
  // revisit fast path (dim % 4 == 0 shapes): recB's key/slot are
  // content-static across revisits and its val column is rewritten by
  // pass A each visit via posAB, so recB never needs a rebuild.  recA
  // additionally caches the VAL column (the ratings), so its reuse is
  // gated on the ratings tensor's identity (``vals`` holds the
  // reference — a different ratings buffer re-scatters recA only).
  torch::Tensor recA, recB, posAB, vals;
  // touched-rows L2-decay extents (huge sparse-touched user tables):
  // depend only on recB's static key column — computed once per batch
  torch::Tensor trows, tcnts, ntouch;
};
// exact composite key (no hash collisions): the held tensor references
// inside each entry keep the keyed storages alive, so a pointer can
// never alias a different live tensor.
using PosCacheKey = std::tuple<int64_t, const void*, const void*, int64_t>;
static std::map<PosCacheKey, PosCacheEntry> g_pos_cache;

static PosCacheKey pos_cache_key(int64_t tag, const torch::Tensor& a,
                                 const torch::Tensor& b, int64_t n) {
  return {tag, a.data_ptr(), b.data_ptr(), n};
}

// analytic L2 decay of the touched user rows: per-row counts when the
// histogram exists (LDS-sized tables), else the (cached) touched-rows
// extents of the user-sorted records
static void decay_users(torch::Tensor& user_table, torch::Tensor& recB,
                        torch::Tensor& cntB, torch::Tensor& histB,
                        int64_t n, int64_t U, int64_t dim, double lr,
                        double reg, int64_t reuse_tag,
                        const PosCacheKey& ck) {
  auto opts_i32 = torch::dtype(torch::kInt32)
                      .device(user_table.device());
  if (cntB.defined() || histB.defined()) {
    auto& cc = cntB.defined() ? cntB : histB;
    hipLaunchKernelGGL(k_scale_rows_by_count,
                       dim3(grid_for(U * dim, 256)), dim3(256), 0,
                       cur_stream(), user_table.data_ptr<float>(),
                       cc.data_ptr<int>(), U, dim, (float)(-lr * reg));
    return;
  }
  // huge sparse-touched table: decay only the rows this batch
  // touches, from the sorted recs' segment extents.  The extents
  // depend only on recB's STATIC key column, so revisits reuse
  // them from the position cache (k_seg_extents was 60% of the
  // HBM-regime mf32 round — profiles/PERF_NOTES.md round 2)
  torch::Tensor trows, tcnts, ntouch;
  PosCacheEntry* e = nullptr;
  if (reuse_tag) {
    auto it = g_pos_cache.find(ck);
    if (it != g_pos_cache.end()) e = &it->second;
  }
  if (e && e->trows.defined()) {
    trows = e->trows; tcnts = e->tcnts; ntouch = e->ntouch;
  } else {
    trows = torch::empty({n}, opts_i32);
    tcnts = torch::empty({n}, opts_i32);
    ntouch = torch::zeros({1}, opts_i32);
    hipLaunchKernelGGL(k_seg_extents, dim3(grid_for(n, 256)),
                       dim3(256), 0, cur_stream(),
                       (const int4*)recB.data_ptr<int>(), n,
                       trows.data_ptr<int>(), tcnts.data_ptr<int>(),
                       ntouch.data_ptr<int>());
    if (e) { e->trows = trows; e->tcnts = tcnts; e->ntouch = ntouch; }
  }
  if (dim % 4 == 0)
    hipLaunchKernelGGL(k_scale_touched_rows_vec4,
                       dim3(grid_for(n * 32, 256)), dim3(256), 0,
                       cur_stream(),
                       (float4*)user_table.data_ptr<float>(),
                       trows.data_ptr<int>(), tcnts.data_ptr<int>(),
                       ntouch.data_ptr<int>(), dim / 4,
                       (float)(-lr * reg));
  else
    hipLaunchKernelGGL(k_scale_touched_rows,
                       dim3(grid_for(n * dim, 256)), dim3(256), 0,
                       cur_stream(), user_table.data_ptr<float>(),
                       trows.data_ptr<int>(), tcnts.data_ptr<int>(),
                       ntouch.data_ptr<int>(), dim,
                       (float)(-lr * reg));
}

torch::Tensor mf_sgd_step(torch::Tensor user_table, torch::Tensor user_rows,
                          torch::Tensor item_vecs, torch::Tensor item_slots,
                          torch::Tensor ratings, double lr, double reg,
                          int64_t reuse_tag, bool mean_users) {
  CHECK_DEV(user_table); CHECK_CONT(user_table);
  CHECK_DEV(item_vecs); CHECK_CONT(item_vecs);
  CHECK_CONT(user_rows); CHECK_CONT(item_slots); CHECK_CONT(ratings);
  int64_t n = ratings.numel(), dim = user_table.size(1);
  int64_t m = item_vecs.size(0), U = user_table.size(0);
  TORCH_CHECK(item_vecs.size(1) == dim);
  if (!n) return torch::zeros_like(item_vecs);
  // register-tiled walks cover dim % 4 == 0 up to 2048; anything else
  // takes the per-record atomic envelope walk (dim4 = 0 signal)
  int64_t dim4 = (dim % 4 == 0 && dim <= 2048) ? dim / 4 : 0;
  if (item_slots.dtype() != user_rows.dtype()) {
    // harmonize to int32 (both ranges are table heights < 2^31)
    if (item_slots.dtype() != torch::kInt32)
      item_slots = item_slots.to(torch::kInt32);
    if (user_rows.dtype() != torch::kInt32)
      user_rows = user_rows.to(torch::kInt32);
  }

  auto opts_i32 = torch::dtype(torch::kInt32).device(ratings.device());

  // cached-position fast path (epoch resweeps): one scatter per side
  // replaces histogram + scan + counting sort
  PosCacheEntry* hit = nullptr;
  PosCacheKey ck{};
  if (reuse_tag) {
    ck = pos_cache_key(reuse_tag, item_slots, user_rows, n);
    auto it = g_pos_cache.find(ck);
    if (it != g_pos_cache.end()) hit = &it->second;
  }

  // revisit fast path: both record arrays come straight from the
  // cache; pass A scatters the coefficient into recB.z via posAB, so
  // neither array needs a rebuild (the two position scatters were 39%
  // of the shipped mf32 round — profiles/PERF_NOTES.md round 2)
  const bool emit_ok = dim4 != 0;
  torch::Tensor recA;
  torch::Tensor posA, posB;
  torch::Tensor histB;
  const bool same_vals =
      hit && hit->vals.defined() &&
      hit->vals.data_ptr() == ratings.data_ptr() &&
      hit->vals.numel() == ratings.numel();
  const bool recA_cached =
      hit && emit_ok && hit->recA.defined() && same_vals;
  if (recA_cached) {
    recA = hit->recA;
  } else if (hit) {
    recA = torch::empty({n, 4}, opts_i32);
    if (item_slots.dtype() == torch::kInt32)
      hipLaunchKernelGGL((k_apply_pos<int>), dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         item_slots.data_ptr<int>(),
                         user_rows.data_ptr<int>(),
                         ratings.data_ptr<float>(),
                         hit->posA.data_ptr<int>(),
                         (int4*)recA.data_ptr<int>(), n);
    else
      hipLaunchKernelGGL((k_apply_pos<int64_t>), dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         item_slots.data_ptr<int64_t>(),
                         user_rows.data_ptr<int64_t>(),
                         ratings.data_ptr<float>(),
                         hit->posA.data_ptr<int>(),
                         (int4*)recA.data_ptr<int>(), n);
  } else {
    bool smallA = m <= PACK_MAX_M, smallB = U <= PACK_MAX_M;
    torch::Tensor histA;
    // histograms: one fused read when both key ranges are LDS-sized
    if (smallA && smallB && m + U <= 2 * PACK_MAX_M) {
      auto hist_both = torch::zeros({m + U}, opts_i32);  // one fill
      histA = hist_both.narrow(0, 0, m);
      histB = hist_both.narrow(0, m, U);
      int hg = (int)std::min<int64_t>((n + 4095) / 4096, 512);
      size_t lds = (size_t)(m + U) * sizeof(int);
      if (item_slots.dtype() == torch::kInt32)
        hipLaunchKernelGGL((k_dual_hist<int>), dim3(std::max(hg, 1)),
                           dim3(256), lds, cur_stream(),
                           item_slots.data_ptr<int>(),
                           user_rows.data_ptr<int>(),
                           histA.data_ptr<int>(), histB.data_ptr<int>(),
                           n, (int)m, (int)U);
      else
        hipLaunchKernelGGL((k_dual_hist<int64_t>), dim3(std::max(hg, 1)),
                           dim3(256), lds, cur_stream(),
                           item_slots.data_ptr<int64_t>(),
                           user_rows.data_ptr<int64_t>(),
                           histA.data_ptr<int>(), histB.data_ptr<int>(),
                           n, (int)m, (int)U);
    } else if (smallB) {
      histB = torch::zeros({U}, opts_i32);
      int hg = (int)std::min<int64_t>((n + 4095) / 4096, 512);
      if (user_rows.dtype() == torch::kInt32)
        hipLaunchKernelGGL((k_key_hist<int>), dim3(std::max(hg, 1)),
                           dim3(256), (size_t)U * sizeof(int),
                           cur_stream(), user_rows.data_ptr<int>(),
                           histB.data_ptr<int>(), n, (int)U);
      else
        hipLaunchKernelGGL((k_key_hist<int64_t>), dim3(std::max(hg, 1)),
                           dim3(256), (size_t)U * sizeof(int),
                           cur_stream(), user_rows.data_ptr<int64_t>(),
                           histB.data_ptr<int>(), n, (int)U);
    }
    if (reuse_tag) {
      posA = torch::empty({n}, opts_i32);
      posB = torch::empty({n}, opts_i32);
    }
    recA = build_rec(item_slots, user_rows, ratings, m, histA, posA);
  }

  // caching + emit: recB (key/slot static; val rewritten by pass A)
  // and the composite posAB are prepared BEFORE pass A on the first
  // visit, so every visit runs the same emit-into-recB walk
  torch::Tensor recB, cntB, posAB;
  bool have_recB = false;
  if (emit_ok && reuse_tag && (!hit || hit->recB.defined())) {
    if (hit) {
      // (a rebuilt recA — fresh ratings buffer — needs no fixup: the
      // coefficient now travels through the sequential e column and
      // recB.w, both untouched by the rebuild)
      recB = hit->recB;
      posAB = hit->posAB;
      cntB = hit->cntB;
    } else {
      TORCH_CHECK(histB.defined() || m > PACK_MAX_M || U > PACK_MAX_M,
                  "histB expected for LDS-sized shapes");
      recB = build_rec_swap(recA, U, histB, posB);
      // each recB record carries its pass-A position in .w (posB is
      // indexed by recA position), so pass B gathers its coefficient
      // from the sequential e column with no extra index load
      hipLaunchKernelGGL(k_pack_wposB, dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         (int4*)recB.data_ptr<int>(),
                         posB.data_ptr<int>(), n);
      posAB = posB;
      cntB = histB;
      if (g_pos_cache.size() >= 32) g_pos_cache.clear();
      g_pos_cache[ck] = {item_slots, user_rows, posA, posB, histB,
                         recA, recB, posAB, ratings, {}, {}, {}};
    }
    have_recB = true;
  }

  // pass A: recs by item slot; fused e + item-delta accumulation
  // (the dim4==0 envelope walk scatters with per-record atomics and
  // has no replica spreading)
  int64_t reps = dim4 ? delta_reps(n, m, dim) : 1;
  torch::Tensor deltas;
  // pass-B-recomputes-e configuration: when the user table is small
  // enough for a side accumulator (it is for every cache-resident and
  // HBM-regime rank-32 shape; not for the 51 GB mf128 partition), the
  // coefficient is RE-derived inside pass B from the same operands —
  // bitwise-identical dot (products commute, identical lane split and
  // reduction order) — so neither walk pays any cross-order
  // permutation traffic for e.  Race-free because pass B then flushes
  // into the zeroed accumulator, never mutating the rows it reads.
  // recB.z carries the ratings captured at BUILD time, so the
  // recompute path additionally requires the current ratings buffer to
  // be that same tensor (first visit, or identity match); a fresh
  // ratings buffer falls back to the e-column path, whose coefficients
  // always come from pass A's fresh recA.z.
  const bool recompute_e =
      have_recB && !mean_users && (!hit || same_vals) &&
      U * dim * 4 <= (256ll << 20);
  torch::Tensor ecol_t;
  float* ecol = nullptr;
  if (have_recB && !recompute_e) {
    ecol_t = torch::empty({n}, ratings.options());
    ecol = ecol_t.data_ptr<float>();
  }
  if (reps > 1) {
    auto rep_buf = torch::zeros({reps * m, dim}, item_vecs.options());
    seg_fused<0, true, true>(rep_buf, recA, item_vecs, user_table, dim4,
                             lr, -lr * reg, 0.0, m * dim, (int)reps - 1,
                             ecol, /*write_g=*/!recompute_e);
    deltas = rep_buf.view({reps, m, dim}).sum(0);
  } else {
    deltas = torch::zeros_like(item_vecs);
    seg_fused<0, true, true>(deltas, recA, item_vecs, user_table, dim4,
                             lr, -lr * reg, 0.0, 0, 7, ecol,
                             /*write_g=*/!recompute_e);
  }

  // pass B: recs by user row; apply user updates in place.  The L2
  // term u -= lr*reg*cnt*u0 is applied analytically first (see
  // k_scale_rows_by_count), then the delta term lr*sum(e*v) accumulates
  // with no SELF read.
  if (!have_recB) {
    if (hit) {
      cntB = hit->cntB;
      recB = torch::empty({n, 4}, opts_i32);
      hipLaunchKernelGGL(k_apply_pos_swap, dim3(grid_for(n, 256)),
                         dim3(256), 0, cur_stream(),
                         (const int4*)recA.data_ptr<int>(),
                         hit->posB.data_ptr<int>(),
                         (int4*)recB.data_ptr<int>(), n);
    } else {
      TORCH_CHECK(histB.defined() || !mean_users,
                  "mean_users needs per-user counts (table too large "
                  "for the LDS histogram)");
      recB = build_rec_swap(recA, U, histB, posB);
      if (reuse_tag) {
        if (g_pos_cache.size() >= 32) g_pos_cache.clear();
        g_pos_cache[ck] = {item_slots, user_rows, posA, posB, histB,
                           {}, {}, {}, {}, {}, {}, {}};
      }
    }
  }
  if (mean_users) {
    TORCH_CHECK(cntB.defined() || histB.defined(),
                "MF mean path needs per-user counts");
    auto& cc = cntB.defined() ? cntB : histB;
    auto acc = torch::zeros_like(user_table);
    seg_fused<2, false, true>(acc, recB, user_table, item_vecs, dim4,
                              lr, 0.0, 0.0, 0, 7, ecol);
    hipLaunchKernelGGL(k_apply_mean_update,
                       dim3(grid_for(U * dim, 256)), dim3(256), 0,
                       cur_stream(), user_table.data_ptr<float>(),
                       acc.data_ptr<float>(), cc.data_ptr<int>(), U, dim,
                       (float)(lr * reg));
    return deltas;
  }
  if (reg != 0.0 && !recompute_e)
    decay_users(user_table, recB, cntB, histB, n, U, dim, lr, reg,
                reuse_tag, ck);
  if (recompute_e) {
    // order matters: the walk reads u0 rows (pre-decay, pre-update) to
    // re-derive e, so it runs BEFORE the analytic L2 decay applied
    // above would be wrong — hence the decay was applied above only
    // for the non-recompute path; here: accumulate first, then decay,
    // then add.
    auto acc = torch::zeros_like(user_table);
    seg_fused<0, false, true>(acc, recB, user_table, item_vecs, dim4,
                              lr, 0.0, 0.0, 0, 7, nullptr,
                              /*write_g=*/false);
    if (reg != 0.0)
      decay_users(user_table, recB, cntB, histB, n, U, dim, lr, reg,
                  reuse_tag, ck);
    user_table.add_(acc);
    return deltas;
  }
  seg_fused<2, false, false>(user_table, recB, user_table, item_vecs,
                             dim4, lr, 0.0, 0.0, 0, 7, ecol);
  return deltas;
}

// K8: fused SGNS step over ONE shared row buffer (center and context
// slots index disjoint rows of `vecs` — the model maps context ids to
// [V, 2V)).  Returns the combined deltas buffer.
//   pass C (recs by center): g = lr*(y - sigmoid(c.x)) cached-keyed
//     dot; center deltas g*x accumulated; g persisted
//   pass D (recs by context): context deltas g*c accumulated into the
//     same buffer
torch::Tensor sgns_step(torch::Tensor vecs, torch::Tensor center_slots,
                        torch::Tensor ctx_slots, torch::Tensor labels,
                        double lr) {
  CHECK_DEV(vecs); CHECK_CONT(vecs);
  CHECK_CONT(center_slots); CHECK_CONT(ctx_slots); CHECK_CONT(labels);
  int64_t n = labels.numel(), dim = vecs.size(1), m = vecs.size(0);
  auto deltas = torch::zeros_like(vecs);
  if (!n) return deltas;
  int64_t dim4 = (dim % 4 == 0 && dim <= 2048) ? dim / 4 : 0;
  auto recC = build_rec(center_slots, ctx_slots, labels, m);
  seg_fused<1, false, true>(deltas, recC, vecs, vecs, dim4, 1.0, 0.0,
                            lr);
  torch::Tensor histD;
  if (m <= PACK_MAX_M) {
    auto opts_i32 = torch::dtype(torch::kInt32).device(vecs.device());
    histD = torch::zeros({m}, opts_i32);
    int hg = (int)std::min<int64_t>((n + 4095) / 4096, 512);
    if (ctx_slots.dtype() == torch::kInt32)
      hipLaunchKernelGGL((k_key_hist<int>), dim3(std::max(hg, 1)),
                         dim3(256), (size_t)m * sizeof(int), cur_stream(),
                         ctx_slots.data_ptr<int>(), histD.data_ptr<int>(),
                         n, (int)m);
    else
      hipLaunchKernelGGL((k_key_hist<int64_t>), dim3(std::max(hg, 1)),
                         dim3(256), (size_t)m * sizeof(int), cur_stream(),
                         ctx_slots.data_ptr<int64_t>(),
                         histD.data_ptr<int>(), n, (int)m);
  }
  auto recD = build_rec_swap(recC, m, histD);
  seg_fused<2, false, true>(deltas, recD, vecs, vecs, dim4, 1.0, 0.0,
                            0.0);
  return deltas;
}

std::vector<torch::Tensor> hash_coalesce(torch::Tensor ids) {
  CHECK_DEV(ids); CHECK_CONT(ids);
  TORCH_CHECK(ids.dtype() == torch::kInt64, "ids must be int64");
  int64_t n = ids.numel();
  auto opts_i32 = torch::dtype(torch::kInt32).device(ids.device());
  if (!n)
    return {torch::empty({0}, ids.options()), torch::empty({0}, opts_i32)};
  // pow2 capacity >= 2n => load factor <= 0.5, ~2 probes expected
  int64_t cap = 1024;
  while (cap < 2 * n) cap <<= 1;
  TORCH_CHECK(cap <= (1ll << 31), "batch too large for hash_coalesce");
  auto keys = torch::empty({cap}, ids.options());
  hipMemsetAsync(keys.data_ptr(), 0xFF, cap * 8, cur_stream());
  auto pos = torch::empty({n}, opts_i32);
  hipLaunchKernelGGL(k_hash_insert, dim3(grid_for(n, 256)), dim3(256), 0,
                     cur_stream(), ids.data_ptr<int64_t>(), n,
                     (long long*)keys.data_ptr<int64_t>(),
                     pos.data_ptr<int>(), cap - 1);
  int64_t nchunks = (cap + 1023) / 1024;
  auto counts = torch::zeros({nchunks}, opts_i32);
  hipLaunchKernelGGL(k_hash_count, dim3(grid_for(cap, 256)), dim3(256),
                     0, cur_stream(),
                     (const long long*)keys.data_ptr<int64_t>(), cap,
                     counts.data_ptr<int>());
  auto cursor = exclusive_starts(counts).contiguous();
  // total uniques = last start + last count (device->host: the caller
  // needs the size to slice uids; torch.unique syncs the same way)
  int last[2];
  hipMemcpyAsync(&last[0], cursor.data_ptr<int>() + nchunks - 1, 4,
                 hipMemcpyDeviceToHost, cur_stream());
  hipMemcpyAsync(&last[1], counts.data_ptr<int>() + nchunks - 1, 4,
                 hipMemcpyDeviceToHost, cur_stream());
  auto slotmap = torch::empty({cap}, opts_i32);
  auto uids = torch::empty({n}, ids.options());
  hipLaunchKernelGGL(k_hash_assign, dim3(grid_for(cap, 256)), dim3(256),
                     0, cur_stream(),
                     (const long long*)keys.data_ptr<int64_t>(), cap,
                     cursor.data_ptr<int>(), slotmap.data_ptr<int>(),
                     uids.data_ptr<int64_t>());
  auto slots = torch::empty({n}, opts_i32);
  hipLaunchKernelGGL(k_hash_slots, dim3(grid_for(n, 256)), dim3(256), 0,
                     cur_stream(), pos.data_ptr<int>(),
                     slotmap.data_ptr<int>(), slots.data_ptr<int>(), n);
  hipStreamSynchronize(cur_stream());
  int64_t m = (int64_t)last[0] + (int64_t)last[1];
  // keys+slotmap stay alive with the caller for hash_lookup resolution
  return {uids.narrow(0, 0, m), slots, keys, slotmap};
}

torch::Tensor hash_lookup(torch::Tensor keys, torch::Tensor slotmap,
                          torch::Tensor ids) {
  CHECK_DEV(keys); CHECK_CONT(keys); CHECK_CONT(slotmap); CHECK_CONT(ids);
  TORCH_CHECK(ids.dtype() == torch::kInt64);
  int64_t n = ids.numel(), cap = keys.numel();
  auto slots = torch::empty({n}, torch::dtype(torch::kInt32)
                                     .device(ids.device()));
  if (!n) return slots;
  hipLaunchKernelGGL(k_hash_lookup, dim3(grid_for(n, 256)), dim3(256),
                     0, cur_stream(), ids.data_ptr<int64_t>(), n,
                     (const long long*)keys.data_ptr<int64_t>(),
                     slotmap.data_ptr<int>(), cap - 1,
                     slots.data_ptr<int>());
  return slots;
}

std::vector<torch::Tensor> sgns_expand(torch::Tensor centers,
                                       torch::Tensor contexts, int64_t k,
                                       int64_t V, int64_t base_key) {
  CHECK_DEV(centers); CHECK_CONT(centers); CHECK_CONT(contexts);
  TORCH_CHECK(centers.dtype() == torch::kInt64 &&
              contexts.dtype() == torch::kInt64);
  int64_t n = centers.numel();
  auto opts = centers.options();
  auto oc = torch::empty({n * (k + 1)}, opts);
  auto ox = torch::empty({n * (k + 1)}, opts);
  auto ol = torch::empty({n * (k + 1)}, opts.dtype(torch::kFloat32));
  if (!n) return {oc, ox, ol};
  hipLaunchKernelGGL(k_sgns_expand, dim3(grid_for(n * (k + 1), 256)),
                     dim3(256), 0, cur_stream(),
                     centers.data_ptr<int64_t>(),
                     contexts.data_ptr<int64_t>(), n, k, V, base_key,
                     oc.data_ptr<int64_t>(), ox.data_ptr<int64_t>(),
                     ol.data_ptr<float>());
  return {oc, ox, ol};
}

// per-static-batch cache for the slot-sorted linear step: the record
// array (slot, instance, x) and the per-instance ||x||^2 are fully
// static; only the pulled weights and labels' effect change per visit
struct LinCacheEntry {
  torch::Tensor slots, xvals, rec, n2;
};
static std::map<PosCacheKey, LinCacheEntry> g_lin_cache;

static int lin_chunk() {
  static int c = env_int("FPS_LIN_CHUNK", 64);
  return c;
}

static std::vector<torch::Tensor> linear_step_cached(
    torch::Tensor indptr, torch::Tensor feat_slots, torch::Tensor x_vals,
    torch::Tensor labels, torch::Tensor weights, int mode, float C,
    int64_t reuse_tag) {
  int64_t B = indptr.numel() - 1, n = x_vals.numel();
  int64_t m = weights.numel();
  auto opts_i32 = torch::dtype(torch::kInt32).device(x_vals.device());
  PosCacheKey ck{reuse_tag, feat_slots.data_ptr(), x_vals.data_ptr(), n};
  auto it = g_lin_cache.find(ck);
  LinCacheEntry* e = it != g_lin_cache.end() ? &it->second : nullptr;
  torch::Tensor rec, n2;
  if (e) {
    rec = e->rec;
    n2 = e->n2;
  } else {
    auto seg = torch::empty({n}, opts_i32);
    hipLaunchKernelGGL(k_lin_seg, dim3(grid_for(B * WAVE, 256)),
                       dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int64_t>(), seg.data_ptr<int>(),
                       B);
    n2 = torch::empty({B}, x_vals.options());
    hipLaunchKernelGGL(k_lin_n2, dim3(grid_for(B * WAVE, 256)),
                       dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int64_t>(),
                       x_vals.data_ptr<float>(), n2.data_ptr<float>(),
                       B);
    auto fs32 = feat_slots.dtype() == torch::kInt32
                    ? feat_slots : feat_slots.to(torch::kInt32);
    rec = build_rec(fs32, seg, x_vals, m);
    if (g_lin_cache.size() >= 32) g_lin_cache.clear();
    g_lin_cache[ck] = {feat_slots, x_vals, rec, n2};
  }
  auto margins = torch::zeros({B}, x_vals.options());
  int chunk = lin_chunk();
  int64_t nchunks = (n + chunk - 1) / chunk;
  hipLaunchKernelGGL(k_lin_margin, dim3(grid_for(nchunks, 256)),
                     dim3(256), 0, cur_stream(),
                     (const int4*)rec.data_ptr<int>(),
                     weights.data_ptr<float>(),
                     margins.data_ptr<float>(), n, chunk);
  auto coeff = torch::empty({B}, x_vals.options());
  hipLaunchKernelGGL(k_lin_coeff, dim3(grid_for(B, 256)), dim3(256), 0,
                     cur_stream(), margins.data_ptr<float>(),
                     n2.data_ptr<float>(), labels.data_ptr<float>(),
                     coeff.data_ptr<float>(), B, mode, C);
  auto deltas = torch::zeros({m}, weights.options());
  hipLaunchKernelGGL(k_lin_delta, dim3(grid_for(nchunks, 256)),
                     dim3(256), 0, cur_stream(),
                     (const int4*)rec.data_ptr<int>(),
                     coeff.data_ptr<float>(), deltas.data_ptr<float>(),
                     n, chunk, 1.0f);
  return {margins, deltas};
}

static int pa_mode(const std::string& variant) {
  if (variant == "PA") return 0;
  if (variant == "PA-I") return 1;
  if (variant == "PA-II") return 2;
  TORCH_CHECK(false, "unknown PA variant ", variant);
}

std::vector<torch::Tensor> pa_step(torch::Tensor indptr,
                                   torch::Tensor feat_slots,
                                   torch::Tensor x_vals,
                                   torch::Tensor labels,
                                   torch::Tensor weights,
                                   std::string variant, double C) {
  CHECK_DEV(weights); CHECK_CONT(weights);
  CHECK_CONT(indptr); CHECK_CONT(feat_slots); CHECK_CONT(x_vals);
  int64_t B = indptr.numel() - 1;
  auto margins = torch::zeros({B}, x_vals.options());
  if (!B) return {margins, torch::zeros_like(weights)};
  int64_t m = weights.numel();
  static int64_t lin_reps = env_int("FPS_LINEAR_REPS", 8);
  int64_t reps = (x_vals.numel() >= lin_reps * m &&
                  lin_reps * m * 4 <= (256ll << 20)) ? lin_reps : 1;
  auto dbuf = torch::zeros({reps * m}, weights.options());
  auto launch = [&](auto* slots) {
    hipLaunchKernelGGL(k_sparse_linear_step, dim3(grid_for(B * WAVE, 256)),
                       dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int64_t>(), slots,
                       x_vals.data_ptr<float>(), labels.data_ptr<float>(),
                       weights.data_ptr<float>(), margins.data_ptr<float>(),
                       dbuf.data_ptr<float>(), B, pa_mode(variant),
                       (float)C, reps > 1 ? m : 0, (int)reps - 1);
  };
  if (feat_slots.dtype() == torch::kInt32) launch(feat_slots.data_ptr<int>());
  else launch(feat_slots.data_ptr<int64_t>());
  auto deltas = reps > 1 ? dbuf.view({reps, m}).sum(0) : dbuf;
  return {margins, deltas};
}

std::vector<torch::Tensor> lr_step(torch::Tensor indptr,
                                   torch::Tensor feat_slots,
                                   torch::Tensor x_vals,
                                   torch::Tensor labels,
                                   torch::Tensor weights,
                                   int64_t reuse_tag) {
  CHECK_DEV(weights); CHECK_CONT(weights);
  int64_t B = indptr.numel() - 1;
  if (B && reuse_tag) {
    auto out = linear_step_cached(indptr, feat_slots, x_vals, labels,
                                  weights, /*mode=*/3, 0.f, reuse_tag);
    out[0].sigmoid_();
    return out;
  }
  auto probs = torch::zeros({B}, x_vals.options());
  if (!B) return {probs, torch::zeros_like(weights)};
  int64_t m = weights.numel();
  static int64_t lin_reps = env_int("FPS_LINEAR_REPS", 8);
  int64_t reps = (x_vals.numel() >= lin_reps * m &&
                  lin_reps * m * 4 <= (256ll << 20)) ? lin_reps : 1;
  auto gbuf = torch::zeros({reps * m}, weights.options());
  auto launch = [&](auto* slots) {
    hipLaunchKernelGGL(k_sparse_linear_step, dim3(grid_for(B * WAVE, 256)),
                       dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int64_t>(), slots,
                       x_vals.data_ptr<float>(), labels.data_ptr<float>(),
                       weights.data_ptr<float>(), probs.data_ptr<float>(),
                       gbuf.data_ptr<float>(), B, /*mode=*/3, 0.f,
                       reps > 1 ? m : 0, (int)reps - 1);
  };
  if (feat_slots.dtype() == torch::kInt32) launch(feat_slots.data_ptr<int>());
  else launch(feat_slots.data_ptr<int64_t>());
  auto grads = reps > 1 ? gbuf.view({reps, m}).sum(0) : gbuf;
  // kernel writes margins; convert to probabilities in-place
  probs.sigmoid_();
  return {probs, grads};
}

std::vector<torch::Tensor> pa_multiclass_step(
    torch::Tensor indptr, torch::Tensor feat_slots, torch::Tensor x_vals,
    torch::Tensor labels, torch::Tensor weights, std::string variant,
    double C) {
  CHECK_DEV(weights); CHECK_CONT(weights);
  int64_t B = indptr.numel() - 1, nc = weights.size(1);
  auto scores = torch::zeros({B, nc}, x_vals.options());
  if (!B) return {scores, torch::zeros_like(weights)};
  // replica-spread the delta scatter like the binary kernel when the
  // batch is dense enough to make hot-feature atomics the bottleneck
  int64_t mw = weights.numel();
  int64_t reps = (x_vals.numel() * nc >= 8 * mw &&
                  8 * mw * 4 <= (256ll << 20)) ? 8 : 1;
  auto dbuf = torch::zeros({reps * mw}, weights.options());
  hipLaunchKernelGGL(k_pa_multiclass_step, dim3(grid_for(B * WAVE, 256)),
                     dim3(256), 0, cur_stream(),
                     indptr.data_ptr<int64_t>(),
                     feat_slots.data_ptr<int64_t>(),
                     x_vals.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     weights.data_ptr<float>(), scores.data_ptr<float>(),
                     dbuf.data_ptr<float>(), B, nc, pa_mode(variant),
                     (float)C, reps > 1 ? mw : 0);
  auto deltas = (reps > 1 ? dbuf.view({reps, mw}).sum(0) : dbuf)
                    .view_as(weights);
  return {scores, deltas};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gather", &gather, "K3 ps_gather");
  m.def("apply_add", &apply_add, "K1 ps_apply_sgd (delta add)",
        py::arg("table"), py::arg("rows"), py::arg("deltas"),
        py::arg("unique") = false);
  m.def("apply_adagrad_sorted", &apply_adagrad_sorted,
        "K2 ps_apply_adagrad over a row-sorted batch");
  m.def("hash_init", &hash_init, "K6 deterministic per-id init");
  m.def("hash_init_rows", &hash_init_rows, "K6 slab init");
  m.def("mf_sgd_step", &mf_sgd_step, "K4 fused MF SGD step",
        py::arg("user_table"), py::arg("user_rows"), py::arg("item_vecs"),
        py::arg("item_slots"), py::arg("ratings"), py::arg("lr"),
        py::arg("reg") = 0.0, py::arg("reuse_tag") = 0,
        py::arg("mean_users") = false);
  m.def("sgns_step", &sgns_step, "K8 fused SGNS step");
  m.def("sgns_expand", &sgns_expand,
        "K5/K8 fused pair expansion + hash-RNG negative sampling");
  m.def("hash_coalesce", &hash_coalesce,
        "K10b hash-based id coalescing (unsorted-unique uids + slots)");
  m.def("hash_lookup", &hash_lookup,
        "K10b slot resolution against an existing coalescer table");
  m.def("pa_step", &pa_step, "K7 PA binary step");
  m.def("pa_multiclass_step", &pa_multiclass_step, "K7 PA multiclass");
  m.def("lr_step", &lr_step, "K9 sparse logistic step",
        py::arg("indptr"), py::arg("feat_slots"), py::arg("x_vals"),
        py::arg("labels"), py::arg("weights"), py::arg("reuse_tag") = 0);
}
