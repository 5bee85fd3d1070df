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
//  * memory-bound throughout: all row traffic is vectorized float4
//    where dim % 4 == 0 (Guideline 13); grids are capped and
//    grid-strided (Guideline 11).
//  * duplicate keys inside a push batch accumulate with device-scope f32
//    atomics (correct for '+' semantics; Adagrad instead runs on a
//    sorted batch with one segment leader per distinct row, because its
//    update is non-linear in the aggregated gradient).
//  * numerics: f32 master values end to end; the semantics oracle is
//    ops/reference.py and tests compare against it elementwise.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define WAVE 64
#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be a device tensor")
#define CHECK_CONT(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

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
template <int GP>
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int off = GP / 2; off > 0; off >>= 1)
    v += __shfl_xor(v, off, GP);
  return v;
}

// ---------------------------------------------------------------------
// K4/K8 shared machinery.  The zipfian workloads make naive per-sample
// atomics serialize on hot rows (measured: the first-cut one-kernel MF
// step ran at ~300 GB/s, 70% of step time — profiles/
// mf32_r01_baseline_atomics.md), so the fused steps are decomposed into
//   (1) k_pair_coeff  — per-sample dot + scalar coefficient, no writes
//                       beyond one float per sample;
//   (2) k_seg_accum   — segmented scatter-accumulate over KEY-SORTED
//                       samples: each lane group walks a contiguous
//                       chunk, accumulates same-key rows in registers,
//                       and issues ONE atomicAdd per (segment x chunk)
//                       — the LDS/register pre-aggregation SURVEY.md
//                       §7.3 calls for.
// Sorting is rocPRIM radix sort via torch::argsort in the host wrapper.
// ---------------------------------------------------------------------

// coeff[j] = mode 0: s_in[j] - A[arows[j]] . B[brows[j]]        (MF: e)
//            mode 1: lr * (s_in[j] - sigmoid(A[arows[j]].B[brows[j]]))
//                                                               (SGNS: g)
template <int GP>
__global__ __launch_bounds__(256) void k_pair_coeff(
    const float* __restrict__ A, const int64_t* __restrict__ arows,
    const float* __restrict__ B, const int64_t* __restrict__ brows,
    const float* __restrict__ s_in, float* __restrict__ coeff,
    int64_t n, int64_t dim, int mode, float lr) {
  constexpr int EPL = 8;
  const int lane = threadIdx.x % GP;
  const int64_t groups_per_block = blockDim.x / GP;
  const int64_t g0 = (int64_t)blockIdx.x * groups_per_block +
                     threadIdx.x / GP;
  const int64_t gstride = (int64_t)gridDim.x * groups_per_block;
  for (int64_t g = g0; g < n; g += gstride) {
    const int64_t ar = arows[g] * dim, br = brows[g] * dim;
    float dot = 0.f;
#pragma unroll
    for (int e = 0; e < EPL; ++e) {
      int c = lane + e * GP;
      if (c < dim) dot += A[ar + c] * B[br + c];
    }
    dot = group_sum<GP>(dot);
    if (lane == 0) {
      if (mode == 0) coeff[g] = s_in[g] - dot;
      else coeff[g] = lr * (s_in[g] - 1.0f / (1.0f + __expf(-dot)));
    }
  }
}

// float4-vectorized variants (dim % 4 == 0): GP lanes of float4 cover
// the row in 16-B coalesced accesses; 3-5 shfl reductions instead of 5-6
// and 4x fewer index loads per element.
template <int GP, int EPL, class TIdx>
__global__ __launch_bounds__(256) void k_pair_coeff_v4(
    const float4* __restrict__ A, const TIdx* __restrict__ arows,
    const float4* __restrict__ B, const TIdx* __restrict__ brows,
    const float* __restrict__ s_in, float* __restrict__ coeff,
    int64_t n, int64_t dim4, int mode, float lr) {
  const int lane = threadIdx.x % GP;
  const int64_t groups_per_block = blockDim.x / GP;
  const int64_t g0 = (int64_t)blockIdx.x * groups_per_block +
                     threadIdx.x / GP;
  const int64_t gstride = (int64_t)gridDim.x * groups_per_block;
  for (int64_t g = g0; g < n; g += gstride) {
    const int64_t ar = (int64_t)arows[g] * dim4;
    const int64_t br = (int64_t)brows[g] * dim4;
    float dot = 0.f;
#pragma unroll
    for (int e = 0; e < EPL; ++e) {
      int c = lane + e * GP;
      if (c < dim4) {
        float4 a = A[ar + c], b = B[br + c];
        dot += a.x * b.x + a.y * b.y + a.z * b.z + a.w * b.w;
      }
    }
    dot = group_sum<GP>(dot);
    if (lane == 0) {
      if (mode == 0) coeff[g] = s_in[g] - dot;
      else coeff[g] = lr * (s_in[g] - 1.0f / (1.0f + __expf(-dot)));
    }
  }
}

// REPS > 1 spreads each flush over REPS replica copies of `out`
// (replica = blockIdx % REPS; caller sums them afterwards): a zipfian
// hot key otherwise serializes thousands of same-address atomic flushes.
template <int GP, int EPL, bool SELF, class TIdx>
__global__ __launch_bounds__(256) void k_seg_accum_v4(
    float4* __restrict__ out, const TIdx* __restrict__ keys,
    const TIdx* __restrict__ slots, const float4* __restrict__ src,
    const float4* __restrict__ self, const float* __restrict__ coeff,
    int64_t n, int64_t dim4, float scale, float a, int chunk,
    int64_t rep_stride) {
  // rep_stride==0 -> single copy; reps fixed at 8 when used (& 7).
  // self[] reads stay un-replicated; only the flush target spreads.
  const int64_t rep_off = (int64_t)(blockIdx.x & 7) * rep_stride;
  const int lane = threadIdx.x % GP;
  const int64_t groups_per_block = blockDim.x / GP;
  const int64_t g0 = (int64_t)blockIdx.x * groups_per_block +
                     threadIdx.x / GP;
  const int64_t gstride = (int64_t)gridDim.x * groups_per_block;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t cidx = g0; cidx < nchunks; cidx += gstride) {
    const int64_t start = cidx * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    float4 acc[EPL];
#pragma unroll
    for (int e = 0; e < EPL; ++e) acc[e] = make_float4(0, 0, 0, 0);
    int cnt = 0;
    TIdx cur = keys[start];
    for (int64_t j = start; j < end; ++j) {
      const TIdx k = keys[j];
      if (k != cur) {
        const int64_t orow = (int64_t)cur * dim4;
#pragma unroll
        for (int e = 0; e < EPL; ++e) {
          int c = lane + e * GP;
          if (c < dim4) {
            float4 t = acc[e];
            t.x *= scale; t.y *= scale; t.z *= scale; t.w *= scale;
            if (SELF) {
              float4 sv = self[orow + c];
              float ac = a * cnt;
              t.x += ac * sv.x; t.y += ac * sv.y;
              t.z += ac * sv.z; t.w += ac * sv.w;
            }
            float* o = (float*)&out[rep_off + orow + c];
            atomicAdd(o + 0, t.x); atomicAdd(o + 1, t.y);
            atomicAdd(o + 2, t.z); atomicAdd(o + 3, t.w);
          }
          acc[e] = make_float4(0, 0, 0, 0);
        }
        cnt = 0;
        cur = k;
      }
      const float gj = coeff[j];
      const int64_t srow = (int64_t)slots[j] * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        if (c < dim4) {
          float4 sv = src[srow + c];
          acc[e].x += gj * sv.x; acc[e].y += gj * sv.y;
          acc[e].z += gj * sv.z; acc[e].w += gj * sv.w;
        }
      }
      ++cnt;
    }
    const int64_t orow = (int64_t)cur * dim4;
#pragma unroll
    for (int e = 0; e < EPL; ++e) {
      int c = lane + e * GP;
      if (c < dim4) {
        float4 t = acc[e];
        t.x *= scale; t.y *= scale; t.z *= scale; t.w *= scale;
        if (SELF) {
          float4 sv = self[orow + c];
          float ac = a * cnt;
          t.x += ac * sv.x; t.y += ac * sv.y;
          t.z += ac * sv.z; t.w += ac * sv.w;
        }
        float* o = (float*)&out[rep_off + orow + c];
        atomicAdd(o + 0, t.x); atomicAdd(o + 1, t.y);
        atomicAdd(o + 2, t.z); atomicAdd(o + 3, t.w);
      }
    }
  }
}

// out[key_j] += sum_j scale*coeff[j]*src[slots[j]]  (+ a*cnt*self[key])
// over samples sorted by key; one atomic flush per segment per chunk.
template <int GP, bool SELF>
__global__ __launch_bounds__(256) void k_seg_accum(
    float* __restrict__ out, const int64_t* __restrict__ keys,
    const int64_t* __restrict__ slots, const float* __restrict__ src,
    const float* __restrict__ self, const float* __restrict__ coeff,
    int64_t n, int64_t dim, float scale, float a, int chunk) {
  constexpr int EPL = 8;
  const int lane = threadIdx.x % GP;
  const int64_t groups_per_block = blockDim.x / GP;
  const int64_t g0 = (int64_t)blockIdx.x * groups_per_block +
                     threadIdx.x / GP;
  const int64_t gstride = (int64_t)gridDim.x * groups_per_block;
  const int64_t nchunks = (n + chunk - 1) / chunk;
  for (int64_t cidx = g0; cidx < nchunks; cidx += gstride) {
    const int64_t start = cidx * chunk;
    const int64_t end = min(n, start + (int64_t)chunk);
    float acc[EPL];
#pragma unroll
    for (int e = 0; e < EPL; ++e) acc[e] = 0.f;
    int cnt = 0;
    int64_t cur = keys[start];
    for (int64_t j = start; j < end; ++j) {
      const int64_t k = keys[j];
      if (k != cur) {
        const int64_t orow = cur * dim;
#pragma unroll
        for (int e = 0; e < EPL; ++e) {
          int c = lane + e * GP;
          if (c < dim) {
            float t = scale * acc[e];
            if (SELF) t += a * cnt * self[orow + c];
            atomicAdd(&out[orow + c], t);
          }
          acc[e] = 0.f;
        }
        cnt = 0;
        cur = k;
      }
      const float gj = coeff[j];
      const int64_t srow = slots[j] * dim;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        if (c < dim) acc[e] += gj * src[srow + c];
      }
      ++cnt;
    }
    const int64_t orow = cur * dim;
#pragma unroll
    for (int e = 0; e < EPL; ++e) {
      int c = lane + e * GP;
      if (c < dim) {
        float t = scale * acc[e];
        if (SELF) t += a * cnt * self[orow + c];
        atomicAdd(&out[orow + c], t);
      }
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
__global__ __launch_bounds__(256) void k_sparse_linear_step(
    const int64_t* __restrict__ indptr,
    const int64_t* __restrict__ feat_slots,
    const float* __restrict__ x_vals, const float* __restrict__ labels,
    const float* __restrict__ weights, float* __restrict__ margins,
    float* __restrict__ deltas, int64_t B, int mode, float C,
    int64_t rep_stride) {
  const int64_t rep_off = (int64_t)(blockIdx.x & 7) * rep_stride;
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
      m += weights[feat_slots[j]] * x;
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
      atomicAdd(&deltas[rep_off + feat_slots[j]], coeff * x_vals[j]);
  }
}

// ---------------------------------------------------------------------
// K7 (multiclass): one wave per instance; lane c owns class c
// (n_classes <= 64).  scores[c] = sum_j w[slot_j][c] * x_j  — lanes read
// consecutive class columns of each weight row (coalesced).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_pa_multiclass_step(
    const int64_t* __restrict__ indptr,
    const int64_t* __restrict__ feat_slots,
    const float* __restrict__ x_vals, const int64_t* __restrict__ labels,
    const float* __restrict__ weights, float* __restrict__ scores,
    float* __restrict__ deltas, int64_t B, int64_t nc, int mode, float C) {
  const int lane = threadIdx.x % WAVE;
  const int64_t waves_per_block = blockDim.x / WAVE;
  const int64_t w0 = (int64_t)blockIdx.x * waves_per_block +
                     threadIdx.x / WAVE;
  const int64_t wstride = (int64_t)gridDim.x * waves_per_block;
  for (int64_t i = w0; i < B; i += wstride) {
    const int64_t lo = indptr[i], hi = indptr[i + 1];
    float sc = 0.f, n2 = 0.f;
    for (int64_t j = lo; j < hi; ++j) {
      float x = x_vals[j];
      if (lane < nc) sc += weights[feat_slots[j] * nc + lane] * x;
      if (lane == 0) n2 += x * x;
    }
    n2 = __shfl(n2, 0);
    if (lane < nc) scores[i * nc + lane] = sc;
    const int64_t y = labels[i];
    if (y < 0) continue;  // unlabeled
    // max over wrong classes + the true-class score, via wave reduction
    float sy = __shfl(sc, (int)y);
    float wrong = (lane < nc && lane != y) ? sc : -INFINITY;
    int ridx = lane;
#pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) {
      float o = __shfl_xor(wrong, off);
      int oi = __shfl_xor(ridx, off);
      if (o > wrong) { wrong = o; ridx = oi; }
    }
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
      atomicAdd(&deltas[feat_slots[j] * nc + y], t);
      atomicAdd(&deltas[feat_slots[j] * nc + ridx], -t);
    }
  }
}

// =====================================================================
// Host wrappers
// =====================================================================

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor gather(torch::Tensor table, torch::Tensor rows) {
  CHECK_DEV(table); CHECK_DEV(rows); CHECK_CONT(table); CHECK_CONT(rows);
  TORCH_CHECK(table.dtype() == torch::kFloat32 &&
              rows.dtype() == torch::kInt64);
  int64_t n = rows.numel(), dim = table.size(1);
  auto out = torch::empty({n, dim}, table.options());
  if (!n) return out;
  if (dim % 4 == 0) {
    int64_t dim4 = dim / 4;
    hipLaunchKernelGGL(k_gather_vec4, dim3(grid_for(n * dim4, 256)),
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
    hipLaunchKernelGGL(k_apply_add_unique_vec4,
                       dim3(grid_for(n * dim4, 256)), dim3(256), 0,
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
// K10 bucket_pack: counting-sort a batch by a small-range key (m <=
// 16384: 2*m*4 B of dynamic LDS), fusing the payload permutation.
// Replaces torch.argsort + 3x index_select in the MF hot path (measured:
// rocPRIM merge sort was ~50% of step GPU time —
// profiles/mf32_r01_baseline_atomics.md follow-up).  Order within a key
// is unspecified (the consumers only need grouping, not stability).
// ---------------------------------------------------------------------
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

template <class TK, class TP>
__global__ __launch_bounds__(256) void k_bucket_pack(
    const TK* __restrict__ keys, const TP* __restrict__ p0,
    const float* __restrict__ p1, int* __restrict__ gcursor,
    int* __restrict__ okeys, int* __restrict__ op0,
    float* __restrict__ op1, int64_t n, int m, int chunk) {
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
      okeys[pos] = k;
      op0[pos] = (int)p0[j];
      op1[pos] = p1[j];
    }
    __syncthreads();
  }
}


// counting-sort pack (K10) for small key ranges; returns int32
// (sorted_keys, sorted_p0, sorted_p1) plus the histogram start offsets.
static constexpr int PACK_MAX_M = 16384;
static constexpr int PACK_CHUNK = 2048;

template <class TK, class TP>
static std::vector<torch::Tensor> bucket_pack_t(torch::Tensor keys,
                                                torch::Tensor p0,
                                                torch::Tensor p1, int m) {
  int64_t n = keys.numel();
  auto opts_i32 = torch::dtype(torch::kInt32).device(keys.device());
  auto hist = torch::zeros({m}, opts_i32);
  size_t lds1 = (size_t)m * sizeof(int);
  int hist_grid = (int)std::min<int64_t>((n + 4095) / 4096, 512);
  hipLaunchKernelGGL((k_key_hist<TK>), dim3(std::max(hist_grid, 1)),
                     dim3(256), lds1, cur_stream(), keys.data_ptr<TK>(),
                     hist.data_ptr<int>(), n, m);
  auto starts = torch::cumsum(hist, 0, torch::kInt32) - hist;
  auto cursor = starts.contiguous();
  auto okeys = torch::empty({n}, opts_i32);
  auto op0 = torch::empty({n}, opts_i32);
  auto op1 = torch::empty_like(p1);
  int64_t nchunks = (n + PACK_CHUNK - 1) / PACK_CHUNK;
  int grid = (int)std::min<int64_t>(nchunks, 2048);
  hipLaunchKernelGGL((k_bucket_pack<TK, TP>), dim3(grid), dim3(256),
                     2 * (size_t)m * sizeof(int), cur_stream(),
                     keys.data_ptr<TK>(), p0.data_ptr<TP>(),
                     p1.data_ptr<float>(), cursor.data_ptr<int>(),
                     okeys.data_ptr<int>(), op0.data_ptr<int>(),
                     op1.data_ptr<float>(), n, m, PACK_CHUNK);
  return {okeys, op0, op1};
}

// sort (keys, index payload, f32 payload) by key: counting sort into
// int32 when the key range m is small, int64 radix sort otherwise.
static std::vector<torch::Tensor> sort_by_key(torch::Tensor keys,
                                              torch::Tensor p0,
                                              torch::Tensor p1,
                                              int64_t m) {
  if (m > 0 && m <= PACK_MAX_M) {
    if (keys.dtype() == torch::kInt32)
      return bucket_pack_t<int, int>(keys, p0, p1, (int)m);
    return bucket_pack_t<int64_t, int64_t>(keys, p0, p1, (int)m);
  }
  auto order = torch::argsort(keys);
  return {keys.index_select(0, order).contiguous(),
          p0.index_select(0, order).contiguous(),
          p1.index_select(0, order).contiguous()};
}

// host-side lane-group dispatch shared by the pair/seg kernels
template <class F>
static void by_group(int64_t dim, F f) {
  if (dim <= 16) f(std::integral_constant<int, 16>{});
  else if (dim <= 256) f(std::integral_constant<int, 32>{});
  else { TORCH_CHECK(dim <= 512, "row dim > 512 unsupported");
         f(std::integral_constant<int, 64>{}); }
}

// (GP, EPL) choice: EPL*GP >= dim4 with GP as small as practical —
// a wave then holds 64/GP independent samples, which is what hides the
// dependent slot->row load latency of the segment walk (measured: GP=64
// on dim=300 left one sample in flight per wave, 1.4 ms per SGNS
// seg_accum; see profiles/).
template <class F>
static void by_group4(int64_t dim4, F f) {
  using namespace std;
  if (dim4 <= 8) f(integral_constant<int, 8>{}, integral_constant<int, 1>{});
  else if (dim4 <= 16) f(integral_constant<int, 8>{}, integral_constant<int, 2>{});
  else if (dim4 <= 32) f(integral_constant<int, 8>{}, integral_constant<int, 4>{});
  else if (dim4 <= 48) f(integral_constant<int, 16>{}, integral_constant<int, 3>{});
  else if (dim4 <= 80) f(integral_constant<int, 16>{}, integral_constant<int, 5>{});
  else { TORCH_CHECK(dim4 <= 128, "row dim > 512 unsupported");
         f(integral_constant<int, 32>{}, integral_constant<int, 4>{}); }
}

static constexpr int SEG_CHUNK = 64;

// seg_accum over key-sorted samples.  When `reps` > 1, `out` is a
// (reps*m, dim) replica buffer the caller sums afterwards (hot-key
// atomic spread); reps is only used on the non-SELF... both paths
// support it via rep_stride = m*dim (SELF reads self[] un-replicated).
static void seg_accum(torch::Tensor out, torch::Tensor keys,
                      torch::Tensor slots, torch::Tensor src,
                      const torch::Tensor* self, torch::Tensor coeff,
                      double scale, double a, int64_t rep_stride = 0) {
  int64_t n = keys.numel(), dim = src.size(1);
  if (!n) return;
  int64_t nchunks = (n + SEG_CHUNK - 1) / SEG_CHUNK;
  TORCH_CHECK(dim % 4 == 0, "seg_accum requires dim % 4 == 0");
  int64_t dim4 = dim / 4;
  bool i32 = keys.dtype() == torch::kInt32;
  by_group4(dim4, [&](auto tag, auto etag) {
    constexpr int GP = decltype(tag)::value;
    constexpr int EPL = decltype(etag)::value;
    dim3 grid(grid_for(nchunks * GP, 256));
    auto launch = [&](auto self_tag, auto idx_tag) {
      constexpr bool SELF = decltype(self_tag)::value;
      using TIdx = typename decltype(idx_tag)::type;
      hipLaunchKernelGGL((k_seg_accum_v4<GP, EPL, SELF, TIdx>), grid,
                         dim3(256), 0, cur_stream(),
                         (float4*)out.data_ptr<float>(),
                         keys.data_ptr<TIdx>(), slots.data_ptr<TIdx>(),
                         (const float4*)src.data_ptr<float>(),
                         SELF ? (const float4*)self->data_ptr<float>()
                              : nullptr,
                         coeff.data_ptr<float>(), n, dim4, (float)scale,
                         (float)a, SEG_CHUNK, rep_stride / 4);
    };
    using TI32 = std::integral_constant<int, 0>;
    if (self) {
      if (i32) launch(std::true_type{}, std::common_type<int>{});
      else launch(std::true_type{}, std::common_type<int64_t>{});
    } else {
      if (i32) launch(std::false_type{}, std::common_type<int>{});
      else launch(std::false_type{}, std::common_type<int64_t>{});
    }
  });
}

static void pair_coeff(torch::Tensor A, torch::Tensor arows,
                       torch::Tensor B, torch::Tensor brows,
                       torch::Tensor s_in, torch::Tensor coeff, int mode,
                       double lr) {
  int64_t n = coeff.numel(), dim = A.size(1);
  if (!n) return;
  if (dim % 4 == 0) {
    int64_t dim4 = dim / 4;
    bool i32 = arows.dtype() == torch::kInt32;
    by_group4(dim4, [&](auto tag, auto etag) {
      constexpr int GP = decltype(tag)::value;
      constexpr int EPL = decltype(etag)::value;
      auto launch = [&](auto idx_tag) {
        using TIdx = typename decltype(idx_tag)::type;
        hipLaunchKernelGGL((k_pair_coeff_v4<GP, EPL, TIdx>),
                           dim3(grid_for(n * GP, 256)), dim3(256), 0,
                           cur_stream(),
                           (const float4*)A.data_ptr<float>(),
                           arows.data_ptr<TIdx>(),
                           (const float4*)B.data_ptr<float>(),
                           brows.data_ptr<TIdx>(),
                           s_in.data_ptr<float>(), coeff.data_ptr<float>(),
                           n, dim4, mode, (float)lr);
      };
      if (i32) launch(std::common_type<int>{});
      else launch(std::common_type<int64_t>{});
    });
    return;
  }
  TORCH_CHECK(arows.dtype() == torch::kInt64,
              "scalar pair_coeff path needs int64 indices");
  by_group(dim, [&](auto tag) {
    constexpr int GP = decltype(tag)::value;
    hipLaunchKernelGGL((k_pair_coeff<GP>),
                       dim3(grid_for(n * GP, 256)), dim3(256), 0,
                       cur_stream(), A.data_ptr<float>(),
                       arows.data_ptr<int64_t>(), B.data_ptr<float>(),
                       brows.data_ptr<int64_t>(), s_in.data_ptr<float>(),
                       coeff.data_ptr<float>(), n, dim, mode, (float)lr);
  });
}

// replica count heuristic for the hot-key delta flush
static int64_t delta_reps(int64_t n, int64_t m, int64_t dim) {
  if (n < 8 * m) return 1;
  int64_t r = 8;
  if (r * m * dim * 4 > (64 << 20)) return 1;  // cap replica buffer 64 MB
  return r;
}

// K4: e = r - u0.v computed on the item-sorted batch (exact batch-
// snapshot semantics: every rating sees pre-update u0 and the pulled v);
// item deltas = lr*(e*u0 - reg*v) segment-accumulated by item; user
// updates = lr*(e*v - reg*u0) segment-accumulated by user.
torch::Tensor mf_sgd_step(torch::Tensor user_table, torch::Tensor user_rows,
                          torch::Tensor item_vecs, torch::Tensor item_slots,
                          torch::Tensor ratings, double lr, double reg) {
  CHECK_DEV(user_table); CHECK_CONT(user_table);
  CHECK_DEV(item_vecs); CHECK_CONT(item_vecs);
  CHECK_CONT(user_rows); CHECK_CONT(item_slots); CHECK_CONT(ratings);
  int64_t n = ratings.numel(), dim = user_table.size(1);
  int64_t m = item_vecs.size(0);
  TORCH_CHECK(item_vecs.size(1) == dim);
  if (!n) return torch::zeros_like(item_vecs);
  // pass A: counting-sort by item (K10), compute e, accumulate deltas
  auto packA = sort_by_key(item_slots, user_rows, ratings, m);
  auto& s_slots = packA[0];
  auto& s_urows = packA[1];
  auto& s_r = packA[2];
  auto e = torch::empty_like(s_r);
  pair_coeff(user_table, s_urows, item_vecs, s_slots, s_r, e, 0, 0.0);
  int64_t reps = (dim % 4 == 0) ? delta_reps(n, m, dim) : 1;
  torch::Tensor deltas;
  if (reps > 1) {
    auto rep_buf = torch::zeros({reps * m, dim}, item_vecs.options());
    seg_accum(rep_buf, s_slots, s_urows, user_table, &item_vecs, e, lr,
              -lr * reg, m * dim);
    deltas = rep_buf.view({reps, m, dim}).sum(0);
  } else {
    deltas = torch::zeros_like(item_vecs);
    seg_accum(deltas, s_slots, s_urows, user_table, &item_vecs, e, lr,
              -lr * reg);
  }
  // pass B: counting-sort by user, apply user updates in place
  auto packB = sort_by_key(s_urows, s_slots, e, user_table.size(0));
  seg_accum(user_table, packB[0], packB[1], item_vecs, &user_table,
            packB[2], lr, -lr * reg);
  return deltas;
}

// K8: g = lr*(label - sigmoid(v.u)); center and context deltas each
// segment-accumulated on their own sorted order.
std::vector<torch::Tensor> sgns_step(
    torch::Tensor center_vecs, torch::Tensor center_slots,
    torch::Tensor ctx_vecs, torch::Tensor ctx_slots, torch::Tensor labels,
    double lr) {
  CHECK_DEV(center_vecs); CHECK_CONT(center_vecs);
  CHECK_DEV(ctx_vecs); CHECK_CONT(ctx_vecs);
  int64_t n = labels.numel();
  auto cd = torch::zeros_like(center_vecs);
  auto od = torch::zeros_like(ctx_vecs);
  if (!n) return {cd, od};
  auto g = torch::empty_like(labels);
  pair_coeff(center_vecs, center_slots, ctx_vecs, ctx_slots, labels, g, 1,
             lr);
  auto packC = sort_by_key(center_slots, ctx_slots, g,
                           center_vecs.size(0));
  seg_accum(cd, packC[0], packC[1], ctx_vecs, nullptr, packC[2], 1.0, 0.0);
  auto packO = sort_by_key(ctx_slots, center_slots, g, ctx_vecs.size(0));
  seg_accum(od, packO[0], packO[1], center_vecs, nullptr, packO[2], 1.0,
            0.0);
  return {cd, od};
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
  int64_t reps = (x_vals.numel() >= 8 * m &&
                  8 * m * 4 <= (256ll << 20)) ? 8 : 1;
  auto dbuf = torch::zeros({reps * m}, weights.options());
  hipLaunchKernelGGL(k_sparse_linear_step, dim3(grid_for(B * WAVE, 256)),
                     dim3(256), 0, cur_stream(),
                     indptr.data_ptr<int64_t>(),
                     feat_slots.data_ptr<int64_t>(),
                     x_vals.data_ptr<float>(), labels.data_ptr<float>(),
                     weights.data_ptr<float>(), margins.data_ptr<float>(),
                     dbuf.data_ptr<float>(), B, pa_mode(variant),
                     (float)C, reps > 1 ? m : 0);
  auto deltas = reps > 1 ? dbuf.view({reps, m}).sum(0) : dbuf;
  return {margins, deltas};
}

std::vector<torch::Tensor> lr_step(torch::Tensor indptr,
                                   torch::Tensor feat_slots,
                                   torch::Tensor x_vals,
                                   torch::Tensor labels,
                                   torch::Tensor weights) {
  CHECK_DEV(weights); CHECK_CONT(weights);
  int64_t B = indptr.numel() - 1;
  auto probs = torch::zeros({B}, x_vals.options());
  if (!B) return {probs, torch::zeros_like(weights)};
  int64_t m = weights.numel();
  int64_t reps = (x_vals.numel() >= 8 * m &&
                  8 * m * 4 <= (256ll << 20)) ? 8 : 1;
  auto gbuf = torch::zeros({reps * m}, weights.options());
  hipLaunchKernelGGL(k_sparse_linear_step, dim3(grid_for(B * WAVE, 256)),
                     dim3(256), 0, cur_stream(),
                     indptr.data_ptr<int64_t>(),
                     feat_slots.data_ptr<int64_t>(),
                     x_vals.data_ptr<float>(), labels.data_ptr<float>(),
                     weights.data_ptr<float>(), probs.data_ptr<float>(),
                     gbuf.data_ptr<float>(), B, /*mode=*/3, 0.f,
                     reps > 1 ? m : 0);
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
  TORCH_CHECK(nc <= 64, "n_classes > 64 unsupported");
  auto scores = torch::zeros({B, nc}, x_vals.options());
  auto deltas = torch::zeros_like(weights);
  if (!B) return {scores, deltas};
  hipLaunchKernelGGL(k_pa_multiclass_step, dim3(grid_for(B * WAVE, 256)),
                     dim3(256), 0, cur_stream(),
                     indptr.data_ptr<int64_t>(),
                     feat_slots.data_ptr<int64_t>(),
                     x_vals.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     weights.data_ptr<float>(), scores.data_ptr<float>(),
                     deltas.data_ptr<float>(), B, nc, pa_mode(variant),
                     (float)C);
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
  m.def("mf_sgd_step", &mf_sgd_step, "K4 fused MF SGD step");
  m.def("sgns_step", &sgns_step, "K8 fused SGNS step");
  m.def("pa_step", &pa_step, "K7 PA binary step");
  m.def("pa_multiclass_step", &pa_multiclass_step, "K7 PA multiclass");
  m.def("lr_step", &lr_step, "K9 sparse logistic step");
}
