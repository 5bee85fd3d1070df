// MFMA A/B microbenchmark (VERDICT round-1 item #2): settle "MFMA for
// the rank-k products" with data on the exact hot shape.
//
// The MF/SGNS hot phase computes, per key-sorted record j,
//     e_j = val_j - dot(keyed[key_j], other[slot_j])        (dim = 128 / 300)
// i.e. a batched GEMV (one shared keyed row per segment, a gathered
// `other` row per record) — NOT a GEMM.  The most favorable possible
// MFMA formulation computes 16 records' dots per v_mfma_f32_16x16x4_f32
// chain by staging A = 16 gathered `other` rows and B = 16 keyed rows in
// LDS and extracting diag(C): both operands are "useful", but only 16
// of the 256 outputs are — a structural 1/16 FLOP efficiency — and
// gfx950's f32-input MFMA runs at the f32 VECTOR rate (155 TF, equal to
// VALU; MI355X_MICROARCH.md "Matrix cores"), so the ceiling of this
// formulation is 155/16 ≈ 9.7 TF versus the VALU walk's direct use of
// the vector units on exactly-useful FLOPs.
//
// This file measures that head-to-head (plus a loads-stripped MFMA
// variant to separate fragment-staging cost from MFMA issue cost):
//   k_dot_valu  — the shipped k_seg_fused dot structure (GP-lane groups,
//                 per-segment keyed-row register cache, float4 loads)
//   k_dot_mfma  — LDS-staged 16-record fragments + 16x16x4 f32 MFMA
//                 k-loop, diag extraction, same e output
//
// Build:  hipcc --offload-arch=gfx950 -O3 tools/mfma_ab.hip -o mfma_ab
// Run:    ./mfma_ab [dim]          (dim = 128 | 300; default both)
// Output: per-kernel ms over the same data + max |Δe| cross-check.
// Results + the keep/reject decision: profiles/mfma_ab_r02.md.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <cmath>
#include <vector>

#define WAVE 64
#define CHECK(x)                                                        \
  do {                                                                  \
    hipError_t e_ = (x);                                                \
    if (e_ != hipSuccess) {                                             \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_), \
              __FILE__, __LINE__);                                      \
      exit(1);                                                          \
    }                                                                   \
  } while (0)

typedef float v4f __attribute__((ext_vector_type(4)));

static float bits_to_float(int b) { float f; memcpy(&f, &b, 4); return f; }
static int float_to_bits(float f) { int b; memcpy(&b, &f, 4); return b; }

template <int GP>
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int off = GP / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, GP);
  return v;
}

// ------------------------------------------------------------------
// VALU walk: the shipped k_seg_fused dot structure (MODE 0, stripped of
// the delta accumulation so both kernels do identical work: dot + e
// writeback).  GP lanes per record-chunk walk, keyed row cached in
// registers per segment, float4 traffic.
// ------------------------------------------------------------------
template <int GP, int EPL>
__global__ __launch_bounds__(256) void k_dot_valu(
    int4* __restrict__ rec, const float4* __restrict__ keyed,
    const float4* __restrict__ other, long n, int dim4, int chunk) {
  const int lane = threadIdx.x % GP;
  const long gpb = blockDim.x / GP;
  const long g0 = (long)blockIdx.x * gpb + threadIdx.x / GP;
  const long gstride = (long)gridDim.x * gpb;
  const long nchunks = (n + chunk - 1) / chunk;
  for (long cidx = g0; cidx < nchunks; cidx += gstride) {
    const long start = cidx * chunk;
    const long end = start + chunk < n ? start + chunk : n;
    float4 kv[EPL];
    int cur = rec[start].x;
    auto load_kv = [&]() {
      const long kr = (long)cur * dim4;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        kv[e] = (c < dim4) ? keyed[kr + c] : make_float4(0, 0, 0, 0);
      }
    };
    load_kv();
    for (long j = start; j < end; ++j) {
      const int4 rj = rec[j];
      if (rj.x != cur) {
        cur = rj.x;
        load_kv();
      }
      const long sr = (long)rj.y * dim4;
      float dot = 0.f;
#pragma unroll
      for (int e = 0; e < EPL; ++e) {
        int c = lane + e * GP;
        if (c < dim4) {
          float4 sv = other[sr + c];
          dot += kv[e].x * sv.x + kv[e].y * sv.y + kv[e].z * sv.z +
                 kv[e].w * sv.w;
        }
      }
      dot = group_sum<GP>(dot);
      if (lane == 0) rec[j].z = __float_as_int(__int_as_float(rj.z) - dot);
    }
  }
}

// ------------------------------------------------------------------
// MFMA variant: one wave processes 16 records per iteration.
//  1. cooperative LDS stage of A (16 gathered `other` rows) and B (16
//     keyed rows), coalesced float loads, rows padded to kill the
//     16-way bank conflict of the fragment reads
//  2. k-loop of dim/4 v_mfma_f32_16x16x4_f32 into one f32x4 accumulator
//  3. diag extraction: lane l holds C[row=(l>>4)*4+r][col=l&15]; the
//     row==col element is record `row`'s dot
// STRIP_LOADS=1 replaces the gathers with constant fragments to isolate
// MFMA issue cost from staging cost.
// ------------------------------------------------------------------
template <int DIM, int PAD, int WPB, bool STRIP_LOADS>
__global__ __launch_bounds__(WPB * WAVE) void k_dot_mfma(
    int4* __restrict__ rec, const float* __restrict__ keyed,
    const float* __restrict__ other, long n) {
  __shared__ float lds[WPB][2][16][DIM + PAD];
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  float(*A)[DIM + PAD] = lds[wave][0];
  float(*B)[DIM + PAD] = lds[wave][1];
  const long ngroups = (n + 15) / 16;
  // block-uniform outer loop so __syncthreads() (fencing each wave's
  // LDS staging) is reached by every wave every iteration
  for (long base = (long)blockIdx.x * WPB; base < ngroups;
       base += (long)gridDim.x * WPB) {
    const long g = base + wave;
    const bool active = g < ngroups;
    const long j0 = g * 16;
    const int cnt = active ? (int)(n - j0 < 16 ? n - j0 : 16) : 0;
    if (!STRIP_LOADS && active) {
      // stage fragments: 16 rows x DIM floats each, coalesced
      for (int idx = lane; idx < 16 * DIM; idx += WAVE) {
        const int row = idx / DIM, col = idx - row * DIM;
        const int4 rj = rec[j0 + (row < cnt ? row : cnt - 1)];
        A[row][col] = other[(long)rj.y * DIM + col];
        B[row][col] = keyed[(long)rj.x * DIM + col];
      }
    }
    if (!STRIP_LOADS) __syncthreads();
    if (active) {
      using f32x4 = __attribute__((ext_vector_type(4))) float;
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const int arow = lane & 15, ak = lane >> 4;  // A[i][k] / B[k][j]
#pragma unroll 4
      for (int kk = 0; kk < DIM / 4; ++kk) {
        float a, b;
        if (STRIP_LOADS) {
          a = (float)(lane + kk);
          b = (float)(lane - kk);
        } else {
          a = A[arow][kk * 4 + ak];
          b = B[arow][kk * 4 + ak];  // B fragment: B[k][j], j=lane&15
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
      }
      const int col = lane & 15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (lane >> 4) * 4 + r;
        if (row == col && row < cnt) {
          const long j = j0 + row;
          const int4 rj = rec[j];
          rec[j].z = __float_as_int(__int_as_float(rj.z) - acc[r]);
        }
      }
    }
    if (!STRIP_LOADS) __syncthreads();
  }
}

// ------------------------------------------------------------------
static double bench(void (*launch)(void*), void* arg, int iters) {
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  launch(arg);  // warm
  CHECK(hipDeviceSynchronize());
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i) launch(arg);
  CHECK(hipEventRecord(b));
  CHECK(hipEventSynchronize(b));
  float ms;
  CHECK(hipEventElapsedTime(&ms, a, b));
  CHECK(hipEventDestroy(a));
  CHECK(hipEventDestroy(b));
  return ms / iters;
}

struct Ctx {
  int4* rec;
  int4* rec_work;
  float* keyed;
  float* other;
  long n;
  int dim;
};

template <int DIM>
static void run_dim(long n, long m, long U) {
  printf("== dim %d: n=%ld records, m=%ld keys (zipf-sorted), U=%ld rows\n",
         DIM, n, m, U);
  std::vector<int4> h_rec(n);
  // zipf-ish keys, sorted (the walk's contract); slots uniform
  srand(42);
  std::vector<int> keys(n);
  for (long j = 0; j < n; ++j) {
    double u = (rand() + 1.0) / (RAND_MAX + 2.0);
    keys[j] = (int)(m * u * u * u) % (int)m;
  }
  std::sort(keys.begin(), keys.end());
  for (long j = 0; j < n; ++j)
    h_rec[j] = make_int4(keys[j], rand() % (int)U,
                         float_to_bits((float)(rand() % 5 - 2)),
                         0);
  std::vector<float> h_keyed(m * DIM), h_other(U * DIM);
  for (auto& v : h_keyed) v = (rand() % 1000 - 500) / 500.0f;
  for (auto& v : h_other) v = (rand() % 1000 - 500) / 500.0f;

  Ctx c;
  c.n = n;
  c.dim = DIM;
  CHECK(hipMalloc(&c.rec, n * sizeof(int4)));
  CHECK(hipMalloc(&c.rec_work, n * sizeof(int4)));
  CHECK(hipMalloc(&c.keyed, m * DIM * sizeof(float)));
  CHECK(hipMalloc(&c.other, U * DIM * sizeof(float)));
  CHECK(hipMemcpy(c.rec, h_rec.data(), n * sizeof(int4),
                  hipMemcpyHostToDevice));
  CHECK(hipMemcpy(c.keyed, h_keyed.data(), m * DIM * sizeof(float),
                  hipMemcpyHostToDevice));
  CHECK(hipMemcpy(c.other, h_other.data(), U * DIM * sizeof(float),
                  hipMemcpyHostToDevice));

  auto reset = [&]() {
    CHECK(hipMemcpy(c.rec_work, c.rec, n * sizeof(int4),
                    hipMemcpyDeviceToDevice));
  };

  constexpr int GP = 16, EPL = (DIM + 63) / 64;  // 128->2, 300->5 float4s
  const int chunk = 256;
  auto launch_valu = [](void* p) {
    Ctx* c = (Ctx*)p;
    long nchunks = (c->n + 255) / 256;
    int grid = (int)std::min<long>((nchunks * GP + 255) / 256, 2048);
    hipLaunchKernelGGL((k_dot_valu<GP, EPL>), dim3(grid), dim3(256), 0, 0,
                       c->rec_work, (const float4*)c->keyed,
                       (const float4*)c->other, c->n, c->dim / 4, 256);
  };
  constexpr int PAD = 5;  // odd stride => 16 consecutive rows hit 16 distinct banks
  constexpr int WPB = (2 * 16 * (DIM + PAD) * 4 * 4 <= 160 * 1024) ? 4 : 2;
  auto launch_mfma = [](void* p) {
    Ctx* c = (Ctx*)p;
    long ngroups = (c->n + 15) / 16;
    int grid = (int)std::min<long>((ngroups + WPB - 1) / WPB, 2048);
    hipLaunchKernelGGL((k_dot_mfma<DIM, PAD, WPB, false>), dim3(grid),
                       dim3(WPB * WAVE), 0, 0, c->rec_work, c->keyed,
                       c->other, c->n);
  };
  auto launch_mfma_noload = [](void* p) {
    Ctx* c = (Ctx*)p;
    long ngroups = (c->n + 15) / 16;
    int grid = (int)std::min<long>((ngroups + WPB - 1) / WPB, 2048);
    hipLaunchKernelGGL((k_dot_mfma<DIM, PAD, WPB, true>), dim3(grid),
                       dim3(WPB * WAVE), 0, 0, c->rec_work, c->keyed,
                       c->other, c->n);
  };

  // correctness cross-check: run both once, compare e
  reset();
  launch_valu(&c);
  CHECK(hipDeviceSynchronize());
  std::vector<int4> e_valu(n);
  CHECK(hipMemcpy(e_valu.data(), c.rec_work, n * sizeof(int4),
                  hipMemcpyDeviceToHost));
  reset();
  launch_mfma(&c);
  CHECK(hipDeviceSynchronize());
  std::vector<int4> e_mfma(n);
  CHECK(hipMemcpy(e_mfma.data(), c.rec_work, n * sizeof(int4),
                  hipMemcpyDeviceToHost));
  double max_d = 0, max_a = 0;
  for (long j = 0; j < n; ++j) {
    float a = bits_to_float(e_valu[j].z);
    float b = bits_to_float(e_mfma[j].z);
    max_d = std::max(max_d, (double)fabsf(a - b));
    max_a = std::max(max_a, (double)fabsf(a));
  }
  printf("   cross-check: max|e_valu - e_mfma| = %.3e (max|e| %.3e)\n",
         max_d, max_a);

  reset();
  double t_valu = bench(launch_valu, &c, 20);
  reset();
  double t_mfma = bench(launch_mfma, &c, 20);
  reset();
  double t_mfma_nl = bench(launch_mfma_noload, &c, 20);
  double flop = 2.0 * n * DIM;
  printf("   VALU walk        : %8.3f ms  (%6.1f GFLOP/s useful)\n",
         t_valu, flop / t_valu / 1e6);
  printf("   MFMA 16x16x4 diag: %8.3f ms  (%6.1f GFLOP/s useful, "
         "%6.1f GFLOP/s issued incl. 16x waste)\n",
         t_mfma, flop / t_mfma / 1e6, 16 * flop / t_mfma / 1e6);
  printf("   MFMA loads-strip : %8.3f ms  (staging cost = %.3f ms)\n",
         t_mfma_nl, t_mfma - t_mfma_nl);
  printf("   => MFMA/VALU = %.2fx %s\n", t_mfma / t_valu,
         t_mfma > t_valu ? "(MFMA slower)" : "(MFMA faster)");

  CHECK(hipFree(c.rec));
  CHECK(hipFree(c.rec_work));
  CHECK(hipFree(c.keyed));
  CHECK(hipFree(c.other));
}

int main(int argc, char** argv) {
  int dim = argc > 1 ? atoi(argv[1]) : 0;
  // mf128 shape: long-tail keys, 512-B rows; SGNS-300 shape
  if (dim == 0 || dim == 128) run_dim<128>(1 << 22, 100000, 1 << 20);
  if (dim == 0 || dim == 300) run_dim<300>(1 << 21, 100000, 1 << 20);
  return 0;
}
