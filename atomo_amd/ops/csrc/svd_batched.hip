// Batched SVD-encode kernels for MI355X (gfx950).
//
// The SVD encode path (SURVEY §2.10 row "LA.svd") is restructured for the
// GPU as: per-layer Gram matrices -> batched eigensolves -> selection
// writing the wire packets.  These kernels batch the two small-fold GPU
// phases over ALL small layers of a model in ONE launch each, driven by a
// descriptor table, so a ~60-layer model costs 2 launches instead of ~250:
//
//   batched_gram_kernel   G_l = A_l^T A_l (tall, n<=64) or A_l A_l^T
//                         (wide, m<=64); LDS-staged row chunks, per-thread
//                         pair accumulation, one atomicAdd per pair/chunk.
//   batched_sel_kernel    tall factor = A_l @ sel_l (r_hat read on-device
//                         from the staged selection, so launch geometry is
//                         step-invariant) written STRAIGHT into the wire
//                         region; tile 0 also scatters header/s/small-factor.
//
// Layers with a big small-dim (> 64, the randomized-solver folds) run
// their Grams and selection as shape-grouped rocBLAS bmms in python
// (svd_encoder._compute_big_grams / _sel_big_folds); odd zero-padded
// folds stay on per-layer rocBLAS GEMMs.  The staged facT/sel blocks use
// a FIXED r_max stride, zero past the sampled r_hat, so the fixed-shape
// GEMM consumers need no host sync.
//
// Descriptor layout (int64, per layer, GD_N words):
//   [0] a_off      offset of the 2-D fold (m x n, row-major) in flat_grad
//   [1] m          fold rows
//   [2] n          fold cols
//   [3] is_tall    1 if m >= n (gram/factor over columns), else 0
//   [4] gram_off   offset of G (sm x sm) in the gram buffer
//   [5] wire_off   offset of this layer's packet in the wire buffer
//   [6] stage_off  offset of this layer's staged selection
//                  [r_hat | s_wire(r_max) | facT(r_max*sm) | sel(sm*r_max)]
//   [7] r_max      wire budget
//
// Work maps: precomputed (tile -> layer, chunk) pairs, static per model.

#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE 64
#define GD_N 8
// rows of the long axis consumed per tile
#define GRAM_CHUNK 256
#define SEL_ROWS_PER_THREAD 4
#define SEL_CHUNK (256 * SEL_ROWS_PER_THREAD)

namespace {

// ---------------------------------------------------------------------------
// batched Gram: one workgroup per (layer, long-axis chunk)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) batched_gram_kernel(
    const float* __restrict__ flat, float* __restrict__ grams,
    const int64_t* __restrict__ desc, const int32_t* __restrict__ work,
    int n_tiles) {
  __shared__ float lds[GRAM_CHUNK * 64];  // chunk rows x sm (sm <= 64)
  for (int tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    const int layer = work[2 * tile];
    const int chunk = work[2 * tile + 1];
    const int64_t* d = desc + (int64_t)layer * GD_N;
    const int64_t a_off = d[0];
    const int m = (int)d[1], n = (int)d[2];
    const bool is_tall = d[3] != 0;
    const int sm = is_tall ? n : m;
    const int tall = is_tall ? m : n;
    float* G = grams + d[4];
    const int t0 = chunk * GRAM_CHUNK;
    const int t1 = min(t0 + GRAM_CHUNK, tall);
    const int rows = t1 - t0;
    // stage A2'[t0:t1, :] into LDS (A2'[t,s] = A[t,s] tall / A[s,t] wide)
    const float* A = flat + a_off;
    if (is_tall) {
      // contiguous block of rows: coalesced copy of rows*n floats
      const int64_t base = (int64_t)t0 * n;
      for (int i = threadIdx.x; i < rows * n; i += blockDim.x)
        lds[i] = A[base + i];  // lds[(t-t0)*n + s]
    } else {
      // A2'[t,s] = A[s*n + t], t in [t0,t1), s in [0,m)
      for (int i = threadIdx.x; i < rows * m; i += blockDim.x) {
        const int t = i / m, s = i % m;  // lds[t*sm + s] layout (row-major)
        lds[(int64_t)t * m + s] = A[(int64_t)s * n + t0 + t];
      }
    }
    __syncthreads();
    // each thread accumulates a strided subset of (i,j) pairs
    const int pairs = sm * sm;
    for (int p = threadIdx.x; p < pairs; p += blockDim.x) {
      const int i = p / sm, j = p % sm;
      if (j < i) continue;  // symmetric: do upper triangle, mirror below
      float acc = 0.f;
      const float* l = lds;
      for (int t = 0; t < rows; ++t, l += sm) acc = fmaf(l[i], l[j], acc);
      atomicAdd(&G[(int64_t)i * sm + j], acc);
      if (j != i) atomicAdd(&G[(int64_t)j * sm + i], acc);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// batched selection GEMM + packet scatter
// tall case: uT(r,t) = sum_s A[t,s] * sel[s,r]   -> wire[wo+1 + r*m + t]
// wide case: vT(r,t) = sum_s A[s,t] * sel[s,r]   -> wire[wo+1+r_max*(m+1)+r*n+t]
// ---------------------------------------------------------------------------
template <int RCAP>
__device__ __forceinline__ void sel_rows(const float* __restrict__ A,
                                         float* __restrict__ wire,
                                         const float* __restrict__ sel_lds,
                                         int t_begin, int tall, int m, int n,
                                         bool is_tall, int r_hat, int r_max,
                                         int64_t t_dst) {
  const int sm = is_tall ? n : m;
  for (int rr = 0; rr < SEL_ROWS_PER_THREAD; ++rr) {
    const int t = t_begin + rr * (int)blockDim.x + (int)threadIdx.x;
    if (t >= tall) break;
    float acc[RCAP];
#pragma unroll
    for (int r = 0; r < RCAP; ++r) acc[r] = 0.f;
    if (is_tall) {
      const float* a_row = A + (int64_t)t * n;
      for (int s = 0; s < sm; ++s) {
        const float av = a_row[s];
        const float* sl = sel_lds + (int64_t)s * r_max;
#pragma unroll
        for (int r = 0; r < RCAP; ++r)
          acc[r] = fmaf(av, (r < r_hat) ? sl[r] : 0.f, acc[r]);
      }
    } else {
      for (int s = 0; s < sm; ++s) {
        const float av = A[(int64_t)s * n + t];
        const float* sl = sel_lds + (int64_t)s * r_max;
#pragma unroll
        for (int r = 0; r < RCAP; ++r)
          acc[r] = fmaf(av, (r < r_hat) ? sl[r] : 0.f, acc[r]);
      }
    }
    for (int r = 0; r < r_hat; ++r)
      wire[t_dst + (int64_t)r * tall + t] = acc[r];
  }
}
__global__ void __launch_bounds__(256) batched_sel_kernel(
    const float* __restrict__ flat, float* __restrict__ wire,
    const float* __restrict__ stage, const int64_t* __restrict__ desc,
    const int32_t* __restrict__ work, int n_tiles, int sel_elems) {
  // dynamic LDS: [sel (sel_elems floats)] [cached_layer (1 int)]
  extern __shared__ __attribute__((aligned(16))) float sel_lds[];
  int* cached_layer_p = (int*)(sel_lds + sel_elems);
#define cached_layer_s (*cached_layer_p)
  if (threadIdx.x == 0) cached_layer_s = -1;
  __syncthreads();
  for (int tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    const int layer = work[2 * tile];
    const int chunk = work[2 * tile + 1];
    const int64_t* d = desc + (int64_t)layer * GD_N;
    const int64_t a_off = d[0];
    const int m = (int)d[1], n = (int)d[2];
    const bool is_tall = d[3] != 0;
    const int sm = is_tall ? n : m;
    const int tall = is_tall ? m : n;
    const int64_t wo = d[5], so = d[6];
    const int r_max = (int)d[7];
    const int r_hat = (int)stage[so];
    // (re)load this layer's selection into LDS
    if (cached_layer_s != layer) {
      __syncthreads();
      const float* sel = stage + so + 1 + (int64_t)r_max * (1 + sm);
      for (int i = threadIdx.x; i < sm * r_max; i += blockDim.x)
        sel_lds[i] = sel[i];  // stored (sm, r_max) row-major, 0 past r_hat
      if (threadIdx.x == 0) cached_layer_s = layer;
      __syncthreads();
    }
    if (chunk == 0) {
      // packet scatter: header, s_wire, small factor
      if (threadIdx.x == 0) wire[wo] = stage[so];
      for (int k = threadIdx.x; k < r_hat; k += blockDim.x)
        wire[wo + 1 + (int64_t)r_max * m + k] = stage[so + 1 + k];
      const float* facT = stage + so + 1 + r_max;  // (r_hat, sm) row-major
      // tall: facT is vT (sm == n); wide: facT is uT (sm == m)
      const int64_t f_dst =
          is_tall ? wo + 1 + (int64_t)r_max * (m + 1) : wo + 1;
      for (int i = threadIdx.x; i < r_hat * sm; i += blockDim.x)
        wire[f_dst + i] = facT[i];
    }
    if (r_hat == 0) continue;
    // tall-factor GEMM for this chunk of t (RCAP templated so the
    // accumulators stay in registers — §5.4 rule 20)
    const float* A = flat + a_off;
    const int64_t t_dst = is_tall ? wo + 1 : wo + 1 + (int64_t)r_max * (m + 1);
    const int t_begin = chunk * SEL_CHUNK;
    if (r_hat <= 4)
      sel_rows<4>(A, wire, sel_lds, t_begin, tall, m, n, is_tall, r_hat,
                  r_max, t_dst);
    else if (r_hat <= 8)
      sel_rows<8>(A, wire, sel_lds, t_begin, tall, m, n, is_tall, r_hat,
                  r_max, t_dst);
    else if (r_hat <= 16)
      sel_rows<16>(A, wire, sel_lds, t_begin, tall, m, n, is_tall, r_hat,
                   r_max, t_dst);
    else
      sel_rows<32>(A, wire, sel_lds, t_begin, tall, m, n, is_tall, r_hat,
                   r_max, t_dst);
  }
#undef cached_layer_s
}

inline int grid_for_tiles(int tiles) {
  int g = tiles;
  if (g > 8192) g = 8192;
  if (g < 1) g = 1;
  return g;
}

}  // namespace

extern "C" {

void atomo_batched_gram_launch(const float* flat, float* grams,
                               const int64_t* desc, const int32_t* work,
                               int n_tiles, hipStream_t stream) {
  hipLaunchKernelGGL(batched_gram_kernel, dim3(grid_for_tiles(n_tiles)),
                     dim3(256), 0, stream, flat, grams, desc, work, n_tiles);
}

void atomo_batched_sel_launch(const float* flat, float* wire,
                              const float* stage, const int64_t* desc,
                              const int32_t* work, int n_tiles, int sel_elems,
                              hipStream_t stream) {
  // +16 for the cached-layer word; +128 because sel_rows' predicated
  // (r < r_hat ? ...) reads may touch up to RCAP-1 floats past the last
  // staged element (values never used, but keep them in-bounds)
  const size_t lds = (size_t)sel_elems * 4 + 16 + 128;
  hipLaunchKernelGGL(batched_sel_kernel, dim3(grid_for_tiles(n_tiles)),
                     dim3(256), lds, stream, flat, wire, stage, desc, work,
                     n_tiles, sel_elems);
}

}  // extern "C"
