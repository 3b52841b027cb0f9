// Batched symmetric eigensolvers (cyclic parallel Jacobi) + selection-stage
// builder for MI355X (gfx950).
//
// Solves the per-layer Gram matrices ON DEVICE so the SVD-encode path never
// ships Grams to the host (reference equivalent: numpy LA.svd per layer,
// codings/svd.py:95).  Two variants:
//
//   jacobi_eigh_kernel      sm <= 64 (or <= 128 via the JMAX=128 variant):
//                           G and V in LDS (row stride JMAX+1 keeps column
//                           walks conflict-free), 256 threads/WG, warm-start
//                           pre-rotation from the previous step's basis.
//   jacobi_eigh_big_kernel  global-memory variant (G in its Gram slot, V in
//                           scratch).  Measured latency-bound and unused by
//                           default — bigger folds go to batched hipSOLVER
//                           syevd (see svd_encoder.py routing).
//
// Both: each Jacobi round applies all N/2 disjoint plane rotations in two
// barrier-separated phases (rows = J^T G, then cols = .J and V.J); pairs
// follow the round-robin tournament schedule; early exit when the
// off-diagonal Frobenius norm drops below 1e-10 x ||G||_F^2.  Eigenvalues
// sort descending; eigenvectors overwrite the Gram slot.
//
// build_stage_kernel gathers the host-sampled atom selection (idx, probs
// per layer) into the staged wire factors:
//   s_wire[r] = sqrt(eval[idx_r]) / p_r
//   facT[r][k] = V[k][idx_r]                  (the small wire factor)
//   sel[k][r]  = V[k][idx_r] / sqrt(eval)     (A @ sel = tall factor)

#include <hip/hip_runtime.h>

#include <cstdint>

#define GD_N 8
#define SWEEPS 10
// 256 threads/WG: measured faster than single-wave — the warm
// pre-rotation (two N^3 LDS matmuls) is throughput-bound and wants the
// threads; the dense small-batch variant below is the opposite case
#define JTHREADS 256
#define JBIG_THREADS 512
#define R_CAP 32
#define SEL_ROW 65  // [r_hat | idx*32 | probs*32]

namespace {

__device__ __forceinline__ void rot_params(float app, float aqq, float apq,
                                           float* c, float* s) {
  *c = 1.f;
  *s = 0.f;
  if (fabsf(apq) > 1e-12f) {
    const float tau = (aqq - app) / (2.f * apq);
    const float t =
        (tau >= 0.f ? 1.f : -1.f) / (fabsf(tau) + sqrtf(1.f + tau * tau));
    *c = rsqrtf(1.f + t * t);
    *s = t * (*c);
  }
}

// ---------------------------------------------------------------------------
// small variant: LDS-resident
// ---------------------------------------------------------------------------
// vwarm: persistent per-layer eigenvector basis from the PREVIOUS step.
// Gradients (hence Grams) evolve slowly, so pre-rotating B = Vw^T G Vw
// leaves B nearly diagonal and the sweep loop exits after ~1-2 sweeps
// instead of ~6-8 from scratch.  The final basis is V = Vw . R and is
// written back to vwarm for the next step.
template <int JMAX, bool WARM>
__global__ void __launch_bounds__(JTHREADS) jacobi_eigh_kernel(
    float* __restrict__ grams, float* __restrict__ evals,
    const int64_t* __restrict__ desc, const int64_t* __restrict__ eval_offs,
    const int32_t* __restrict__ rows_list, int n_mats,
    float* __restrict__ vwarm, const int64_t* __restrict__ vwarm_offs,
    int save_warm, int max_sweeps) {
  constexpr int JSTRIDE = JMAX + 1;
  __shared__ float G[JMAX * JSTRIDE];
  __shared__ float V[JMAX * JSTRIDE];
  __shared__ float T[WARM ? JMAX * JSTRIDE : 1];
  __shared__ float cs[JMAX / 2], sn[JMAX / 2];
  __shared__ int pp[JMAX / 2], qq[JMAX / 2];
  __shared__ int order[JMAX];
  __shared__ float offsq[JTHREADS / 64];
  __shared__ int done_s;

  if (blockIdx.x >= (unsigned)n_mats) return;
  const int row = rows_list[blockIdx.x];
  const int64_t* d = desc + (int64_t)row * GD_N;
  const int m = (int)d[1], n = (int)d[2];
  const bool is_tall = d[3] != 0;
  const int sm = is_tall ? n : m;
  float* Gg = grams + d[4];
  float* ev = evals + eval_offs[row];
  float* Vw = (WARM || save_warm) ? vwarm + vwarm_offs[blockIdx.x] : nullptr;
  const int tid = threadIdx.x;
  const int N = (sm + 1) & ~1;  // even-padded

  float fro2 = 0.f;
  for (int i = tid; i < N * N; i += JTHREADS) {
    const int r = i / N, c = i % N;
    const float g = (r < sm && c < sm) ? Gg[r * sm + c] : 0.f;
    G[r * JSTRIDE + c] = g;
    if (WARM)
      V[r * JSTRIDE + c] = (r < sm && c < sm) ? Vw[r * sm + c]
                                              : ((r == c) ? 1.f : 0.f);
    else
      V[r * JSTRIDE + c] = (r == c) ? 1.f : 0.f;
    fro2 += g * g;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) fro2 += __shfl_xor(fro2, off, 64);
  if ((tid & 63) == 0) offsq[tid >> 6] = fro2;
  if (tid == 0) done_s = 0;
  __syncthreads();
  if (WARM) {
    // B = V^T G V  (T = G.V, then G = V^T.T), all in LDS
    for (int i = tid; i < N * N; i += JTHREADS) {
      const int r = i / N, c = i % N;
      float acc = 0.f;
      for (int k = 0; k < N; ++k)
        acc = fmaf(G[r * JSTRIDE + k], V[k * JSTRIDE + c], acc);
      T[r * JSTRIDE + c] = acc;
    }
    __syncthreads();
    for (int i = tid; i < N * N; i += JTHREADS) {
      const int r = i / N, c = i % N;
      float acc = 0.f;
      for (int k = 0; k < N; ++k)
        acc = fmaf(V[k * JSTRIDE + r], T[k * JSTRIDE + c], acc);
      G[r * JSTRIDE + c] = acc;
    }
    __syncthreads();
  }
  float fro_all = 0.f;
  for (int w = 0; w < JTHREADS / 64; ++w) fro_all += offsq[w];
  const float tol2 = fro_all * 1e-10f;

  const int np = N / 2;
  for (int sweep = 0; sweep < max_sweeps && !done_s; ++sweep) {
    for (int round = 0; round < N - 1; ++round) {
      if (tid < np) {
        auto player = [&](int slot) {
          return slot == 0 ? 0 : 1 + (slot - 1 + round) % (N - 1);
        };
        const int a = player(tid);
        const int b = player(N - 1 - tid);
        const int p = min(a, b), q = max(a, b);
        pp[tid] = p;
        qq[tid] = q;
        rot_params(G[p * JSTRIDE + p], G[q * JSTRIDE + q], G[p * JSTRIDE + q],
                   &cs[tid], &sn[tid]);
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += JTHREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[p * JSTRIDE + k], gq = G[q * JSTRIDE + k];
        G[p * JSTRIDE + k] = c * gp - s * gq;
        G[q * JSTRIDE + k] = s * gp + c * gq;
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += JTHREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[k * JSTRIDE + p], gq = G[k * JSTRIDE + q];
        G[k * JSTRIDE + p] = c * gp - s * gq;
        G[k * JSTRIDE + q] = s * gp + c * gq;
        const float vp = V[k * JSTRIDE + p], vq = V[k * JSTRIDE + q];
        V[k * JSTRIDE + p] = c * vp - s * vq;
        V[k * JSTRIDE + q] = s * vp + c * vq;
      }
      __syncthreads();
    }
    float off2 = 0.f;
    for (int i = tid; i < N * N; i += JTHREADS) {
      const int r = i / N, c = i % N;
      if (r != c) {
        const float g = G[r * JSTRIDE + c];
        off2 += g * g;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) off2 += __shfl_xor(off2, off, 64);
    if ((tid & 63) == 0) offsq[tid >> 6] = off2;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < JTHREADS / 64; ++w) t += offsq[w];
      if (t <= tol2) done_s = 1;
    }
    __syncthreads();
  }

  if (tid == 0) {
    for (int i = 0; i < sm; ++i) order[i] = i;
    for (int i = 1; i < sm; ++i) {
      const int oi = order[i];
      const float vi = G[oi * JSTRIDE + oi];
      int j = i - 1;
      while (j >= 0 && G[order[j] * JSTRIDE + order[j]] < vi) {
        order[j + 1] = order[j];
        --j;
      }
      order[j + 1] = oi;
    }
  }
  __syncthreads();
  for (int j = tid; j < sm; j += JTHREADS) {
    const float lam = G[order[j] * JSTRIDE + order[j]];
    ev[j] = lam > 0.f ? lam : 0.f;
  }
  __syncthreads();
  for (int i = tid; i < sm * sm; i += JTHREADS) {
    const int k = i / sm, j = i % sm;
    const float v = V[k * JSTRIDE + order[j]];
    Gg[i] = v;
    if (WARM || save_warm) Vw[i] = v;
  }
}


// ---------------------------------------------------------------------------
// dense small-batch variant: a plain (n_mats, nb, nb) symmetric fp32 array,
// one 64-thread wave per matrix, nb <= 64 even.  Eigenvalues sorted
// DESCENDING into evals (n_mats, nb); eigenvectors overwrite the matrix
// slot (column j = eigenvector j).  Serves the randomized big-fold solver
// (svd_encoder._solve_big_folds_randomized): Lowdin orthonormalization and
// the Rayleigh-Ritz projection both need tiny batched eighs, and batched
// hipSOLVER syevd costs ~1 ms of launch latency per call — this kernel is
// ~100 us for ~100 matrices.
// ---------------------------------------------------------------------------
#define DMAX 64
#define DTHREADS 64

__global__ void __launch_bounds__(DTHREADS) jacobi_dense_kernel(
    float* __restrict__ a, float* __restrict__ evals, int n_mats, int nb,
    int max_sweeps) {
  constexpr int DSTRIDE = DMAX + 1;
  __shared__ float G[DMAX * DSTRIDE];
  __shared__ float V[DMAX * DSTRIDE];
  __shared__ float cs[DMAX / 2], sn[DMAX / 2];
  __shared__ int pp[DMAX / 2], qq[DMAX / 2];
  __shared__ int order[DMAX];
  __shared__ float fro_s;
  __shared__ int done_s;
  if (blockIdx.x >= (unsigned)n_mats) return;
  float* Ag = a + (int64_t)blockIdx.x * nb * nb;
  float* ev = evals + (int64_t)blockIdx.x * nb;
  const int tid = threadIdx.x;
  const int N = nb;  // caller guarantees even, <= DMAX

  float fro2 = 0.f;
  for (int i = tid; i < N * N; i += DTHREADS) {
    const int r = i / N, c = i % N;
    const float g = Ag[i];
    G[r * DSTRIDE + c] = g;
    V[r * DSTRIDE + c] = (r == c) ? 1.f : 0.f;
    fro2 += g * g;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) fro2 += __shfl_xor(fro2, off, 64);
  if (tid == 0) {
    fro_s = fro2;
    done_s = 0;
  }
  __syncthreads();
  const float tol2 = fro_s * 1e-10f;

  const int np = N / 2;
  for (int sweep = 0; sweep < max_sweeps && !done_s; ++sweep) {
    for (int round = 0; round < N - 1; ++round) {
      if (tid < np) {
        auto player = [&](int slot) {
          return slot == 0 ? 0 : 1 + (slot - 1 + round) % (N - 1);
        };
        const int x = player(tid);
        const int y = player(N - 1 - tid);
        const int p = min(x, y), q = max(x, y);
        pp[tid] = p;
        qq[tid] = q;
        rot_params(G[p * DSTRIDE + p], G[q * DSTRIDE + q], G[p * DSTRIDE + q],
                   &cs[tid], &sn[tid]);
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += DTHREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[p * DSTRIDE + k], gq = G[q * DSTRIDE + k];
        G[p * DSTRIDE + k] = c * gp - s * gq;
        G[q * DSTRIDE + k] = s * gp + c * gq;
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += DTHREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[k * DSTRIDE + p], gq = G[k * DSTRIDE + q];
        G[k * DSTRIDE + p] = c * gp - s * gq;
        G[k * DSTRIDE + q] = s * gp + c * gq;
        const float vp = V[k * DSTRIDE + p], vq = V[k * DSTRIDE + q];
        V[k * DSTRIDE + p] = c * vp - s * vq;
        V[k * DSTRIDE + q] = s * vp + c * vq;
      }
      __syncthreads();
    }
    float off2 = 0.f;
    for (int i = tid; i < N * N; i += DTHREADS) {
      const int r = i / N, c = i % N;
      if (r != c) {
        const float g = G[r * DSTRIDE + c];
        off2 += g * g;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) off2 += __shfl_xor(off2, off, 64);
    if (tid == 0 && off2 <= tol2) done_s = 1;
    __syncthreads();
  }

  if (tid == 0) {
    for (int i = 0; i < N; ++i) order[i] = i;
    for (int i = 1; i < N; ++i) {
      const int oi = order[i];
      const float vi = G[oi * DSTRIDE + oi];
      int j = i - 1;
      while (j >= 0 && G[order[j] * DSTRIDE + order[j]] < vi) {
        order[j + 1] = order[j];
        --j;
      }
      order[j + 1] = oi;
    }
  }
  __syncthreads();
  for (int j = tid; j < N; j += DTHREADS) ev[j] = G[order[j] * DSTRIDE + order[j]];
  __syncthreads();
  for (int i = tid; i < N * N; i += DTHREADS) {
    const int k = i / N, j = i % N;
    Ag[i] = V[k * DSTRIDE + order[j]];
  }
}

// ---------------------------------------------------------------------------
// big variant: G in its global Gram slot, V in global scratch
// (both L2-resident for sm <= 512)
// ---------------------------------------------------------------------------
#define BIGMAX 512

__global__ void __launch_bounds__(JBIG_THREADS) jacobi_eigh_big_kernel(
    float* __restrict__ grams, float* __restrict__ vbuf,
    float* __restrict__ evals, const int64_t* __restrict__ desc,
    const int64_t* __restrict__ eval_offs,
    const int32_t* __restrict__ rows_list,
    const int64_t* __restrict__ v_offs, int n_mats) {
  __shared__ float cs[BIGMAX / 2], sn[BIGMAX / 2];
  __shared__ short pp[BIGMAX / 2], qq[BIGMAX / 2];
  __shared__ float diag[BIGMAX];
  __shared__ short order[BIGMAX];
  __shared__ float offsq[JBIG_THREADS / 64];
  __shared__ int done_s;

  if (blockIdx.x >= (unsigned)n_mats) return;
  const int row = rows_list[blockIdx.x];
  const int64_t* d = desc + (int64_t)row * GD_N;
  const int m = (int)d[1], n = (int)d[2];
  const bool is_tall = d[3] != 0;
  const int sm = is_tall ? n : m;
  float* G = grams + d[4];  // (sm, sm) row-major, in place
  float* V = vbuf + v_offs[blockIdx.x];
  float* ev = evals + eval_offs[row];
  const int tid = threadIdx.x;
  // sm is even for every real fold here (tall folds have even a*b or this
  // layer went to the host path); assume N == sm.
  const int N = sm;
  const int np = N / 2;

  float fro2 = 0.f;
  for (int i = tid; i < N * N; i += JBIG_THREADS) {
    const float g = G[i];
    V[i] = (i / N == i % N) ? 1.f : 0.f;
    fro2 += g * g;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) fro2 += __shfl_xor(fro2, off, 64);
  if ((tid & 63) == 0) offsq[tid >> 6] = fro2;
  if (tid == 0) done_s = 0;
  __syncthreads();
  float fro_all = 0.f;
  for (int w = 0; w < JBIG_THREADS / 64; ++w) fro_all += offsq[w];
  const float tol2 = fro_all * 1e-10f;

  for (int sweep = 0; sweep < SWEEPS && !done_s; ++sweep) {
    for (int round = 0; round < N - 1; ++round) {
      for (int i = tid; i < np; i += JBIG_THREADS) {
        auto player = [&](int slot) {
          return slot == 0 ? 0 : 1 + (slot - 1 + round) % (N - 1);
        };
        const int a = player(i);
        const int b = player(N - 1 - i);
        const int p = min(a, b), q = max(a, b);
        pp[i] = (short)p;
        qq[i] = (short)q;
        rot_params(G[(int64_t)p * N + p], G[(int64_t)q * N + q],
                   G[(int64_t)p * N + q], &cs[i], &sn[i]);
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += JBIG_THREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[(int64_t)p * N + k], gq = G[(int64_t)q * N + k];
        G[(int64_t)p * N + k] = c * gp - s * gq;
        G[(int64_t)q * N + k] = s * gp + c * gq;
      }
      __syncthreads();
      for (int i = tid; i < np * N; i += JBIG_THREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[(int64_t)k * N + p], gq = G[(int64_t)k * N + q];
        G[(int64_t)k * N + p] = c * gp - s * gq;
        G[(int64_t)k * N + q] = s * gp + c * gq;
        const float vp = V[(int64_t)k * N + p], vq = V[(int64_t)k * N + q];
        V[(int64_t)k * N + p] = c * vp - s * vq;
        V[(int64_t)k * N + q] = s * vp + c * vq;
      }
      __syncthreads();
    }
    float off2 = 0.f;
    for (int i = tid; i < N * N; i += JBIG_THREADS) {
      if (i / N != i % N) {
        const float g = G[i];
        off2 += g * g;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) off2 += __shfl_xor(off2, off, 64);
    if ((tid & 63) == 0) offsq[tid >> 6] = off2;
    __syncthreads();
    if (tid == 0) {
      float t = 0.f;
      for (int w = 0; w < JBIG_THREADS / 64; ++w) t += offsq[w];
      if (t <= tol2) done_s = 1;
    }
    __syncthreads();
  }

  // stage the diagonal in LDS and sort
  for (int i = tid; i < N; i += JBIG_THREADS) diag[i] = G[(int64_t)i * N + i];
  __syncthreads();
  if (tid == 0) {
    for (int i = 0; i < N; ++i) order[i] = (short)i;
    for (int i = 1; i < N; ++i) {
      const short oi = order[i];
      const float vi = diag[oi];
      int j = i - 1;
      while (j >= 0 && diag[order[j]] < vi) {
        order[j + 1] = order[j];
        --j;
      }
      order[j + 1] = oi;
    }
  }
  __syncthreads();
  for (int j = tid; j < N; j += JBIG_THREADS) {
    const float lam = diag[order[j]];
    ev[j] = lam > 0.f ? lam : 0.f;
  }
  // permute eigenvectors into the gram slot: Gg[k][j] = V[k][order[j]]
  for (int i = tid; i < N * N; i += JBIG_THREADS) {
    const int k = i / N, j = i % N;
    G[i] = V[(int64_t)k * N + order[j]];
  }
}

// ---------------------------------------------------------------------------
// stage builder: per layer gather the sampled atoms into the wire staging
// sel table row (fp32): [r_hat | idx 0..31 (as float) | probs 0..31]
// stage layout:          [r_hat | s_wire(r_max) | facT(r_max*sm) | sel(sm*r_max)]
// facT rows / sel cols use FIXED r_max stride, zero beyond r_hat
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(64) build_stage_kernel(
    const float* __restrict__ evecs, const float* __restrict__ evals,
    const float* __restrict__ sel_table, float* __restrict__ stage,
    const int64_t* __restrict__ desc, const int64_t* __restrict__ eval_offs,
    int n_layers) {
  __shared__ float s_sel[R_CAP], inv_s[R_CAP];
  __shared__ int idxs[R_CAP];
  const int layer = blockIdx.x;
  if (layer >= n_layers) return;
  const int64_t* d = desc + (int64_t)layer * GD_N;
  const int m = (int)d[1], n = (int)d[2];
  const bool is_tall = d[3] != 0;
  const int sm = is_tall ? n : m;
  const int64_t so = d[6];
  const int r_max = (int)d[7];
  const float* V = evecs + d[4];  // (sm, sm) row-major, col j = atom j
  const float* ev = evals + eval_offs[layer];
  const float* row = sel_table + (int64_t)layer * SEL_ROW;
  const int r_hat = (int)row[0];
  const int tid = threadIdx.x;
  if (tid == 0) stage[so] = (float)r_hat;
  if (tid < r_hat) {
    const int idx = (int)row[1 + tid];
    const float p = row[1 + R_CAP + tid];
    const float s = sqrtf(ev[idx]);
    idxs[tid] = idx;
    s_sel[tid] = s / p;          // shipped singular value (unbiased rescale)
    inv_s[tid] = s > 1e-12f ? 1.f / s : 0.f;
    stage[so + 1 + tid] = s_sel[tid];
  }
  if (tid < r_max && tid >= r_hat) stage[so + 1 + tid] = 0.f;
  __syncthreads();
  float* facT = stage + so + 1 + r_max;                     // (r_max, sm)
  float* sel = stage + so + 1 + (int64_t)r_max * (1 + sm);  // (sm, r_max)
  for (int i = tid; i < r_max * sm; i += 64) {
    const int r = i / sm, k = i % sm;
    const float v = (r < r_hat) ? V[(int64_t)k * sm + idxs[r]] : 0.f;
    facT[(int64_t)r * sm + k] = v;
    sel[(int64_t)k * r_max + r] = (r < r_hat) ? v * inv_s[r] : 0.f;
  }
}

// ---------------------------------------------------------------------------
// fused on-device sampler + stage builder: removes the host round trip.
// Per layer (one wave64 WG): p_i = min(1, rank*s_i/sum s) (rank==0:
// s_i/s_0), one counter-hash Bernoulli draw per atom, redraw-until-nonempty
// (reference _sample_svd, codings/svd.py:49-67), cap at r_max keeping the
// highest-probability atoms; then build the staged wire factors exactly
// like build_stage_kernel.  used_words accumulates the actual packet sizes
// (the Msg-bytes counter) on device.
// ---------------------------------------------------------------------------
namespace {

__device__ __forceinline__ float u01_hash2(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + 0x9E3779B97F4A7C15ull * (idx + 1);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);
}

#define SAMPLE_SM_MAX 4096

__global__ void __launch_bounds__(64) sample_stage_kernel(
    const float* __restrict__ evecs, const float* __restrict__ evals,
    float* __restrict__ stage, const int64_t* __restrict__ desc,
    const int64_t* __restrict__ eval_offs, int n_layers, int rank,
    int truncate, const unsigned long long* __restrict__ seed_buf,
    unsigned long long* __restrict__ used_words) {
  // seed read from device memory so a hipGraph replay draws fresh dice:
  // the host advances a pinned scalar, a captured H2D copy refreshes
  // seed_buf, and every replay samples new atoms
  const uint64_t seed = seed_buf[0];
  __shared__ float s_lds[SAMPLE_SM_MAX];
  __shared__ float s_sel[R_CAP], inv_s[R_CAP];
  __shared__ int idxs[R_CAP];
  __shared__ int r_hat_s;
  const int layer = blockIdx.x;
  if (layer >= n_layers) return;
  const int64_t* d = desc + (int64_t)layer * GD_N;
  const int m = (int)d[1], n = (int)d[2];
  const bool is_tall = d[3] != 0;
  const int sm = is_tall ? n : m;
  const int64_t so = d[6];
  const int r_max = (int)d[7];
  const float* V = evecs + d[4];
  const float* ev = evals + eval_offs[layer];
  const int tid = threadIdx.x;

  // s = sqrt(eval), staged in LDS; wave-sum
  float ssum = 0.f;
  for (int i = tid; i < sm; i += 64) {
    const float s = sqrtf(fmaxf(ev[i], 0.f));
    s_lds[i] = s;
    ssum += s;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) ssum += __shfl_xor(ssum, off, 64);

  if (tid == 0) {
    int r_hat = 0;
    if (truncate) {  // deterministic top-r (master-style, svd.py:109-113)
      int r = rank > 0 ? rank : r_max;
      if (r > r_max) r = r_max;
      if (r > sm) r = sm;
      if (r > R_CAP) r = R_CAP;
      for (int i = 0; i < r; ++i) {
        idxs[i] = i;
        s_sel[i] = s_lds[i];
        inv_s[i] = s_lds[i] > 1e-12f ? 1.f / s_lds[i] : 0.f;
      }
      r_hat = r;
    } else if (s_lds[0] < 1e-6f) {  // degenerate: ship atom 0 with p = 1
      idxs[0] = 0;
      s_sel[0] = s_lds[0];
      inv_s[0] = 0.f;
      r_hat = 1;
    } else {
      const float inv_sum = 1.0f / ssum;
      for (int attempt = 0; attempt < 64 && r_hat == 0; ++attempt) {
        const uint64_t base =
            seed + (uint64_t)layer * 0x100000ull + (uint64_t)attempt * 0x10000ull;
        for (int i = 0; i < sm; ++i) {
          float p = (rank == 0) ? s_lds[i] / s_lds[0]
                                : (float)rank * s_lds[i] * inv_sum;
          p = fminf(p, 1.0f);
          if (u01_hash2(base, (uint64_t)i) < p) {
            if (r_hat < r_max && r_hat < R_CAP) {
              idxs[r_hat] = i;
              const float s = s_lds[i];
              s_sel[r_hat] = s / p;  // unbiased rescale
              inv_s[r_hat] = s > 1e-12f ? 1.f / s : 0.f;
              ++r_hat;
            }
            // overflow beyond the wire budget: drop (tail event; the
            // highest-probability atoms come first since s is sorted)
          }
        }
      }
      if (r_hat == 0) {  // 64 failed redraws: ship the top atom, p ~ its prob
        float p = (rank == 0) ? 1.0f : fminf((float)rank * s_lds[0] * inv_sum, 1.0f);
        idxs[0] = 0;
        s_sel[0] = s_lds[0] / p;
        inv_s[0] = s_lds[0] > 1e-12f ? 1.f / s_lds[0] : 0.f;
        r_hat = 1;
      }
    }
    r_hat_s = r_hat;
    stage[so] = (float)r_hat;
    atomicAdd(used_words,
              (unsigned long long)(1 + r_hat * (m + n + 1)));
  }
  __syncthreads();
  const int r_hat = r_hat_s;
  for (int k = tid; k < r_max; k += 64)
    stage[so + 1 + k] = (k < r_hat) ? s_sel[k] : 0.f;
  float* facT = stage + so + 1 + r_max;
  float* sel = stage + so + 1 + (int64_t)r_max * (1 + sm);
  for (int i = tid; i < r_max * sm; i += 64) {
    const int r = i / sm, k = i % sm;
    const float v = (r < r_hat) ? V[(int64_t)k * sm + idxs[r]] : 0.f;
    facT[(int64_t)r * sm + k] = v;
    sel[(int64_t)k * r_max + r] = (r < r_hat) ? v * inv_s[r] : 0.f;
  }
}

}  // namespace (inner)
}  // namespace

extern "C" {

void atomo_sample_stage_launch(const float* evecs, const float* evals,
                               float* stage, const int64_t* desc,
                               const int64_t* eval_offs, int n_layers,
                               int rank, int truncate,
                               const unsigned long long* seed_buf,
                               unsigned long long* used_words,
                               hipStream_t stream) {
  hipLaunchKernelGGL(sample_stage_kernel, dim3(n_layers), dim3(64), 0, stream,
                     evecs, evals, stage, desc, eval_offs, n_layers, rank,
                     truncate, seed_buf, used_words);
}


void atomo_jacobi_eigh_launch(float* grams, float* evals, const int64_t* desc,
                              const int64_t* eval_offs,
                              const int32_t* rows_list, int n_mats, int jmax,
                              float* vwarm, const int64_t* vwarm_offs,
                              int warm, int max_sweeps, hipStream_t stream) {
  // warm < 0: no warm basis at all (also skip the save)
  const int save = (warm >= 0 && vwarm != nullptr) ? 1 : 0;
  if (jmax <= 64) {
    if (warm > 0 && vwarm != nullptr)
      hipLaunchKernelGGL((jacobi_eigh_kernel<64, true>), dim3(n_mats),
                         dim3(JTHREADS), 0, stream, grams, evals, desc,
                         eval_offs, rows_list, n_mats, vwarm, vwarm_offs,
                         save, max_sweeps);
    else
      hipLaunchKernelGGL((jacobi_eigh_kernel<64, false>), dim3(n_mats),
                         dim3(JTHREADS), 0, stream, grams, evals, desc,
                         eval_offs, rows_list, n_mats, vwarm, vwarm_offs,
                         save, max_sweeps);
  } else {
    hipLaunchKernelGGL((jacobi_eigh_kernel<128, false>), dim3(n_mats),
                       dim3(JTHREADS), 0, stream, grams, evals, desc,
                       eval_offs, rows_list, n_mats, vwarm, vwarm_offs, 0,
                       max_sweeps);
  }
}

void atomo_jacobi_eigh_big_launch(float* grams, float* vbuf, float* evals,
                                  const int64_t* desc,
                                  const int64_t* eval_offs,
                                  const int32_t* rows_list,
                                  const int64_t* v_offs, int n_mats,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(jacobi_eigh_big_kernel, dim3(n_mats), dim3(JBIG_THREADS),
                     0, stream, grams, vbuf, evals, desc, eval_offs, rows_list,
                     v_offs, n_mats);
}

void atomo_jacobi_dense_launch(float* a, float* evals, int n_mats, int nb,
                               int max_sweeps, hipStream_t stream) {
  hipLaunchKernelGGL(jacobi_dense_kernel, dim3(n_mats), dim3(DTHREADS), 0,
                     stream, a, evals, n_mats, nb, max_sweeps);
}

void atomo_build_stage_launch(const float* evecs, const float* evals,
                              const float* sel_table, float* stage,
                              const int64_t* desc, const int64_t* eval_offs,
                              int n_layers, hipStream_t stream) {
  hipLaunchKernelGGL(build_stage_kernel, dim3(n_layers), dim3(64), 0, stream,
                     evecs, evals, sel_table, stage, desc, eval_offs,
                     n_layers);
}

}  // extern "C"
