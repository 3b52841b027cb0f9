// Batched symmetric eigensolver (cyclic parallel Jacobi) + selection-stage
// builder for MI355X (gfx950).
//
// Solves the per-layer Gram matrices G (sm x sm, sm <= 64) ON DEVICE so the
// SVD-encode path never ships Grams to the host (reference equivalent:
// numpy LA.svd per layer, codings/svd.py:95).  One wave64 workgroup per
// matrix; G and V live in LDS (row stride 65 to keep column walks
// conflict-free); each Jacobi round applies all N/2 disjoint plane
// rotations in two barrier-separated phases (rows = J^T G, then cols = .J
// and V.J); pairs follow the round-robin tournament schedule.  Fixed sweep
// count (machine-eps convergence for n <= 64 needs ~6; we run 10).
// Eigenvalues are sorted descending and the eigenvectors overwrite the Gram
// slot in-place.
//
// build_stage_kernel then gathers the host-sampled atom selection
// (idx, probs per layer) into the staged wire factors:
//   s_wire[r] = sqrt(eval[idx_r]) / p_r
//   facT[r][k] = V[k][idx_r]                  (the small wire factor)
//   sel[k][r]  = V[k][idx_r] / sqrt(eval)     (A @ sel = tall factor)

#include <hip/hip_runtime.h>

#include <cstdint>

#define GD_N 8
#define JMAX 64
#define JSTRIDE 65
#define SWEEPS 10
#define JTHREADS 256

namespace {

__global__ void __launch_bounds__(JTHREADS) jacobi_eigh_kernel(
    float* __restrict__ grams, float* __restrict__ evals,
    const int64_t* __restrict__ desc, const int64_t* __restrict__ eval_offs,
    int n_layers) {
  __shared__ float G[JMAX * JSTRIDE];
  __shared__ float V[JMAX * JSTRIDE];
  __shared__ float cs[JMAX / 2], sn[JMAX / 2];
  __shared__ int pp[JMAX / 2], qq[JMAX / 2];
  __shared__ int order[JMAX];
  __shared__ float offsq[JTHREADS / 64];
  __shared__ int done_s;

  const int layer = blockIdx.x;
  if (layer >= n_layers) return;
  const int64_t* d = desc + (int64_t)layer * GD_N;
  const int m = (int)d[1], n = (int)d[2];
  const bool is_tall = d[3] != 0;
  const int sm = is_tall ? n : m;
  float* Gg = grams + d[4];
  float* ev = evals + eval_offs[layer];
  const int tid = threadIdx.x;
  const int N = (sm + 1) & ~1;  // even-padded

  // load G, init V = I (pad rows/cols zero)
  float fro2 = 0.f;
  for (int i = tid; i < N * N; i += JTHREADS) {
    const int r = i / N, c = i % N;
    const float g = (r < sm && c < sm) ? Gg[r * sm + c] : 0.f;
    G[r * JSTRIDE + c] = g;
    V[r * JSTRIDE + c] = (r == c) ? 1.f : 0.f;
    fro2 += g * g;
  }
  // block-reduce the Frobenius norm (convergence scale)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) fro2 += __shfl_xor(fro2, off, 64);
  if ((tid & 63) == 0) offsq[tid >> 6] = fro2;
  if (tid == 0) done_s = 0;
  __syncthreads();
  float fro_all = 0.f;
  for (int w = 0; w < JTHREADS / 64; ++w) fro_all += offsq[w];
  const float tol2 = fro_all * 1e-13f;

  const int np = N / 2;
  for (int sweep = 0; sweep < SWEEPS && !done_s; ++sweep) {
    for (int round = 0; round < N - 1; ++round) {
      // tournament pairing: slot 0 fixed, others rotate
      if (tid < np) {
        auto player = [&](int slot) {
          return slot == 0 ? 0 : 1 + (slot - 1 + round) % (N - 1);
        };
        int a = player(tid);
        int b = player(N - 1 - tid);
        const int p = min(a, b), q = max(a, b);
        pp[tid] = p;
        qq[tid] = q;
        const float app = G[p * JSTRIDE + p];
        const float aqq = G[q * JSTRIDE + q];
        const float apq = G[p * JSTRIDE + q];
        float c = 1.f, s = 0.f;
        if (fabsf(apq) > 1e-12f) {
          const float tau = (aqq - app) / (2.f * apq);
          const float t =
              (tau >= 0.f ? 1.f : -1.f) / (fabsf(tau) + sqrtf(1.f + tau * tau));
          c = rsqrtf(1.f + t * t);
          s = t * c;
        }
        cs[tid] = c;
        sn[tid] = s;
      }
      __syncthreads();
      // phase 1: rows p,q <- J^T G
      for (int i = tid; i < np * N; i += JTHREADS) {
        const int pr = i / N, k = i % N;
        const int p = pp[pr], q = qq[pr];
        const float c = cs[pr], s = sn[pr];
        const float gp = G[p * JSTRIDE + k], gq = G[q * JSTRIDE + k];
        G[p * JSTRIDE + k] = c * gp - s * gq;
        G[q * JSTRIDE + k] = s * gp + c * gq;
      }
      __syncthreads();
      // phase 2: cols p,q <- G J ; V <- V J
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
    // convergence: off-diagonal Frobenius^2 below tolerance -> stop
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

  // sort eigenvalues descending (insertion sort by one lane; sm <= 64)
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
  // write sorted evals and evecs (evecs overwrite the gram slot, row-major
  // [k][j] = V[k][order[j]])
  for (int j = tid; j < sm; j += JTHREADS) {
    const float lam = G[order[j] * JSTRIDE + order[j]];
    ev[j] = lam > 0.f ? lam : 0.f;
  }
  __syncthreads();
  for (int i = tid; i < sm * sm; i += JTHREADS) {
    const int k = i / sm, j = i % sm;
    Gg[i] = V[k * JSTRIDE + order[j]];
  }
}

// ---------------------------------------------------------------------------
// stage builder: per layer gather the sampled atoms into the wire staging
// sel table row (fp32): [r_hat | idx 0..15 (as float) | probs 0..15]
// stage layout:          [r_hat | s_wire(r_max) | facT(r_max*sm) | sel(sm*r_max)]
// ---------------------------------------------------------------------------
#define SEL_ROW 33

__global__ void __launch_bounds__(64) build_stage_kernel(
    const float* __restrict__ evecs, const float* __restrict__ evals,
    const float* __restrict__ sel_table, float* __restrict__ stage,
    const int64_t* __restrict__ desc, const int64_t* __restrict__ eval_offs,
    int n_layers) {
  __shared__ float s_sel[16], inv_s[16];
  __shared__ int idxs[16];
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
    const float p = row[17 + tid];
    const float s = sqrtf(ev[idx]);
    idxs[tid] = idx;
    s_sel[tid] = s / p;          // shipped singular value (unbiased rescale)
    inv_s[tid] = s > 1e-12f ? 1.f / s : 0.f;
    stage[so + 1 + tid] = s_sel[tid];
  }
  __syncthreads();
  float* facT = stage + so + 1 + r_max;                  // (r_hat, sm)
  float* sel = stage + so + 1 + (int64_t)r_max * (1 + sm);  // (sm, r_hat)
  for (int i = tid; i < r_hat * sm; i += 64) {
    const int r = i / sm, k = i % sm;
    const float v = V[(int64_t)k * sm + idxs[r]];
    facT[(int64_t)r * sm + k] = v;
    sel[(int64_t)k * r_hat + r] = v * inv_s[r];
  }
}

}  // namespace

extern "C" {

void atomo_jacobi_eigh_launch(float* grams, float* evals, const int64_t* desc,
                              const int64_t* eval_offs, int n_layers,
                              hipStream_t stream) {
  hipLaunchKernelGGL(jacobi_eigh_kernel, dim3(n_layers), dim3(JTHREADS), 0, stream,
                     grams, evals, desc, eval_offs, n_layers);
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
