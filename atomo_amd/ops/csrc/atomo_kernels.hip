// atomo_amd HIP/CDNA4 kernels for MI355X (gfx950).
//
// Hand-written device code for the reference's compute hot spots
// (reference routines in /root/reference/src/codings/qsgd.py,
// codings/svd.py, optim/sgd.py — semantics only, no code carried over):
//
//   qsgd_pack_kernel         one wave64 per bucket: shfl L2-norm (or
//                            terngrad clip+max) reduction, counter-hash
//                            stochastic rounding, LDS-staged (1+q)-bit pack.
//   qsgd_pack/unpack_batched descriptor-table variants: ONE launch covers
//                            every layer of the model.
//   qsgd_unpack_acc_kernel   one thread per packed word, accumulate into
//                            the PS aggregation buffer.
//   svd_decode_acc_kernel    fused u.diag(s).vT over W workers' packets,
//                            one read-modify-write sweep of the output.
//   svd_decode_batched_kernel one launch over all layers x workers.
//   fused_sgd/adam_kernel    scale + weight-decay + momentum/Adam + apply,
//                            one flat sweep of the parameter buffer.
//
// All kernels are memory-bound sweeps: tiled for 64-wide wavefronts,
// grid-stride with a grid cap so the 256-CU / 8-XCD chip fills without
// launch storms.  Python bindings live in bindings.cpp.

#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE 64

namespace {

inline int grid_for(int64_t work, int block) {
  int64_t g = (work + block - 1) / block;
  if (g > 16384) g = 16384;
  if (g < 1) g = 1;
  return (int)g;
}

// counter-based RNG: splitmix64 hash -> uniform float in [0, 1)
__device__ __forceinline__ float u01_hash(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + 0x9E3779B97F4A7C15ull * (idx + 1);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);  // top 24 bits
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// ---------------------------------------------------------------------------
// QSGD pack: wire region = [norms (nb fp32)] [packed (nb*wpb u32)]
// element j of a bucket -> word j/epw, bits (j%epw)*(1+q); epw = 32/(1+q)
// ---------------------------------------------------------------------------
__global__ void qsgd_pack_kernel(const float* __restrict__ grad,
                                 float* __restrict__ norms,
                                 uint32_t* __restrict__ packed,
                                 int64_t numel, int bucket_size, int qlevel,
                                 bool terngrad, uint64_t seed, int n_buckets,
                                 int wpb) {
  extern __shared__ __attribute__((aligned(16))) uint8_t lds_codes[];
  const int waves_per_wg = blockDim.x / WAVE;
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int bits = 1 + qlevel;
  const int epw = 32 / bits;
  const int s_levels = (1 << qlevel) - 1;
  uint8_t* my_codes = lds_codes + (size_t)wave_id * bucket_size;

  for (int bucket = blockIdx.x * waves_per_wg + wave_id; bucket < n_buckets;
       bucket += gridDim.x * waves_per_wg) {
    const int64_t base = (int64_t)bucket * bucket_size;
    // --- pass 1: bucket statistics (wave-parallel, strided by lane) ---
    float ssq = 0.f, sum = 0.f;
    for (int e = lane; e < bucket_size; e += WAVE) {
      const int64_t g = base + e;
      const float w = (g < numel) ? grad[g] : 0.f;
      ssq += w * w;
      sum += w;
    }
    float norm, limit = 0.f;
    if (terngrad) {
      // clip at 2.5 sigma (population std), then inf-norm of the clipped
      const float n_inv = 1.0f / (float)bucket_size;
      const float mean = wave_reduce_sum(sum) * n_inv;
      const float var = fmaxf(wave_reduce_sum(ssq) * n_inv - mean * mean, 0.f);
      limit = 2.5f * sqrtf(var);
      float cmax = 0.f;
      for (int e = lane; e < bucket_size; e += WAVE) {
        const int64_t g = base + e;
        float w = (g < numel) ? grad[g] : 0.f;
        w = fminf(fmaxf(w, -limit), limit);
        cmax = fmaxf(cmax, fabsf(w));
      }
      norm = wave_reduce_max(cmax);
    } else {
      norm = sqrtf(wave_reduce_sum(ssq));
    }
    if (lane == 0) norms[bucket] = norm;
    const float inv_norm_s = (norm > 1e-30f) ? ((float)s_levels / norm) : 0.f;

    // --- pass 2: quantize into LDS byte codes ---
    for (int e = lane; e < bucket_size; e += WAVE) {
      const int64_t g = base + e;
      float w = (g < numel) ? grad[g] : 0.f;
      if (terngrad) w = fminf(fmaxf(w, -limit), limit);
      const float scaled = fabsf(w) * inv_norm_s;
      int xi = (int)scaled;  // floor (scaled >= 0)
      const float frac = scaled - (float)xi;
      // unbiased stochastic rounding: round up with probability frac
      if (u01_hash(seed, (uint64_t)g) < frac) xi += 1;
      if (xi > s_levels) xi = s_levels;
      const int sign = (w < 0.f) ? 1 : 0;
      my_codes[e] = (uint8_t)((sign << qlevel) | xi);
    }
    // same-wave cross-lane LDS hand-off: order write -> read
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "workgroup");
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
    // --- pass 3: shift-or epw codes per u32 word ---
    uint32_t* out = packed + (int64_t)bucket * wpb;
    for (int wrd = lane; wrd < wpb; wrd += WAVE) {
      uint32_t acc = 0;
      const int e0 = wrd * epw;
      for (int k = 0; k < epw; ++k) {
        const int e = e0 + k;
        const uint32_t c = (e < bucket_size) ? my_codes[e] : 0u;
        acc |= c << (k * bits);
      }
      out[wrd] = acc;
    }
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "workgroup");
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
  }
}

// ---------------------------------------------------------------------------
// QSGD unpack + accumulate: one thread per packed word
// ---------------------------------------------------------------------------
__global__ void qsgd_unpack_acc_kernel(const float* __restrict__ norms,
                                       const uint32_t* __restrict__ packed,
                                       float* __restrict__ out, int64_t numel,
                                       int bucket_size, int qlevel,
                                       int n_buckets, int wpb) {
  const int bits = 1 + qlevel;
  const int epw = 32 / bits;
  const int s_levels = (1 << qlevel) - 1;
  const uint32_t ximask = (1u << qlevel) - 1u;
  const int64_t n_words = (int64_t)n_buckets * wpb;
  for (int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; w < n_words;
       w += (int64_t)gridDim.x * blockDim.x) {
    const int bucket = (int)(w / wpb);
    const int wib = (int)(w % wpb);
    const float scale = norms[bucket] / (float)s_levels;
    const uint32_t word = packed[w];
    const int64_t ebase = (int64_t)bucket * bucket_size + (int64_t)wib * epw;
    for (int k = 0; k < epw; ++k) {
      const uint32_t c = (word >> (k * bits)) & ((1u << bits) - 1u);
      const int64_t e = ebase + k;
      if (e < numel && (int64_t)(wib)*epw + k < bucket_size) {
        const float xi = (float)(c & ximask);
        const float sgn = (c >> qlevel) ? -1.f : 1.f;
        out[e] += sgn * xi * scale;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Batched QSGD over ALL layers in one launch (descriptor-table driven, like
// the SVD-encode kernels).  Descriptor row (int64, QD_N words):
//   [0] grad_off (into flat_grad)  [1] numel  [2] wire_off  [3] n_buckets
//   [4] words_per_bucket
// pack work map: (layer, local bucket) per wave; unpack work map:
// (layer, 256-word chunk) per workgroup.
// ---------------------------------------------------------------------------
#define QD_N 5

__global__ void qsgd_pack_batched_kernel(
    const float* __restrict__ flat, float* __restrict__ wire,
    const int64_t* __restrict__ desc, const int32_t* __restrict__ work,
    int n_tiles, int bucket_size, int qlevel, bool terngrad,
    const unsigned long long* __restrict__ seed_buf) {
  // seed from device memory: a captured H2D copy refreshes it per graph
  // replay so every replay rolls fresh stochastic-rounding dice
  const uint64_t seed = seed_buf[0];
  extern __shared__ uint8_t lds_codes[];
  const int waves_per_wg = blockDim.x / WAVE;
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int bits = 1 + qlevel;
  const int epw = 32 / bits;
  const int s_levels = (1 << qlevel) - 1;
  uint8_t* my_codes = lds_codes + (size_t)wave_id * bucket_size;

  for (int tile = blockIdx.x * waves_per_wg + wave_id;
       tile < n_tiles; tile += gridDim.x * waves_per_wg) {
    const int layer = work[2 * tile];
    const int bucket = work[2 * tile + 1];
    const int64_t* d = desc + (int64_t)layer * QD_N;
    const float* grad = flat + d[0];
    const int64_t numel = d[1];
    const int wpb = (int)d[4];
    float* norms = wire + d[2];
    uint32_t* packed = reinterpret_cast<uint32_t*>(norms + d[3]);
    const int64_t base = (int64_t)bucket * bucket_size;
    float ssq = 0.f, sum = 0.f;
    for (int e = lane; e < bucket_size; e += WAVE) {
      const int64_t g = base + e;
      const float w = (g < numel) ? grad[g] : 0.f;
      ssq += w * w;
      sum += w;
    }
    float norm, limit = 0.f;
    if (terngrad) {
      const float n_inv = 1.0f / (float)bucket_size;
      const float mean = wave_reduce_sum(sum) * n_inv;
      const float var = fmaxf(wave_reduce_sum(ssq) * n_inv - mean * mean, 0.f);
      limit = 2.5f * sqrtf(var);
      float cmax = 0.f;
      for (int e = lane; e < bucket_size; e += WAVE) {
        const int64_t g = base + e;
        float w = (g < numel) ? grad[g] : 0.f;
        w = fminf(fmaxf(w, -limit), limit);
        cmax = fmaxf(cmax, fabsf(w));
      }
      norm = wave_reduce_max(cmax);
    } else {
      norm = sqrtf(wave_reduce_sum(ssq));
    }
    if (lane == 0) norms[bucket] = norm;
    const float inv_norm_s = (norm > 1e-30f) ? ((float)s_levels / norm) : 0.f;
    for (int e = lane; e < bucket_size; e += WAVE) {
      const int64_t g = base + e;
      float w = (g < numel) ? grad[g] : 0.f;
      if (terngrad) w = fminf(fmaxf(w, -limit), limit);
      const float scaled = fabsf(w) * inv_norm_s;
      int xi = (int)scaled;
      const float frac = scaled - (float)xi;
      if (u01_hash(seed + (uint64_t)layer * 0x1000000000000ull, (uint64_t)g) <
          frac)
        xi += 1;
      if (xi > s_levels) xi = s_levels;
      my_codes[e] = (uint8_t)((((w < 0.f) ? 1 : 0) << qlevel) | xi);
    }
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "workgroup");
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
    uint32_t* out = packed + (int64_t)bucket * wpb;
    for (int wrd = lane; wrd < wpb; wrd += WAVE) {
      uint32_t acc = 0;
      const int e0 = wrd * epw;
      for (int k = 0; k < epw; ++k) {
        const int e = e0 + k;
        const uint32_t c = (e < bucket_size) ? my_codes[e] : 0u;
        acc |= c << (k * bits);
      }
      out[wrd] = acc;
    }
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "workgroup");
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "workgroup");
  }
}

#define QUNPACK_CHUNK 256

__global__ void qsgd_unpack_batched_kernel(
    const float* __restrict__ wire, float* __restrict__ agg,
    const int64_t* __restrict__ desc, const int32_t* __restrict__ work,
    int n_tiles, int bucket_size, int qlevel) {
  const int bits = 1 + qlevel;
  const int epw = 32 / bits;
  const int s_levels = (1 << qlevel) - 1;
  const uint32_t ximask = (1u << qlevel) - 1u;
  for (int tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    const int layer = work[2 * tile];
    const int chunk = work[2 * tile + 1];
    const int64_t* d = desc + (int64_t)layer * QD_N;
    const int64_t numel = d[1];
    const int nb = (int)d[3];
    const int wpb = (int)d[4];
    const float* norms = wire + d[2];
    const uint32_t* packed = reinterpret_cast<const uint32_t*>(norms + nb);
    float* out = agg + d[0];
    const int64_t n_words = (int64_t)nb * wpb;
    const int64_t w0 = (int64_t)chunk * QUNPACK_CHUNK;
    const int64_t w1 = min(w0 + QUNPACK_CHUNK, n_words);
    for (int64_t w = w0 + threadIdx.x; w < w1; w += blockDim.x) {
      const int bucket = (int)(w / wpb);
      const int wib = (int)(w % wpb);
      const float scale = norms[bucket] / (float)s_levels;
      const uint32_t word = packed[w];
      const int64_t ebase =
          (int64_t)bucket * bucket_size + (int64_t)wib * epw;
      for (int k = 0; k < epw; ++k) {
        const uint32_t c = (word >> (k * bits)) & ((1u << bits) - 1u);
        const int64_t e = ebase + k;
        if (e < numel && (int64_t)wib * epw + k < bucket_size) {
          const float xi = (float)(c & ximask);
          const float sgn = (c >> qlevel) ? -1.f : 1.f;
          out[e] += sgn * xi * scale;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// SVD decode + accumulate over W packets in one output sweep.
// Packet layout (fp32): [r_hat | uT (r_max x m) | s (r_max) | vT (r_max x n)]
// out2d(m, n) += sum_w sum_r u_w[i,r] * s_w[r] * vT_w[r,j]
// One thread per output element: writes coalesced; uT[r*m+i] is broadcast
// within a j-group and coalesced across i-groups; s/vT rows are L1/L2-hot.
// ---------------------------------------------------------------------------
__global__ void svd_decode_acc_kernel(const float* __restrict__ regions,
                                      float* __restrict__ out, int W,
                                      int64_t stride, int m, int n,
                                      int r_max) {
  const int64_t total = (int64_t)m * n;
  const int64_t s_off = 1 + (int64_t)r_max * m;
  const int64_t v_off = s_off + r_max;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (int64_t)gridDim.x * blockDim.x) {
    const int i = (int)(t / n);
    const int j = (int)(t % n);
    float acc = 0.f;
    for (int w = 0; w < W; ++w) {
      const float* reg = regions + (int64_t)w * stride;
      const int r_hat = (int)reg[0];
      const float* uT = reg + 1;
      const float* s = reg + s_off;
      const float* vT = reg + v_off;
      for (int r = 0; r < r_hat; ++r) {
        acc = fmaf(uT[(int64_t)r * m + i] * s[r], vT[(int64_t)r * n + j], acc);
      }
    }
    out[t] += acc;
  }
}

// ---------------------------------------------------------------------------
// Batched SVD decode over ALL layers and ALL workers in one launch.
// Descriptor row (int64, DD_N words): [agg_off, m, n, r_max, wire_off]
// Work map: (layer, chunk of DEC_CHUNK output elements).
// agg[agg_off + t] += sum_w sum_r u_w[i,r] s_w[r] vT_w[r,j], t=(i,j)
// ---------------------------------------------------------------------------
#define DD_N 5
#define DEC_CHUNK 1024

__global__ void svd_decode_batched_kernel(
    const float* __restrict__ stacked, int64_t row_stride, int W,
    float* __restrict__ agg, const int64_t* __restrict__ desc,
    const int32_t* __restrict__ work, int n_tiles) {
  for (int tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
    const int layer = work[2 * tile];
    const int chunk = work[2 * tile + 1];
    const int64_t* d = desc + (int64_t)layer * DD_N;
    const int64_t agg_off = d[0];
    const int m = (int)d[1], n = (int)d[2];
    const int r_max = (int)d[3];
    const int64_t wo = d[4];
    const int64_t total = (int64_t)m * n;
    const int64_t s_off = wo + 1 + (int64_t)r_max * m;
    const int64_t v_off = s_off + r_max;
    const int64_t t0 = (int64_t)chunk * DEC_CHUNK;
    const int64_t t1 = min(t0 + DEC_CHUNK, total);
    for (int64_t t = t0 + threadIdx.x; t < t1; t += blockDim.x) {
      const int i = (int)(t / n);
      const int j = (int)(t % n);
      float acc = 0.f;
      for (int w = 0; w < W; ++w) {
        const float* row = stacked + (int64_t)w * row_stride;
        const int r_hat = (int)row[wo];
        const float* uT = row + wo + 1;
        const float* s = row + s_off;
        const float* vT = row + v_off;
        for (int r = 0; r < r_hat; ++r)
          acc = fmaf(uT[(int64_t)r * m + i] * s[r], vT[(int64_t)r * n + j],
                     acc);
      }
      agg[agg_off + t] += acc;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused SGD: p -= lr * step(grad*scale + wd*p, momentum buffer)
// ---------------------------------------------------------------------------
__global__ void fused_sgd_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ buf, int64_t n, float lr,
                                 float momentum, float weight_decay,
                                 bool nesterov, float dampening,
                                 float grad_scale,
                                 const float* __restrict__ lr_dev) {
  // lr from device memory when provided: a captured pinned-memory copy
  // refreshes it per hipGraph replay, so lr shrinkage needs no recapture
  if (lr_dev != nullptr) lr = lr_dev[0];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float pi = p[i];
    float d = g[i] * grad_scale + weight_decay * pi;
    if (momentum != 0.f) {
      float b = buf[i] * momentum + (1.f - dampening) * d;
      buf[i] = b;
      d = nesterov ? d + momentum * b : b;
    }
    p[i] = pi - lr * d;
  }
}

// ---------------------------------------------------------------------------
// Fused Adam: one flat sweep (reference math optim/adam.py:37-93, incl.
// amsgrad), with the PS's 1/num_workers fold into grad_scale.
// ---------------------------------------------------------------------------
__global__ void fused_adam_kernel(float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ exp_avg,
                                  float* __restrict__ exp_avg_sq,
                                  float* __restrict__ max_exp_avg_sq,
                                  int64_t n, float lr, float beta1,
                                  float beta2, float eps, float weight_decay,
                                  float bias1, float bias2, float grad_scale) {
  const float step_size = lr / bias1;
  const float inv_bias2 = 1.0f / bias2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float pi = p[i];
    const float gi = g[i] * grad_scale + weight_decay * pi;
    const float m = exp_avg[i] * beta1 + (1.f - beta1) * gi;
    float v = exp_avg_sq[i] * beta2 + (1.f - beta2) * gi * gi;
    exp_avg[i] = m;
    exp_avg_sq[i] = v;
    if (max_exp_avg_sq != nullptr) {
      v = fmaxf(max_exp_avg_sq[i], v);
      max_exp_avg_sq[i] = v;
    }
    const float denom = sqrtf(v * inv_bias2) + eps;
    p[i] = pi - step_size * m / denom;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// extern "C" launchers (bindings.cpp provides the torch glue)
// ---------------------------------------------------------------------------
extern "C" {

void atomo_fused_adam_launch(float* p, const float* g, float* exp_avg,
                             float* exp_avg_sq, float* max_exp_avg_sq,
                             int64_t n, float lr, float beta1, float beta2,
                             float eps, float weight_decay, float bias1,
                             float bias2, float grad_scale,
                             hipStream_t stream) {
  const int block = 256;
  const int grid = grid_for(n, block);
  hipLaunchKernelGGL(fused_adam_kernel, dim3(grid), dim3(block), 0, stream, p,
                     g, exp_avg, exp_avg_sq, max_exp_avg_sq, n, lr, beta1,
                     beta2, eps, weight_decay, bias1, bias2, grad_scale);
}

void atomo_qsgd_pack_launch(const float* grad, float* norms, uint32_t* packed,
                            int64_t numel, int bucket_size, int qlevel,
                            bool terngrad, uint64_t seed, int n_buckets,
                            int wpb, hipStream_t stream) {
  const int block = 256;
  const int waves_per_wg = block / WAVE;
  const size_t lds = (size_t)waves_per_wg * bucket_size;
  const int grid = grid_for((int64_t)n_buckets * WAVE, block);
  hipLaunchKernelGGL(qsgd_pack_kernel, dim3(grid), dim3(block), lds, stream,
                     grad, norms, packed, numel, bucket_size, qlevel, terngrad,
                     seed, n_buckets, wpb);
}

void atomo_qsgd_unpack_acc_launch(const float* norms, const uint32_t* packed,
                                  float* out, int64_t numel, int bucket_size,
                                  int qlevel, int n_buckets, int wpb,
                                  hipStream_t stream) {
  const int block = 256;
  const int grid = grid_for((int64_t)n_buckets * wpb, block);
  hipLaunchKernelGGL(qsgd_unpack_acc_kernel, dim3(grid), dim3(block), 0,
                     stream, norms, packed, out, numel, bucket_size, qlevel,
                     n_buckets, wpb);
}

void atomo_svd_decode_acc_launch(const float* regions, float* out, int W,
                                 int64_t stride, int m, int n, int r_max,
                                 hipStream_t stream) {
  const int block = 256;
  const int grid = grid_for((int64_t)m * n, block);
  hipLaunchKernelGGL(svd_decode_acc_kernel, dim3(grid), dim3(block), 0, stream,
                     regions, out, W, stride, m, n, r_max);
}

void atomo_qsgd_pack_batched_launch(const float* flat, float* wire,
                                    const int64_t* desc, const int32_t* work,
                                    int n_tiles, int bucket_size, int qlevel,
                                    bool terngrad,
                                    const unsigned long long* seed_buf,
                                    hipStream_t stream) {
  const int block = 256;
  const int waves_per_wg = block / WAVE;
  const size_t lds = (size_t)waves_per_wg * bucket_size;
  const int grid = grid_for((int64_t)n_tiles * WAVE, block);
  hipLaunchKernelGGL(qsgd_pack_batched_kernel, dim3(grid), dim3(block), lds,
                     stream, flat, wire, desc, work, n_tiles, bucket_size,
                     qlevel, terngrad, seed_buf);
}

void atomo_qsgd_unpack_batched_launch(const float* wire, float* agg,
                                      const int64_t* desc,
                                      const int32_t* work, int n_tiles,
                                      int bucket_size, int qlevel,
                                      hipStream_t stream) {
  const int grid = grid_for(n_tiles, 1) > 8192 ? 8192
                   : (n_tiles < 1 ? 1 : n_tiles);
  hipLaunchKernelGGL(qsgd_unpack_batched_kernel, dim3(grid), dim3(256), 0,
                     stream, wire, agg, desc, work, n_tiles, bucket_size,
                     qlevel);
}

void atomo_svd_decode_batched_launch(const float* stacked, int64_t row_stride,
                                     int W, float* agg, const int64_t* desc,
                                     const int32_t* work, int n_tiles,
                                     hipStream_t stream) {
  const int grid = n_tiles > 8192 ? 8192 : (n_tiles < 1 ? 1 : n_tiles);
  hipLaunchKernelGGL(svd_decode_batched_kernel, dim3(grid), dim3(256), 0,
                     stream, stacked, row_stride, W, agg, desc, work, n_tiles);
}

void atomo_fused_sgd_launch(float* p, const float* g, float* buf, int64_t n,
                            float lr, float momentum, float weight_decay,
                            bool nesterov, float dampening, float grad_scale,
                            const float* lr_dev, hipStream_t stream) {
  const int block = 256;
  const int grid = grid_for(n, block);
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(grid), dim3(block), 0, stream, p,
                     g, buf, n, lr, momentum, weight_decay, nesterov,
                     dampening, grad_scale, lr_dev);
}

}  // extern "C"
