// K12: fused optimizer over ONE flat parameter buffer.
//
// Replaces tf.clip_by_global_norm + RMSProp/Adam apply (reference
// agent/impala.py:96-100) with two launches per step:
//   1. drla_sq_norm        — grid-stride float4 squared-norm reduction
//                            (wave shuffle -> LDS -> one atomic per block)
//   2. drla_{rmsprop,adam} — clip-scale folded into the update, one fused
//                            HBM pass over params+grads+state.
// The clip scale is computed on-device from the norm result (no host sync):
// scale = clip / max(norm, clip), passed as a 1-element tensor.

#include "drla_common.h"

extern "C" __global__ void drla_sq_norm(
    const float* __restrict__ x, float* __restrict__ out, long long n) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  float acc = 0.0f;
  const long long n4 = n / 4;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  for (long long k = i; k < n4; k += stride) {
    float4 v = x4[k];
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  // scalar tail handled by the first lanes
  for (long long k = n4 * 4 + i; k < n; k += stride) acc += x[k] * x[k];
  // wave64 reduce
  for (int off = DRLA_WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, DRLA_WAVE);
  __shared__ float lds[DRLA_BLOCK / DRLA_WAVE];
  const int wave = threadIdx.x / DRLA_WAVE;
  const int lane = threadIdx.x % DRLA_WAVE;
  if (lane == 0) lds[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    for (int w = 0; w < blockDim.x / DRLA_WAVE; ++w) s += lds[w];
    // 2048 blocks atomically adding ONE address serialize at ~28 us
    // (measured r22/r23 — load rewrites changed nothing); spreading the
    // block partials over 16 cache lines cuts that 16x, and the 16-way
    // final sum folds into drla_clip_scale at the consumer for free.
    atomicAdd(out + (blockIdx.x & (DRLA_NORM_SLOTS - 1)) * 16, s);
  }
}

// sq_norm_buf: DRLA_NORM_SLOTS partial sums, one per 64 B cache line;
// computes the tf.clip_by_global_norm factor in-kernel (no host round trip).
__device__ __forceinline__ float drla_clip_scale(const float* sq_norm_buf,
                                                 float clip) {
  if (clip <= 0.0f) return 1.0f;
  float sq = 0.0f;
  for (int s = 0; s < DRLA_NORM_SLOTS; ++s) sq += sq_norm_buf[s * 16];
  const float norm = sqrtf(sq);
  return norm > clip ? clip / norm : 1.0f;
}

// lr comes through a 1-element device buffer so the kernel is
// hipGraph-replay safe with a per-step decayed LR (the host rewrites the
// buffer before each replay; a by-value lr would be baked into the graph).
extern "C" __global__ void drla_rmsprop_step(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ ms, const float* __restrict__ sq_norm_buf,
    float clip, const float* __restrict__ lr_buf, float rho, float eps,
    long long n) {
  const float scale = drla_clip_scale(sq_norm_buf, clip);
  const float lr = *lr_buf;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    const float gc = g[i] * scale;
    const float m = rho * ms[i] + (1.0f - rho) * gc * gc;
    ms[i] = m;
    p[i] -= lr * gc * rsqrtf(m + eps);
  }
}

// ---- bf16-model / fp32-master variants (mixed precision) ----
// model params + grads are bf16 (raw ushort), master + state fp32; the
// update runs on the master and writes the rounded bf16 copy the model
// reads — one fused HBM pass, no per-layer weight casts anywhere else.

__device__ __forceinline__ float drla_bf16_to_f32(unsigned short u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __uint_as_float(x);
}

extern "C" __global__ void drla_sq_norm_bf16(
    const unsigned short* __restrict__ x, float* __restrict__ out,
    long long n) {
  // load as uint4 (8 bf16 = 16 B/lane; an ext_vector ushort4 load was
  // compiling to scalar u16 loads — 290 GB/s vs ~4 TB/s)
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  float acc = 0.0f;
  const long long n8 = n / 8;
  const uint4* x8 = reinterpret_cast<const uint4*>(x);
  for (long long k = i; k < n8; k += stride) {
    const uint4 v = x8[k];
    // named components (no address-of: a pointer into a local can force
    // the vector to scratch — guide §5.4 rule 20)

    float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
    a0 = drla_bf16_to_f32((unsigned short)(v.x & 0xFFFF));
    a1 = drla_bf16_to_f32((unsigned short)(v.x >> 16));
    acc += a0 * a0 + a1 * a1;
    a2 = drla_bf16_to_f32((unsigned short)(v.y & 0xFFFF));
    a3 = drla_bf16_to_f32((unsigned short)(v.y >> 16));
    acc += a2 * a2 + a3 * a3;
    a0 = drla_bf16_to_f32((unsigned short)(v.z & 0xFFFF));
    a1 = drla_bf16_to_f32((unsigned short)(v.z >> 16));
    acc += a0 * a0 + a1 * a1;
    a2 = drla_bf16_to_f32((unsigned short)(v.w & 0xFFFF));
    a3 = drla_bf16_to_f32((unsigned short)(v.w >> 16));
    acc += a2 * a2 + a3 * a3;

  }
  for (long long k = n8 * 8 + i; k < n; k += stride) {
    float a = drla_bf16_to_f32(x[k]);
    acc += a * a;
  }
  for (int off = DRLA_WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, DRLA_WAVE);
  __shared__ float lds[DRLA_BLOCK / DRLA_WAVE];
  const int wave = threadIdx.x / DRLA_WAVE;
  const int lane = threadIdx.x % DRLA_WAVE;
  if (lane == 0) lds[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.0f;
    for (int w = 0; w < blockDim.x / DRLA_WAVE; ++w) s += lds[w];
    atomicAdd(out + (blockIdx.x & (DRLA_NORM_SLOTS - 1)) * 16, s);
  }
}

extern "C" __global__ void drla_rmsprop_step_bf16(
    unsigned short* __restrict__ p_bf16, const unsigned short* __restrict__ g,
    float* __restrict__ master, float* __restrict__ ms,
    const float* __restrict__ sq_norm_buf, float clip,
    const float* __restrict__ lr_buf, float rho, float eps, long long n) {
  const float scale = drla_clip_scale(sq_norm_buf, clip);
  const float lr = *lr_buf;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    const float gc = drla_bf16_to_f32(g[i]) * scale;
    const float m = rho * ms[i] + (1.0f - rho) * gc * gc;
    ms[i] = m;
    const float w = master[i] - lr * gc * rsqrtf(m + eps);
    master[i] = w;
    p_bf16[i] = drla_f32_to_bf16(w);
  }
}

extern "C" __global__ void drla_adam_step_bf16(
    unsigned short* __restrict__ p_bf16, const unsigned short* __restrict__ g,
    float* __restrict__ master, float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ sq_norm_buf, float clip,
    const float* __restrict__ lr_buf, float beta1, float beta2, float eps,
    long long n) {
  const float scale = drla_clip_scale(sq_norm_buf, clip);
  const float lr_t = *lr_buf;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    const float gc = drla_bf16_to_f32(g[i]) * scale;
    const float mi = beta1 * m[i] + (1.0f - beta1) * gc;
    const float vi = beta2 * v[i] + (1.0f - beta2) * gc * gc;
    m[i] = mi;
    v[i] = vi;
    const float w = master[i] - lr_t * mi / (sqrtf(vi) + eps);
    master[i] = w;
    p_bf16[i] = drla_f32_to_bf16(w);
  }
}

extern "C" __global__ void drla_adam_step(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const float* __restrict__ sq_norm_buf, float clip,
    const float* __restrict__ lr_buf, float beta1, float beta2, float eps,
    long long n) {
  const float scale = drla_clip_scale(sq_norm_buf, clip);
  const float lr_t = *lr_buf;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    const float gc = g[i] * scale;
    const float mi = beta1 * m[i] + (1.0f - beta1) * gc;
    const float vi = beta2 * v[i] + (1.0f - beta2) * gc * gc;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr_t * mi / (sqrtf(vi) + eps);
  }
}

// K12b: scatter-mode grad gather — pack the autograd-owned grad tensors
// (stable hipGraph-pool addresses) into the flat bucket in ONE launch,
// replacing ~20 per-param AccumulateGrad add kernels. Slots are 8-element
// aligned (optim.py flatten_dense_params), so a vec8 chunk never crosses a
// slot; source pool allocations are >=256 B aligned.
typedef __attribute__((ext_vector_type(4))) unsigned int drla_u32x4;

extern "C" __global__ void drla_grad_gather(
    const unsigned long long* __restrict__ srcs,  // [nseg] device pointers
    const long long* __restrict__ offs,           // [nseg] padded starts
    const long long* __restrict__ sizes,          // [nseg] true numels
    unsigned short* __restrict__ dst, int nseg, long long chunks,
    float* __restrict__ norm_ws, int norm_n) {
  // piggyback: zero the persistent sq-norm partial buffer (consumed by
  // the sq_norm launch that follows on this stream) — saves a fill kernel
  if (norm_ws && blockIdx.x == 0 && (int)threadIdx.x < norm_n) {
    norm_ws[threadIdx.x] = 0.0f;
  }
  for (long long c = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       c < chunks; c += gridDim.x * (long long)blockDim.x) {
    const long long i = c * 8;
    int lo = 0, hi = nseg - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (offs[mid] <= i) lo = mid; else hi = mid - 1;
    }
    const long long local = i - offs[lo];
    const long long n = sizes[lo];
    if (local >= n) continue;  // alignment hole: flat stays zero
    const unsigned short* src =
        reinterpret_cast<const unsigned short*>(srcs[lo]);
    if (local + 8 <= n) {
      *reinterpret_cast<drla_u32x4*>(dst + i) =
          *reinterpret_cast<const drla_u32x4*>(src + local);
    } else {
      for (int e = 0; e < 8 && local + e < n; ++e) dst[i + e] = src[local + e];
    }
  }
}
