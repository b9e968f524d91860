// K11: batch frame ingest — uint8 -> float/255.
//
// HBM-bound: 1 byte in, 4 (f32) or 2 (bf16) bytes out per element. Each
// lane consumes 16 input bytes per iteration (uchar16 as uint4) and writes
// 64/32 bytes, so a wave moves 1 KiB of input per instruction — the
// coalescing sweet spot (guide §2). Grid-stride, blocks of 256 (4 waves).

#include "drla_common.h"

extern "C" __global__ void drla_u8_normalize_f32(
    const uchar4* __restrict__ in, float4* __restrict__ out, long long n4) {
  const float inv = 1.0f / 255.0f;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n4; i += stride) {
    uchar4 v = in[i];
    out[i] = make_float4(v.x * inv, v.y * inv, v.z * inv, v.w * inv);
  }
}

typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;
typedef __attribute__((ext_vector_type(4))) unsigned short ushort4v;

extern "C" __global__ void drla_u8_normalize_bf16(
    const uchar4* __restrict__ in, ushort4v* __restrict__ out, long long n4) {
  const float inv = 1.0f / 255.0f;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n4; i += stride) {
    uchar4 v = in[i];
    ushort4v o;
    o.x = drla_f32_to_bf16(v.x * inv);
    o.y = drla_f32_to_bf16(v.y * inv);
    o.z = drla_f32_to_bf16(v.z * inv);
    o.w = drla_f32_to_bf16(v.w * inv);
    out[i] = o;
  }
}

// scalar tail for sizes not divisible by 4
extern "C" __global__ void drla_u8_normalize_f32_tail(
    const unsigned char* __restrict__ in, float* __restrict__ out,
    long long start, long long n) {
  long long i = start + blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i < n) out[i] = in[i] * (1.0f / 255.0f);
}

extern "C" __global__ void drla_u8_normalize_bf16_tail(
    const unsigned char* __restrict__ in, unsigned short* __restrict__ out,
    long long start, long long n) {
  long long i = start + blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i < n) out[i] = drla_f32_to_bf16(in[i] * (1.0f / 255.0f));
}

// K2 backward: action-embedding table gradient.
// torch's embedding_backward_feature_kernel costs ~38us/step on the tiny
// [A<=64, 256] table; here: per-(n,h) atomicAdd into an f32 scratch
// (A*H distinct addresses, deep L2 combining), then one cast kernel.
extern "C" __global__ void drla_embed_bwd_scatter(
    const long long* __restrict__ indices,      // [N]
    const unsigned short* __restrict__ gout16,  // [N,H] bf16 (nullable)
    const float* __restrict__ gout32,           // [N,H] f32 (nullable)
    float* __restrict__ scratch,                // [A,H] f32, zeroed
    long long N, int H, long long nrows) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = N * H;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < total; i += stride) {
    const long long n = i / H;
    const int h = i - n * H;
    float g;
    if (gout16) {
      unsigned int x = ((unsigned int)gout16[i]) << 16;
      g = __uint_as_float(x);
    } else {
      g = gout32[i];
    }
    // clamp external indices (see drla_clamp_idx rationale)
    const long long row = drla_clamp_idx((int)indices[n], (int)nrows);
    atomicAdd(&scratch[row * H + h], g);
  }
}

// variant that zeroes the f32 source as it reads (persistent-scratch
// consumers: embed_bwd keeps one zero-between-calls scratch)
extern "C" __global__ void drla_f32_to_bf16_zero_kernel(
    float* __restrict__ src, unsigned short* __restrict__ dst,
    long long n) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < n; i += stride) {
    dst[i] = drla_f32_to_bf16(src[i]);
    src[i] = 0.0f;
  }
}
