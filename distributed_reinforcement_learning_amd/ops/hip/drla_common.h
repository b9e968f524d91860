// Common helpers for the drla gfx950 kernels.
// CDNA4: wave64, 256 CUs / 8 XCDs; memory-bound kernels follow the
// grid-stride + vectorized-access pattern (guide §6 G11/G13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DRLA_WAVE 64
// squared-norm partials: 16 slots, each on its own 64 B cache line
// (buffer length = 16*16 floats; slot s lives at [s*16])
#define DRLA_NORM_SLOTS 16
#define DRLA_BLOCK 256
// cap grids at ~8 blocks/CU x 256 CUs and grid-stride the rest
#define DRLA_MAX_BLOCKS 2048

static inline int drla_grid(long long work_items, int block = DRLA_BLOCK) {
  long long blocks = (work_items + block - 1) / block;
  if (blocks > DRLA_MAX_BLOCKS) blocks = DRLA_MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

// Defensive clamp for indices that arrive from EXTERNAL data (actor
// rings, replay payloads): an out-of-range action must not become a GPU
// aperture fault that takes the whole learner down — corrupt input
// yields a wrong-but-bounded read instead.
__device__ __forceinline__ int drla_clamp_idx(int a, int n) {
  return a < 0 ? 0 : (a >= n ? n - 1 : a);
}

__device__ __forceinline__ float drla_sigmoid(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// round-to-nearest-even f32 -> bf16 raw bits. gfx950 has a hardware
// packed convert (v_cvt_pk_bf16_f32, 1 VALU op); the bit-math expansion
// costs ~6 ops per value and showed up in the staging-heavy kernels
// (PMC p31: conv_fwd_l1 at 3.7e8 VALU insts).
__device__ __forceinline__ unsigned short drla_f32_to_bf16(float f) {
  __hip_bfloat162 v = __float22bfloat162_rn(float2{f, f});
  return *reinterpret_cast<unsigned short*>(&v);
}

// packed pair -> one u32 (low = a, high = b); single v_cvt_pk_bf16_f32
__device__ __forceinline__ unsigned int drla_f32x2_to_bf16x2(float a,
                                                             float b) {
  __hip_bfloat162 v = __float22bfloat162_rn(float2{a, b});
  return *reinterpret_cast<unsigned int*>(&v);
}
