// K10: GPU prioritized-replay segment tree.
//
// Same array layout as the CPU replay (replay/sum_tree.py, mirroring
// reference buffer_queue.py:326-371): tree[0 .. 2*cap-2] f32, leaves at
// cap-1. Payloads live in preallocated HBM tensors (replay/gpu_memory.py),
// so learner-side priorities NEVER round-trip to the host — TD errors from
// the loss kernels feed update_batch directly (SURVEY §7 build plan item 7).
//
// update: one thread per index — atomicExch the leaf (duplicate updates
// serialize into a consistent delta chain), then atomicAdd the delta up to
// the root. O(log cap) atomics per update, batched.
// sample: one thread per query — root-to-leaf descent.

#include "drla_common.h"

extern "C" __global__ void drla_per_update(
    float* __restrict__ tree, const long long* __restrict__ idxs,
    const float* __restrict__ prios, int n, long long cap) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  long long idx = idxs[i];
  const float p = prios[i];
  const float old = atomicExch(&tree[idx], p);
  float delta = p - old;
  while (idx != 0) {
    idx = (idx - 1) >> 1;
    atomicAdd(&tree[idx], delta);
  }
}

// Periodic interior rebuild: millions of float32 atomicAdd delta
// propagations drift the interior sums away from the true leaf sums
// (the CPU twin replay/sum_tree.py uses float64), slowly biasing the
// stratified sampling. One launch per tree level, host-driven
// (replay/gpu_memory.py rebuild()): tree[i] = tree[2i+1] + tree[2i+2]
// for i in [first, first+count).
extern "C" __global__ void drla_per_rebuild_level(
    float* __restrict__ tree, long long first, long long count) {
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i >= count) return;
  const long long idx = first + i;
  tree[idx] = tree[2 * idx + 1] + tree[2 * idx + 2];
}

// One-kernel replay-batch gather: replaces the per-field index_select
// chain (7 launches for an R2D2 sample at ~5-9 us each) with a single
// launch copying every field of every sampled row. blockIdx.y = field;
// fields whose row stride is 16-byte-divisible copy as uint4 (the frame /
// hidden-state payloads), the tiny remainder fields copy bytewise.
extern "C" __global__ void drla_multi_gather(
    const long long* __restrict__ rows,            // [B] sampled rows
    const unsigned long long* __restrict__ srcs,   // [F] payload ptrs
    const unsigned long long* __restrict__ dsts,   // [F] output ptrs
    const long long* __restrict__ fbytes,          // [F] bytes per row
    int B) {
  const int f = blockIdx.y;
  const long long fb = fbytes[f];
  const char* __restrict__ src = reinterpret_cast<const char*>(srcs[f]);
  char* __restrict__ dst = reinterpret_cast<char*>(dsts[f]);
  const long long cpr = (fb + 15) >> 4;  // 16B chunks per row
  const long long total = (long long)B * cpr;
  const bool vec = (fb & 15) == 0;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += gridDim.x * (long long)blockDim.x) {
    const long long r = i / cpr;
    const long long off = (i - r * cpr) << 4;
    const char* s = src + rows[r] * fb + off;
    char* d = dst + r * fb + off;
    if (vec) {
      *reinterpret_cast<uint4*>(d) = *reinterpret_cast<const uint4*>(s);
    } else {
      const int n = (int)(fb - off < 16 ? fb - off : 16);
      for (int k = 0; k < n; ++k) d[k] = s[k];
    }
  }
}

extern "C" __global__ void drla_per_sample(
    const float* __restrict__ tree, const float* __restrict__ s,
    const float* __restrict__ n_entries,  // [1] device buffer (the host
                                          // advances it; capture-safe)
    long long* __restrict__ out_idx, float* __restrict__ out_prio, int n,
    long long cap) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const long long tree_len = 2 * cap - 1;
  float rem = s[i];
  long long idx = 0;
  while (true) {
    const long long left = 2 * idx + 1;
    if (left >= tree_len) break;
    const float ls = tree[left];
    if (rem <= ls) {
      idx = left;
    } else {
      rem -= ls;
      idx = left + 1;
    }
  }
  // float32 rounding at a segment boundary can walk past the last
  // WRITTEN leaf onto a zero-priority slot -> stale payload and an
  // infinite IS weight ((n*0)^-beta); clamp into the populated range
  const long long last = cap - 2 + (long long)n_entries[0];
  if (idx > last) idx = last;
  out_idx[i] = idx;
  out_prio[i] = tree[idx];
}
