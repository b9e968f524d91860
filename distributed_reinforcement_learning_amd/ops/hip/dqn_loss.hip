// K8: fused n-step(=1) double-DQN target + IS-weighted
// TD loss, forward + closed-form backward.
//
//   a* = argmax_a Qm(s',a);  y = r + disc * Qt(s',a*)   (reference
//   agent/apex.py:60-69);   td = y - Qm(s,a)
//   loss = mean_b w_b * td_b^2;  dQm(s,a) = gloss * w * (-2 td) / B
//
// One thread per batch row; replaces the ~12 eager launches of the torch
// composition with one tiny kernel each way.

#include "drla_common.h"

typedef unsigned short bf16raw;

__device__ __forceinline__ float dq_ld(const bf16raw* p16, const float* p32,
                                       long long i) {
  if (p16) {
    unsigned int x = ((unsigned int)p16[i]) << 16;
    return __uint_as_float(x);
  }
  return p32[i];
}

extern "C" __global__ void drla_dqn_loss_fwd(
    const bf16raw* __restrict__ mq16, const float* __restrict__ mq32,
    const float* __restrict__ next_main, const float* __restrict__ next_tgt,
    const int* __restrict__ actions, const float* __restrict__ rewards,
    const float* __restrict__ discounts, const float* __restrict__ weights,
    float* __restrict__ loss,      // [1], zeroed by caller
    float* __restrict__ td_out,    // [B] signed td
    int B, int A) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const long long base = (long long)b * A;
  int astar = 0;
  float best = next_main[base];
  for (int k = 1; k < A; ++k) {
    const float v = next_main[base + k];
    if (v > best) { best = v; astar = k; }
  }
  const float y = rewards[b] + discounts[b] * next_tgt[base + astar];
  const float sav = dq_ld(mq16, mq32,
                          base + drla_clamp_idx(actions[b], A));
  const float td = y - sav;
  td_out[b] = td;
  atomicAdd(loss, weights[b] * td * td / B);
}

extern "C" __global__ void drla_dqn_loss_bwd(
    const float* __restrict__ td, const int* __restrict__ actions,
    const float* __restrict__ weights, const float* __restrict__ gloss,
    bf16raw* __restrict__ dmq16, float* __restrict__ dmq32, int B, int A) {
  const long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  if (i >= (long long)B * A) return;
  const int b = i / A;
  const int k = i - (long long)b * A;
  const float g = (k == drla_clamp_idx(actions[b], A))
                      ? gloss[0] * weights[b] * (-2.0f * td[b]) / B
                      : 0.0f;
  if (dmq16) dmq16[i] = drla_f32_to_bf16(g);
  else dmq32[i] = g;
}
