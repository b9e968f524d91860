// K9: fused R2D2 sequence-TD tail, forward + closed-form
// backward.
//
// Consumes the post-burn-in Q windows [B,W,A] (main, grad; target,
// no-grad) plus RAW rewards/done and computes, per (b, t<W-1):
//   a*    = argmax_a Qm[b,t+1,a]                 (double-DQN action)
//   nsav  = Qt[b,t+1,a*]
//   y     = h( h^-1(nsav) * disc_t + clip(r_t) ) (value rescaling,
//            reference optimizer/burn_in.py:23-32; agent/r2d2.py:62-93)
//   td    = y - Qm[b,t,act_t]
// then per sequence: unweighted_b = mean_t td^2, td_out_b = |mean_t td|,
// and loss = mean_b w_b * unweighted_b.
//
// Replaces the ~18 eager torch launches of the tail (clip, discounts,
// slices, two gathers, argmax, h/h^-1, squares, three means, weighting)
// with one kernel each way — the R2D2 step's remaining torch glue
// (VERDICT r1 weak #5 / item 7).
//
// One block per sequence b; W-1 <= 1024 lanes; LDS tree-reduce for the
// two per-sequence sums; block 0 lane 0 has already zeroed loss via the
// caller (torch::zeros({1})).
//
// Backward: only Qm[b, t<W-1, act_t] carries grad (target path is
// detached, argmax is piecewise-constant):
//   dQm[b,t,a] = gloss * w_b * (-2 td_t) / ((W-1) * B) * [a == act_t]

#include "drla_common.h"

typedef unsigned short r2_bf16;

__device__ __forceinline__ float r2_ld(const r2_bf16* p16, const float* p32,
                                       long long i) {
  if (p16) {
    unsigned int x = ((unsigned int)p16[i]) << 16;
    return __uint_as_float(x);
  }
  return p32[i];
}

__device__ __forceinline__ float r2_clip(float r, int mode) {
  if (mode == 0) return fminf(1.0f, fmaxf(-1.0f, r));   // abs_one
  if (mode == 1) {                                      // soft_asymmetric
    const float sq = tanhf(r / 5.0f);
    return (r < 0.0f ? 0.3f * sq : sq) * 5.0f;
  }
  return r;                                             // none
}

__device__ __forceinline__ float r2_h(float x) {
  const float eps = 1e-3f;
  const float s = (x > 0.0f) - (x < 0.0f);
  return s * (sqrtf(fabsf(x) + 1.0f) - 1.0f) + eps * x;
}

__device__ __forceinline__ float r2_hinv(float x) {
  const float eps = 1e-3f;
  const float s = (x > 0.0f) - (x < 0.0f);
  const float t = (sqrtf(1.0f + 4.0f * eps * (fabsf(x) + 1.0f + eps))
                   - 1.0f) / (2.0f * eps);
  return s * (t * t - 1.0f);
}

extern "C" __global__ void drla_r2d2_loss_fwd(
    const r2_bf16* __restrict__ mq16, const float* __restrict__ mq32,
    const r2_bf16* __restrict__ tq16, const float* __restrict__ tq32,
    const int* __restrict__ actions,     // [B,W]
    const float* __restrict__ rewards,   // [B,W] raw
    const unsigned char* __restrict__ done,  // [B,W]
    const float* __restrict__ weights,   // [B]
    float gamma, int clip_mode,
    float* __restrict__ loss,    // [1], zeroed by caller
    float* __restrict__ td_st,   // [B,W-1] signed td stash (backward)
    float* __restrict__ td_out,  // [B] |mean td| (priorities)
    int B, int W, int A) {
  const int b = blockIdx.x;
  const int t = threadIdx.x;          // [0, W-1)
  const int T1 = W - 1;
  __shared__ float sq[1024];
  __shared__ float sm[1024];
  float td = 0.0f;
  if (t < T1) {
    const long long rowN = ((long long)b * W + t + 1) * A;
    int astar = 0;
    float best = r2_ld(mq16, mq32, rowN);
    for (int a = 1; a < A; ++a) {
      const float v = r2_ld(mq16, mq32, rowN + a);
      if (v > best) { best = v; astar = a; }
    }
    const float nsav = r2_ld(tq16, tq32, rowN + astar);
    const long long bt = (long long)b * W + t;
    const float disc = done[bt] ? 0.0f : gamma;
    const float y = r2_h(r2_hinv(nsav) * disc
                         + r2_clip(rewards[bt], clip_mode));
    const float sav = r2_ld(mq16, mq32,
                            ((long long)b * W + t) * A + actions[bt]);
    td = y - sav;
    td_st[(long long)b * T1 + t] = td;
  }
  sq[t] = td * td;
  sm[t] = td;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (t < s) { sq[t] += sq[t + s]; sm[t] += sm[t + s]; }
    __syncthreads();
  }
  if (t == 0) {
    atomicAdd(loss, weights[b] * sq[0] / T1 / B);
    td_out[b] = fabsf(sm[0] / T1);
  }
}

extern "C" __global__ void drla_r2d2_loss_bwd(
    const float* __restrict__ td_st,   // [B,W-1]
    const int* __restrict__ actions,   // [B,W]
    const float* __restrict__ weights, // [B]
    const float* __restrict__ gloss,   // [1]
    r2_bf16* __restrict__ dmq16, float* __restrict__ dmq32,
    int B, int W, int A) {
  const long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = (long long)B * W * A;
  if (i >= total) return;
  const int a = i % A;
  const long long bt = i / A;
  const int t = bt % W;
  const int b = bt / W;
  const int T1 = W - 1;
  float g = 0.0f;
  if (t < T1 && a == actions[bt]) {
    g = gloss[0] * weights[b]
        * (-2.0f * td_st[(long long)b * T1 + t]) / T1 / B;
  }
  if (dmq16) dmq16[i] = drla_f32_to_bf16(g);
  else dmq32[i] = g;
}
