// K6: fused A2C 1-step-TD loss pipeline, forward +
// closed-form backward — the A3C learner's whole post-network math
// (reference optimizer/a2c.py:3-26 + the agent's clip/discount glue,
// agent/a3c.py:39-51) in one kernel each way:
//
//   p      = softmax(logits_n)          (stashed)
//   adv    = clip(r_n) + disc_n * V'(n) - V(n)   (disc = (1-done)*gamma)
//   pi     = -mean_n adv * log(p[a_n] + 1e-8)    (adv detached;
//            log pi(a) — the repo's documented fix over the reference's
//            raw-probability multiply, algorithms/a2c.py)
//   base   = mean_n adv^2                        (V' detached)
//   ent    = mean_n sum_j p_j log p_j            (negative entropy)
//   total  = pi + c_bl*base + c_ent*ent
//
// Backward (same softmax/entropy forms as drla_vtrace_loss_bwd):
//   dlogits_n = [gpi * -adv*w*(onehot - p) + ge * p*(log p - E)] / N
//   dV_n      = gb * (-2 adv) / N

#include "drla_common.h"

typedef unsigned short a2c_bf16;

__device__ __forceinline__ float a2c_ld(const a2c_bf16* p16,
                                        const float* p32, long long i) {
  if (p16) {
    unsigned int x = ((unsigned int)p16[i]) << 16;
    return __uint_as_float(x);
  }
  return p32[i];
}

__device__ __forceinline__ float a2c_clip(float r, int mode) {
  if (mode == 0) return fminf(1.0f, fmaxf(-1.0f, r));
  if (mode == 1) {
    const float sq = tanhf(r / 5.0f);
    return (r < 0.0f ? 0.3f * sq : sq) * 5.0f;
  }
  return r;
}

extern "C" __global__ void drla_a2c_loss_fwd(
    const a2c_bf16* __restrict__ lg16, const float* __restrict__ lg32,
    const float* __restrict__ value,       // [N]
    const float* __restrict__ next_value,  // [N]
    const int* __restrict__ actions,       // [N]
    const float* __restrict__ rewards,     // [N] raw
    const unsigned char* __restrict__ done,
    float gamma, int clip_mode, float c_bl, float c_ent,
    float* __restrict__ losses,   // [4] pi, base, ent, total (zeroed)
    float* __restrict__ p_stash,  // [N,A]
    float* __restrict__ adv_st,   // [N]
    int N, int A) {
  const int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  const long long base = (long long)n * A;
  float mx = -1e30f;
  for (int k = 0; k < A; ++k)
    mx = fmaxf(mx, a2c_ld(lg16, lg32, base + k));
  float z = 0.0f;
  for (int k = 0; k < A; ++k)
    z += __expf(a2c_ld(lg16, lg32, base + k) - mx);
  const float inv_z = 1.0f / z;
  float ent = 0.0f;
  for (int k = 0; k < A; ++k) {
    const float p = __expf(a2c_ld(lg16, lg32, base + k) - mx) * inv_z;
    p_stash[base + k] = p;
    // clamp inside the log: p == 0 after softmax underflow would make
    // 0 * logf(0) NaN; the correct limit is 0
    ent += p * __logf(fmaxf(p, 1e-30f));
  }
  const float disc = done[n] ? 0.0f : gamma;
  const float adv = a2c_clip(rewards[n], clip_mode)
                    + disc * next_value[n] - value[n];
  adv_st[n] = adv;
  const float pi =
      -adv * __logf(p_stash[base + drla_clamp_idx(actions[n], A)]
                    + 1e-8f) / N;
  const float bl = adv * adv / N;
  const float en = ent / N;
  atomicAdd(&losses[0], pi);
  atomicAdd(&losses[1], bl);
  atomicAdd(&losses[2], en);
  atomicAdd(&losses[3], pi + c_bl * bl + c_ent * en);
}

extern "C" __global__ void drla_a2c_loss_bwd(
    const float* __restrict__ p_stash, const float* __restrict__ adv_st,
    const int* __restrict__ actions, const float* __restrict__ grad3,
    int from_total, float c_bl, float c_ent,
    a2c_bf16* __restrict__ dlg16, float* __restrict__ dlg32,
    float* __restrict__ dvalue, int N, int A) {
  const float gpi = grad3[0];
  const float gb = from_total ? gpi * c_bl : grad3[1];
  const float ge = from_total ? gpi * c_ent : grad3[2];
  const int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  const long long base = (long long)n * A;
  const int a = drla_clamp_idx(actions[n], A);
  const float adv = adv_st[n];
  const float sa = p_stash[base + a];
  const float w = sa / (sa + 1e-8f);
  float E = 0.0f;
  for (int k = 0; k < A; ++k) {
    const float p = p_stash[base + k];
    E += p * __logf(fmaxf(p, 1e-30f));  // 0*log(0) -> 0, not NaN
  }
  for (int k = 0; k < A; ++k) {
    const float p = p_stash[base + k];
    const float onehot = (k == a) ? 1.0f : 0.0f;
    const float d = (gpi * (-adv * w * (onehot - p))
                     + ge * p * (__logf(fmaxf(p, 1e-30f)) - E)) / N;
    if (dlg16) dlg16[base + k] = drla_f32_to_bf16(d);
    else dlg32[base + k] = d;
  }
  dvalue[n] = gb * (-2.0f * adv) / N;
}
