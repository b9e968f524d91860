// K5+K7+K13 fused: the ENTIRE IMPALA loss pipeline in two
// kernels.
//
// Forward: one workgroup per batch row (rows are fully independent —
// softmax over A per (b,t) -> rho = pi[a]/mu[a] -> two V-trace reverse scans
// (first + middle windows) -> pg advantage -> the three losses, accumulated
// with one atomicAdd triple per block). Semantics: reference
// agent/impala.py:63-100 + optimizer/vtrace.py:29-126 (sum reductions).
//
// Backward (closed form — everything but log pi(a) and the entropy is
// stop-gradient in the reference):
//   dlogits[b,t,:] = gpi * -adv * sa/(sa+eps) * (onehot - s)
//                  + ge  * s * (log s - sum_j s_j log s_j)   for t < T-2
//   dvalue[b,t]    = gb * (v - vs)                           for t < T-2
//
// Replaces ~90 eager launches (~400 us/step measured,
// profiles/impala_bench_r06_bf16_kernels.md) with two small kernels.

#include "drla_common.h"

typedef unsigned short bf16raw;

__device__ __forceinline__ float vt_ld(const bf16raw* p, long long i) {
  unsigned int x = ((unsigned int)p[i]) << 16;
  return __uint_as_float(x);
}

#define VT_MAX_T 128  // trajectory-length cap for the LDS scan buffers

// grid = B blocks of 256; losses[4] must be ZERO on entry (atomicAdd).
// Reward clipping (clip_mode: 0 = abs_one, 1 = soft_asymmetric, 2 = none,
// reference agent/impala.py:45-49) and discounts = (1-done)*gamma are
// computed IN-KERNEL from the raw reward/done rows — the python-side
// clamp/not/cast/mul prep kernels disappear. losses[3] accumulates the
// combined total = pi + c_bl*baseline + c_ent*entropy so the loss-combine
// arithmetic costs nothing host-side either.
extern "C" __global__ __launch_bounds__(256)
void drla_vtrace_loss_fwd(
    const bf16raw* __restrict__ logits_bf16,   // [B,T,A] (nullable)
    const float* __restrict__ logits_f32,      // [B,T,A] (nullable)
    const float* __restrict__ value,           // [B,T]
    const float* __restrict__ mu,              // [B,T,A]
    const int* __restrict__ actions,           // [B,T]
    const float* __restrict__ rewards,         // [B,T] RAW
    const unsigned char* __restrict__ done,    // [B,T] bool
    float gamma, int clip_mode, float c_bl, float c_ent,
    float* __restrict__ p_stash,               // [B,T,A] softmax out
    float* __restrict__ vs_stash,              // [B,T-2]
    float* __restrict__ adv_stash,             // [B,T-2]
    float* __restrict__ losses,                // [4]: pi, base, ent, total
    int B, int T, int A) {
  __shared__ float rho[VT_MAX_T];
  __shared__ float vsm[VT_MAX_T];
  __shared__ float rc[VT_MAX_T];
  __shared__ float gc[VT_MAX_T];
  __shared__ float red[3 * 4];

  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int Tp = T - 2;
  const long long row = (long long)b * T;

  // phase A: softmax + rho, one thread per t (multi-pass over the
  // L2-resident row — a local A-array would be runtime-indexed and spill,
  // guide §5.4 rule 20); reward clip + discount computed alongside
  for (int t = tid; t < T; t += blockDim.x) {
    {
      const float rr = rewards[row + t];
      float rcl;
      if (clip_mode == 0) {
        rcl = fminf(1.0f, fmaxf(-1.0f, rr));
      } else if (clip_mode == 1) {
        const float sq = tanhf(rr * 0.2f);
        rcl = ((rr < 0.0f) ? 0.3f * sq : sq) * 5.0f;
      } else {
        rcl = rr;
      }
      rc[t] = rcl;
      gc[t] = done[row + t] ? 0.0f : gamma;
    }
    const long long base = (row + t) * (long long)A;
    float mx = -1e30f;
    for (int k = 0; k < A; ++k) {
      const float x = logits_bf16 ? vt_ld(logits_bf16, base + k)
                                  : logits_f32[base + k];
      mx = fmaxf(mx, x);
    }
    float denom = 0.0f;
    for (int k = 0; k < A; ++k) {
      const float x = logits_bf16 ? vt_ld(logits_bf16, base + k)
                                  : logits_f32[base + k];
      denom += __expf(x - mx);
    }
    const float inv = 1.0f / denom;
    for (int k = 0; k < A; ++k) {
      const float x = logits_bf16 ? vt_ld(logits_bf16, base + k)
                                  : logits_f32[base + k];
      p_stash[base + k] = __expf(x - mx) * inv;
    }
    const int a = drla_clamp_idx(actions[row + t], A);
    rho[t] = p_stash[base + a] / mu[base + a];
  }
  __syncthreads();

  // phase B: the two reverse scans, serial in T (thread 0)
  if (tid == 0) {
    const float* v = value + row;
    const float* r = rc;
    const float* g = gc;
    // middle-window scan -> vs_plus_1 (bootstrap v[T-1])
    float acc = 0.0f;
    for (int t = Tp - 1; t >= 0; --t) {
      const float c = fminf(1.0f, rho[t + 1]);
      const float delta = c * (r[t + 1] + g[t + 1] * v[t + 2] - v[t + 1]);
      acc = delta + g[t + 1] * c * acc;
      vsm[t] = acc + v[t + 1];
    }
    // first-window scan -> vs (bootstrap v[T-2])
    acc = 0.0f;
    for (int t = Tp - 1; t >= 0; --t) {
      const float c = fminf(1.0f, rho[t]);
      const float delta = c * (r[t] + g[t] * v[t + 1] - v[t]);
      acc = delta + g[t] * c * acc;
      const float vs = acc + v[t];
      vs_stash[b * Tp + t] = vs;
      adv_stash[b * Tp + t] = c * (r[t] + g[t] * vsm[t] - v[t]);
    }
  }
  __syncthreads();

  // phase C: loss sums over this row's first window, (t, k) split across
  // threads for the entropy inner loop
  float pi_l = 0.0f, base_l = 0.0f, ent_l = 0.0f;
  for (int i = tid; i < Tp * A; i += blockDim.x) {
    const int t = i / A;
    const int k = i - t * A;
    const long long pbase = (row + t) * (long long)A;
    const float p = p_stash[pbase + k];
    // clamp inside the log: softmax underflow gives p == 0 exactly and
    // 0 * logf(0) is NaN — the correct limit of p*log p is 0
    ent_l += p * __logf(fmaxf(p, 1e-30f));  // negative entropy
                                            // (reference vtrace.py:120)
    if (k == 0) {
      const float adv = adv_stash[b * Tp + t];
      pi_l -= __logf(p_stash[pbase + drla_clamp_idx(
                  actions[row + t], A)] + 1e-8f) * adv;
      const float diff = vs_stash[b * Tp + t] - value[row + t];
      base_l += 0.5f * diff * diff;
    }
  }
  for (int off = DRLA_WAVE / 2; off > 0; off >>= 1) {
    pi_l += __shfl_down(pi_l, off, DRLA_WAVE);
    base_l += __shfl_down(base_l, off, DRLA_WAVE);
    ent_l += __shfl_down(ent_l, off, DRLA_WAVE);
  }
  const int wave = tid / DRLA_WAVE;
  const int lane = tid % DRLA_WAVE;
  if (lane == 0) {
    red[wave] = pi_l;
    red[4 + wave] = base_l;
    red[8 + wave] = ent_l;
  }
  __syncthreads();
  if (tid == 0) {
    float s0 = 0, s1 = 0, s2 = 0;
    for (int w = 0; w < (int)(blockDim.x / DRLA_WAVE); ++w) {
      s0 += red[w];
      s1 += red[4 + w];
      s2 += red[8 + w];
    }
    atomicAdd(&losses[0], s0);
    atomicAdd(&losses[1], s1);
    atomicAdd(&losses[2], s2);
    atomicAdd(&losses[3], s0 + c_bl * s1 + c_ent * s2);
  }
}

// one thread per (b,t): writes the full A-row of dlogits + dvalue.
// gpi/gb/ge are the upstream gradients of the three loss outputs
// (1, baseline_coef, entropy_coef when total.backward() is called).
// from_total != 0: grad3 is the 1-element upstream grad of the COMBINED
// total and the per-loss grads are (gt, gt*c_bl, gt*c_ent) — no host-side
// stack/scale kernels for the standard total.backward() path.
extern "C" __global__ void drla_vtrace_loss_bwd(
    const float* __restrict__ p_stash, const float* __restrict__ vs_stash,
    const float* __restrict__ adv_stash, const float* __restrict__ value,
    const int* __restrict__ actions, const float* __restrict__ grad3,
    int from_total, float c_bl, float c_ent,
    bf16raw* __restrict__ dlogits_bf16, float* __restrict__ dlogits_f32,
    float* __restrict__ dvalue, int B, int T, int A) {
  const float gpi = grad3[0];
  const float gb = from_total ? gpi * c_bl : grad3[1];
  const float ge = from_total ? gpi * c_ent : grad3[2];
  const int Tp = T - 2;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = (long long)B * T;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < total; i += stride) {
    const int b = i / T;
    const int t = i - (long long)b * T;
    const long long pbase = i * A;
    if (t >= Tp) {
      for (int k = 0; k < A; ++k) {
        if (dlogits_bf16) dlogits_bf16[pbase + k] = 0;
        else dlogits_f32[pbase + k] = 0.0f;
      }
      dvalue[i] = 0.0f;
      continue;
    }
    const int wi = b * Tp + t;
    const int a = drla_clamp_idx(actions[i], A);
    const float adv = adv_stash[wi];
    const float sa = p_stash[pbase + a];
    const float w = sa / (sa + 1e-8f);
    // row entropy term E = sum_j s_j log s_j
    float E = 0.0f;
    for (int k = 0; k < A; ++k) {
      const float s = p_stash[pbase + k];
      E += s * __logf(fmaxf(s, 1e-30f));  // 0*log(0) -> 0, not NaN
    }
    for (int k = 0; k < A; ++k) {
      const float s = p_stash[pbase + k];
      const float onehot = (k == a) ? 1.0f : 0.0f;
      const float d_pi = -adv * w * (onehot - s);
      const float d_ent = s * (__logf(fmaxf(s, 1e-30f)) - E);
      const float d = gpi * d_pi + ge * d_ent;
      if (dlogits_bf16) dlogits_bf16[pbase + k] = drla_f32_to_bf16(d);
      else dlogits_f32[pbase + k] = d;
    }
    dvalue[i] = gb * (value[i] - vs_stash[wi]);
  }
}
