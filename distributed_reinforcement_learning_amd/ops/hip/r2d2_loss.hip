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
                            ((long long)b * W + t) * A
                                + drla_clamp_idx(actions[bt], A));
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

// No-grad dueling head over the post-burn-in window, ONE launch:
//   x = relu(h @ Wt^T + bt);  y = x @ Wo^T + bo;  q = y[:A] - y[A]
// replacing the 5-kernel torch chain (cast, trunk addmm, relu, out addmm,
// slice-sub) on the R2D2 target-net / TD-scoring paths. Weights stream
// through LDS once per block; h rows come straight from the seq-recurrence
// kernel's [B,L,H] output with the window slice resolved by index math
// (srow = b*L + burn + t — no strided-copy kernel).
// blockDim 256 = 2 rows in flight x 128 lanes; nn.Linear weight layout
// [out][in] row-major.
extern "C" __global__ void drla_dueling_head_fwd(
    const float* __restrict__ h,       // [B, L, IN]
    const r2_bf16* __restrict__ Wt,    // [MID][IN]
    const r2_bf16* __restrict__ bt,    // [MID]
    const r2_bf16* __restrict__ Wo,    // [AO][MID]
    const r2_bf16* __restrict__ bo,    // [AO]
    r2_bf16* __restrict__ q,           // [B, W, AO-1]
    int B, int L, int burn, int IN, int MID, int AO) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  r2_bf16* wt = reinterpret_cast<r2_bf16*>(smem);          // [MID*IN]
  r2_bf16* wo = wt + MID * IN;                             // [AO*MID]
  float* xb = reinterpret_cast<float*>(wo + AO * MID);     // [2][MID]
  float* hb = xb + 2 * MID;                                // [2][IN]
  float* yb = hb + 2 * IN;                                 // [2][AO]

  const int tid = threadIdx.x;
  for (int i = tid; i < MID * IN; i += blockDim.x) wt[i] = Wt[i];
  for (int i = tid; i < AO * MID; i += blockDim.x) wo[i] = Wo[i];
  __syncthreads();

  const int W = L - burn;
  const long long npairs = ((long long)B * W + 1) / 2;
  const int half = tid / 128;   // which of the 2 in-flight rows
  const int lane = tid % 128;
  const int A = AO - 1;
  for (long long p = blockIdx.x; p < npairs; p += gridDim.x) {
    const long long n = 2 * p + half;
    const bool live = n < (long long)B * W;
    if (live) {
      const long long b = n / W;
      const long long srow = (b * L + burn + (n - b * W)) * IN;
      for (int i = lane; i < IN; i += 128) hb[half * IN + i] = h[srow + i];
    }
    __syncthreads();
    if (live) {
      for (int o = lane; o < MID; o += 128) {
        float acc = r2_ld(bt, nullptr, o);
        const r2_bf16* wrow = wt + o * IN;
        for (int i = 0; i < IN; ++i)
          acc = fmaf(hb[half * IN + i], r2_ld(wrow, nullptr, i), acc);
        xb[half * MID + o] = fmaxf(acc, 0.0f);
      }
    }
    __syncthreads();
    if (live && lane < AO) {
      float acc = r2_ld(bo, nullptr, lane);
      const r2_bf16* wrow = wo + lane * MID;
      for (int i = 0; i < MID; ++i)
        acc = fmaf(xb[half * MID + i], r2_ld(wrow, nullptr, i), acc);
      yb[half * AO + lane] = acc;
    }
    __syncthreads();
    if (live && lane < A) {
      q[n * A + lane] =
          drla_f32_to_bf16(yb[half * AO + lane] - yb[half * AO + A]);
    }
    __syncthreads();
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
  if (t < T1 && a == drla_clamp_idx(actions[bt], A)) {
    g = gloss[0] * weights[b]
        * (-2.0f * td_st[(long long)b * T1 + t]) / T1 / B;
  }
  if (dmq16) dmq16[i] = drla_f32_to_bf16(g);
  else dmq32[i] = g;
}

// ---------------------------------------------------------------------------
// GRAD-CARRYING dueling head (the R2D2 trained window): q = y[:A] - y[A],
// y = relu(h @ Wt^T + bt) @ Wo^T + bo. The torch chain costs ~80 us/step
// in ~13 launches at [B*W=128..640, 64]; this trio does fwd in 1 launch
// and bwd in 2 (chain + finalize).
//
// bwd math (dq [N,A] from the K9 kernel):
//   dz[:A] = dq,  dz[A] = -sum_a dq[a]          (the dueling subtract)
//   dx     = (dz @ Wo) * [x > 0]
//   dh     = dx @ Wt
//   dWo    = dz^T @ x,  dbo = colsum dz
//   dWt    = dx^T @ h,  dbt = colsum dx
// Weight-grad partials accumulate in per-block LDS tiles (the whole
// dWt/dWo fit: [MID][IN]+[AO][MID] f32 <= ~50 KB at the R2D2 shape) and
// merge with ONE atomicAdd per element per block into an f32 workspace;
// the finalize kernel emits the four bf16 grads.
// ---------------------------------------------------------------------------

extern "C" __global__ void drla_dhead_train_fwd(
    const float* __restrict__ h,       // [N, IN]
    const r2_bf16* __restrict__ Wt,    // [MID][IN]
    const r2_bf16* __restrict__ bt,    // [MID]
    const r2_bf16* __restrict__ Wo,    // [AO][MID]
    const r2_bf16* __restrict__ bo,    // [AO]
    r2_bf16* __restrict__ q,           // [N, AO-1]
    r2_bf16* __restrict__ x_st,        // [N, MID] post-ReLU stash
    int N, int IN, int MID, int AO) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  r2_bf16* wt = reinterpret_cast<r2_bf16*>(smem);
  r2_bf16* wo = wt + MID * IN;
  float* xb = reinterpret_cast<float*>(wo + AO * MID);     // [2][MID]
  float* hb = xb + 2 * MID;                                // [2][IN]
  float* yb = hb + 2 * IN;                                 // [2][AO]

  const int tid = threadIdx.x;
  for (int i = tid; i < MID * IN; i += blockDim.x) wt[i] = Wt[i];
  for (int i = tid; i < AO * MID; i += blockDim.x) wo[i] = Wo[i];
  __syncthreads();

  const long long npairs = ((long long)N + 1) / 2;
  const int half = tid / 128;
  const int lane = tid % 128;
  const int A = AO - 1;
  for (long long p = blockIdx.x; p < npairs; p += gridDim.x) {
    const long long n = 2 * p + half;
    const bool live = n < (long long)N;
    if (live) {
      for (int i = lane; i < IN; i += 128)
        hb[half * IN + i] = h[n * IN + i];
    }
    __syncthreads();
    if (live) {
      for (int o = lane; o < MID; o += 128) {
        float acc = r2_ld(bt, nullptr, o);
        const r2_bf16* wrow = wt + o * IN;
        for (int i = 0; i < IN; ++i)
          acc = fmaf(hb[half * IN + i], r2_ld(wrow, nullptr, i), acc);
        const float x = fmaxf(acc, 0.0f);
        xb[half * MID + o] = x;
        x_st[n * MID + o] = drla_f32_to_bf16(x);
      }
    }
    __syncthreads();
    if (live && lane < AO) {
      float acc = r2_ld(bo, nullptr, lane);
      const r2_bf16* wrow = wo + lane * MID;
      for (int i = 0; i < MID; ++i)
        acc = fmaf(xb[half * MID + i], r2_ld(wrow, nullptr, i), acc);
      yb[half * AO + lane] = acc;
    }
    __syncthreads();
    if (live && lane < A) {
      q[n * A + lane] =
          drla_f32_to_bf16(yb[half * AO + lane] - yb[half * AO + A]);
    }
    __syncthreads();
  }
}

extern "C" __global__ void drla_dhead_train_bwd(
    const r2_bf16* __restrict__ dq,    // [N, AO-1]
    const r2_bf16* __restrict__ x_st,  // [N, MID]
    const float* __restrict__ h,       // [N, IN]
    const r2_bf16* __restrict__ Wt,    // [MID][IN]
    const r2_bf16* __restrict__ Wo,    // [AO][MID]
    float* __restrict__ dh,            // [N, IN] f32
    float* __restrict__ ws,            // [MID*IN + MID + AO*MID + AO],
                                       // zeroed: dWt | dbt | dWo | dbo
    int N, int IN, int MID, int AO) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  r2_bf16* wt = reinterpret_cast<r2_bf16*>(smem);          // [MID*IN]
  r2_bf16* wo = wt + MID * IN;                             // [AO*MID]
  float* lWt = reinterpret_cast<float*>(wo + AO * MID);    // [MID*IN]
  float* lWo = lWt + MID * IN;                             // [AO*MID]
  float* lbt = lWo + AO * MID;                             // [MID]
  float* lbo = lbt + MID;                                  // [AO]
  float* dzb = lbo + AO;                                   // [AO]
  float* dxb = dzb + AO;                                   // [MID]
  float* xb = dxb + MID;                                   // [MID]
  float* hbuf = xb + MID;                                  // [IN]

  const int tid = threadIdx.x;
  for (int i = tid; i < MID * IN; i += blockDim.x) {
    wt[i] = Wt[i];
    lWt[i] = 0.0f;
  }
  for (int i = tid; i < AO * MID; i += blockDim.x) {
    wo[i] = Wo[i];
    lWo[i] = 0.0f;
  }
  for (int i = tid; i < MID; i += blockDim.x) lbt[i] = 0.0f;
  for (int i = tid; i < AO; i += blockDim.x) lbo[i] = 0.0f;
  __syncthreads();

  const int A = AO - 1;
  const int rows_per_block = (N + gridDim.x - 1) / gridDim.x;
  const long long n0 = (long long)blockIdx.x * rows_per_block;
  const long long n1 = min((long long)N, n0 + rows_per_block);
  for (long long n = n0; n < n1; ++n) {
    // dz (+ dbo) and stage x/h
    if (tid < AO) {
      float v;
      if (tid < A) {
        v = r2_ld(dq, nullptr, n * A + tid);
      } else {
        float s = 0.0f;
        for (int a = 0; a < A; ++a)
          s += r2_ld(dq, nullptr, n * A + a);
        v = -s;
      }
      dzb[tid] = v;
      lbo[tid] += v;
    }
    for (int o = tid; o < MID; o += blockDim.x)
      xb[o] = r2_ld(x_st, nullptr, n * MID + o);
    for (int i = tid; i < IN; i += blockDim.x) hbuf[i] = h[n * IN + i];
    __syncthreads();
    // dx (+ dbt, dWo)
    for (int o = tid; o < MID; o += blockDim.x) {
      float acc = 0.0f;
      for (int a = 0; a < AO; ++a) {
        const float dz = dzb[a];
        acc = fmaf(dz, r2_ld(wo, nullptr, a * MID + o), acc);
        lWo[a * MID + o] = fmaf(dz, xb[o], lWo[a * MID + o]);
      }
      const float dx = (xb[o] > 0.0f) ? acc : 0.0f;
      dxb[o] = dx;
      lbt[o] += dx;
    }
    __syncthreads();
    // dh and the dWt rank-1 update
    for (int i = tid; i < IN; i += blockDim.x) {
      float acc = 0.0f;
      for (int o = 0; o < MID; ++o)
        acc = fmaf(dxb[o], r2_ld(wt, nullptr, o * IN + i), acc);
      dh[n * IN + i] = acc;
    }
    for (int oi = tid; oi < MID * IN; oi += blockDim.x) {
      const int o = oi / IN;
      const int i = oi - o * IN;
      lWt[oi] = fmaf(dxb[o], hbuf[i], lWt[oi]);
    }
    __syncthreads();
  }
  // merge block-local partials: one atomicAdd per element per block
  for (int i = tid; i < MID * IN; i += blockDim.x)
    atomicAdd(&ws[i], lWt[i]);
  for (int i = tid; i < MID; i += blockDim.x)
    atomicAdd(&ws[MID * IN + i], lbt[i]);
  for (int i = tid; i < AO * MID; i += blockDim.x)
    atomicAdd(&ws[MID * IN + MID + i], lWo[i]);
  for (int i = tid; i < AO; i += blockDim.x)
    atomicAdd(&ws[MID * IN + MID + AO * MID + i], lbo[i]);
}

extern "C" __global__ void drla_dhead_train_fin(
    const float* __restrict__ ws, r2_bf16* __restrict__ dWt,
    r2_bf16* __restrict__ dbt, r2_bf16* __restrict__ dWo,
    r2_bf16* __restrict__ dbo, int IN, int MID, int AO) {
  const int total = MID * IN + MID + AO * MID + AO;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    const r2_bf16 v = drla_f32_to_bf16(ws[i]);
    if (i < MID * IN) dWt[i] = v;
    else if (i < MID * IN + MID) dbt[i - MID * IN] = v;
    else if (i < MID * IN + MID + AO * MID)
      dWo[i - MID * IN - MID] = v;
    else dbo[i - MID * IN - MID - AO * MID] = v;
  }
}
