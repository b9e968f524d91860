// K5: fused V-trace reverse scan.
//
// vs_minus_v[b,t] = delta[b,t] + discount[b,t] * c[b,t] * vs_minus_v[b,t+1]
// — sequential in T, parallel in B. One lane owns one batch row; at the
// reference shape ([B=32, T=18], ~2 KB inputs) the whole scan is one wave and
// entirely latency-bound, replacing the reference's T-step tf.scan graph
// (vtrace.py:88-100). Row-major [B,T] layout: a lane walks its row backwards
// with stride 1 (per-lane sequential, L2-resident).

#include "drla_common.h"

extern "C" __global__ void drla_vtrace_scan(
    const float* __restrict__ deltas, const float* __restrict__ discounts,
    const float* __restrict__ cs, float* __restrict__ out, int B, int T) {
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const float* dl = deltas + (long long)b * T;
  const float* dc = discounts + (long long)b * T;
  const float* cc = cs + (long long)b * T;
  float* o = out + (long long)b * T;
  float acc = 0.0f;
  for (int t = T - 1; t >= 0; --t) {
    acc = fmaf(dc[t] * cc[t], acc, dl[t]);
    o[t] = acc;
  }
}
