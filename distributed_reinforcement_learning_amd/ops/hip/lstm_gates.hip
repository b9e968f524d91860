// K3: fused LSTM-cell elementwise tail, forward + backward.
//
// The cell GEMM ([x,h] @ W) runs on MFMA via hipBLASLt; this kernel fuses the
// remaining 7 elementwise ops (2x sigmoid, 2x tanh, blend, output) into ONE
// HBM pass each way, instead of ~10 eager launches. Gate order [i, g, f, o]
// with TF forget_bias added to f (models/blocks.LSTMCellTF; reference
// impala_actor_critic.py:18-25 uses tf LSTMCell semantics).
//
// Layout: gates [N, 4H] row-major, states [N, H]. One lane per (n, h) cell;
// fully coalesced; stash holds the post-activation gates for backward.

#include "drla_common.h"

extern "C" __global__ void drla_lstm_tail_fwd(
    const float* __restrict__ gates, const float* __restrict__ c_prev,
    float* __restrict__ new_h, float* __restrict__ new_c,
    float* __restrict__ stash,  // [N,4H] activated gates (i,g,f,o)
    float forget_bias, long long N, int H) {
  long long idx = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = N * H;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; idx < total; idx += stride) {
    const long long n = idx / H;
    const int h = idx - n * H;
    const long long g0 = n * 4LL * H + h;
    const float i_s = drla_sigmoid(gates[g0]);
    const float g_t = tanhf(gates[g0 + H]);
    const float f_s = drla_sigmoid(gates[g0 + 2 * H] + forget_bias);
    const float o_s = drla_sigmoid(gates[g0 + 3 * H]);
    const float c_new = f_s * c_prev[idx] + i_s * g_t;
    new_c[idx] = c_new;
    new_h[idx] = o_s * tanhf(c_new);
    stash[g0] = i_s;
    stash[g0 + H] = g_t;
    stash[g0 + 2 * H] = f_s;
    stash[g0 + 3 * H] = o_s;
  }
}

extern "C" __global__ void drla_lstm_tail_bwd(
    const float* __restrict__ grad_h, const float* __restrict__ grad_c,
    const float* __restrict__ stash, const float* __restrict__ c_prev,
    const float* __restrict__ new_c, float* __restrict__ grad_gates,
    float* __restrict__ grad_c_prev, long long N, int H) {
  long long idx = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = N * H;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; idx < total; idx += stride) {
    const long long n = idx / H;
    const int h = idx - n * H;
    const long long g0 = n * 4LL * H + h;
    const float i_s = stash[g0];
    const float g_t = stash[g0 + H];
    const float f_s = stash[g0 + 2 * H];
    const float o_s = stash[g0 + 3 * H];
    const float tc = tanhf(new_c[idx]);
    const float dh = grad_h[idx];
    const float d_tc = dh * o_s * (1.0f - tc * tc) + grad_c[idx];
    grad_c_prev[idx] = d_tc * f_s;
    const float di = d_tc * g_t;
    const float dg = d_tc * i_s;
    const float df = d_tc * c_prev[idx];
    const float do_ = dh * tc;
    grad_gates[g0] = di * i_s * (1.0f - i_s);
    grad_gates[g0 + H] = dg * (1.0f - g_t * g_t);
    grad_gates[g0 + 2 * H] = df * f_s * (1.0f - f_s);
    grad_gates[g0 + 3 * H] = do_ * o_s * (1.0f - o_s);
  }
}
