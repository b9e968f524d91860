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

typedef unsigned short lstm_bf16;

__device__ __forceinline__ float lstm_b2f(lstm_bf16 u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __uint_as_float(x);
}


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
    float* __restrict__ grad_c_prev, long long N, int H,
    long long gh_stride) {
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
    const float dh = grad_h[n * gh_stride + h];
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


// bf16-gates variants: the gate GEMM (addmm) emits bf16; reading it
// directly (and emitting bf16 grad_gates) removes the [N,4H] f32 cast
// kernel on each side of the tail (~2 x 4.7 us/step).
extern "C" __global__ void drla_lstm_tail_fwd_bf16(
    const unsigned short* __restrict__ gates,
    const float* __restrict__ c_prev, float* __restrict__ new_h,
    float* __restrict__ new_c, float* __restrict__ stash, float forget_bias,
    long long N, int H) {
  long long idx = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = N * H;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; idx < total; idx += stride) {
    const long long n = idx / H;
    const int h = idx - n * H;
    const long long g0 = n * 4LL * H + h;
    const float i_s = drla_sigmoid(lstm_b2f(gates[g0]));
    const float g_t = tanhf(lstm_b2f(gates[g0 + H]));
    const float f_s = drla_sigmoid(lstm_b2f(gates[g0 + 2 * H])
                                   + forget_bias);
    const float o_s = drla_sigmoid(lstm_b2f(gates[g0 + 3 * H]));
    const float c_new = f_s * c_prev[idx] + i_s * g_t;
    new_c[idx] = c_new;
    new_h[idx] = o_s * tanhf(c_new);
    stash[g0] = i_s;
    stash[g0 + H] = g_t;
    stash[g0 + 2 * H] = f_s;
    stash[g0 + 3 * H] = o_s;
  }
}

extern "C" __global__ void drla_lstm_tail_bwd_bf16(
    const float* __restrict__ grad_h, const float* __restrict__ grad_c,
    const float* __restrict__ stash, const float* __restrict__ c_prev,
    const float* __restrict__ new_c, unsigned short* __restrict__ grad_gates,
    float* __restrict__ grad_c_prev, long long N, int H,
    long long gh_stride) {
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
    const float dh = grad_h[n * gh_stride + h];
    const float d_tc = dh * o_s * (1.0f - tc * tc) + grad_c[idx];
    grad_c_prev[idx] = d_tc * f_s;
    const float di = d_tc * g_t;
    const float dg = d_tc * i_s;
    const float df = d_tc * c_prev[idx];
    const float do_ = dh * tc;
    grad_gates[g0] = drla_f32_to_bf16(di * i_s * (1.0f - i_s));
    grad_gates[g0 + H] = drla_f32_to_bf16(dg * (1.0f - g_t * g_t));
    grad_gates[g0 + 2 * H] = drla_f32_to_bf16(df * f_s * (1.0f - f_s));
    grad_gates[g0 + 3 * H] = drla_f32_to_bf16(do_ * o_s * (1.0f - o_s));
  }
}

// K3 sequence form (SURVEY §5.7a + BASELINE "burn_in hidden-state
// recompute"): the WHOLE no-grad LSTM unroll in ONE kernel.
//
// The x-projection (feat @ Wx + b) has no recurrence and runs as one
// batched MFMA GEMM outside; only the tiny h @ Wh chain is sequential.
// One block per batch row: Wh [H][4H] staged in LDS once, then L steps of
// {gates_h = h @ Wh + xgates[t]; cell tail; done-mask reset} with h/c in
// LDS. blockDim = 4H (H=64 -> 256 threads); consecutive lanes read
// consecutive Wh bytes (conflict-free).
//
// Replaces the reference's per-timestep replica chain
// (r2d2_lstm.py:67-114): ~5 launches/step -> 1 launch per sequence.

extern "C" __global__ void drla_lstm_seq_fwd(
    const lstm_bf16* __restrict__ xg16,  // [B,L,4H] (nullable)
    const float* __restrict__ xg32,      // [B,L,4H] (nullable)
    const lstm_bf16* __restrict__ Wh,    // [H][4H] bf16
    const float* __restrict__ h0,        // [B,H]
    const float* __restrict__ c0,        // [B,H]
    const unsigned char* __restrict__ done,  // [B,L]
    float* __restrict__ h_out,           // [B,L,H]
    float* __restrict__ h_fin,           // [B,H]
    float* __restrict__ c_fin,           // [B,H]
    float forget_bias, int B, int L, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  lstm_bf16* wh = reinterpret_cast<lstm_bf16*>(smem);        // [H*4H]
  float* hbuf = reinterpret_cast<float*>(wh + H * 4 * H);    // [H]
  float* cbuf = hbuf + H;                                    // [H]

  const int b = blockIdx.x;
  const int j = threadIdx.x;          // gate index in [0, 4H)
  const int G = 4 * H;

  for (int i = j; i < H * G; i += blockDim.x) wh[i] = Wh[i];
  if (j < H) {
    hbuf[j] = h0[(long long)b * H + j];
    cbuf[j] = c0[(long long)b * H + j];
  }
  __syncthreads();

  for (int t = 0; t < L; ++t) {
    // gates_h[j] = sum_k h[k] * Wh[k][j]
    float g = 0.0f;
    for (int k = 0; k < H; ++k) {
      g = fmaf(hbuf[k], lstm_b2f(wh[k * G + j]), g);
    }
    const long long xbase = ((long long)b * L + t) * G + j;
    g += xg16 ? lstm_b2f(xg16[xbase]) : xg32[xbase];
    __shared__ float gates[1024];
    gates[j] = g;
    __syncthreads();
    if (j < H) {
      const float i_s = drla_sigmoid(gates[j]);
      const float g_t = tanhf(gates[H + j]);
      const float f_s = drla_sigmoid(gates[2 * H + j] + forget_bias);
      const float o_s = drla_sigmoid(gates[3 * H + j]);
      const float c_new = f_s * cbuf[j] + i_s * g_t;
      const float h_new = o_s * tanhf(c_new);
      h_out[((long long)b * L + t) * H + j] = h_new;
      // done-mask reset applies to the CARRY, not the emitted h
      const float keep = done[(long long)b * L + t] ? 0.0f : 1.0f;
      hbuf[j] = h_new * keep;
      cbuf[j] = c_new * keep;
    }
    __syncthreads();
  }
  if (j < H) {
    h_fin[(long long)b * H + j] = hbuf[j];
    c_fin[(long long)b * H + j] = cbuf[j];
  }
}
