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

// grad_c may be null (the cell's new_c is unused downstream — IMPALA's
// batched unroll consumes only new_h — so autograd has no c-grad; the
// null saves a [N,H] zero-fill per step)
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
    const float gc = grad_c ? grad_c[idx] : 0.0f;
    const float d_tc = dh * o_s * (1.0f - tc * tc) + gc;
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
    const float gc = grad_c ? grad_c[idx] : 0.0f;
    const float d_tc = dh * o_s * (1.0f - tc * tc) + gc;
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

// ---------------------------------------------------------------------------
// K3 sequence form WITH GRADIENTS: the R2D2 trained window currently loops
// L per-step cell calls (~6 launches each + backward). This pair runs the
// whole recurrence as one kernel each way; the x-projection GEMM, dWh GEMM
// and bias reduce stay outside (MFMA-shaped, autograd/hipBLASLt).
// Stashes: activated gates [B,L,4H] f32, c_prev [B,L,H] f32 and h_prev
// [B,L,H] bf16 (consumed by the dWh GEMM).
// ---------------------------------------------------------------------------

extern "C" __global__ void drla_lstm_seq_train_fwd(
    const lstm_bf16* __restrict__ xg16,  // [B,L,4H]
    const lstm_bf16* __restrict__ Wh,    // [H][4H]
    const float* __restrict__ h0,        // [B,H]
    const float* __restrict__ c0,        // [B,H]
    const unsigned char* __restrict__ done,  // [B,L]
    float* __restrict__ h_out,           // [B,L,H]
    float* __restrict__ h_fin, float* __restrict__ c_fin,  // [B,H]
    float* __restrict__ acts,            // [B,L,4H] i,g,f,o activated
    float* __restrict__ c_prev_st,       // [B,L,H]
    lstm_bf16* __restrict__ h_prev_st,   // [B,L,H]
    float forget_bias, int B, int L, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  lstm_bf16* wh = reinterpret_cast<lstm_bf16*>(smem);
  float* hbuf = reinterpret_cast<float*>(wh + H * 4 * H);
  float* cbuf = hbuf + H;

  const int b = blockIdx.x;
  const int j = threadIdx.x;
  const int G = 4 * H;

  for (int i = j; i < H * G; i += blockDim.x) wh[i] = Wh[i];
  if (j < H) {
    hbuf[j] = h0[(long long)b * H + j];
    cbuf[j] = c0[(long long)b * H + j];
  }
  __syncthreads();

  for (int t = 0; t < L; ++t) {
    float g = 0.0f;
    for (int k = 0; k < H; ++k) {
      g = fmaf(hbuf[k], lstm_b2f(wh[k * G + j]), g);
    }
    const long long xbase = ((long long)b * L + t) * G + j;
    g += lstm_b2f(xg16[xbase]);
    __shared__ float gates[1024];
    gates[j] = g;
    __syncthreads();
    if (j < H) {
      const long long sb = ((long long)b * L + t) * H + j;
      h_prev_st[sb] = drla_f32_to_bf16(hbuf[j]);
      c_prev_st[sb] = cbuf[j];
      const float i_s = drla_sigmoid(gates[j]);
      const float g_t = tanhf(gates[H + j]);
      const float f_s = drla_sigmoid(gates[2 * H + j] + forget_bias);
      const float o_s = drla_sigmoid(gates[3 * H + j]);
      acts[xbase] = i_s;
      acts[((long long)b * L + t) * G + H + j] = g_t;
      acts[((long long)b * L + t) * G + 2 * H + j] = f_s;
      acts[((long long)b * L + t) * G + 3 * H + j] = o_s;
      const float c_new = f_s * cbuf[j] + i_s * g_t;
      const float h_new = o_s * tanhf(c_new);
      h_out[sb] = h_new;
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

extern "C" __global__ void drla_lstm_seq_train_bwd(
    const float* __restrict__ dh_out,    // [B,L,H]
    const float* __restrict__ dh_fin,    // [B,H] or null
    const float* __restrict__ dc_fin,    // [B,H] or null
    const float* __restrict__ acts,      // [B,L,4H]
    const float* __restrict__ c_prev_st, // [B,L,H]
    const lstm_bf16* __restrict__ Wh,    // [H][4H]
    const unsigned char* __restrict__ done,  // [B,L]
    lstm_bf16* __restrict__ dxg,         // [B,L,4H] gate-preact grads
    float* __restrict__ dh0, float* __restrict__ dc0,  // [B,H]
    int B, int L, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  lstm_bf16* wh = reinterpret_cast<lstm_bf16*>(smem);
  float* dhc = reinterpret_cast<float*>(wh + H * 4 * H);  // [H]
  float* dcc = dhc + H;                                   // [H]

  const int b = blockIdx.x;
  const int j = threadIdx.x;
  const int G = 4 * H;

  for (int i = j; i < H * G; i += blockDim.x) wh[i] = Wh[i];
  if (j < H) {
    dhc[j] = dh_fin ? dh_fin[(long long)b * H + j] : 0.0f;
    dcc[j] = dc_fin ? dc_fin[(long long)b * H + j] : 0.0f;
  }
  __syncthreads();

  __shared__ float dg4[1024];
  for (int t = L - 1; t >= 0; --t) {
    const float keep = done[(long long)b * L + t] ? 0.0f : 1.0f;
    if (j < H) {
      const long long ab = ((long long)b * L + t) * G;
      const long long sb = ((long long)b * L + t) * H + j;
      const float i_s = acts[ab + j];
      const float g_t = acts[ab + H + j];
      const float f_s = acts[ab + 2 * H + j];
      const float o_s = acts[ab + 3 * H + j];
      const float cp = c_prev_st[sb];
      const float c_new = f_s * cp + i_s * g_t;
      const float tc = tanhf(c_new);
      // carry grads were stored PRE-mask; the forward applied keep_t to
      // this step's outputs before carrying them into t+1
      const float dh_new = dh_out[sb] + dhc[j] * keep;
      const float d_tc = dh_new * o_s * (1.0f - tc * tc) + dcc[j] * keep;
      const float di = d_tc * g_t;
      const float dg = d_tc * i_s;
      const float df = d_tc * cp;
      const float do_ = dh_new * tc;
      dg4[j] = di * i_s * (1.0f - i_s);
      dg4[H + j] = dg * (1.0f - g_t * g_t);
      dg4[2 * H + j] = df * f_s * (1.0f - f_s);
      dg4[3 * H + j] = do_ * o_s * (1.0f - o_s);
      dcc[j] = d_tc * f_s;  // pre-mask carry for t-1
    }
    __syncthreads();
    dxg[((long long)b * L + t) * G + j] = drla_f32_to_bf16(dg4[j]);
    if (j < H) {
      float s = 0.0f;
      for (int l = 0; l < G; ++l) {
        s = fmaf(dg4[l], lstm_b2f(wh[j * G + l]), s);
      }
      dhc[j] = s;  // pre-mask carry for t-1 (h0 enters unmasked)
    }
    __syncthreads();
  }
  if (j < H) {
    dh0[(long long)b * H + j] = dhc[j];
    dc0[(long long)b * H + j] = dcc[j];
  }
}
