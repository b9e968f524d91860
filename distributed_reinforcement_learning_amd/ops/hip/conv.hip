// K1: hand-written implicit-GEMM conv stack for the Atari
// model on gfx950 — MFMA 16x16x32 bf16, LDS-tiled, NHWC, VALID padding.
//
// Forward kernels fuse: (layer 1) uint8 -> /255 normalize on load, and
// (all layers) bias + ReLU epilogue. Weights are consumed exactly as torch
// stores them channels_last: W[co][kh][kw][ci] == W[co][k] row-major with
// k = (kh*KW + kw)*CI + ci — each LDS B-tile row is one contiguous read.
//
// GEMM view per layer (reference model/impala_actor_critic.py:5-10):
//   L1: M = N*20*20, K = 8*8*C  (C=4: u8 input),  CO = 32
//   L2: M = N*9*9,   K = 4*4*32,                  CO = 64
//   L3: M = N*7*7,   K = 3*3*64,                  CO = 64
//
// Block: 256 threads (4 waves), BM = 128 rows x BN = CO cols, BK = 32.
// Per wave: 32 rows x CO cols as 2 x (CO/16) mfma_f32_16x16x32_bf16
// fragments. A-tile [128][32+8] and B-tile [CO][32+8] live in LDS with
// +8 bf16 row padding (conflict-free ds_read_b128, guide §6 G4).
//
// Fragment maps (verified on-box by drla_mfma_probe + the parity tests):
//   A: lane l holds A[l&15][(l>>4)*8 + e], e = 0..7   (8 bf16 = 4 VGPRs)
//   B: lane l holds B[(l>>4)*8 + e][l&15]             (8 bf16)
//   D: lane l, reg r -> row (l>>4)*4 + r, col l&15    (4 f32)

#include "drla_common.h"

typedef unsigned short bf16raw;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

__device__ __forceinline__ float cv_bf2f(bf16raw u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __uint_as_float(x);
}

// ---------------------------------------------------------------------------
// layout probe: D = A @ B for one 16x16x32 tile, used by tests to pin the
// fragment maps empirically (guide §3: check with ASYMMETRIC operands).
// A, B row-major [16][32] / [32][16] bf16; D [16][16] f32.
// ---------------------------------------------------------------------------
extern "C" __global__ void drla_mfma_probe(const bf16raw* __restrict__ A,
                                           const bf16raw* __restrict__ B,
                                           float* __restrict__ D) {
  const int l = threadIdx.x;
  bf16x8 a_frag, b_frag;
  for (int e = 0; e < 8; ++e) {
    a_frag[e] = (short)A[(l & 15) * 32 + ((l >> 4) * 8 + e)];
    b_frag[e] = (short)B[((l >> 4) * 8 + e) * 16 + (l & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) {
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
  }
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <typename IN_T, int CI, int CO, int KH, int KW, int STRIDE, int HI,
          int WI, int HO, int WO>
__device__ void conv_fwd_impl(const IN_T* __restrict__ in,
                              const bf16raw* __restrict__ w,   // [CO][K]
                              const bf16raw* __restrict__ bias,  // [CO]
                              bf16raw* __restrict__ out,       // [M][CO]
                              int batch,
                              IN_T* __restrict__ x_stash = nullptr) {
  constexpr int K = KH * KW * CI;
  constexpr int BM = 128;
  constexpr int BK = 32;
  constexpr int APAD = 8;
  constexpr int NFRAG = CO / 16;
  const int M = batch * HO * WO;

  __shared__ bf16raw Abuf[BM][BK + APAD];
  __shared__ bf16raw Bbuf[CO][BK + APAD];
  __shared__ float BiasBuf[CO];

  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int row0 = blockIdx.x * BM;

  if (tid < CO) BiasBuf[tid] = cv_bf2f(bias[tid]);

  f32x4 acc[2][NFRAG];
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < NFRAG; ++ni)
      acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // per-thread A staging assignment: thread t fills row (t>>1),
  // k-halves (t&1)*16..+16 of the 128x32 tile
  const int a_row = tid >> 1;
  const int a_k0 = (tid & 1) * 16;
  const int gm = row0 + a_row;
  // decode output coordinate once
  const int n_idx = gm / (HO * WO);
  const int rem = gm - n_idx * (HO * WO);
  const int ho = rem / WO;
  const int wo = rem - ho * WO;
  const long long in_base =
      ((long long)n_idx * HI + ho * STRIDE) * WI + wo * STRIDE;

  // software-pipelined K-loop: chunk k0+BK's global loads issue into
  // registers while chunk k0's MFMA runs (same structure as the wgrad —
  // the per-chunk load-latency chain is what sets kernel time here)
  unsigned int ra4u_0, ra4u_1, ra4u_2, ra4u_3;
  unsigned int ra1u[4];
  bf16x8 rab0 = {0, 0, 0, 0, 0, 0, 0, 0}, rab1 = rab0;
  uint4 rbw4;
  uint2 rbw2;
  constexpr int THREADS_PER_CO = 256 / CO;       // 8 (CO=32) or 4
  constexpr int KCHUNK = BK / THREADS_PER_CO;    // 4 or 8
  const int b_co = tid / THREADS_PER_CO;
  const int b_kpart = (tid % THREADS_PER_CO) * KCHUNK;

  auto load_chunk = [&](int k0) {
    if (gm < M) {
      if constexpr (CI == 4) {
        // the thread's 4 taps share kh and have CONSECUTIVE kw (16
        // k-values = 4 taps x 4 ci; tap0 % 8 is always 0 or 4), so the
        // four uchar4 gathers are 16 contiguous 16 B-aligned bytes
        const int kk = k0 + a_k0;
        const int kh = kk / (KW * CI);
        const int kw = (kk - kh * KW * CI) / CI;
        const uint4 packed = *reinterpret_cast<const uint4*>(
            in + (in_base + (long long)kh * WI + kw) * CI);
        ra4u_0 = packed.x; ra4u_1 = packed.y;
        ra4u_2 = packed.z; ra4u_3 = packed.w;
      } else if constexpr (CI == 1) {
        // 16 consecutive k = two full kw rows of 8 contiguous bytes
        // (4-byte aligned): 4 uint loads instead of 16 ubyte gathers
        const int kh0 = (k0 + a_k0) / KW;
#pragma unroll
        for (int rrow = 0; rrow < 2; ++rrow) {
          const unsigned char* src =
              in + in_base + (long long)(kh0 + rrow) * WI;
          ra1u[rrow * 2 + 0] = *reinterpret_cast<const unsigned int*>(src);
          ra1u[rrow * 2 + 1] =
              *reinterpret_cast<const unsigned int*>(src + 4);
        }
      } else {
        const int kk = k0 + a_k0;
        const int kh = kk / (KW * CI);
        const int kwci = kk - kh * KW * CI;
        const int kw = kwci / CI;
        const int ci = kwci - kw * CI;
        const bf16raw* src = reinterpret_cast<const bf16raw*>(in) +
                             (in_base + (long long)kh * WI + kw) * CI + ci;
        rab0 = *reinterpret_cast<const bf16x8*>(src);
        rab1 = *reinterpret_cast<const bf16x8*>(src + 8);
      }
    }
    const bf16raw* src = w + (long long)b_co * K + k0 + b_kpart;
    if constexpr (KCHUNK == 8) {
      rbw4 = *reinterpret_cast<const uint4*>(src);
    } else {
      rbw2 = *reinterpret_cast<const uint2*>(src);
    }
  };

  auto store_chunk = [&]() {
    if (gm < M) {
      if constexpr (CI == 4) {
        const float sc = 1.0f / 255.0f;
#define DRLA_CF_PUT(t, u)                                               \
        {                                                               \
          const int o = a_k0 + (t) * 4;                                 \
          Abuf[a_row][o + 0] = drla_f32_to_bf16((float)((u) & 0xFF) * sc); \
          Abuf[a_row][o + 1] =                                          \
              drla_f32_to_bf16((float)(((u) >> 8) & 0xFF) * sc);        \
          Abuf[a_row][o + 2] =                                          \
              drla_f32_to_bf16((float)(((u) >> 16) & 0xFF) * sc);       \
          Abuf[a_row][o + 3] = drla_f32_to_bf16((float)((u) >> 24) * sc); \
        }
        DRLA_CF_PUT(0, ra4u_0); DRLA_CF_PUT(1, ra4u_1);
        DRLA_CF_PUT(2, ra4u_2); DRLA_CF_PUT(3, ra4u_3);
#undef DRLA_CF_PUT
      } else if constexpr (CI == 1) {
        const float sc = 1.0f / 255.0f;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          const unsigned int u = ra1u[q];
          Abuf[a_row][a_k0 + q * 4 + 0] =
              drla_f32_to_bf16((float)(u & 0xFF) * sc);
          Abuf[a_row][a_k0 + q * 4 + 1] =
              drla_f32_to_bf16((float)((u >> 8) & 0xFF) * sc);
          Abuf[a_row][a_k0 + q * 4 + 2] =
              drla_f32_to_bf16((float)((u >> 16) & 0xFF) * sc);
          Abuf[a_row][a_k0 + q * 4 + 3] =
              drla_f32_to_bf16((float)(u >> 24) * sc);
        }
      } else {
        *reinterpret_cast<bf16x8*>(&Abuf[a_row][a_k0]) = rab0;
        *reinterpret_cast<bf16x8*>(&Abuf[a_row][a_k0 + 8]) = rab1;
      }
    } else {
      for (int t = 0; t < 16; t += 8) {
        *reinterpret_cast<uint4*>(&Abuf[a_row][a_k0 + t]) = uint4{0, 0, 0, 0};
      }
    }
    if constexpr (KCHUNK == 8) {
      *reinterpret_cast<uint4*>(&Bbuf[b_co][b_kpart]) = rbw4;
    } else {
      *reinterpret_cast<uint2*>(&Bbuf[b_co][b_kpart]) = rbw2;
    }
  };

  load_chunk(0);
  for (int k0 = 0; k0 < K; k0 += BK) {
    store_chunk();
    __syncthreads();
    if (k0 + BK < K) load_chunk(k0 + BK);
    // ---- MFMA: wave covers rows [wave*32, wave*32+32) x CO ----
    for (int mi = 0; mi < 2; ++mi) {
      const int arow = wave * 32 + mi * 16 + (lane & 15);
      const bf16x8 a_frag =
          *reinterpret_cast<const bf16x8*>(&Abuf[arow][(lane >> 4) * 8]);
      for (int ni = 0; ni < NFRAG; ++ni) {
        const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            &Bbuf[ni * 16 + (lane & 15)][(lane >> 4) * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: bias + ReLU, bf16 NHWC store ----
  for (int mi = 0; mi < 2; ++mi) {
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int col = ni * 16 + (lane & 15);
      const float b = BiasBuf[col];
      for (int r = 0; r < 4; ++r) {
        const int grow = row0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (grow < M) {
          const float v = fmaxf(acc[mi][ni][r] + b, 0.0f);
          out[(long long)grow * CO + col] = drla_f32_to_bf16(v);
        }
      }
    }
  }

  // pass-through input stash (graphed-IMPALA l1): the overlapped H2D of
  // the NEXT batch rewrites the static input during the backward, but
  // the l1 wgrad re-reads the input — stashing it here (bundled stores,
  // no extra launch) moves that read inside the forward for ~4 us
  // instead of a ~20 us standalone clone. Byte count is 16-divisible for
  // every supported geometry (binding checks).
  if (x_stash) {
    const long long n16 =
        (long long)batch * HI * WI * CI * (long long)sizeof(IN_T) / 16;
    const uint4* src = reinterpret_cast<const uint4*>(in);
    uint4* dst = reinterpret_cast<uint4*>(x_stash);
    const long long nthreads = gridDim.x * (long long)blockDim.x;
    for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
         i < n16; i += nthreads) {
      dst[i] = src[i];
    }
  }
}

// b_frag reads above: B fragment needs B[k][col] where the LDS image is
// Bbuf[co][k] — i.e. we feed mfma(A, B^T-read) which computes A @ B with
// B[k][col] = Bbuf[col][k]. The probe/parity tests pin this down.

extern "C" __global__ __launch_bounds__(256) void drla_conv_fwd_l1(
    const unsigned char* in, const bf16raw* w, const bf16raw* bias,
    bf16raw* out, int batch, unsigned char* x_stash) {
  conv_fwd_impl<unsigned char, 4, 32, 8, 8, 4, 84, 84, 20, 20>(
      in, w, bias, out, batch, x_stash);
}

extern "C" __global__ __launch_bounds__(256) void drla_conv_fwd_l1_c1(
    const unsigned char* in, const bf16raw* w, const bf16raw* bias,
    bf16raw* out, int batch, unsigned char* x_stash) {
  // R2D2's single-channel POMDP frames: CI=1 -> K=64; stage per-tap scalars
  conv_fwd_impl<unsigned char, 1, 32, 8, 8, 4, 84, 84, 20, 20>(
      in, w, bias, out, batch, x_stash);
}

extern "C" __global__ __launch_bounds__(256) void drla_conv_fwd_l2(
    const bf16raw* in, const bf16raw* w, const bf16raw* bias, bf16raw* out,
    int batch) {
  conv_fwd_impl<bf16raw, 32, 64, 4, 4, 2, 20, 20, 9, 9>(in, w, bias, out,
                                                        batch);
}

extern "C" __global__ __launch_bounds__(256) void drla_conv_fwd_l3(
    const bf16raw* in, const bf16raw* w, const bf16raw* bias, bf16raw* out,
    int batch) {
  conv_fwd_impl<bf16raw, 64, 64, 3, 3, 1, 9, 9, 7, 7>(in, w, bias, out,
                                                      batch);
}

// ---------------------------------------------------------------------------
// backward helpers
// ---------------------------------------------------------------------------

// dY_masked = dY * (Y > 0) with dbias partials FUSED into the same pass.
// Thread-to-column pinning: with blockDim (256) and the grid both multiples
// of CO (CO in {32,64}), a grid-stride walk keeps every thread on ONE co
// column — local accumulate, LDS-reduce per co, one atomicAdd per (block,
// co). 640 blocks adding the SAME [CO] addresses serialize (~10 us,
// r24/r25 — same pathology as sq_norm), so partials spread over 16 slots
// of a persistent [16][64] buffer; drla_wgrad_finalize (which always runs
// next in the conv backward) does the 16-way sum, emits bf16 dbias, and
// re-zeroes the slots.
// dy may be an N-strided view (a slice of the fused xh gradient):
// row_elems = elements per logical row, n_stride = elements between row
// starts; row_elems == n_stride means contiguous. Chunks never straddle
// rows (row_elems % 8 == 0).
extern "C" __global__ void drla_relu_mask_bwd(
    const bf16raw* __restrict__ dy, const bf16raw* __restrict__ y,
    bf16raw* __restrict__ out, float* __restrict__ dbias_slots,
    long long n, int CO, long long n_stride, long long row_elems) {
  // vectorized: each thread moves 8 bf16 per iteration (uint4), which pins
  // it to ONE 8-column group because stride*8 % CO == 0 (CO in {32,64},
  // blockDim 256). Per-thread acc[8] -> LDS -> one atomicAdd per (block,
  // column).
  const long long n8 = n / 8;
  const long long stride = gridDim.x * (long long)blockDim.x;
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const int tid = threadIdx.x;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const uint4* dy4 = reinterpret_cast<const uint4*>(dy);
  const uint4* y4 = reinterpret_cast<const uint4*>(y);
  uint4* out4 = reinterpret_cast<uint4*>(out);
  // named .x/.y/.z/.w only — a pointer into a local vector can force it
  // to scratch (guide §5.4 rule 20; cost measured on sq_norm_bf16)
#define DRLA_RM_LANE(dw, yw, ow, a0, a1)                              \
  {                                                                   \
    const bf16raw d0 = (bf16raw)((dw) & 0xFFFF);                      \
    const bf16raw d1 = (bf16raw)((dw) >> 16);                         \
    const bf16raw m0 =                                                \
        (cv_bf2f((bf16raw)((yw) & 0xFFFF)) > 0.0f) ? d0 : (bf16raw)0; \
    const bf16raw m1 =                                                \
        (cv_bf2f((bf16raw)((yw) >> 16)) > 0.0f) ? d1 : (bf16raw)0;    \
    (ow) = ((unsigned int)m1 << 16) | m0;                             \
    (a0) += cv_bf2f(m0);                                              \
    (a1) += cv_bf2f(m1);                                              \
  }
  const long long row8 = row_elems / 8;
  const bool strided = (n_stride != row_elems);
  for (; i < n8; i += stride) {
    long long si = i;
    if (strided) {
      const long long r = i / row8;
      si = (r * n_stride + (i - r * row8) * 8) / 8;
    }
    const uint4 dv = dy4[si];
    const uint4 yv = y4[i];
    uint4 ov;
    DRLA_RM_LANE(dv.x, yv.x, ov.x, acc[0], acc[1]);
    DRLA_RM_LANE(dv.y, yv.y, ov.y, acc[2], acc[3]);
    DRLA_RM_LANE(dv.z, yv.z, ov.z, acc[4], acc[5]);
    DRLA_RM_LANE(dv.w, yv.w, ov.w, acc[6], acc[7]);
    out4[i] = ov;
  }
#undef DRLA_RM_LANE
  __shared__ float red[DRLA_BLOCK * 8];
  for (int e = 0; e < 8; ++e) red[tid * 8 + e] = acc[e];
  __syncthreads();
  // column j gets acc[j%8] of every thread whose group (tid*8)%CO == j-j%8
  if (tid < CO) {
    const int g = tid / 8;
    const int e = tid % 8;
    // threads with in-block index t where (t*8)%CO == g*8, i.e.
    // t = g + k*(CO/8)
    float s = 0.0f;
    for (int t = g; t < DRLA_BLOCK; t += CO / 8) s += red[t * 8 + e];
    atomicAdd(&dbias_slots[(blockIdx.x & 15) * 64 + tid], s);
  }
}

// cast + transpose the f32 [K][CO] wgrad scratch into bf16 [CO][K]
// (the channels_last weight-grad layout), re-zeroing the persistent
// scratch as it reads (zero-between-calls invariant, no fill kernel).
extern "C" __global__ void drla_wgrad_finalize(
    float* __restrict__ scratch, bf16raw* __restrict__ dw, int K,
    int CO, float* __restrict__ dbias_slots,
    bf16raw* __restrict__ dbias) {
  // bias-grad epilogue: 16-way slot sum from relu_mask_bwd, bf16 out,
  // slots re-zeroed (persistent zero-between-calls buffer)
  if (dbias_slots && blockIdx.x == 0 && (int)threadIdx.x < CO) {
    float s = 0.0f;
    for (int t = 0; t < 16; ++t) {
      s += dbias_slots[t * 64 + threadIdx.x];
      dbias_slots[t * 64 + threadIdx.x] = 0.0f;
    }
    dbias[threadIdx.x] = drla_f32_to_bf16(s);
  }
  long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
  const long long total = (long long)K * CO;
  const long long stride = gridDim.x * (long long)blockDim.x;
  for (; i < total; i += stride) {
    const int co = i / K;
    const int k = i - (long long)co * K;
    const long long si = (long long)k * CO + co;
    dw[i] = drla_f32_to_bf16(scratch[si]);
    scratch[si] = 0.0f;
  }
}

// ---------------------------------------------------------------------------
// wgrad: dW^T[k][co] = sum_m A_im2col[m][k] * dY[m][co]
// grid: (K/64, S); each block owns 64 k-rows x CO cols and a slice of
// M; partials atomicAdd into the f32 [K][CO] scratch. (A no-atomic
// [S][K][CO] slab variant measured WORSE across the board — the S-deep
// finalize reduce is the new bottleneck — so atomics stay; the real cost
// is the per-chunk global-load latency chain, hence the prefetch below.)
// ---------------------------------------------------------------------------

template <typename IN_T, int CI, int CO, int KH, int KW, int STRIDE, int HI,
          int WI, int HO, int WO>
__device__ void conv_wgrad_impl(const IN_T* __restrict__ in,
                                const bf16raw* __restrict__ dy,  // [M][CO]
                                float* __restrict__ scratch,     // [K][CO]
                                int batch) {
  constexpr int K = KH * KW * CI;
  constexpr int BKM = 64;   // m-rows per chunk (2 MFMA k-steps)
  constexpr int BKK = 64;   // k-cols per block
  constexpr int PAD = 8;
  constexpr int NFRAG = CO / 16;
  constexpr int CO_PER_T = (CO * 32) / 256;  // B staging: co per thread/half
  const int M = batch * HO * WO;

  // Both images staged TRANSPOSED so the MFMA fragments are single
  // ds_read_b128s, and SOFTWARE-PIPELINED: chunk m+1's global loads issue
  // into registers while chunk m's MFMA runs — the per-chunk load-latency
  // chain (the measured bottleneck: time scaled with M/split, not with
  // atomics or LDS ops) overlaps compute.
  // DOUBLE-BUFFERED images: chunk t+1's LDS stores overlap chunk t's
  // MFMAs (different buffer), leaving ONE barrier per chunk instead of
  // two — with the register prefetch this removes the last serial
  // stage of the per-chunk latency chain.
  __shared__ bf16raw Am[2][BKK][BKM + PAD];   // A image: [k][m]
  __shared__ bf16raw BmT[2][CO][BKM + PAD];   // dY image: [co][m]

  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int k_row0 = blockIdx.x * BKK;

  f32x4 acc[NFRAG];
  for (int ni = 0; ni < NFRAG; ++ni) acc[ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_per_split = (M + gridDim.y - 1) / gridDim.y;
  const int m_begin = blockIdx.y * m_per_split;
  const int m_end = min(M, m_begin + m_per_split);

  // staging assignment: thread t -> A row t>>2 with k-chunk (t&3)*16; the
  // k-position decode is m-invariant, so tap offsets precompute once
  const int a_m = tid >> 2;
  const int a_k = (tid & 3) * 16;
  long long a_off0 = 0, a_off1 = 0;
  long long a_offc[16];
  if constexpr (CI == 4) {
    // the thread's 4 taps are pixel-consecutive within one kh row
    // (tap0 % 8 is 0 or 4), so one base offset covers the uint4 load
    const int kb = k_row0 + a_k;
    const int kh = kb / (KW * CI);
    a_off0 = (long long)kh * WI + (kb - kh * KW * CI) / CI;
  } else if constexpr (CI == 1) {
    // the contiguous uint row loads below read all 16 taps without a
    // per-tap range check — only sound when every k-block is fully
    // in-range (the 8x8x1 instantiation: K=64 fills one BKK block)
    static_assert(K % BKK == 0,
                  "CI==1 contiguous row staging requires K % BKK == 0");
#pragma unroll
    for (int t = 0; t < 16; ++t) {
      const int k2 = k_row0 + a_k + t;
      const int kh = k2 / KW;
      a_offc[t] = (k2 < K) ? (long long)kh * WI + (k2 - kh * KW) : -1;
    }
  } else {
    const int kk0 = k_row0 + a_k;
    const int kh0 = kk0 / (KW * CI);
    const int kwci0 = kk0 - kh0 * KW * CI;
    a_off0 = (kk0 < K)
        ? ((long long)kh0 * WI + kwci0 / CI) * CI + (kwci0 % CI) : -1;
    const int kk1 = kk0 + 8;
    const int kh1 = kk1 / (KW * CI);
    const int kwci1 = kk1 - kh1 * KW * CI;
    a_off1 = (kk1 < K)
        ? ((long long)kh1 * WI + kwci1 / CI) * CI + (kwci1 % CI) : -1;
  }
  const int b_lm0 = tid / (CO / CO_PER_T);
  const int b_co0 = (tid % (CO / CO_PER_T)) * CO_PER_T;

  // register prefetch state
  unsigned int ra4u_0, ra4u_1, ra4u_2, ra4u_3;
  unsigned int ra1u[4];
  bf16x8 rab0 = {0, 0, 0, 0, 0, 0, 0, 0}, rab1 = rab0;
  bf16x8 rb8_0 = rab0, rb8_1 = rab0;
  uint2 rb4_0 = {0, 0}, rb4_1 = {0, 0};
  bool a_live = false, b_live0 = false, b_live1 = false;

  auto load_chunk = [&](int m0) {
    const int m = m0 + a_m;
    a_live = (m < m_end);
    if (a_live) {
      const int n_idx = m / (HO * WO);
      const int rem = m - n_idx * (HO * WO);
      const int ho = rem / WO;
      const int wo = rem - ho * WO;
      const long long base =
          ((long long)n_idx * HI + ho * STRIDE) * WI + wo * STRIDE;
      if constexpr (CI == 4) {
        // taps 0..3 of this thread share kh with consecutive kw (see the
        // fwd note): one contiguous 16 B-aligned uint4 replaces the four
        // uchar4 gathers
        const uint4 packed = *reinterpret_cast<const uint4*>(
            in + (base + a_off0) * CI);
        ra4u_0 = packed.x; ra4u_1 = packed.y;
        ra4u_2 = packed.z; ra4u_3 = packed.w;
      } else if constexpr (CI == 1) {
        // two contiguous 8-byte kw rows (see the fwd note)
#pragma unroll
        for (int rrow = 0; rrow < 2; ++rrow) {
          const unsigned char* src = in + base + a_offc[rrow * 8];
          ra1u[rrow * 2 + 0] = *reinterpret_cast<const unsigned int*>(src);
          ra1u[rrow * 2 + 1] =
              *reinterpret_cast<const unsigned int*>(src + 4);
        }
      } else {
        const bf16raw* inb = reinterpret_cast<const bf16raw*>(in);
        if (a_off0 >= 0) {
          rab0 = *reinterpret_cast<const bf16x8*>(inb + base * CI + a_off0);
        }
        if (a_off1 >= 0) {
          rab1 = *reinterpret_cast<const bf16x8*>(inb + base * CI + a_off1);
        }
      }
    }
    const int mb0 = m0 + b_lm0;
    b_live0 = (mb0 < m_end);
    if (b_live0) {
      if constexpr (CO_PER_T == 8) {
        rb8_0 = *reinterpret_cast<const bf16x8*>(dy + (long long)mb0 * CO +
                                                 b_co0);
      } else {
        rb4_0 = *reinterpret_cast<const uint2*>(dy + (long long)mb0 * CO +
                                                b_co0);
      }
    }
    const int mb1 = m0 + b_lm0 + 32;
    b_live1 = (mb1 < m_end);
    if (b_live1) {
      if constexpr (CO_PER_T == 8) {
        rb8_1 = *reinterpret_cast<const bf16x8*>(dy + (long long)mb1 * CO +
                                                 b_co0);
      } else {
        rb4_1 = *reinterpret_cast<const uint2*>(dy + (long long)mb1 * CO +
                                                b_co0);
      }
    }
  };

  auto store_chunk = [&](int buf) {
    bf16raw (*AmB)[BKM + PAD] = Am[buf];
    bf16raw (*BmTB)[BKM + PAD] = BmT[buf];
    const int lm = a_m;
    if (a_live) {
      if constexpr (CI == 4) {
        const float sc = 1.0f / 255.0f;
#define DRLA_WG_PUT(st, u)                                              \
        {                                                               \
          const int o = a_k + ((st) >> 1) * 8 + ((st) & 1) * 4;         \
          AmB[o + 0][lm] = drla_f32_to_bf16((float)((u) & 0xFF) * sc);  \
          AmB[o + 1][lm] =                                              \
              drla_f32_to_bf16((float)(((u) >> 8) & 0xFF) * sc);        \
          AmB[o + 2][lm] =                                              \
              drla_f32_to_bf16((float)(((u) >> 16) & 0xFF) * sc);       \
          AmB[o + 3][lm] = drla_f32_to_bf16((float)((u) >> 24) * sc);   \
        }
        DRLA_WG_PUT(0, ra4u_0); DRLA_WG_PUT(1, ra4u_1);
        DRLA_WG_PUT(2, ra4u_2); DRLA_WG_PUT(3, ra4u_3);
#undef DRLA_WG_PUT
      } else if constexpr (CI == 1) {
        const float sc1 = 1.0f / 255.0f;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          const unsigned int u = ra1u[q];
          AmB[a_k + q * 4 + 0][lm] =
              drla_f32_to_bf16((float)(u & 0xFF) * sc1);
          AmB[a_k + q * 4 + 1][lm] =
              drla_f32_to_bf16((float)((u >> 8) & 0xFF) * sc1);
          AmB[a_k + q * 4 + 2][lm] =
              drla_f32_to_bf16((float)((u >> 16) & 0xFF) * sc1);
          AmB[a_k + q * 4 + 3][lm] =
              drla_f32_to_bf16((float)(u >> 24) * sc1);
        }
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          AmB[a_k + e][lm] = rab0[e];
          AmB[a_k + 8 + e][lm] = rab1[e];
        }
      }
    } else {
#pragma unroll
      for (int t = 0; t < 16; ++t) AmB[a_k + t][lm] = 0;
    }
    if constexpr (CO_PER_T == 8) {
      const bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      const bf16x8 v0 = b_live0 ? rb8_0 : z;
      const bf16x8 v1 = b_live1 ? rb8_1 : z;
      const int lm1 = b_lm0 + 32;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        BmTB[b_co0 + e][b_lm0] = v0[e];
        BmTB[b_co0 + e][lm1] = v1[e];
      }
    } else {
      const uint2 z = {0, 0};
      const uint2 v0 = b_live0 ? rb4_0 : z;
      const uint2 v1 = b_live1 ? rb4_1 : z;
      const int lm1 = b_lm0 + 32;
      BmTB[b_co0 + 0][b_lm0] = (bf16raw)(v0.x & 0xFFFF);
      BmTB[b_co0 + 1][b_lm0] = (bf16raw)(v0.x >> 16);
      BmTB[b_co0 + 2][b_lm0] = (bf16raw)(v0.y & 0xFFFF);
      BmTB[b_co0 + 3][b_lm0] = (bf16raw)(v0.y >> 16);
      BmTB[b_co0 + 0][lm1] = (bf16raw)(v1.x & 0xFFFF);
      BmTB[b_co0 + 1][lm1] = (bf16raw)(v1.x >> 16);
      BmTB[b_co0 + 2][lm1] = (bf16raw)(v1.y & 0xFFFF);
      BmTB[b_co0 + 3][lm1] = (bf16raw)(v1.y >> 16);
    }
  };

  // ping-pong: store chunk t+1 into the other buffer while chunk t's
  // MFMAs read this one; ONE barrier per chunk
  load_chunk(m_begin);
  store_chunk(0);
  load_chunk(m_begin + BKM);
  __syncthreads();
  int nchunks = 0;
  for (int m0 = m_begin; m0 < m_end; m0 += BKM) ++nchunks;
  for (int t = 0; t < nchunks; ++t) {
    const int cur = t & 1;
    if (t + 1 < nchunks) {
      store_chunk(1 - cur);
      load_chunk(m_begin + (long long)(t + 2) * BKM);
    }
    for (int kk = 0; kk < BKM; kk += 32) {
      const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
          &Am[cur][wave * 16 + (lane & 15)][kk + (lane >> 4) * 8]);
      for (int ni = 0; ni < NFRAG; ++ni) {
        const int co = ni * 16 + (lane & 15);
        const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            &BmT[cur][co][kk + (lane >> 4) * 8]);
        acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag,
                                                          acc[ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  for (int ni = 0; ni < NFRAG; ++ni) {
    const int co = ni * 16 + (lane & 15);
    for (int r = 0; r < 4; ++r) {
      const int k = k_row0 + wave * 16 + (lane >> 4) * 4 + r;
      if (k < K) atomicAdd(&scratch[(long long)k * CO + co], acc[ni][r]);
    }
  }
}

extern "C" __global__ __launch_bounds__(256) void drla_conv_wgrad_l1(
    const unsigned char* in, const bf16raw* dy, float* scratch, int batch) {
  conv_wgrad_impl<unsigned char, 4, 32, 8, 8, 4, 84, 84, 20, 20>(
      in, dy, scratch, batch);
}
extern "C" __global__ __launch_bounds__(256) void drla_conv_wgrad_l1_c1(
    const unsigned char* in, const bf16raw* dy, float* scratch, int batch) {
  conv_wgrad_impl<unsigned char, 1, 32, 8, 8, 4, 84, 84, 20, 20>(
      in, dy, scratch, batch);
}
extern "C" __global__ __launch_bounds__(256) void drla_conv_wgrad_l2(
    const bf16raw* in, const bf16raw* dy, float* scratch, int batch) {
  conv_wgrad_impl<bf16raw, 32, 64, 4, 4, 2, 20, 20, 9, 9>(in, dy,
                                                              scratch, batch);
}
extern "C" __global__ __launch_bounds__(256) void drla_conv_wgrad_l3(
    const bf16raw* in, const bf16raw* dy, float* scratch, int batch) {
  conv_wgrad_impl<bf16raw, 64, 64, 3, 3, 1, 9, 9, 7, 7>(in, dy, scratch,
                                                            batch);
}

// ---------------------------------------------------------------------------
// dgrad: dX[n,hi,wi,ci] = sum_{kh,kw,co valid} dY[n,ho',wo',co] *
//        W[co][(kh*KW+kw)*CI+ci],  ho' = (hi-kh)/STRIDE (when integral).
// Implicit GEMM: M2 = N*HI*WI rows, Kg = KH*KW*CO, CI cols.
// ---------------------------------------------------------------------------

template <int CI, int CO, int KH, int KW, int STRIDE, int HI, int WI, int HO,
          int WO>
__device__ void conv_dgrad_impl(const bf16raw* __restrict__ dy,  // [M][CO]
                                const bf16raw* __restrict__ w,   // [CO][K]
                                bf16raw* __restrict__ dx,        // [M2][CI]
                                int batch) {
  constexpr int K = KH * KW * CI;
  constexpr int Kg = KH * KW * CO;
  constexpr int BM = 128;
  constexpr int BK = 32;   // kg per step: one tap x 32 co
  constexpr int PAD = 8;
  constexpr int NFRAG = CI / 16;
  const int M2 = batch * HI * WI;

  __shared__ bf16raw Ad[BM][BK + PAD];    // dY gather: [m2][kg_local]
  __shared__ bf16raw Bd[CI][BK + PAD];    // W^T image: [ci][co_local]

  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int row0 = blockIdx.x * BM;

  f32x4 acc[2][NFRAG];
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < NFRAG; ++ni)
      acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // per-thread A staging: row = tid>>1, kg-half = (tid&1)*16
  const int a_row = tid >> 1;
  const int a_k0 = (tid & 1) * 16;
  const int m2 = row0 + a_row;
  const int n_idx = m2 / (HI * WI);
  const int rem = m2 - n_idx * (HI * WI);
  const int hi = rem / WI;
  const int wi = rem - hi * WI;

  // software-pipelined: next tap-chunk's globals prefetch into registers
  // during this chunk's MFMA (see wgrad note)
  uint4 rda0, rda1;
  bf16raw rbw[8];
  constexpr int CI_PER_T = (CI * BK) / 256;  // 4 (CI=32) or 8 (CI=64)
  const int b_col = tid % BK;
  const int b_ci0 = (tid / BK) * CI_PER_T;

  auto load_chunk = [&](int kg0) {
    const int tap = kg0 / CO;
    const int co0 = kg0 - tap * CO;
    const int kh = tap / KW;
    const int kw = tap - kh * KW;
    bool valid = (m2 < M2);
    int ho = 0, wo = 0;
    if (valid) {
      const int hh = hi - kh;
      const int ww = wi - kw;
      valid = hh >= 0 && ww >= 0 && (hh % STRIDE) == 0 && (ww % STRIDE) == 0;
      if (valid) {
        ho = hh / STRIDE;
        wo = ww / STRIDE;
        valid = ho < HO && wo < WO;
      }
    }
    if (valid) {
      const bf16raw* src =
          dy + ((long long)(n_idx * HO + ho) * WO + wo) * CO + co0 + a_k0;
      rda0 = *reinterpret_cast<const uint4*>(src);
      rda1 = *reinterpret_cast<const uint4*>(src + 8);
    } else {
      rda0 = uint4{0, 0, 0, 0};
      rda1 = uint4{0, 0, 0, 0};
    }
    const bf16raw* src = w + (long long)(co0 + b_col) * K + tap * CI + b_ci0;
#pragma unroll
    for (int e = 0; e < CI_PER_T; ++e) rbw[e] = src[e];
  };

  auto store_chunk = [&]() {
    *reinterpret_cast<uint4*>(&Ad[a_row][a_k0]) = rda0;
    *reinterpret_cast<uint4*>(&Ad[a_row][a_k0 + 8]) = rda1;
#pragma unroll
    for (int e = 0; e < CI_PER_T; ++e) Bd[b_ci0 + e][b_col] = rbw[e];
  };

  load_chunk(0);
  for (int kg0 = 0; kg0 < Kg; kg0 += BK) {
    store_chunk();
    __syncthreads();
    if (kg0 + BK < Kg) load_chunk(kg0 + BK);

    for (int mi = 0; mi < 2; ++mi) {
      const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
          &Ad[wave * 32 + mi * 16 + (lane & 15)][(lane >> 4) * 8]);
      for (int ni = 0; ni < NFRAG; ++ni) {
        const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            &Bd[ni * 16 + (lane & 15)][(lane >> 4) * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  for (int mi = 0; mi < 2; ++mi) {
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int ci = ni * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r) {
        const int grow = row0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (grow < M2) {
          dx[(long long)grow * CI + ci] = drla_f32_to_bf16(acc[mi][ni][r]);
        }
      }
    }
  }
}

// stride-2 dgrad, parity-class decomposition: for fixed (hi%2, wi%2) only
// taps with kh≡hi, kw≡wi (mod 2) can contribute, so 4 of the 16 taps are
// live — the generic kernel burns 3/4 of its k-loop staging zeros. Blocks
// are grouped by parity class (blockIdx.y) so the whole block walks just
// the 4 live taps: 8 BK=32 iterations instead of 32.
template <int CI, int CO, int KH, int KW, int HI, int WI, int HO, int WO>
__device__ void conv_dgrad_s2_impl(const bf16raw* __restrict__ dy,
                                   const bf16raw* __restrict__ w,
                                   bf16raw* __restrict__ dx, int batch) {
  constexpr int K = KH * KW * CI;
  constexpr int BM = 128;
  constexpr int BK = 32;
  constexpr int PAD = 8;
  constexpr int NFRAG = CI / 16;
  constexpr int H2 = HI / 2, W2 = WI / 2;   // per-class extent
  const int ph = blockIdx.y >> 1, pw = blockIdx.y & 1;
  const int Mc = batch * H2 * W2;           // rows in this class

  __shared__ bf16raw Ad[BM][BK + PAD];
  __shared__ bf16raw Bd[CI][BK + PAD];

  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int row0 = blockIdx.x * BM;

  f32x4 acc[2][NFRAG];
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < NFRAG; ++ni)
      acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int a_row = tid >> 1;
  const int a_k0 = (tid & 1) * 16;
  const int mc = row0 + a_row;
  const int n_idx = mc / (H2 * W2);
  const int rem = mc - n_idx * (H2 * W2);
  const int hi = (rem / W2) * 2 + ph;
  const int wi = (rem - (rem / W2) * W2) * 2 + pw;

  uint4 rda0, rda1;
  bf16raw rbw[8];
  constexpr int CI_PER_T = (CI * BK) / 256;
  const int b_col = tid % BK;
  const int b_ci0 = (tid / BK) * CI_PER_T;
  constexpr int NIT = 4 * (CO / BK);

  auto load_chunk = [&](int it) {
    const int t = it / (CO / BK);           // live tap 0..3
    const int co0 = (it - t * (CO / BK)) * BK;
    const int kh = ph + 2 * (t >> 1);
    const int kw = pw + 2 * (t & 1);
    bool valid = (mc < Mc);
    int ho = 0, wo = 0;
    if (valid) {
      const int hh = hi - kh;
      const int ww = wi - kw;
      valid = hh >= 0 && ww >= 0;
      if (valid) {
        ho = hh >> 1;
        wo = ww >> 1;
        valid = ho < HO && wo < WO;
      }
    }
    if (valid) {
      const bf16raw* src =
          dy + ((long long)(n_idx * HO + ho) * WO + wo) * CO + co0 + a_k0;
      rda0 = *reinterpret_cast<const uint4*>(src);
      rda1 = *reinterpret_cast<const uint4*>(src + 8);
    } else {
      rda0 = uint4{0, 0, 0, 0};
      rda1 = uint4{0, 0, 0, 0};
    }
    const int tap = kh * KW + kw;
    const bf16raw* src = w + (long long)(co0 + b_col) * K + tap * CI + b_ci0;
#pragma unroll
    for (int e = 0; e < CI_PER_T; ++e) rbw[e] = src[e];
  };

  auto store_chunk = [&]() {
    *reinterpret_cast<uint4*>(&Ad[a_row][a_k0]) = rda0;
    *reinterpret_cast<uint4*>(&Ad[a_row][a_k0 + 8]) = rda1;
#pragma unroll
    for (int e = 0; e < CI_PER_T; ++e) Bd[b_ci0 + e][b_col] = rbw[e];
  };

  load_chunk(0);
  for (int it = 0; it < NIT; ++it) {
    store_chunk();
    __syncthreads();
    if (it + 1 < NIT) load_chunk(it + 1);
    for (int mi = 0; mi < 2; ++mi) {
      const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
          &Ad[wave * 32 + mi * 16 + (lane & 15)][(lane >> 4) * 8]);
      for (int ni = 0; ni < NFRAG; ++ni) {
        const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            &Bd[ni * 16 + (lane & 15)][(lane >> 4) * 8]);
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  for (int mi = 0; mi < 2; ++mi) {
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int ci = ni * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r) {
        const int growc = row0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + r;
        if (growc < Mc) {
          const int gn = growc / (H2 * W2);
          const int grem = growc - gn * (H2 * W2);
          const int ghi = (grem / W2) * 2 + ph;
          const int gwi = (grem - (grem / W2) * W2) * 2 + pw;
          dx[(((long long)gn * HI + ghi) * WI + gwi) * CI + ci] =
              drla_f32_to_bf16(acc[mi][ni][r]);
        }
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(256) void drla_conv_dgrad_l2(
    const bf16raw* dy, const bf16raw* w, bf16raw* dx, int batch) {
  conv_dgrad_s2_impl<32, 64, 4, 4, 20, 20, 9, 9>(dy, w, dx, batch);
}
extern "C" __global__ __launch_bounds__(256) void drla_conv_dgrad_l3(
    const bf16raw* dy, const bf16raw* w, bf16raw* dx, int batch) {
  conv_dgrad_impl<64, 64, 3, 3, 1, 9, 9, 7, 7>(dy, w, dx, batch);
}
