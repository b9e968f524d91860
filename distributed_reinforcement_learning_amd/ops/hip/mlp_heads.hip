// K4 fused: BOTH IMPALA heads — policy (256->256->256->A)
// and value (256->256->256->1) — in ONE kernel each way.
//
// The heads are tiny (M = B*T = 640 rows, K = 256): every torch layer is a
// latency-bound ~5 us launch, and fwd+bwd spend ~200 us/step in ~30 small
// kernels (GEMMs, ReLU, bias grads, casts — profiles r17). Rows are
// independent, so one block walks its 32 rows through all six layers with
// weights streamed from L2 into LDS k-chunks; backward fuses the dgrad
// chain + ReLU masks + bias partials, leaving only the six efficient
// [256 x M x 256] wgrad GEMMs to hipBLASLt.
//
// Layouts: torch Linear weight is [out, in] row-major; act/dz images in LDS
// are [32 row][256 col] bf16 with +8 padding. MFMA 16x16x32 bf16, fragment
// maps as conv.hip.

#include "drla_common.h"

typedef unsigned short bf16raw;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

#define MH_HID 256
#define MH_BM 32
#define MH_PAD 8
#define MH_LD (MH_HID + MH_PAD)

__device__ __forceinline__ float mh_b2f(bf16raw u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __uint_as_float(x);
}

// One dense layer: act_out[r][o] = act_in[r][:] . W[o][:] + b[o], optional
// ReLU; nout <= 256. All 4 waves stage; waves whose 64-col quarter is
// beyond nout skip compute. act images [MH_BM][MH_LD].
template <bool RELU>
__device__ void mh_layer(const bf16raw (*act_in)[MH_LD],
                         bf16raw (*act_out)[MH_LD],
                         const bf16raw* __restrict__ W,   // [nout][256]
                         const float* __restrict__ bias,  // [nout]
                         bf16raw (*wbuf)[MH_LD],          // [32][MH_LD]
                         bf16raw* __restrict__ stash,     // [N,256] or null
                         int row0, int N, int nout) {
  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int col0 = wave * 64;
  const bool live = col0 < nout;

  f32x4 acc[2][4];
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < 4; ++ni)
      acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < MH_HID; k0 += 32) {
    // stage W k-chunk as wbuf[k][out-col]: W rows are in-dim contiguous, so
    // thread t loads W[out = t>>1][k0 + (t&1)*16 ..+16] (2 x vec8) for
    // out-cols t>>1 and t>>1 + 128
    for (int half = 0; half < 2; ++half) {
      const int oc = (tid >> 1) + half * 128;
      const int ks = (tid & 1) * 16;
      if (oc < nout) {
        const bf16raw* src = W + (long long)oc * MH_HID + k0 + ks;
        // transposed write: wbuf[k][oc]
        for (int e = 0; e < 16; ++e) {
          wbuf[ks + e][oc] = src[e];
        }
      } else {
        for (int e = 0; e < 16; ++e) wbuf[ks + e][oc] = 0;
      }
    }
    __syncthreads();
    if (live) {
      for (int mi = 0; mi < 2; ++mi) {
        const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
            &act_in[mi * 16 + (lane & 15)][k0 + (lane >> 4) * 8]);
        for (int ni = 0; ni < 4; ++ni) {
          bf16x8 b_frag;
          const int oc = col0 + ni * 16 + (lane & 15);
          for (int e = 0; e < 8; ++e) {
            b_frag[e] = (short)wbuf[(lane >> 4) * 8 + e][oc];
          }
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[mi][ni], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }
  // epilogue: bias (+ReLU), write act_out + stash
  if (live) {
    for (int ni = 0; ni < 4; ++ni) {
      const int oc = col0 + ni * 16 + (lane & 15);
      const float b = (oc < nout) ? bias[oc] : 0.0f;
      for (int mi = 0; mi < 2; ++mi) {
        for (int r = 0; r < 4; ++r) {
          const int lrow = mi * 16 + (lane >> 4) * 4 + r;
          float v = acc[mi][ni][r] + b;
          if (RELU) v = fmaxf(v, 0.0f);
          const bf16raw bv = drla_f32_to_bf16(v);
          act_out[lrow][oc] = bv;
          if (stash && row0 + lrow < N && oc < nout) {
            stash[(long long)(row0 + lrow) * nout + oc] = bv;
          }
        }
      }
    }
  }
  __syncthreads();
}

extern "C" __global__ __launch_bounds__(256) void drla_mlp_heads_fwd(
    const float* __restrict__ h,        // [N,256]
    const bf16raw* __restrict__ W1p, const float* __restrict__ b1p,
    const bf16raw* __restrict__ W2p, const float* __restrict__ b2p,
    const bf16raw* __restrict__ W3p, const float* __restrict__ b3p,
    const bf16raw* __restrict__ W1v, const float* __restrict__ b1v,
    const bf16raw* __restrict__ W2v, const float* __restrict__ b2v,
    const bf16raw* __restrict__ W3v, const float* __restrict__ b3v,
    bf16raw* __restrict__ logits,       // [N,A]
    float* __restrict__ value,          // [N]
    bf16raw* __restrict__ stash,    // [N,5*256]: a1p,a2p,a1v,a2v,h_bf16
    int N, int A) {
  __shared__ bf16raw hb[MH_BM][MH_LD];
  __shared__ bf16raw acta[MH_BM][MH_LD];
  __shared__ bf16raw actb[MH_BM][MH_LD];
  __shared__ bf16raw wbuf[32][MH_LD];

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;

  // load h (f32 -> bf16): thread t loads row t>>3, cols (t&7)*32..+32
  {
    const int r = tid >> 3;
    const int c0 = (tid & 7) * 32;
    const int grow = row0 + r;
    for (int c = 0; c < 32; ++c) {
      const float v = (grow < N) ? h[(long long)grow * MH_HID + c0 + c]
                                 : 0.0f;
      const bf16raw bv = drla_f32_to_bf16(v);
      hb[r][c0 + c] = bv;
      if (grow < N) {
        stash[4 * (long long)N * MH_HID + (long long)grow * MH_HID
              + c0 + c] = bv;
      }
    }
  }
  __syncthreads();

  const long long soff = (long long)N * MH_HID;
  // policy head
  mh_layer<true>(hb, acta, W1p, b1p, wbuf, stash, row0, N, MH_HID);
  mh_layer<true>(acta, actb, W2p, b2p, wbuf, stash + soff, row0, N, MH_HID);
  mh_layer<false>(actb, acta, W3p, b3p, wbuf, nullptr, row0, N, A);
  // write logits from acta[:, :A]
  {
    const int r = tid >> 3;
    const int c = tid & 7;  // A <= 32: two col passes of 8? A<=32 -> 4 each
    for (int cc = c; cc < A; cc += 8) {
      if (row0 + r < N) {
        logits[(long long)(row0 + r) * A + cc] = acta[r][cc];
      }
    }
  }
  __syncthreads();
  // value head
  mh_layer<true>(hb, acta, W1v, b1v, wbuf, stash + 2 * soff, row0, N,
                 MH_HID);
  mh_layer<true>(acta, actb, W2v, b2v, wbuf, stash + 3 * soff, row0, N,
                 MH_HID);
  mh_layer<false>(actb, acta, W3v, b3v, wbuf, nullptr, row0, N, 1);
  {
    const int r = tid;
    if (r < MH_BM && row0 + r < N) {
      value[row0 + r] = mh_b2f(acta[r][0]);
    }
  }
}

// ---------------------------------------------------------------------------
// backward: fused dgrad chain + ReLU masks + bias partials for both heads.
// Leaves dz1..dz2 per head in global memory for the (efficient) hipBLASLt
// wgrad GEMMs; dh accumulates both heads.
// ---------------------------------------------------------------------------

// da_out = dz_in @ W  (A_op = dz [m][k=o], B_op = W [o][i], K = kdim),
// then epilogue: v = da * mask(astash > 0) (if MASKED) -> dz_next LDS +
// global + bias partials into colsum LDS; else v -> dh f32 global (ADD if
// ACCUM).
template <bool MASKED, bool ACCUM>
__device__ void mh_dgrad_layer(const bf16raw (*dz_in)[MH_LD],
                               bf16raw (*dz_out)[MH_LD],
                               const bf16raw* __restrict__ W,  // [kdim][256]
                               bf16raw (*wbuf)[MH_LD],
                               const bf16raw* __restrict__ astash,  // [N,256]
                               bf16raw* __restrict__ dz_global,     // [N,256]
                               float* __restrict__ dh_global,       // [N,256]
                               float* __restrict__ colsum,          // [256]
                               int row0, int N, int kdim) {
  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int col0 = wave * 64;

  f32x4 acc[2][4];
  for (int mi = 0; mi < 2; ++mi)
    for (int ni = 0; ni < 4; ++ni)
      acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < kdim; k0 += 32) {
    // stage W[k0..k0+32)[all 256] directly (rows contiguous): thread t ->
    // o-row t>>3, col segment (t&7)*32..+32
    {
      const int o = (tid >> 3);
      const int c0 = (tid & 7) * 32;
      if (k0 + o < kdim) {
        const bf16raw* src = W + (long long)(k0 + o) * MH_HID + c0;
        for (int e = 0; e < 32; e += 8) {
          *reinterpret_cast<uint4*>(&wbuf[o][c0 + e]) =
              *reinterpret_cast<const uint4*>(src + e);
        }
      } else {
        for (int e = 0; e < 32; e += 8) {
          *reinterpret_cast<uint4*>(&wbuf[o][c0 + e]) = uint4{0, 0, 0, 0};
        }
      }
    }
    __syncthreads();
    for (int mi = 0; mi < 2; ++mi) {
      const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
          &dz_in[mi * 16 + (lane & 15)][k0 + (lane >> 4) * 8]);
      for (int ni = 0; ni < 4; ++ni) {
        bf16x8 b_frag;
        const int ic = col0 + ni * 16 + (lane & 15);
        for (int e = 0; e < 8; ++e) {
          b_frag[e] = (short)wbuf[(lane >> 4) * 8 + e][ic];
        }
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  // epilogue
  for (int ni = 0; ni < 4; ++ni) {
    const int ic = col0 + ni * 16 + (lane & 15);
    float csum = 0.0f;
    for (int mi = 0; mi < 2; ++mi) {
      for (int r = 0; r < 4; ++r) {
        const int lrow = mi * 16 + (lane >> 4) * 4 + r;
        const int grow = row0 + lrow;
        float v = acc[mi][ni][r];
        if (MASKED) {
          const float a = (grow < N)
              ? mh_b2f(astash[(long long)grow * MH_HID + ic]) : 0.0f;
          v = (a > 0.0f) ? v : 0.0f;
          const bf16raw bv = drla_f32_to_bf16(v);
          dz_out[lrow][ic] = bv;
          if (grow < N) dz_global[(long long)grow * MH_HID + ic] = bv;
          csum += v;
        } else if (grow < N) {
          if (ACCUM) {
            dh_global[(long long)grow * MH_HID + ic] += v;
          } else {
            dh_global[(long long)grow * MH_HID + ic] = v;
          }
        }
      }
    }
    if (MASKED) atomicAdd(&colsum[ic], csum);
  }
  __syncthreads();
}

extern "C" __global__ __launch_bounds__(256) void drla_mlp_heads_bwd(
    const bf16raw* __restrict__ dlogits,   // [N,A]
    const float* __restrict__ dvalue,      // [N]
    const bf16raw* __restrict__ stash,     // [N,4*256]
    const bf16raw* __restrict__ W1p, const bf16raw* __restrict__ W2p,
    const bf16raw* __restrict__ W3p,       // [A,256]
    const bf16raw* __restrict__ W1v, const bf16raw* __restrict__ W2v,
    const bf16raw* __restrict__ W3v,       // [1,256]
    bf16raw* __restrict__ dz1p, bf16raw* __restrict__ dz2p,  // [N,256]
    bf16raw* __restrict__ dz1v, bf16raw* __restrict__ dz2v,  // [N,256]
    float* __restrict__ dh,                // [N,256]
    float* __restrict__ db1p, float* __restrict__ db2p,
    float* __restrict__ db3p,              // [A]
    float* __restrict__ db1v, float* __restrict__ db2v,
    float* __restrict__ db3v,              // [1]
    int N, int A) {
  __shared__ bf16raw dza[MH_BM][MH_LD];
  __shared__ bf16raw dzb[MH_BM][MH_LD];
  __shared__ bf16raw wbuf[32][MH_LD];
  __shared__ float colsum[MH_HID];

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;
  const long long soff = (long long)N * MH_HID;
  const bf16raw* a1p = stash;
  const bf16raw* a2p = stash + soff;
  const bf16raw* a1v = stash + 2 * soff;
  const bf16raw* a2v = stash + 3 * soff;

  // ---- policy chain ----
  // stage dz3p (zero-padded to 32 cols) + its bias partials
  {
    const int r = tid >> 3;
    const int c = tid & 7;
    for (int cc = c; cc < 32; cc += 8) {
      const int grow = row0 + r;
      bf16raw v = 0;
      if (cc < A && grow < N) v = dlogits[(long long)grow * A + cc];
      dza[r][cc] = v;
    }
  }
  if (tid < MH_HID) colsum[tid] = 0.0f;
  __syncthreads();
  // db3p partials: one thread per col (< A), sum over the block's rows
  if (tid < A) {
    float s = 0.0f;
    for (int r = 0; r < MH_BM; ++r) s += mh_b2f(dza[r][tid]);
    atomicAdd(&db3p[tid], s);
  }
  __syncthreads();
  // kdim = A (not 32): rows >= A of W3p do not exist — the stage must
  // zero-fill them, or garbage (possibly NaN bits) meets the zero-padded
  // dz and 0 * NaN = NaN
  mh_dgrad_layer<true, false>(dza, dzb, W3p, wbuf, a2p, dz2p, nullptr,
                              colsum, row0, N, A);
  // flush db2p
  if (tid < MH_HID) {
    atomicAdd(&db2p[tid], colsum[tid]);
    colsum[tid] = 0.0f;
  }
  __syncthreads();
  mh_dgrad_layer<true, false>(dzb, dza, W2p, wbuf, a1p, dz1p, nullptr,
                              colsum, row0, N, MH_HID);
  if (tid < MH_HID) {
    atomicAdd(&db1p[tid], colsum[tid]);
    colsum[tid] = 0.0f;
  }
  __syncthreads();
  mh_dgrad_layer<false, false>(dza, dzb, W1p, wbuf, nullptr, nullptr, dh,
                               colsum, row0, N, MH_HID);

  // ---- value chain ----
  // da2v[r][i] = dvalue[r] * W3v[0][i], masked by a2v -> dz2v
  {
    const int r = tid >> 3;
    const int c0 = (tid & 7) * 32;
    const int grow = row0 + r;
    const float dv = (grow < N) ? dvalue[grow] : 0.0f;
    for (int c = 0; c < 32; ++c) {
      const int ic = c0 + c;
      float v = dv * mh_b2f(W3v[ic]);
      if (grow < N) {
        const float a = mh_b2f(a2v[(long long)grow * MH_HID + ic]);
        v = (a > 0.0f) ? v : 0.0f;
      } else {
        v = 0.0f;
      }
      const bf16raw bv = drla_f32_to_bf16(v);
      dza[r][ic] = bv;
      if (grow < N) dz2v[(long long)grow * MH_HID + ic] = bv;
      atomicAdd(&colsum[ic], v);
    }
  }
  __syncthreads();
  if (tid < MH_HID) {
    atomicAdd(&db2v[tid], colsum[tid]);
    colsum[tid] = 0.0f;
  }
  // db3v = sum dvalue over block rows
  if (tid == 0) {
    float s = 0.0f;
    for (int r = 0; r < MH_BM && row0 + r < N; ++r) s += dvalue[row0 + r];
    atomicAdd(&db3v[0], s);
  }
  __syncthreads();
  mh_dgrad_layer<true, false>(dza, dzb, W2v, wbuf, a1v, dz1v, nullptr,
                              colsum, row0, N, MH_HID);
  if (tid < MH_HID) {
    atomicAdd(&db1v[tid], colsum[tid]);
  }
  __syncthreads();
  mh_dgrad_layer<false, true>(dzb, dza, W1v, wbuf, nullptr, nullptr, dh,
                              colsum, row0, N, MH_HID);
}
