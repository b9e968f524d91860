// K4 fused: BOTH IMPALA heads — policy (256->256->256->A)
// and value (256->256->256->1) — in ONE kernel each way.
//
// The heads are tiny (M = B*T = 640 rows, K = 256): every torch layer is a
// latency-bound ~5 us launch, and fwd+bwd spend ~200 us/step in ~30 small
// kernels (profiles r17). Rows are independent, so one block walks its 32
// rows through all six layers.
//
// Formulation: TRANSPOSED OUTPUT (D[out][row]) so both MFMA operands are
// vector loads —
//   A_op[m=out][k] = W[out][k]  (torch [out,in] rows, straight from global:
//                                each W element is used once per block, so
//                                LDS staging buys nothing — guide §5 "GEMV /
//                                decode weights" row)
//   B_op[k][col=row] = act[row][k]  (row-major LDS act image, vec8)
// The epilogue un-transposes with scalar LDS writes (once per layer, not
// per k-chunk). The backward dgrad chain uses the same shape on
// Pre-TRANSPOSED weights (built by the autograd wrapper: 4 small .t()
// copies), fusing ReLU masks and bias gradients; only the six
// MFMA-efficient wgrad GEMMs stay on hipBLASLt.

#include "drla_common.h"

typedef unsigned short bf16raw;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

#define MH_HID 256
#define MH_BM 16
#define MH_PAD 8
#define MH_LD (MH_HID + MH_PAD)

__device__ __forceinline__ float mh_b2f(bf16raw u) {
  unsigned int x = ((unsigned int)u) << 16;
  return __uint_as_float(x);
}

// One dense pass: out[r][o] = in[r][:] . M[o][:] (+bias, ReLU / mask),
// M [nout][kdim] row-major global, kdim and nout multiples of 16 (pad rows
// of M beyond its true extent must be readable — wrappers pad layer-3
// operands). act images [MH_BM][MH_LD] bf16.
//
// MASK_STASH != null: v *= (stash > 0), dz-global write + colsum bias
// partials (backward layers). BIAS != null: v += bias, RELU applies
// (forward layers). DH != null: terminal f32 row-grad output (ACCUM adds).
template <bool RELU, bool ACCUM>
__device__ void mh_pass(const bf16raw (*act_in)[MH_LD],
                        bf16raw (*act_out)[MH_LD],
                        const bf16raw* __restrict__ M,   // [nout][kdim]
                        const bf16raw* __restrict__ bias,  // [nout] or null
                        const bf16raw* __restrict__ mask_stash,  // [N,256]
                        bf16raw* __restrict__ dz_global,         // [N,256]
                        float* __restrict__ colsum,              // [256]
                        bf16raw* __restrict__ stash,     // [N,nout] or null
                        float* __restrict__ dh_global,   // [N,256] or null
                        int row0, int N, int nout, int kdim) {
  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int out0 = wave * 64;          // this wave's 64 output columns
  const bool live = out0 < nout;

  f32x4 acc[4];
  for (int ni = 0; ni < 4; ++ni) acc[ni] = {0.f, 0.f, 0.f, 0.f};

  const bf16x8 zero8 = {0, 0, 0, 0, 0, 0, 0, 0};
  if (live) {
    // software-pipelined k-loop: the next chunk's global weight fragments
    // load into registers while this chunk's MFMAs run (the serial
    // load->mfma chain across ~8 chunks x 3 chained passes is what sets
    // these kernels' time — 40-80 blocks can't hide it with occupancy)
    bf16x8 a_cur0, a_cur1, a_cur2, a_cur3;
    const int kf = (lane >> 4) * 8;
#define DRLA_MH_LOAD(ni, dst, k0)                                       \
    {                                                                   \
      const int orow = out0 + (ni) * 16 + (lane & 15);                  \
      dst = (orow < nout)                                               \
          ? *reinterpret_cast<const bf16x8*>(                           \
                M + (long long)orow * kdim + (k0) + kf)                 \
          : zero8;                                                      \
    }
    DRLA_MH_LOAD(0, a_cur0, 0); DRLA_MH_LOAD(1, a_cur1, 0);
    DRLA_MH_LOAD(2, a_cur2, 0); DRLA_MH_LOAD(3, a_cur3, 0);
    for (int k0 = 0; k0 < kdim; k0 += 32) {
      bf16x8 a_nxt0, a_nxt1, a_nxt2, a_nxt3;
      if (k0 + 32 < kdim) {
        DRLA_MH_LOAD(0, a_nxt0, k0 + 32); DRLA_MH_LOAD(1, a_nxt1, k0 + 32);
        DRLA_MH_LOAD(2, a_nxt2, k0 + 32); DRLA_MH_LOAD(3, a_nxt3, k0 + 32);
      }
      const bf16x8 b_frag =
          *reinterpret_cast<const bf16x8*>(&act_in[lane & 15][k0 + kf]);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_cur0, b_frag,
                                                       acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_cur1, b_frag,
                                                       acc[1], 0, 0, 0);
      acc[2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_cur2, b_frag,
                                                       acc[2], 0, 0, 0);
      acc[3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_cur3, b_frag,
                                                       acc[3], 0, 0, 0);
      a_cur0 = a_nxt0; a_cur1 = a_nxt1; a_cur2 = a_nxt2; a_cur3 = a_nxt3;
    }
#undef DRLA_MH_LOAD
  }
  __syncthreads();
  // epilogue: D[out][row] un-transpose -> act_out[row][out] (+stash/dz/dh);
  // per (ni, r) the lane touches ONE output column across its two rows, so
  // bias partials need one LDS atomic per (ni, r).
  if (live) {
    for (int ni = 0; ni < 4; ++ni) {
      for (int r = 0; r < 4; ++r) {
        const int oc = out0 + ni * 16 + (lane >> 4) * 4 + r;
        const int lrow = lane & 15;
        const int grow = row0 + lrow;
        float v = acc[ni][r];
        if (bias) v += (oc < nout) ? mh_b2f(bias[oc]) : 0.0f;
        if (RELU) v = fmaxf(v, 0.0f);
        if (mask_stash) {
          const float a = (grow < N)
              ? mh_b2f(mask_stash[(long long)grow * MH_HID + oc]) : 0.0f;
          v = (a > 0.0f) ? v : 0.0f;
        }
        if (dh_global) {
          // both head chains land here concurrently (blockIdx.y split):
          // accumulate atomically into the zeroed dh buffer
          if (grow < N && oc < nout) {
            atomicAdd(&dh_global[(long long)grow * MH_HID + oc], v);
          }
        } else {
          const bf16raw bv = drla_f32_to_bf16(v);
          act_out[lrow][oc] = bv;
          if (stash && grow < N && oc < nout) {
            stash[(long long)grow * nout + oc] = bv;
          }
          if (dz_global && grow < N && oc < nout) {
            dz_global[(long long)grow * MH_HID + oc] = bv;
          }
          if (colsum && oc < nout && grow < N) {
            atomicAdd(&colsum[oc], v);
          }
        }
      }
    }
  }
  __syncthreads();
}


extern "C" __global__ __launch_bounds__(256) void drla_mlp_heads_fwd(
    const float* __restrict__ h,        // [N,256]
    const bf16raw* __restrict__ W1p, const bf16raw* __restrict__ b1p,
    const bf16raw* __restrict__ W2p, const bf16raw* __restrict__ b2p,
    const bf16raw* __restrict__ W3p, const bf16raw* __restrict__ b3p,
    const bf16raw* __restrict__ W1v, const bf16raw* __restrict__ b1v,
    const bf16raw* __restrict__ W2v, const bf16raw* __restrict__ b2v,
    const bf16raw* __restrict__ W3v, const bf16raw* __restrict__ b3v,
    bf16raw* __restrict__ logits,       // [N,A]
    float* __restrict__ value,          // [N]
    bf16raw* __restrict__ stash,    // [N,5*256]: a1p,a2p,a1v,a2v,h_bf16
    int N, int A) {
  // grid: (ceil(N/16), 2) — blockIdx.y 0 = policy chain, 1 = value chain
  __shared__ bf16raw hb[MH_BM][MH_LD];
  __shared__ bf16raw acta[MH_BM][MH_LD];
  __shared__ bf16raw actb[MH_BM][MH_LD];

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;
  const bool policy = blockIdx.y == 0;
  const long long soff = (long long)N * MH_HID;

  // load h (f32 -> bf16); the policy block stashes the bf16 copy
  {
    const int r = tid >> 4;
    const int c0 = (tid & 15) * 16;
    const int grow = row0 + r;
    for (int c = 0; c < 16; ++c) {
      const float v = (grow < N) ? h[(long long)grow * MH_HID + c0 + c]
                                 : 0.0f;
      const bf16raw bv = drla_f32_to_bf16(v);
      hb[r][c0 + c] = bv;
      if (policy && grow < N) {
        stash[4 * soff + (long long)grow * MH_HID + c0 + c] = bv;
      }
    }
  }
  __syncthreads();

  if (policy) {
    mh_pass<true, false>(hb, acta, W1p, b1p, nullptr, nullptr, nullptr,
                         stash, nullptr, row0, N, MH_HID, MH_HID);
    mh_pass<true, false>(acta, actb, W2p, b2p, nullptr, nullptr, nullptr,
                         stash + soff, nullptr, row0, N, MH_HID, MH_HID);
    mh_pass<false, false>(actb, acta, W3p, b3p, nullptr, nullptr, nullptr,
                          nullptr, nullptr, row0, N, A, MH_HID);
    const int r = tid >> 3;
    const int c = tid & 7;
    if (r < MH_BM) {
      for (int cc = c; cc < A; cc += 8) {
        if (row0 + r < N) {
          logits[(long long)(row0 + r) * A + cc] = acta[r][cc];
        }
      }
    }
  } else {
    mh_pass<true, false>(hb, acta, W1v, b1v, nullptr, nullptr, nullptr,
                         stash + 2 * soff, nullptr, row0, N, MH_HID,
                         MH_HID);
    mh_pass<true, false>(acta, actb, W2v, b2v, nullptr, nullptr, nullptr,
                         stash + 3 * soff, nullptr, row0, N, MH_HID,
                         MH_HID);
    mh_pass<false, false>(actb, acta, W3v, b3v, nullptr, nullptr, nullptr,
                          nullptr, nullptr, row0, N, 1, MH_HID);
    if (tid < MH_BM && row0 + tid < N) {
      value[row0 + tid] = mh_b2f(acta[tid][0]);
    }
  }
}

// backward: takes PRE-TRANSPOSED weights (wrapper-built):
//   wT3p [256,32] = pad(W3p^T), wT2p/wT1p/wT2v/wT1v [256,256] = W^T,
//   W3v [1,256] as-is (value out layer backward is an outer product).
extern "C" __global__ __launch_bounds__(256) void drla_mlp_heads_bwd(
    const bf16raw* __restrict__ dlogits,   // [N,A]
    const float* __restrict__ dvalue,      // [N]
    const bf16raw* __restrict__ stash,     // [N,5*256]
    const bf16raw* __restrict__ wT1p, const bf16raw* __restrict__ wT2p,
    const bf16raw* __restrict__ wT3p,      // [256,32] padded
    const bf16raw* __restrict__ wT1v, const bf16raw* __restrict__ wT2v,
    const bf16raw* __restrict__ W3v,       // [1,256]
    bf16raw* __restrict__ dz1p, bf16raw* __restrict__ dz2p,  // [N,256]
    bf16raw* __restrict__ dz1v, bf16raw* __restrict__ dz2v,  // [N,256]
    float* __restrict__ dh,                // [N,256], ZEROED (atomics)
    float* __restrict__ db1p, float* __restrict__ db2p,
    float* __restrict__ db3p,              // [A]
    float* __restrict__ db1v, float* __restrict__ db2v,
    float* __restrict__ db3v,              // [1]
    int N, int A) {
  // grid: (ceil(N/16), 2) — blockIdx.y 0 = policy chain, 1 = value chain;
  // both accumulate dh atomically.
  __shared__ bf16raw dza[MH_BM][MH_LD];
  __shared__ bf16raw dzb[MH_BM][MH_LD];
  __shared__ float colsum[MH_HID];

  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;
  const long long soff = (long long)N * MH_HID;
  const bf16raw* a1p = stash;
  const bf16raw* a2p = stash + soff;
  const bf16raw* a1v = stash + 2 * soff;
  const bf16raw* a2v = stash + 3 * soff;

  if (tid < MH_HID) colsum[tid] = 0.0f;

  if (blockIdx.y == 0) {
    // ---- policy chain ----
    {
      const int r = tid >> 4;
      const int c = tid & 15;
      for (int cc = c; cc < 32; cc += 16) {
        const int grow = row0 + r;
        bf16raw v = 0;
        if (cc < A && grow < N) v = dlogits[(long long)grow * A + cc];
        dza[r][cc] = v;
      }
    }
    __syncthreads();
    if (tid < A) {
      float s = 0.0f;
      for (int r = 0; r < MH_BM; ++r) {
        if (row0 + r < N) s += mh_b2f(dza[r][tid]);
      }
      atomicAdd(&db3p[tid], s);
    }
    __syncthreads();
    mh_pass<false, false>(dza, dzb, wT3p, nullptr, a2p, dz2p, colsum,
                          nullptr, nullptr, row0, N, MH_HID, 32);
    if (tid < MH_HID) {
      atomicAdd(&db2p[tid], colsum[tid]);
      colsum[tid] = 0.0f;
    }
    __syncthreads();
    mh_pass<false, false>(dzb, dza, wT2p, nullptr, a1p, dz1p, colsum,
                          nullptr, nullptr, row0, N, MH_HID, MH_HID);
    if (tid < MH_HID) atomicAdd(&db1p[tid], colsum[tid]);
    __syncthreads();
    mh_pass<false, false>(dza, dzb, wT1p, nullptr, nullptr, nullptr,
                          nullptr, nullptr, dh, row0, N, MH_HID, MH_HID);
  } else {
    // ---- value chain ----
    // da2v[r][i] = dvalue[r] * W3v[i], masked by a2v
    {
      const int r = tid >> 4;
      const int c0 = (tid & 15) * 16;
      const int grow = row0 + r;
      const float dv = (grow < N) ? dvalue[grow] : 0.0f;
      for (int c = 0; c < 16; ++c) {
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
    if (tid == 0) {
      float s = 0.0f;
      for (int r = 0; r < MH_BM && row0 + r < N; ++r) s += dvalue[row0 + r];
      atomicAdd(&db3v[0], s);
    }
    __syncthreads();
    mh_pass<false, false>(dza, dzb, wT2v, nullptr, a1v, dz1v, colsum,
                          nullptr, nullptr, row0, N, MH_HID, MH_HID);
    if (tid < MH_HID) atomicAdd(&db1v[tid], colsum[tid]);
    __syncthreads();
    mh_pass<false, true>(dzb, dza, wT1v, nullptr, nullptr, nullptr,
                         nullptr, nullptr, dh, row0, N, MH_HID, MH_HID);
  }
}

// ---------------------------------------------------------------------------
// Weight-transpose pack: the dgrad kernel above wants W^T operands, which
// the wrapper used to build with 5 .t().contiguous() copies + an F.pad —
// ~7 eager launches per step at the ~4.5 us small-kernel floor (profile
// r23). One kernel writes the whole packed buffer instead.
// Layout (bf16 elements): wT1p@0, wT2p@65536, wT3p@131072 ([256,32],
// cols >= A zero), wT1v@139264, wT2v@204800, W3v copy @270336; total 270592.
extern "C" __global__ void drla_heads_wt_pack(
    const bf16raw* __restrict__ w1p, const bf16raw* __restrict__ w2p,
    const bf16raw* __restrict__ w3p, const bf16raw* __restrict__ w1v,
    const bf16raw* __restrict__ w2v, const bf16raw* __restrict__ w3v,
    bf16raw* __restrict__ out, int A) {
  const int total = 4 * 65536 + 256 * 32 + 256;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    bf16raw v;
    if (i < 65536) {
      v = w1p[(i & 255) * 256 + (i >> 8)];
    } else if (i < 131072) {
      const int j = i - 65536;
      v = w2p[(j & 255) * 256 + (j >> 8)];
    } else if (i < 139264) {
      const int j = i - 131072;
      const int c = j & 31;
      v = (c < A) ? w3p[c * 256 + (j >> 5)] : (bf16raw)0;
    } else if (i < 204800) {
      const int j = i - 139264;
      v = w1v[(j & 255) * 256 + (j >> 8)];
    } else if (i < 270336) {
      const int j = i - 204800;
      v = w2v[(j & 255) * 256 + (j >> 8)];
    } else {
      v = w3v[i - 270336];
    }
    out[i] = v;
  }
}

// ---------------------------------------------------------------------------
// All six head wgrads in ONE launch: dW = dz^T @ act with K = N rows.
// Replaces 6 hipBLASLt GEMM dispatches (~5/step at ~13 us each, r23).
//
// Jobs (all outputs [M,256] bf16):
//   0: dW1p = dz1p^T @ hb     M=256      3: dW2v = dz2v^T @ a1v  M=256
//   1: dW2p = dz2p^T @ a1p    M=256      4: dW3p = dlogits^T @ a2p  M=A
//   2: dW1v = dz1v^T @ hb     M=256      5: dW3v = dvalue^T @ a2v   M=1
//
// Both MFMA operands are k-major in global (dz[k][m], act[k][n]), so each
// 32k-chunk is staged TRANSPOSED into padded LDS (vec8 global row loads,
// scalar LDS writes) and the fragments read back as vec8 ds_read_b128
// (row stride 40 elems = 80 B keeps 16 B alignment).
#define HW_LDK 40  // 32 + 8 pad

extern "C" __global__ __launch_bounds__(256) void drla_heads_wgrad(
    const bf16raw* __restrict__ dz1p, const bf16raw* __restrict__ dz2p,
    const bf16raw* __restrict__ dz1v, const bf16raw* __restrict__ dz2v,
    const bf16raw* __restrict__ dlogits,  // [N,A]
    const float* __restrict__ dvalue,     // [N]
    const bf16raw* __restrict__ stash,    // [N,5*256]
    bf16raw* __restrict__ dw1p, bf16raw* __restrict__ dw2p,
    bf16raw* __restrict__ dw3p,           // [A,256]
    bf16raw* __restrict__ dw1v, bf16raw* __restrict__ dw2v,
    bf16raw* __restrict__ dw3v,           // [1,256]
    const float* __restrict__ ws_tail,    // f32 bias partials from heads_bwd
    bf16raw* __restrict__ db1p, bf16raw* __restrict__ db2p,
    bf16raw* __restrict__ db3p, bf16raw* __restrict__ db1v,
    bf16raw* __restrict__ db2v, bf16raw* __restrict__ db3v,
    int N, int A) {
  // grid.x = 73: jobs 0-3 are 4x4 tiles of 64x64 (blocks 0..63), job 4 is
  // blocks 64..67 (one m-tile, A <= 32), job 5 blocks 68..71; block 72
  // converts the f32 bias-grad partials to six contiguous bf16 tensors
  // (contiguous outputs avoid AccumulateGrad's clone-a-view kernels).
  const int bid = blockIdx.x;
  if (bid == 72) {
    const int tid = threadIdx.x;
    const int total = 4 * 256 + A + 1;
    for (int i = tid; i < total; i += 256) {
      const bf16raw v = drla_f32_to_bf16(ws_tail[i]);
      if (i < 256)            db1p[i] = v;
      else if (i < 512)       db2p[i - 256] = v;
      else if (i < 512 + A)   db3p[i - 512] = v;
      else if (i < 768 + A)   db1v[i - 512 - A] = v;
      else if (i < 1024 + A)  db2v[i - 768 - A] = v;
      else                    db3v[0] = v;
    }
    return;
  }
  int job, mt, nt;
  if (bid < 64) {
    job = bid >> 4;
    mt = (bid >> 2) & 3;
    nt = bid & 3;
  } else if (bid < 68) {
    job = 4; mt = 0; nt = bid - 64;
  } else {
    job = 5; mt = 0; nt = bid - 68;
  }
  const long long soff = (long long)N * MH_HID;
  const bf16raw* dz;
  const bf16raw* act;
  bf16raw* out;
  int M;
  switch (job) {
    case 0: dz = dz1p; act = stash + 4 * soff; out = dw1p; M = 256; break;
    case 1: dz = dz2p; act = stash;            out = dw2p; M = 256; break;
    case 2: dz = dz1v; act = stash + 4 * soff; out = dw1v; M = 256; break;
    case 3: dz = dz2v; act = stash + 2 * soff; out = dw2v; M = 256; break;
    case 4: dz = dlogits; act = stash + soff;  out = dw3p; M = A;   break;
    default: dz = nullptr; act = stash + 3 * soff; out = dw3v; M = 1;
  }
  const int m0 = mt * 64;
  const int n0 = nt * 64;

  __shared__ bf16raw dzT[64][HW_LDK];   // [m_local][k_local]
  __shared__ bf16raw actT[64][HW_LDK];  // [n_local][k_local]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  f32x4 acc[4];
  for (int ni = 0; ni < 4; ++ni) acc[ni] = {0.f, 0.f, 0.f, 0.f};

  // software-pipelined K(=N)-loop: next chunk's globals prefetch into
  // registers during this chunk's MFMA (same pattern as the conv wgrad)
  const int st_kk = tid >> 3;
  const int st_c0 = (tid & 7) * 8;
  const bf16x8 z8 = {0, 0, 0, 0, 0, 0, 0, 0};
  bf16x8 r_act = z8, r_dz = z8;

  auto load_chunk = [&](int k0) {
    const int k = k0 + st_kk;
    r_act = z8;
    if (k < N) {
      r_act = *reinterpret_cast<const bf16x8*>(
          act + (long long)k * MH_HID + n0 + st_c0);
    }
    if (job < 4) {           // dz [N,256] bf16, vec8
      r_dz = z8;
      if (k < N) {
        r_dz = *reinterpret_cast<const bf16x8*>(
            dz + (long long)k * MH_HID + m0 + st_c0);
      }
    } else if (job == 4) {   // dlogits [N,A], scalar predicated
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int m = st_c0 + e;
        r_dz[e] = (k < N && m < A)
            ? (short)dlogits[(long long)k * A + m] : (short)0;
      }
    } else {                 // dvalue [N] f32, M = 1
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int m = st_c0 + e;
        r_dz[e] = (k < N && m == 0)
            ? (short)drla_f32_to_bf16(dvalue[k]) : (short)0;
      }
    }
  };

  auto store_chunk = [&]() {
    actT[st_c0 + 0][st_kk] = r_act[0]; actT[st_c0 + 1][st_kk] = r_act[1];
    actT[st_c0 + 2][st_kk] = r_act[2]; actT[st_c0 + 3][st_kk] = r_act[3];
    actT[st_c0 + 4][st_kk] = r_act[4]; actT[st_c0 + 5][st_kk] = r_act[5];
    actT[st_c0 + 6][st_kk] = r_act[6]; actT[st_c0 + 7][st_kk] = r_act[7];
    dzT[st_c0 + 0][st_kk] = r_dz[0]; dzT[st_c0 + 1][st_kk] = r_dz[1];
    dzT[st_c0 + 2][st_kk] = r_dz[2]; dzT[st_c0 + 3][st_kk] = r_dz[3];
    dzT[st_c0 + 4][st_kk] = r_dz[4]; dzT[st_c0 + 5][st_kk] = r_dz[5];
    dzT[st_c0 + 6][st_kk] = r_dz[6]; dzT[st_c0 + 7][st_kk] = r_dz[7];
  };

  load_chunk(0);
  for (int k0 = 0; k0 < N; k0 += 32) {
    store_chunk();
    __syncthreads();
    if (k0 + 32 < N) load_chunk(k0 + 32);
    // wave w: m rows [w*16, w*16+16); frags vec8 from LDS
    const bf16x8 a_frag = *reinterpret_cast<const bf16x8*>(
        &dzT[wave * 16 + (lane & 15)][(lane >> 4) * 8]);
    for (int ni = 0; ni < 4; ++ni) {
      const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
          &actT[ni * 16 + (lane & 15)][(lane >> 4) * 8]);
      acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_frag, b_frag, acc[ni], 0, 0, 0);
    }
    __syncthreads();
  }
  // D: lane l reg r -> row (l>>4)*4+r (m), col l&15 (n)
  for (int ni = 0; ni < 4; ++ni) {
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wave * 16 + (lane >> 4) * 4 + r;
      const int n = n0 + ni * 16 + (lane & 15);
      if (m < M) out[(long long)m * MH_HID + n] = drla_f32_to_bf16(acc[ni][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// K2 fused action-embedding MLP: one-hot(prev_action) -> 256 (+bias, ReLU)
// -> 256 (+bias, ReLU), reference model/impala_actor_critic.py:12-16.
// The torch composition (lookup, add, relu, linear, relu + their backwards,
// two bias column-reduces and an embedding scatter) costs ~70 us/step in
// ~12 launches at [640]; these kernels do it in 1 fwd + 3 bwd launches.
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void drla_embed_mlp_fwd(
    const long long* __restrict__ pa,     // [N]
    const bf16raw* __restrict__ table,    // [A,256]
    const bf16raw* __restrict__ b1,       // [256]
    const bf16raw* __restrict__ W2,       // [256,256] (nn.Linear [out,in])
    const bf16raw* __restrict__ b2,       // [256]
    bf16raw* __restrict__ out,            // [N,256] (post-ReLU2)
    bf16raw* __restrict__ a1stash,        // [N,256] (post-ReLU1)
    int N, int A) {
  __shared__ bf16raw a1img[MH_BM][MH_LD];
  __shared__ bf16raw outimg[MH_BM][MH_LD];
  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;
  {
    const int r = tid >> 4;
    const int c0 = (tid & 15) * 16;
    const int grow = row0 + r;
    // clamp: pa arrives from external actor data — an out-of-range
    // index must not become an aperture fault (drla_common.h)
    const long long idx =
        (grow < N) ? drla_clamp_idx((int)pa[grow], A) : 0;
    const bf16raw* trow = table + idx * MH_HID + c0;
    for (int c = 0; c < 16; ++c) {
      const float v =
          fmaxf(mh_b2f(trow[c]) + mh_b2f(b1[c0 + c]), 0.0f);
      const bf16raw bv = drla_f32_to_bf16(v);
      a1img[r][c0 + c] = bv;
      if (grow < N) a1stash[(long long)grow * MH_HID + c0 + c] = bv;
    }
  }
  __syncthreads();
  mh_pass<true, false>(a1img, outimg, W2, b2, nullptr, nullptr, nullptr,
                       out, nullptr, row0, N, MH_HID, MH_HID);
}

// [256,256] transpose pack for the dgrad pass below
extern "C" __global__ void drla_embed_w2t_pack(
    const bf16raw* __restrict__ w2, bf16raw* __restrict__ out) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < 65536;
       i += gridDim.x * blockDim.x) {
    out[i] = w2[(i & 255) * 256 + (i >> 8)];
  }
}

// backward: dz2 = dy * (out>0) (written for the dW2 GEMM), db2 = colsum,
// da1 = (dz2 @ W2) * (a1>0) (written for the table scatter), db1 = colsum.
// dy may be N-strided (a slice of the fused xh gradient). bias_ws: f32
// [2*256] ZERO on entry (atomic partials; db1 at [0:256], db2 at [256:]).
extern "C" __global__ __launch_bounds__(256) void drla_embed_mlp_bwd(
    const bf16raw* __restrict__ dy, long long dy_stride,
    const bf16raw* __restrict__ out, const bf16raw* __restrict__ a1stash,
    const bf16raw* __restrict__ W2T,  // [256,256] pre-transposed
    bf16raw* __restrict__ dz2,        // [N,256]
    bf16raw* __restrict__ da1,        // [N,256]
    float* __restrict__ bias_ws, int N) {
  __shared__ bf16raw dz2img[MH_BM][MH_LD];
  __shared__ bf16raw da1img[MH_BM][MH_LD];
  __shared__ float colsum[MH_HID];
  const int tid = threadIdx.x;
  const int row0 = blockIdx.x * MH_BM;
  if (tid < MH_HID) colsum[tid] = 0.0f;
  __syncthreads();
  {
    const int r = tid >> 4;
    const int c0 = (tid & 15) * 16;
    const int grow = row0 + r;
    float part[16];
    for (int c = 0; c < 16; ++c) {
      float v = 0.0f;
      if (grow < N) {
        const float o =
            mh_b2f(out[(long long)grow * MH_HID + c0 + c]);
        if (o > 0.0f) {
          v = mh_b2f(dy[(long long)grow * dy_stride + c0 + c]);
        }
      }
      const bf16raw bv = drla_f32_to_bf16(v);
      dz2img[r][c0 + c] = bv;
      if (grow < N) dz2[(long long)grow * MH_HID + c0 + c] = bv;
      part[c] = v;
    }
    for (int c = 0; c < 16; ++c) atomicAdd(&colsum[c0 + c], part[c]);
  }
  __syncthreads();
  if (tid < MH_HID) {
    atomicAdd(&bias_ws[MH_HID + tid], colsum[tid]);
    colsum[tid] = 0.0f;
  }
  __syncthreads();
  // dgrad through W2 with the ReLU1 mask and db1 column sums fused
  mh_pass<false, false>(dz2img, da1img, W2T, nullptr, a1stash, da1,
                        colsum, nullptr, nullptr, row0, N, MH_HID, MH_HID);
  if (tid < MH_HID) atomicAdd(&bias_ws[tid], colsum[tid]);
}

// finalize for the fused embed backward: cast the persistent f32 scratch
// ([A*256] table-grad partials ++ [512] bias partials) to three contiguous
// bf16 outputs, re-zeroing the scratch (zero-between-calls invariant).
extern "C" __global__ void drla_embed_finalize(
    float* __restrict__ src, bf16raw* __restrict__ dtable,
    bf16raw* __restrict__ db1, bf16raw* __restrict__ db2,
    long long n_table) {
  const long long total = n_table + 512;
  for (long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       i < total; i += gridDim.x * (long long)blockDim.x) {
    const bf16raw v = drla_f32_to_bf16(src[i]);
    src[i] = 0.0f;
    if (i < n_table) dtable[i] = v;
    else if (i < n_table + 256) db1[i - n_table] = v;
    else db2[i - n_table - 256] = v;
  }
}
