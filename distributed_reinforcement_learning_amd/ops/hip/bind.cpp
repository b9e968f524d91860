// PyTorch bindings for the drla gfx950 kernels (native HIP — no CUDA
// masquerade: ATen/hip API, hipLaunchKernelGGL on the current HIP stream).
// Compiled by hipcc together with the .hip translation units (see
// build_ext.py); loaded as distributed_reinforcement_learning_amd.ops._drla_hip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <tuple>

#include "drla_common.h"

// kernel decls (defined in the .hip TUs)
extern "C" __global__ void drla_u8_normalize_f32(const uchar4*, float4*,
                                                 long long);
typedef __attribute__((ext_vector_type(4))) unsigned short ushort4v;
extern "C" __global__ void drla_u8_normalize_bf16(const uchar4*, ushort4v*,
                                                  long long);
extern "C" __global__ void drla_u8_normalize_f32_tail(const unsigned char*,
                                                      float*, long long,
                                                      long long);
extern "C" __global__ void drla_u8_normalize_bf16_tail(const unsigned char*,
                                                       unsigned short*,
                                                       long long, long long);
// conv stack (conv.hip)
extern "C" __global__ void drla_mfma_probe(const unsigned short*,
                                           const unsigned short*, float*);
extern "C" __global__ void drla_conv_fwd_l1(const unsigned char*,
                                            const unsigned short*,
                                            const unsigned short*,
                                            unsigned short*, int,
                                            unsigned char*);
extern "C" __global__ void drla_conv_fwd_l1_c1(const unsigned char*,
                                               const unsigned short*,
                                               const unsigned short*,
                                               unsigned short*, int,
                                               unsigned char*);
extern "C" __global__ void drla_conv_fwd_l2(const unsigned short*,
                                            const unsigned short*,
                                            const unsigned short*,
                                            unsigned short*, int);
extern "C" __global__ void drla_conv_fwd_l3(const unsigned short*,
                                            const unsigned short*,
                                            const unsigned short*,
                                            unsigned short*, int);
extern "C" __global__ void drla_relu_mask_bwd(const unsigned short*,
                                              const unsigned short*,
                                              unsigned short*, float*,
                                              long long, int, long long,
                                              long long);
extern "C" __global__ void drla_wgrad_finalize(float*, unsigned short*,
                                               int, int, float*,
                                               unsigned short*);
extern "C" __global__ void drla_conv_wgrad_l1(const unsigned char*,
                                              const unsigned short*, float*,
                                              int);
extern "C" __global__ void drla_conv_wgrad_l1_c1(const unsigned char*,
                                                 const unsigned short*,
                                                 float*, int);
extern "C" __global__ void drla_conv_wgrad_l2(const unsigned short*,
                                              const unsigned short*, float*,
                                              int);
extern "C" __global__ void drla_conv_wgrad_l3(const unsigned short*,
                                              const unsigned short*, float*,
                                              int);
extern "C" __global__ void drla_conv_dgrad_l2(const unsigned short*,
                                              const unsigned short*,
                                              unsigned short*, int);
extern "C" __global__ void drla_conv_dgrad_l3(const unsigned short*,
                                              const unsigned short*,
                                              unsigned short*, int);
extern "C" __global__ void drla_dqn_loss_fwd(
    const unsigned short*, const float*, const float*, const float*,
    const int*, const float*, const float*, const float*, float*, float*,
    int, int);
extern "C" __global__ void drla_dqn_loss_bwd(
    const float*, const int*, const float*, const float*, unsigned short*,
    float*, int, int);
extern "C" __global__ void drla_a2c_loss_fwd(
    const unsigned short*, const float*, const float*, const float*,
    const int*, const float*, const unsigned char*, float, int, float,
    float, float*, float*, float*, int, int);
extern "C" __global__ void drla_a2c_loss_bwd(
    const float*, const float*, const int*, const float*, int, float,
    float, unsigned short*, float*, float*, int, int);
extern "C" __global__ void drla_r2d2_loss_fwd(
    const unsigned short*, const float*, const unsigned short*,
    const float*, const int*, const float*, const unsigned char*,
    const float*, float, int, float*, float*, float*, int, int, int);
extern "C" __global__ void drla_dueling_head_fwd(
    const float*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, unsigned short*, int,
    int, int, int, int, int);
extern "C" __global__ void drla_dhead_train_fwd(
    const float*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, unsigned short*,
    unsigned short*, int, int, int, int);
extern "C" __global__ void drla_dhead_train_bwd(
    const unsigned short*, const unsigned short*, const float*,
    const unsigned short*, const unsigned short*, float*, float*, int,
    int, int, int);
extern "C" __global__ void drla_dhead_train_fin(
    const float*, unsigned short*, unsigned short*, unsigned short*,
    unsigned short*, int, int, int);
extern "C" __global__ void drla_r2d2_loss_bwd(
    const float*, const int*, const float*, const float*, unsigned short*,
    float*, int, int, int);
extern "C" __global__ void drla_per_update(float*, const long long*,
                                           const float*, int, long long);
extern "C" __global__ void drla_per_sample(const float*, const float*,
                                           const float*, long long*,
                                           float*, int, long long);
extern "C" __global__ void drla_per_rebuild_level(float*, long long,
                                                  long long);
extern "C" __global__ void drla_multi_gather(
    const long long*, const unsigned long long*, const unsigned long long*,
    const long long*, int);
extern "C" __global__ void drla_embed_bwd_scatter(
    const long long*, const unsigned short*, const float*, float*,
    long long, int, long long);
extern "C" __global__ void drla_f32_to_bf16_zero_kernel(float*,
                                                        unsigned short*,
                                                        long long);

extern "C" __global__ void drla_vtrace_scan(const float*, const float*,
                                            const float*, float*, int, int);
extern "C" __global__ void drla_vtrace_loss_fwd(
    const unsigned short*, const float*, const float*, const float*,
    const int*, const float*, const unsigned char*, float, int, float,
    float, float*, float*, float*, float*, int, int, int);
extern "C" __global__ void drla_vtrace_loss_bwd(
    const float*, const float*, const float*, const float*, const int*,
    const float*, int, float, float, unsigned short*, float*, float*, int,
    int, int);
extern "C" __global__ void drla_lstm_tail_fwd(const float*, const float*,
                                              float*, float*, float*, float,
                                              long long, int);
extern "C" __global__ void drla_lstm_tail_bwd(const float*, const float*,
                                              const float*, const float*,
                                              const float*, float*, float*,
                                              long long, int, long long);
extern "C" __global__ void drla_lstm_tail_fwd_bf16(
    const unsigned short*, const float*, float*, float*, float*, float,
    long long, int);
extern "C" __global__ void drla_lstm_tail_bwd_bf16(
    const float*, const float*, const float*, const float*, const float*,
    unsigned short*, float*, long long, int, long long);
extern "C" __global__ void drla_mlp_heads_fwd(
    const float*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, unsigned short*, float*, unsigned short*, int,
    int);
extern "C" __global__ void drla_mlp_heads_bwd(
    const unsigned short*, const float*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    unsigned short*, unsigned short*, unsigned short*, unsigned short*,
    float*, float*, float*, float*, float*, float*, float*, int, int);
extern "C" __global__ void drla_heads_wt_pack(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const unsigned short*,
    unsigned short*, int);
extern "C" __global__ void drla_heads_wgrad(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, const float*,
    const unsigned short*, unsigned short*, unsigned short*, unsigned short*,
    unsigned short*, unsigned short*, unsigned short*, const float*,
    unsigned short*, unsigned short*, unsigned short*, unsigned short*,
    unsigned short*, unsigned short*, int, int);
extern "C" __global__ void drla_embed_mlp_fwd(
    const long long*, const unsigned short*, const unsigned short*,
    const unsigned short*, const unsigned short*, unsigned short*,
    unsigned short*, int, int);
extern "C" __global__ void drla_embed_w2t_pack(const unsigned short*,
                                               unsigned short*);
extern "C" __global__ void drla_embed_mlp_bwd(
    const unsigned short*, long long, const unsigned short*,
    const unsigned short*, const unsigned short*, unsigned short*,
    unsigned short*, float*, int);
extern "C" __global__ void drla_embed_finalize(float*, unsigned short*,
                                               unsigned short*,
                                               unsigned short*, long long);
extern "C" __global__ void drla_grad_gather(
    const unsigned long long*, const long long*, const long long*,
    unsigned short*, int, long long, float*, int);
extern "C" __global__ void drla_lstm_seq_train_fwd(
    const unsigned short*, const unsigned short*, const float*, const float*,
    const unsigned char*, float*, float*, float*, float*, float*,
    unsigned short*, float, int, int, int);
extern "C" __global__ void drla_lstm_seq_train_bwd(
    const float*, const float*, const float*, const float*, const float*,
    const unsigned short*, const unsigned char*, unsigned short*, float*,
    float*, int, int, int);
extern "C" __global__ void drla_lstm_seq_fwd(
    const unsigned short*, const float*, const unsigned short*, const float*,
    const float*, const unsigned char*, float*, float*, float*, float, int,
    int, int);
extern "C" __global__ void drla_sq_norm(const float*, float*, long long);
extern "C" __global__ void drla_sq_norm_bf16(const unsigned short*, float*,
                                             long long);
extern "C" __global__ void drla_rmsprop_step_bf16(
    unsigned short*, const unsigned short*, float*, float*, const float*,
    float, const float*, float, float, long long);
extern "C" __global__ void drla_adam_step_bf16(
    unsigned short*, const unsigned short*, float*, float*, float*,
    const float*, float, const float*, float, float, float, long long);
extern "C" __global__ void drla_rmsprop_step(float*, const float*, float*,
                                             const float*, float,
                                             const float*, float, float,
                                             long long);
extern "C" __global__ void drla_adam_step(float*, const float*, float*,
                                          float*, const float*, float,
                                          const float*, float, float, float,
                                          long long);

namespace {

hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check_gpu_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

torch::Tensor normalize_frames_f32(torch::Tensor frames) {
  check_gpu_contig(frames, "frames");
  TORCH_CHECK(frames.scalar_type() == torch::kUInt8, "frames must be uint8");
  auto out = torch::empty_like(frames, frames.options().dtype(torch::kFloat));
  const long long n = frames.numel();
  const long long n4 = n / 4;
  if (n4 > 0) {
    hipLaunchKernelGGL(drla_u8_normalize_f32, dim3(drla_grid(n4)),
                       dim3(DRLA_BLOCK), 0, cur_stream(),
                       reinterpret_cast<const uchar4*>(frames.data_ptr<uint8_t>()),
                       reinterpret_cast<float4*>(out.data_ptr<float>()), n4);
  }
  if (n % 4) {
    hipLaunchKernelGGL(drla_u8_normalize_f32_tail, dim3(1), dim3(DRLA_BLOCK),
                       0, cur_stream(), frames.data_ptr<uint8_t>(),
                       out.data_ptr<float>(), n4 * 4, n);
  }
  return out;
}

torch::Tensor normalize_frames_bf16(torch::Tensor frames) {
  check_gpu_contig(frames, "frames");
  TORCH_CHECK(frames.scalar_type() == torch::kUInt8, "frames must be uint8");
  auto out = torch::empty_like(frames,
                               frames.options().dtype(torch::kBFloat16));
  const long long n = frames.numel();
  const long long n4 = n / 4;
  if (n4 > 0) {
    hipLaunchKernelGGL(drla_u8_normalize_bf16, dim3(drla_grid(n4)),
                       dim3(DRLA_BLOCK), 0, cur_stream(),
                       reinterpret_cast<const uchar4*>(frames.data_ptr<uint8_t>()),
                       reinterpret_cast<ushort4v*>(out.data_ptr()), n4);
  }
  if (n % 4) {
    hipLaunchKernelGGL(drla_u8_normalize_bf16_tail, dim3(1), dim3(DRLA_BLOCK),
                       0, cur_stream(), frames.data_ptr<uint8_t>(),
                       reinterpret_cast<unsigned short*>(out.data_ptr()),
                       n4 * 4, n);
  }
  return out;
}

torch::Tensor vtrace_scan(torch::Tensor deltas, torch::Tensor discounts,
                          torch::Tensor cs) {
  for (auto* t : {&deltas, &discounts, &cs}) {
    check_gpu_contig(*t, "vtrace input");
    TORCH_CHECK(t->scalar_type() == torch::kFloat, "vtrace wants float32");
  }
  const int B = deltas.size(0);
  const int T = deltas.size(1);
  auto out = torch::empty_like(deltas);
  hipLaunchKernelGGL(drla_vtrace_scan, dim3(drla_grid(B)), dim3(DRLA_BLOCK),
                     0, cur_stream(), deltas.data_ptr<float>(),
                     discounts.data_ptr<float>(), cs.data_ptr<float>(),
                     out.data_ptr<float>(), B, T);
  return out;
}

// ---- conv stack ----------------------------------------------------------
namespace convcfg {
struct Layer { int ci, co, kh, kw, stride, hi, wi, ho, wo; };
static const Layer L[4] = {
    {4, 32, 8, 8, 4, 84, 84, 20, 20},   // 0: l1 (u8 x4)
    {1, 32, 8, 8, 4, 84, 84, 20, 20},   // 1: l1_c1 (u8 x1)
    {32, 64, 4, 4, 2, 20, 20, 9, 9},    // 2: l2
    {64, 64, 3, 3, 1, 9, 9, 7, 7},      // 3: l3
};
}  // namespace convcfg

static const unsigned short* u16p(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
static unsigned short* u16pm(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  check_gpu_contig(A, "A");
  check_gpu_contig(B, "B");
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(drla_mfma_probe, dim3(1), dim3(64), 0, cur_stream(),
                     u16p(A), u16p(B), D.data_ptr<float>());
  return D;
}

std::tuple<torch::Tensor, torch::Tensor> conv_fwd(
    int layer, torch::Tensor in, torch::Tensor w, torch::Tensor bias,
    bool stash_input) {
  check_gpu_contig(in, "in");
  check_gpu_contig(w, "w");
  check_gpu_contig(bias, "bias");
  const auto& cfg = convcfg::L[layer];
  TORCH_CHECK(w.scalar_type() == torch::kBFloat16, "weights must be bf16");
  TORCH_CHECK(bias.scalar_type() == torch::kBFloat16, "bias must be bf16");
  const int batch = in.size(0);
  const int M = batch * cfg.ho * cfg.wo;
  auto out = torch::empty({batch, cfg.ho, cfg.wo, cfg.co},
                          in.options().dtype(torch::kBFloat16));
  const int grid = (M + 127) / 128;
  auto* outp = u16pm(out);
  // stashed-input mode (u8 layers only): the kernel bundles a
  // pass-through copy of the input so the backward never re-reads the
  // (possibly upload-overlapped) source buffer
  torch::Tensor x_stash;
  unsigned char* stashp = nullptr;
  if (stash_input) {
    TORCH_CHECK(layer <= 1, "input stash is a u8-layer feature");
    TORCH_CHECK((in.numel() % 16) == 0, "stash wants 16-divisible bytes");
    x_stash = torch::empty_like(in);
    stashp = x_stash.data_ptr<uint8_t>();
  }
  switch (layer) {
    case 0:
      hipLaunchKernelGGL(drla_conv_fwd_l1, dim3(grid), dim3(256), 0,
                         cur_stream(), in.data_ptr<uint8_t>(), u16p(w),
                         u16p(bias), outp, batch, stashp);
      break;
    case 1:
      hipLaunchKernelGGL(drla_conv_fwd_l1_c1, dim3(grid), dim3(256), 0,
                         cur_stream(), in.data_ptr<uint8_t>(), u16p(w),
                         u16p(bias), outp, batch, stashp);
      break;
    case 2:
      hipLaunchKernelGGL(drla_conv_fwd_l2, dim3(grid), dim3(256), 0,
                         cur_stream(), u16p(in), u16p(w),
                         u16p(bias), outp, batch);
      break;
    case 3:
      hipLaunchKernelGGL(drla_conv_fwd_l3, dim3(grid), dim3(256), 0,
                         cur_stream(), u16p(in), u16p(w),
                         u16p(bias), outp, batch);
      break;
    default:
      TORCH_CHECK(false, "bad layer");
  }
  return {out, x_stash};
}

// persistent [16][64] bias-grad partial slots shared by relu_mask_bwd
// (atomic producers, compute stream) and wgrad_finalize (sum + re-zero
// consumer — which may run on the SIDE wgrad stream, see ops/conv_op.py),
// so each conv layer owns its own slot set: a later layer's mask must not
// touch slots a pending finalize still reads.
static torch::Tensor& dbias_slot_buf(const torch::Tensor& like,
                                     int64_t slot_id) {
  static torch::Tensor bufs[4];
  TORCH_CHECK(slot_id >= 0 && slot_id < 4, "bad slot id");
  if (!bufs[slot_id].defined()) {
    bufs[slot_id] =
        torch::zeros({16 * 64}, like.options().dtype(torch::kFloat));
  }
  return bufs[slot_id];
}

torch::Tensor relu_mask_bwd(torch::Tensor dy, torch::Tensor y, int64_t CO,
                            int64_t slot_id) {
  check_gpu_contig(y, "y");
  TORCH_CHECK(dy.is_cuda(), "dy must be on GPU");
  // dy may be an N-strided view (slice of the fused xh grad): inner dims
  // contiguous, only stride(0) loose
  long long n_stride, row_elems;
  if (dy.is_contiguous()) {
    n_stride = row_elems = dy.numel() / dy.size(0);
  } else {
    row_elems = 1;
    for (int d = 1; d < dy.dim(); ++d) {
      TORCH_CHECK(dy.stride(d - 1) >= dy.stride(d), "dy must be row-major");
      row_elems *= dy.size(d);
    }
    TORCH_CHECK(dy.stride(dy.dim() - 1) == 1 &&
                    dy.stride(0) >= row_elems && row_elems % 8 == 0,
                "dy view must be inner-contiguous");
    for (int d = 1; d + 1 < dy.dim(); ++d) {
      TORCH_CHECK(dy.stride(d) == dy.size(d + 1) * dy.stride(d + 1),
                  "dy inner dims must be contiguous");
    }
    n_stride = dy.stride(0);
  }
  auto out = torch::empty(dy.sizes(), dy.options());
  auto& slots = dbias_slot_buf(dy, slot_id);
  const long long n = dy.numel();
  TORCH_CHECK(n % 8 == 0, "relu_mask_bwd wants numel % 8 == 0");
  int grid = drla_grid(n / 8);
  if (grid > 640) grid = 640;
  hipLaunchKernelGGL(drla_relu_mask_bwd, dim3(grid), dim3(DRLA_BLOCK),
                     0, cur_stream(),
                     reinterpret_cast<const unsigned short*>(dy.data_ptr()),
                     u16p(y), u16pm(out), slots.data_ptr<float>(), n,
                     (int)CO, n_stride, row_elems);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> conv_wgrad(int layer,
                                                    torch::Tensor in,
                                                    torch::Tensor dy) {
  check_gpu_contig(in, "in");
  check_gpu_contig(dy, "dy");
  const auto& cfg = convcfg::L[layer];
  const int batch = in.size(0);
  const int K = cfg.kh * cfg.kw * cfg.ci;
  // runtime-tunable M-split (DRLA_WGRAD_SPLIT sweeps it; defaults from the
  // on-box sweeps — atomic-free slab stores make high splits cheap)
  static int split_env = [] {
    const char* e = getenv("DRLA_WGRAD_SPLIT");
    return e ? atoi(e) : 0;
  }();
  const int split = split_env ? split_env : ((layer <= 1) ? 512 : 64);
  // persistent zero-between-calls scratch (finalize re-zeroes on read)
  static torch::Tensor scratch_cache[4];
  if (!scratch_cache[layer].defined()) {
    scratch_cache[layer] =
        torch::zeros({K, cfg.co}, dy.options().dtype(torch::kFloat));
  }
  auto scratch = scratch_cache[layer];
  dim3 grid((K + 63) / 64, split);
  switch (layer) {
    case 0:
      hipLaunchKernelGGL(drla_conv_wgrad_l1, grid, dim3(256), 0,
                         cur_stream(), in.data_ptr<uint8_t>(), u16p(dy),
                         scratch.data_ptr<float>(), batch);
      break;
    case 1:
      hipLaunchKernelGGL(drla_conv_wgrad_l1_c1, grid, dim3(256), 0,
                         cur_stream(), in.data_ptr<uint8_t>(), u16p(dy),
                         scratch.data_ptr<float>(), batch);
      break;
    case 2:
      hipLaunchKernelGGL(drla_conv_wgrad_l2, grid, dim3(256), 0,
                         cur_stream(), u16p(in), u16p(dy),
                         scratch.data_ptr<float>(), batch);
      break;
    case 3:
      hipLaunchKernelGGL(drla_conv_wgrad_l3, grid, dim3(256), 0,
                         cur_stream(), u16p(in), u16p(dy),
                         scratch.data_ptr<float>(), batch);
      break;
    default:
      TORCH_CHECK(false, "bad layer");
  }
  auto dw = torch::empty({cfg.co, K}, dy.options().dtype(torch::kBFloat16));
  auto dbias = torch::empty({cfg.co}, dy.options().dtype(torch::kBFloat16));
  auto& slots = dbias_slot_buf(dy, layer);
  hipLaunchKernelGGL(drla_wgrad_finalize,
                     dim3(drla_grid((long long)K * cfg.co)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     scratch.data_ptr<float>(), u16pm(dw), K, cfg.co,
                     slots.data_ptr<float>(), u16pm(dbias));
  return {dw, dbias};
}

torch::Tensor conv_dgrad(int layer, torch::Tensor dy, torch::Tensor w) {
  check_gpu_contig(dy, "dy");
  check_gpu_contig(w, "w");
  TORCH_CHECK(layer == 2 || layer == 3, "dgrad only for l2/l3");
  const auto& cfg = convcfg::L[layer];
  const int batch = dy.size(0);
  const int M2 = batch * cfg.hi * cfg.wi;
  auto dx = torch::empty({batch, cfg.hi, cfg.wi, cfg.ci},
                         dy.options().dtype(torch::kBFloat16));
  const int grid = (M2 + 127) / 128;
  if (layer == 2) {
    // parity-class kernel: blockIdx.y = (hi%2, wi%2) class, each class
    // covers batch * (hi/2) * (wi/2) rows
    const int grid_c = (batch * (cfg.hi / 2) * (cfg.wi / 2) + 127) / 128;
    hipLaunchKernelGGL(drla_conv_dgrad_l2, dim3(grid_c, 4), dim3(256), 0,
                       cur_stream(), u16p(dy), u16p(w), u16pm(dx), batch);
  } else {
    hipLaunchKernelGGL(drla_conv_dgrad_l3, dim3(grid), dim3(256), 0,
                       cur_stream(), u16p(dy), u16p(w), u16pm(dx), batch);
  }
  return dx;
}

std::tuple<torch::Tensor, torch::Tensor> dqn_loss_fwd(
    torch::Tensor main_q, torch::Tensor next_main, torch::Tensor next_tgt,
    torch::Tensor actions, torch::Tensor rewards, torch::Tensor discounts,
    torch::Tensor weights) {
  for (auto* t : {&main_q, &next_main, &next_tgt, &actions, &rewards,
                  &discounts, &weights})
    check_gpu_contig(*t, "dqn_loss input");
  TORCH_CHECK(actions.scalar_type() == torch::kInt, "actions must be i32");
  const int B = main_q.size(0), A = main_q.size(1);
  const bool bf16 = main_q.scalar_type() == torch::kBFloat16;
  auto fopt = rewards.options().dtype(torch::kFloat);
  auto loss = torch::zeros({1}, fopt);
  auto td = torch::empty({B}, fopt);
  hipLaunchKernelGGL(
      drla_dqn_loss_fwd, dim3((B + 255) / 256), dim3(256), 0, cur_stream(),
      bf16 ? u16p(main_q) : nullptr,
      bf16 ? nullptr : main_q.data_ptr<float>(),
      next_main.data_ptr<float>(), next_tgt.data_ptr<float>(),
      actions.data_ptr<int>(), rewards.data_ptr<float>(),
      discounts.data_ptr<float>(), weights.data_ptr<float>(),
      loss.data_ptr<float>(), td.data_ptr<float>(), B, A);
  return {loss, td};
}

torch::Tensor dqn_loss_bwd(torch::Tensor td, torch::Tensor actions,
                           torch::Tensor weights, torch::Tensor gloss,
                           int64_t A, bool want_bf16) {
  for (auto* t : {&td, &actions, &weights, &gloss})
    check_gpu_contig(*t, "dqn_loss bwd input");
  const int B = td.numel();
  auto dmq = torch::empty(
      {B, A},
      td.options().dtype(want_bf16 ? torch::kBFloat16 : torch::kFloat));
  hipLaunchKernelGGL(
      drla_dqn_loss_bwd, dim3(((long long)B * A + 255) / 256), dim3(256), 0,
      cur_stream(), td.data_ptr<float>(), actions.data_ptr<int>(),
      weights.data_ptr<float>(), gloss.data_ptr<float>(),
      want_bf16 ? u16pm(dmq) : nullptr,
      want_bf16 ? nullptr : dmq.data_ptr<float>(), B, (int)A);
  return dmq;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> a2c_loss_fwd(
    torch::Tensor logits, torch::Tensor value, torch::Tensor next_value,
    torch::Tensor actions, torch::Tensor rewards, torch::Tensor done,
    double gamma, int64_t clip_mode, double c_bl, double c_ent) {
  for (auto* t : {&logits, &value, &next_value, &actions, &rewards, &done})
    check_gpu_contig(*t, "a2c loss input");
  TORCH_CHECK(actions.scalar_type() == torch::kInt, "actions must be i32");
  TORCH_CHECK(done.scalar_type() == torch::kBool, "done must be bool");
  const int N = logits.size(0), A = logits.size(1);
  const bool bf16 = logits.scalar_type() == torch::kBFloat16;
  auto fopt = value.options().dtype(torch::kFloat);
  auto losses = torch::zeros({4}, fopt);
  auto p_stash = torch::empty({N, A}, fopt);
  auto adv_st = torch::empty({N}, fopt);
  hipLaunchKernelGGL(
      drla_a2c_loss_fwd, dim3((N + 255) / 256), dim3(256), 0, cur_stream(),
      bf16 ? u16p(logits) : nullptr,
      bf16 ? nullptr : logits.data_ptr<float>(), value.data_ptr<float>(),
      next_value.data_ptr<float>(), actions.data_ptr<int>(),
      rewards.data_ptr<float>(),
      reinterpret_cast<const unsigned char*>(done.data_ptr<bool>()),
      static_cast<float>(gamma), static_cast<int>(clip_mode),
      static_cast<float>(c_bl), static_cast<float>(c_ent),
      losses.data_ptr<float>(), p_stash.data_ptr<float>(),
      adv_st.data_ptr<float>(), N, A);
  return {losses, p_stash, adv_st};
}

std::tuple<torch::Tensor, torch::Tensor> a2c_loss_bwd(
    torch::Tensor p_stash, torch::Tensor adv_st, torch::Tensor actions,
    torch::Tensor grad3, bool from_total, double c_bl, double c_ent,
    bool want_bf16) {
  for (auto* t : {&p_stash, &adv_st, &actions, &grad3})
    check_gpu_contig(*t, "a2c bwd input");
  const int N = p_stash.size(0), A = p_stash.size(1);
  auto dlg = torch::empty(
      {N, A}, p_stash.options().dtype(want_bf16 ? torch::kBFloat16
                                                : torch::kFloat));
  auto dvalue = torch::empty({N}, p_stash.options());
  hipLaunchKernelGGL(
      drla_a2c_loss_bwd, dim3((N + 255) / 256), dim3(256), 0, cur_stream(),
      p_stash.data_ptr<float>(), adv_st.data_ptr<float>(),
      actions.data_ptr<int>(), grad3.data_ptr<float>(),
      from_total ? 1 : 0, static_cast<float>(c_bl),
      static_cast<float>(c_ent), want_bf16 ? u16pm(dlg) : nullptr,
      want_bf16 ? nullptr : dlg.data_ptr<float>(),
      dvalue.data_ptr<float>(), N, A);
  return {dlg, dvalue};
}

static int next_pow2(int x) {
  int p = 1;
  while (p < x) p <<= 1;
  return p;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> r2d2_loss_fwd(
    torch::Tensor mq, torch::Tensor tq, torch::Tensor actions,
    torch::Tensor rewards, torch::Tensor done, torch::Tensor weights,
    double gamma, int64_t clip_mode) {
  for (auto* t : {&mq, &tq, &actions, &rewards, &done, &weights})
    check_gpu_contig(*t, "r2d2 loss input");
  TORCH_CHECK(actions.scalar_type() == torch::kInt, "actions must be i32");
  TORCH_CHECK(done.scalar_type() == torch::kBool, "done must be bool");
  const int B = mq.size(0), W = mq.size(1), A = mq.size(2);
  TORCH_CHECK(W >= 2 && W <= 1024, "window out of range");
  const bool bf16 = mq.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(tq.scalar_type() == mq.scalar_type(), "mq/tq dtype mismatch");
  auto fopt = rewards.options().dtype(torch::kFloat);
  auto loss = torch::zeros({1}, fopt);
  auto td_st = torch::empty({B, W - 1}, fopt);
  auto td_out = torch::empty({B}, fopt);
  const int block = next_pow2(W - 1);
  hipLaunchKernelGGL(
      drla_r2d2_loss_fwd, dim3(B), dim3(block), 0, cur_stream(),
      bf16 ? u16p(mq) : nullptr, bf16 ? nullptr : mq.data_ptr<float>(),
      bf16 ? u16p(tq) : nullptr, bf16 ? nullptr : tq.data_ptr<float>(),
      actions.data_ptr<int>(), rewards.data_ptr<float>(),
      reinterpret_cast<const unsigned char*>(done.data_ptr<bool>()),
      weights.data_ptr<float>(),
      static_cast<float>(gamma), static_cast<int>(clip_mode),
      loss.data_ptr<float>(), td_st.data_ptr<float>(),
      td_out.data_ptr<float>(), B, W, A);
  return {loss, td_st, td_out};
}

torch::Tensor dueling_head_fwd(torch::Tensor h, torch::Tensor Wt,
                               torch::Tensor bt, torch::Tensor Wo,
                               torch::Tensor bo, int64_t burn) {
  for (auto* t : {&h, &Wt, &bt, &Wo, &bo})
    check_gpu_contig(*t, "dueling head input");
  TORCH_CHECK(h.dim() == 3 && h.scalar_type() == torch::kFloat,
              "h must be [B,L,IN] f32");
  const int B = h.size(0), L = h.size(1), IN = h.size(2);
  const int MID = Wt.size(0), AO = Wo.size(0);
  TORCH_CHECK(Wt.size(1) == IN && Wo.size(1) == MID, "weight shape");
  TORCH_CHECK(IN <= 256 && MID <= 512 && AO <= 64 && AO >= 2,
              "dueling head dims out of range");
  const int W = L - (int)burn;
  auto q = torch::empty({B, W, AO - 1},
                        h.options().dtype(torch::kBFloat16));
  const size_t lds = (size_t)(MID * IN + AO * MID) * 2 +
                     (size_t)(2 * MID + 2 * IN + 2 * AO) * 4;
  TORCH_CHECK(lds <= 160 * 1024,
              "dueling head weights exceed the 160 KB LDS budget");
  const long long npairs = ((long long)B * W + 1) / 2;
  // few blocks, many row-pairs each: every block stages the full weight
  // set in LDS, so the grid must amortize that (~24 KB/block) over rows
  const int gx = (int)std::min<long long>((npairs + 7) / 8, 64);
  hipLaunchKernelGGL(drla_dueling_head_fwd, dim3(gx), dim3(256), lds,
                     cur_stream(), h.data_ptr<float>(), u16p(Wt),
                     u16p(bt), u16p(Wo), u16p(bo), u16pm(q), B, L,
                     (int)burn, IN, MID, AO);
  return q;
}

std::tuple<torch::Tensor, torch::Tensor> dhead_train_fwd(
    torch::Tensor h, torch::Tensor Wt, torch::Tensor bt, torch::Tensor Wo,
    torch::Tensor bo) {
  for (auto* t : {&h, &Wt, &bt, &Wo, &bo})
    check_gpu_contig(*t, "dhead fwd input");
  TORCH_CHECK(h.dim() == 2 && h.scalar_type() == torch::kFloat,
              "h must be [N,IN] f32");
  const int N = h.size(0), IN = h.size(1);
  const int MID = Wt.size(0), AO = Wo.size(0);
  TORCH_CHECK(Wt.size(1) == IN && Wo.size(1) == MID && AO >= 2,
              "dhead weight shape");
  const size_t lds = (size_t)(MID * IN + AO * MID) * 2 +
                     (size_t)(2 * MID + 2 * IN + 2 * AO) * 4;
  TORCH_CHECK(lds <= 160 * 1024, "dhead fwd LDS budget");
  auto q = torch::empty({N, AO - 1},
                        h.options().dtype(torch::kBFloat16));
  auto x_st = torch::empty({N, MID},
                           h.options().dtype(torch::kBFloat16));
  const long long npairs = ((long long)N + 1) / 2;
  const int gx = (int)std::min<long long>((npairs + 7) / 8, 64);
  hipLaunchKernelGGL(drla_dhead_train_fwd, dim3(gx), dim3(256), lds,
                     cur_stream(), h.data_ptr<float>(), u16p(Wt), u16p(bt),
                     u16p(Wo), u16p(bo), u16pm(q), u16pm(x_st), N, IN, MID,
                     AO);
  return {q, x_st};
}

std::vector<torch::Tensor> dhead_train_bwd(
    torch::Tensor dq, torch::Tensor x_st, torch::Tensor h,
    torch::Tensor Wt, torch::Tensor Wo) {
  for (auto* t : {&dq, &x_st, &h, &Wt, &Wo})
    check_gpu_contig(*t, "dhead bwd input");
  const int N = h.size(0), IN = h.size(1);
  const int MID = Wt.size(0), AO = Wo.size(0);
  TORCH_CHECK(dq.scalar_type() == torch::kBFloat16, "dq must be bf16");
  const size_t lds = (size_t)(MID * IN + AO * MID) * 2 +
                     (size_t)(2 * (MID * IN + AO * MID)
                              + 2 * MID + 2 * AO + MID + IN) * 4;
  TORCH_CHECK(lds <= 160 * 1024, "dhead bwd LDS budget");
  auto fopt = h.options();
  auto bopt = h.options().dtype(torch::kBFloat16);
  auto dh = torch::empty({N, IN}, fopt);
  const int ws_n = MID * IN + MID + AO * MID + AO;
  auto ws = torch::zeros({ws_n}, fopt);
  const int gx =
      (int)std::min<long long>(((long long)N + 7) / 8, 64);
  hipLaunchKernelGGL(drla_dhead_train_bwd, dim3(gx), dim3(256), lds,
                     cur_stream(), u16p(dq), u16p(x_st),
                     h.data_ptr<float>(), u16p(Wt), u16p(Wo),
                     dh.data_ptr<float>(), ws.data_ptr<float>(), N, IN,
                     MID, AO);
  auto dWt = torch::empty({MID, IN}, bopt);
  auto dbt = torch::empty({MID}, bopt);
  auto dWo = torch::empty({AO, MID}, bopt);
  auto dbo = torch::empty({AO}, bopt);
  hipLaunchKernelGGL(drla_dhead_train_fin, dim3(drla_grid(ws_n)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     ws.data_ptr<float>(), u16pm(dWt), u16pm(dbt),
                     u16pm(dWo), u16pm(dbo), IN, MID, AO);
  return {dh, dWt, dbt, dWo, dbo};
}

torch::Tensor r2d2_loss_bwd(torch::Tensor td_st, torch::Tensor actions,
                            torch::Tensor weights, torch::Tensor gloss,
                            int64_t W, int64_t A, bool want_bf16) {
  for (auto* t : {&td_st, &actions, &weights, &gloss})
    check_gpu_contig(*t, "r2d2 bwd input");
  const int B = td_st.size(0);
  auto dmq = torch::empty(
      {B, W, A}, td_st.options().dtype(want_bf16 ? torch::kBFloat16
                                                 : torch::kFloat));
  const long long total = (long long)B * W * A;
  hipLaunchKernelGGL(drla_r2d2_loss_bwd, dim3(drla_grid(total)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     td_st.data_ptr<float>(), actions.data_ptr<int>(),
                     weights.data_ptr<float>(), gloss.data_ptr<float>(),
                     want_bf16 ? u16pm(dmq) : nullptr,
                     want_bf16 ? nullptr : dmq.data_ptr<float>(), B,
                     (int)W, (int)A);
  return dmq;
}

void per_update(torch::Tensor tree, torch::Tensor idxs,
                torch::Tensor prios, int64_t cap) {
  for (auto* t : {&tree, &idxs, &prios}) check_gpu_contig(*t, "per tensor");
  TORCH_CHECK(idxs.scalar_type() == torch::kLong, "idxs must be i64");
  const int n = idxs.numel();
  hipLaunchKernelGGL(drla_per_update, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), tree.data_ptr<float>(),
                     reinterpret_cast<const long long*>(
                         idxs.data_ptr<int64_t>()),
                     prios.data_ptr<float>(), n, cap);
}

void per_rebuild(torch::Tensor tree, int64_t cap) {
  // bottom-up level-by-level recompute of the interior sums from the
  // leaves (drift repair for the float32 atomicAdd delta chains); the
  // level loop lives here so Python pays one call
  check_gpu_contig(tree, "tree");
  // interior nodes are [0, cap-2]; process descending ranges [hi>>1, hi):
  // any i >= hi>>1 has children 2i+1 >= hi, already final from the prior
  // launch — correct for any capacity, ~log2(cap) launches
  long long hi = cap - 1;
  while (hi > 0) {
    const long long lo = hi >> 1;
    const long long count = hi - lo;
    hipLaunchKernelGGL(drla_per_rebuild_level,
                       dim3((count + 255) / 256), dim3(256), 0,
                       cur_stream(), tree.data_ptr<float>(), lo, count);
    hi = lo;
  }
}

void multi_gather(torch::Tensor rows, torch::Tensor srcs,
                  torch::Tensor dsts, torch::Tensor fbytes,
                  int64_t max_chunks) {
  for (auto* t : {&rows, &srcs, &dsts, &fbytes})
    check_gpu_contig(*t, "multi_gather table");
  TORCH_CHECK(rows.scalar_type() == torch::kLong, "rows must be i64");
  const int B = rows.numel();
  const int F = srcs.numel();
  const long long total = (long long)B * max_chunks;
  const int gx = (int)std::min<long long>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(drla_multi_gather, dim3(gx, F), dim3(256), 0,
                     cur_stream(),
                     reinterpret_cast<const long long*>(
                         rows.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(
                         srcs.data_ptr<int64_t>()),
                     reinterpret_cast<const unsigned long long*>(
                         dsts.data_ptr<int64_t>()),
                     reinterpret_cast<const long long*>(
                         fbytes.data_ptr<int64_t>()), B);
}

std::tuple<torch::Tensor, torch::Tensor> per_sample(torch::Tensor tree,
                                                    torch::Tensor s,
                                                    torch::Tensor n_entries,
                                                    int64_t cap) {
  check_gpu_contig(tree, "tree");
  check_gpu_contig(s, "s");
  check_gpu_contig(n_entries, "n_entries");
  const int n = s.numel();
  auto idx = torch::empty({n}, s.options().dtype(torch::kLong));
  auto prio = torch::empty({n}, s.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(drla_per_sample, dim3((n + 255) / 256), dim3(256), 0,
                     cur_stream(), tree.data_ptr<float>(),
                     s.data_ptr<float>(), n_entries.data_ptr<float>(),
                     reinterpret_cast<long long*>(idx.data_ptr<int64_t>()),
                     prio.data_ptr<float>(), n, cap);
  return {idx, prio};
}

torch::Tensor embed_bwd(torch::Tensor indices, torch::Tensor grad_out,
                        int64_t num_rows, bool want_bf16) {
  check_gpu_contig(indices, "indices");
  check_gpu_contig(grad_out, "grad_out");
  TORCH_CHECK(indices.scalar_type() == torch::kLong, "indices must be i64");
  const long long N = grad_out.size(0);
  const int H = grad_out.size(1);
  const bool in16 = grad_out.scalar_type() == torch::kBFloat16;
  torch::Tensor scratch;
  if (want_bf16) {
    // persistent zero-between-calls scratch (the zeroing cast below
    // restores the invariant); f32 callers own the returned tensor, so
    // they keep the per-call allocation
    static torch::Tensor emb_scratch;
    if (!emb_scratch.defined() || emb_scratch.size(0) != num_rows ||
        emb_scratch.size(1) != H) {
      emb_scratch = torch::zeros({num_rows, H},
                                 grad_out.options().dtype(torch::kFloat));
    }
    scratch = emb_scratch;
  } else {
    scratch = torch::zeros({num_rows, H},
                           grad_out.options().dtype(torch::kFloat));
  }
  hipLaunchKernelGGL(
      drla_embed_bwd_scatter, dim3(drla_grid(N * H)), dim3(DRLA_BLOCK), 0,
      cur_stream(),
      reinterpret_cast<const long long*>(indices.data_ptr<int64_t>()),
      in16 ? reinterpret_cast<const unsigned short*>(grad_out.data_ptr())
           : nullptr,
      in16 ? nullptr : grad_out.data_ptr<float>(), scratch.data_ptr<float>(),
      N, H, (long long)num_rows);
  if (!want_bf16) return scratch;
  auto out = torch::empty({num_rows, H},
                          grad_out.options().dtype(torch::kBFloat16));
  hipLaunchKernelGGL(drla_f32_to_bf16_zero_kernel,
                     dim3(drla_grid(num_rows * H)), dim3(DRLA_BLOCK), 0,
                     cur_stream(), scratch.data_ptr<float>(),
                     reinterpret_cast<unsigned short*>(out.data_ptr()),
                     (long long)(num_rows * H));
  return out;
}

std::vector<torch::Tensor> vtrace_loss_fwd(
    torch::Tensor logits, torch::Tensor value, torch::Tensor mu,
    torch::Tensor actions, torch::Tensor rewards, torch::Tensor done,
    double gamma, int64_t clip_mode, double c_bl, double c_ent) {
  for (auto* t : {&logits, &value, &mu, &actions, &rewards, &done})
    check_gpu_contig(*t, "vtrace_loss input");
  TORCH_CHECK(actions.scalar_type() == torch::kInt, "actions must be int32");
  TORCH_CHECK(done.scalar_type() == torch::kBool, "done must be bool");
  const int B = logits.size(0), T = logits.size(1), A = logits.size(2);
  TORCH_CHECK(A <= 64, "num_action cap is 64");
  const bool bf16 = logits.scalar_type() == torch::kBFloat16;
  auto fopt = value.options().dtype(torch::kFloat);
  TORCH_CHECK(T <= 128, "trajectory cap is 128 (VT_MAX_T)");
  auto p_stash = torch::empty({B, T, A}, fopt);
  auto vs_stash = torch::empty({B, T - 2}, fopt);
  auto adv_stash = torch::empty({B, T - 2}, fopt);
  auto losses = torch::zeros({4}, fopt);  // blocks atomicAdd into it
  hipLaunchKernelGGL(
      drla_vtrace_loss_fwd, dim3(B), dim3(256), 0, cur_stream(),
      bf16 ? reinterpret_cast<const unsigned short*>(logits.data_ptr())
           : nullptr,
      bf16 ? nullptr : logits.data_ptr<float>(), value.data_ptr<float>(),
      mu.data_ptr<float>(), actions.data_ptr<int>(),
      rewards.data_ptr<float>(),
      reinterpret_cast<const unsigned char*>(done.data_ptr<bool>()),
      static_cast<float>(gamma), (int)clip_mode, static_cast<float>(c_bl),
      static_cast<float>(c_ent), p_stash.data_ptr<float>(),
      vs_stash.data_ptr<float>(), adv_stash.data_ptr<float>(),
      losses.data_ptr<float>(), B, T, A);
  return {losses, p_stash, vs_stash, adv_stash};
}

std::tuple<torch::Tensor, torch::Tensor> vtrace_loss_bwd(
    torch::Tensor p_stash, torch::Tensor vs_stash, torch::Tensor adv_stash,
    torch::Tensor value, torch::Tensor actions, torch::Tensor grad3,
    bool from_total, double c_bl, double c_ent, bool want_bf16) {
  for (auto* t : {&p_stash, &vs_stash, &adv_stash, &value, &actions, &grad3})
    check_gpu_contig(*t, "vtrace_loss bwd input");
  const int B = p_stash.size(0), T = p_stash.size(1), A = p_stash.size(2);
  auto dlogits = torch::empty(
      {B, T, A},
      value.options().dtype(want_bf16 ? torch::kBFloat16 : torch::kFloat));
  auto dvalue = torch::empty({B, T}, value.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(
      drla_vtrace_loss_bwd, dim3(drla_grid((long long)B * T)), dim3(256), 0,
      cur_stream(), p_stash.data_ptr<float>(), vs_stash.data_ptr<float>(),
      adv_stash.data_ptr<float>(), value.data_ptr<float>(),
      actions.data_ptr<int>(), grad3.data_ptr<float>(),
      from_total ? 1 : 0, static_cast<float>(c_bl),
      static_cast<float>(c_ent),
      want_bf16 ? reinterpret_cast<unsigned short*>(dlogits.data_ptr())
                : nullptr,
      want_bf16 ? nullptr : dlogits.data_ptr<float>(),
      dvalue.data_ptr<float>(), B, T, A);
  return {dlogits, dvalue};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> lstm_tail_fwd(
    torch::Tensor gates, torch::Tensor c_prev, double forget_bias) {
  check_gpu_contig(gates, "gates");
  check_gpu_contig(c_prev, "c_prev");
  const bool bf16 = gates.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || gates.scalar_type() == torch::kFloat,
              "gates must be bf16 or float32");
  TORCH_CHECK(c_prev.scalar_type() == torch::kFloat, "c_prev must be float32");
  const long long N = gates.size(0);
  const int H4 = gates.size(1);
  TORCH_CHECK(H4 % 4 == 0, "gates dim1 must be 4*H");
  const int H = H4 / 4;
  TORCH_CHECK(c_prev.size(1) == H, "c_prev width mismatch");
  auto new_h = torch::empty_like(c_prev);
  auto new_c = torch::empty_like(c_prev);
  auto stash = torch::empty({N, (long long)H4},
                            c_prev.options().dtype(torch::kFloat));
  if (bf16) {
    hipLaunchKernelGGL(drla_lstm_tail_fwd_bf16, dim3(drla_grid(N * H)),
                       dim3(DRLA_BLOCK), 0, cur_stream(), u16p(gates),
                       c_prev.data_ptr<float>(), new_h.data_ptr<float>(),
                       new_c.data_ptr<float>(), stash.data_ptr<float>(),
                       static_cast<float>(forget_bias), N, H);
  } else {
    hipLaunchKernelGGL(drla_lstm_tail_fwd, dim3(drla_grid(N * H)),
                       dim3(DRLA_BLOCK), 0, cur_stream(),
                       gates.data_ptr<float>(), c_prev.data_ptr<float>(),
                       new_h.data_ptr<float>(), new_c.data_ptr<float>(),
                       stash.data_ptr<float>(),
                       static_cast<float>(forget_bias), N, H);
  }
  return {new_h, new_c, stash};
}

std::tuple<torch::Tensor, torch::Tensor> lstm_tail_bwd(
    torch::Tensor grad_h, c10::optional<torch::Tensor> grad_c_opt,
    torch::Tensor stash, torch::Tensor c_prev, torch::Tensor new_c,
    bool bf16_gates) {
  for (auto* t : {&stash, &c_prev, &new_c})
    check_gpu_contig(*t, "lstm bwd input");
  // grad_c is null when new_c is unused downstream (saves the zero-fill)
  const float* grad_c_ptr = nullptr;
  if (grad_c_opt.has_value()) {
    check_gpu_contig(*grad_c_opt, "grad_c");
    grad_c_ptr = grad_c_opt->data_ptr<float>();
  }
  // grad_h may be an N-strided [N,H] view (slice of the xh gradient)
  TORCH_CHECK(grad_h.is_cuda() && grad_h.dim() == 2 &&
              grad_h.stride(1) == 1, "grad_h must be row-contiguous");
  const long long gh_stride = grad_h.stride(0);
  const long long N = stash.size(0);
  const int H = stash.size(1) / 4;
  auto grad_c_prev = torch::empty_like(c_prev);
  if (bf16_gates) {
    auto grad_gates = torch::empty(
        {N, (long long)stash.size(1)},
        stash.options().dtype(torch::kBFloat16));
    hipLaunchKernelGGL(drla_lstm_tail_bwd_bf16, dim3(drla_grid(N * H)),
                       dim3(DRLA_BLOCK), 0, cur_stream(),
                       grad_h.data_ptr<float>(), grad_c_ptr,
                       stash.data_ptr<float>(), c_prev.data_ptr<float>(),
                       new_c.data_ptr<float>(), u16pm(grad_gates),
                       grad_c_prev.data_ptr<float>(), N, H, gh_stride);
    return {grad_gates, grad_c_prev};
  }
  auto grad_gates = torch::empty_like(stash);
  hipLaunchKernelGGL(drla_lstm_tail_bwd, dim3(drla_grid(N * H)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     grad_h.data_ptr<float>(), grad_c_ptr,
                     stash.data_ptr<float>(), c_prev.data_ptr<float>(),
                     new_c.data_ptr<float>(), grad_gates.data_ptr<float>(),
                     grad_c_prev.data_ptr<float>(), N, H, gh_stride);
  return {grad_gates, grad_c_prev};
}

std::vector<torch::Tensor> mlp_heads_fwd(
    torch::Tensor h, std::vector<torch::Tensor> weights,
    std::vector<torch::Tensor> biases, int64_t A) {
  check_gpu_contig(h, "h");
  TORCH_CHECK(weights.size() == 6 && biases.size() == 6);
  for (auto& t : weights) check_gpu_contig(t, "W");
  for (auto& t : biases) check_gpu_contig(t, "b");
  const int N = h.size(0);
  TORCH_CHECK(h.size(1) == 256 && A <= 32);
  auto bopt = h.options().dtype(torch::kBFloat16);
  auto logits = torch::empty({N, A}, bopt);
  auto value = torch::empty({N}, h.options().dtype(torch::kFloat));
  auto stash = torch::empty({N, 5 * 256}, bopt);
  hipLaunchKernelGGL(
      drla_mlp_heads_fwd, dim3((N + 15) / 16, 2), dim3(256), 0,
      cur_stream(),
      h.data_ptr<float>(), u16p(weights[0]),
      u16p(biases[0]), u16p(weights[1]),
      u16p(biases[1]), u16p(weights[2]),
      u16p(biases[2]), u16p(weights[3]),
      u16p(biases[3]), u16p(weights[4]),
      u16p(biases[4]), u16p(weights[5]),
      u16p(biases[5]), u16pm(logits),
      value.data_ptr<float>(), u16pm(stash), N, (int)A);
  return {logits, value, stash};
}

std::vector<torch::Tensor> mlp_heads_bwd(
    torch::Tensor dlogits, torch::Tensor dvalue, torch::Tensor stash,
    std::vector<torch::Tensor> weights, int64_t A) {
  check_gpu_contig(dlogits, "dlogits");
  check_gpu_contig(dvalue, "dvalue");
  check_gpu_contig(stash, "stash");
  TORCH_CHECK(weights.size() == 6);
  const int N = dvalue.numel();
  auto bopt = dlogits.options().dtype(torch::kBFloat16);
  auto fopt = dvalue.options().dtype(torch::kFloat);
  auto dz1p = torch::empty({N, 256}, bopt);
  auto dz2p = torch::empty({N, 256}, bopt);
  auto dz1v = torch::empty({N, 256}, bopt);
  auto dz2v = torch::empty({N, 256}, bopt);
  // dh + all six bias grads live in ONE zeroed workspace (atomics targets;
  // one fill kernel instead of seven)
  const long long ws_n = (long long)N * 256 + 4 * 256 + A + 1;
  auto ws = torch::zeros({ws_n}, fopt);
  long long off = 0;
  auto dh = ws.narrow(0, off, (long long)N * 256).view({N, 256});
  off += (long long)N * 256;
  auto db1p = ws.narrow(0, off, 256); off += 256;
  auto db2p = ws.narrow(0, off, 256); off += 256;
  auto db3p = ws.narrow(0, off, A); off += A;
  auto db1v = ws.narrow(0, off, 256); off += 256;
  auto db2v = ws.narrow(0, off, 256); off += 256;
  auto db3v = ws.narrow(0, off, 1);
  hipLaunchKernelGGL(
      drla_mlp_heads_bwd, dim3((N + 15) / 16, 2), dim3(256), 0,
      cur_stream(),
      u16p(dlogits), dvalue.data_ptr<float>(), u16p(stash),
      u16p(weights[0]), u16p(weights[1]), u16p(weights[2]),
      u16p(weights[3]), u16p(weights[4]), u16p(weights[5]), u16pm(dz1p),
      u16pm(dz2p), u16pm(dz1v), u16pm(dz2v), dh.data_ptr<float>(),
      db1p.data_ptr<float>(), db2p.data_ptr<float>(),
      db3p.data_ptr<float>(), db1v.data_ptr<float>(),
      db2v.data_ptr<float>(), db3v.data_ptr<float>(), N, (int)A);
  // ws comes back too so the wrapper can cast the contiguous bias-grad
  // tail (db1p..db3v) to bf16 in ONE kernel instead of six
  return {dz1p, dz2p, dz1v, dz2v, dh,   db1p, db2p, db3p, db1v, db2v,
          db3v, ws};
}

std::tuple<torch::Tensor, torch::Tensor> embed_mlp_fwd(
    torch::Tensor pa, torch::Tensor table, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2) {
  for (auto* t : {&pa, &table, &b1, &w2, &b2})
    check_gpu_contig(*t, "embed fwd input");
  TORCH_CHECK(pa.scalar_type() == torch::kLong, "pa must be int64");
  const int N = pa.numel();
  auto bopt = table.options();
  auto out = torch::empty({N, 256}, bopt);
  auto a1 = torch::empty({N, 256}, bopt);
  hipLaunchKernelGGL(drla_embed_mlp_fwd, dim3((N + 15) / 16), dim3(256), 0,
                     cur_stream(),
                     reinterpret_cast<const long long*>(pa.data_ptr<int64_t>()),
                     u16p(table), u16p(b1), u16p(w2), u16p(b2), u16pm(out),
                     u16pm(a1), N, (int)table.size(0));
  return {out, a1};
}

std::vector<torch::Tensor> embed_mlp_bwd(torch::Tensor dy, torch::Tensor out,
                                         torch::Tensor a1, torch::Tensor w2,
                                         torch::Tensor idx, int64_t A) {
  for (auto* t : {&out, &a1, &w2}) check_gpu_contig(*t, "embed bwd input");
  check_gpu_contig(idx, "idx");
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 2 && dy.stride(1) == 1,
              "dy must be row-contiguous");
  const int N = out.size(0);
  const long long n_table = (long long)A * 256;
  auto bopt = out.options();
  // persistent buffers: W2^T (fully overwritten) and the f32 partial
  // scratch (zero-between-calls; drla_embed_finalize re-zeroes it)
  static torch::Tensor w2t, scratch;
  if (!w2t.defined()) w2t = torch::empty({65536}, bopt);
  if (!scratch.defined() || scratch.numel() != n_table + 512) {
    scratch = torch::zeros({n_table + 512}, bopt.dtype(torch::kFloat));
  }
  hipLaunchKernelGGL(drla_embed_w2t_pack, dim3(drla_grid(65536)),
                     dim3(DRLA_BLOCK), 0, cur_stream(), u16p(w2),
                     u16pm(w2t));
  auto dz2 = torch::empty({N, 256}, bopt);
  auto da1 = torch::empty({N, 256}, bopt);
  float* bias_ws = scratch.data_ptr<float>() + n_table;
  hipLaunchKernelGGL(drla_embed_mlp_bwd, dim3((N + 15) / 16), dim3(256), 0,
                     cur_stream(), u16p(dy), (long long)dy.stride(0),
                     u16p(out), u16p(a1), u16p(w2t), u16pm(dz2), u16pm(da1),
                     bias_ws, N);
  // table scatter into the f32 scratch, then one 3-output finalize
  hipLaunchKernelGGL(
      drla_embed_bwd_scatter, dim3(drla_grid((long long)N * 256)),
      dim3(DRLA_BLOCK), 0, cur_stream(),
      reinterpret_cast<const long long*>(idx.data_ptr<int64_t>()),
      u16p(da1), nullptr, scratch.data_ptr<float>(), N, 256,
      (long long)A);
  auto dtable = torch::empty({A, 256}, bopt);
  auto db1 = torch::empty({256}, bopt);
  auto db2 = torch::empty({256}, bopt);
  hipLaunchKernelGGL(drla_embed_finalize, dim3(drla_grid(n_table + 512)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     scratch.data_ptr<float>(), u16pm(dtable), u16pm(db1),
                     u16pm(db2), n_table);
  return {dz2, dtable, db1, db2};
}

torch::Tensor mlp_heads_pack_wt(std::vector<torch::Tensor> weights,
                                int64_t A) {
  TORCH_CHECK(weights.size() == 6);
  for (auto& w : weights) check_gpu_contig(w, "pack weight");
  auto out = torch::empty({4 * 65536 + 256 * 32 + 256},
                          weights[0].options());
  hipLaunchKernelGGL(drla_heads_wt_pack, dim3(drla_grid(out.numel())),
                     dim3(DRLA_BLOCK), 0, cur_stream(), u16p(weights[0]),
                     u16p(weights[1]), u16p(weights[2]), u16p(weights[3]),
                     u16p(weights[4]), u16p(weights[5]), u16pm(out),
                     (int)A);
  return out;
}

std::vector<torch::Tensor> mlp_heads_wgrad(
    torch::Tensor dz1p, torch::Tensor dz2p, torch::Tensor dz1v,
    torch::Tensor dz2v, torch::Tensor dlogits, torch::Tensor dvalue,
    torch::Tensor stash, torch::Tensor ws, int64_t A) {
  for (auto* t : {&dz1p, &dz2p, &dz1v, &dz2v, &dlogits, &stash, &ws})
    check_gpu_contig(*t, "wgrad input");
  check_gpu_contig(dvalue, "dvalue");
  const int N = dvalue.numel();
  auto bopt = dz1p.options();
  auto dw1p = torch::empty({256, 256}, bopt);
  auto dw2p = torch::empty({256, 256}, bopt);
  auto dw3p = torch::empty({A, 256}, bopt);
  auto dw1v = torch::empty({256, 256}, bopt);
  auto dw2v = torch::empty({256, 256}, bopt);
  auto dw3v = torch::empty({1, 256}, bopt);
  auto db1p = torch::empty({256}, bopt);
  auto db2p = torch::empty({256}, bopt);
  auto db3p = torch::empty({A}, bopt);
  auto db1v = torch::empty({256}, bopt);
  auto db2v = torch::empty({256}, bopt);
  auto db3v = torch::empty({1}, bopt);
  // ws tail = the f32 bias-grad partials written by mlp_heads_bwd
  const float* ws_tail = ws.data_ptr<float>() + (long long)N * 256;
  hipLaunchKernelGGL(drla_heads_wgrad, dim3(73), dim3(256), 0, cur_stream(),
                     u16p(dz1p), u16p(dz2p), u16p(dz1v), u16p(dz2v),
                     u16p(dlogits), dvalue.data_ptr<float>(), u16p(stash),
                     u16pm(dw1p), u16pm(dw2p), u16pm(dw3p), u16pm(dw1v),
                     u16pm(dw2v), u16pm(dw3v), ws_tail, u16pm(db1p),
                     u16pm(db2p), u16pm(db3p), u16pm(db1v), u16pm(db2v),
                     u16pm(db3v), N, (int)A);
  return {dw1p, dw2p, dw3p, dw1v, dw2v, dw3v,
          db1p, db2p, db3p, db1v, db2v, db3v};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> lstm_seq_fwd(
    torch::Tensor xgates, torch::Tensor Wh, torch::Tensor h0,
    torch::Tensor c0, torch::Tensor done, double forget_bias) {
  for (auto* t : {&xgates, &Wh, &h0, &c0, &done})
    check_gpu_contig(*t, "lstm_seq input");
  TORCH_CHECK(Wh.scalar_type() == torch::kBFloat16, "Wh must be bf16");
  TORCH_CHECK(done.scalar_type() == torch::kBool, "done must be bool");
  const int B = xgates.size(0), L = xgates.size(1);
  const int H = xgates.size(2) / 4;
  TORCH_CHECK(4 * H <= 1024, "lstm hidden cap is 256");
  const bool bf16 = xgates.scalar_type() == torch::kBFloat16;
  auto fopt = h0.options().dtype(torch::kFloat);
  auto h_out = torch::empty({B, L, H}, fopt);
  auto h_fin = torch::empty({B, H}, fopt);
  auto c_fin = torch::empty({B, H}, fopt);
  const int lds = H * 4 * H * 2 + 2 * H * 4 + 16;
  hipLaunchKernelGGL(
      drla_lstm_seq_fwd, dim3(B), dim3(4 * H), lds, cur_stream(),
      bf16 ? u16p(xgates) : nullptr,
      bf16 ? nullptr : xgates.data_ptr<float>(), u16p(Wh),
      h0.data_ptr<float>(), c0.data_ptr<float>(),
      reinterpret_cast<const unsigned char*>(done.data_ptr<bool>()),
      h_out.data_ptr<float>(),
      h_fin.data_ptr<float>(), c_fin.data_ptr<float>(),
      static_cast<float>(forget_bias), B, L, H);
  return {h_out, h_fin, c_fin};
}

void grad_gather(torch::Tensor srcs, torch::Tensor offs,
                 torch::Tensor sizes, torch::Tensor dst,
                 c10::optional<torch::Tensor> norm_ws) {
  check_gpu_contig(dst, "dst");
  TORCH_CHECK(dst.scalar_type() == torch::kBFloat16,
              "grad_gather wants a bf16 flat bucket");
  TORCH_CHECK(dst.numel() % 8 == 0, "flat bucket must be 8-aligned");
  for (auto* t : {&srcs, &offs, &sizes}) {
    check_gpu_contig(*t, "gather table");
    TORCH_CHECK(t->scalar_type() == torch::kLong, "table must be int64");
  }
  const long long chunks = dst.numel() / 8;
  float* nw = nullptr;
  int nw_n = 0;
  if (norm_ws.has_value()) {
    check_gpu_contig(*norm_ws, "norm_ws");
    nw = norm_ws->data_ptr<float>();
    nw_n = (int)norm_ws->numel();
  }
  hipLaunchKernelGGL(
      drla_grad_gather, dim3(drla_grid(chunks)), dim3(DRLA_BLOCK), 0,
      cur_stream(),
      reinterpret_cast<const unsigned long long*>(srcs.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(offs.data_ptr<int64_t>()),
      reinterpret_cast<const long long*>(sizes.data_ptr<int64_t>()),
      u16pm(dst), (int)srcs.numel(), chunks, nw, nw_n);
}

std::vector<torch::Tensor> lstm_seq_train_fwd(
    torch::Tensor xg, torch::Tensor Wh, torch::Tensor h0, torch::Tensor c0,
    torch::Tensor done, double forget_bias) {
  for (auto* t : {&xg, &Wh, &h0, &c0, &done})
    check_gpu_contig(*t, "lstm_seq_train input");
  TORCH_CHECK(xg.scalar_type() == torch::kBFloat16 &&
              Wh.scalar_type() == torch::kBFloat16);
  const int B = xg.size(0), L = xg.size(1), H = xg.size(2) / 4;
  TORCH_CHECK(4 * H <= 1024 && (long long)H * 4 * H * 2 <= 120 * 1024,
              "lstm seq-train caps: 4H<=1024 and Wh must fit LDS");
  auto fopt = h0.options().dtype(torch::kFloat);
  auto h_out = torch::empty({B, L, H}, fopt);
  auto h_fin = torch::empty({B, H}, fopt);
  auto c_fin = torch::empty({B, H}, fopt);
  auto acts = torch::empty({B, L, 4 * H}, fopt);
  auto c_prev = torch::empty({B, L, H}, fopt);
  auto h_prev = torch::empty({B, L, H}, xg.options());
  const int lds = H * 4 * H * 2 + 2 * H * 4 + 16;
  hipLaunchKernelGGL(drla_lstm_seq_train_fwd, dim3(B), dim3(4 * H), lds,
                     cur_stream(), u16p(xg), u16p(Wh),
                     h0.data_ptr<float>(), c0.data_ptr<float>(),
                     reinterpret_cast<const unsigned char*>(
                         done.data_ptr<bool>()),
                     h_out.data_ptr<float>(), h_fin.data_ptr<float>(),
                     c_fin.data_ptr<float>(), acts.data_ptr<float>(),
                     c_prev.data_ptr<float>(), u16pm(h_prev),
                     static_cast<float>(forget_bias), B, L, H);
  return {h_out, h_fin, c_fin, acts, c_prev, h_prev};
}

std::vector<torch::Tensor> lstm_seq_train_bwd(
    torch::Tensor dh_out, c10::optional<torch::Tensor> dh_fin,
    c10::optional<torch::Tensor> dc_fin, torch::Tensor acts,
    torch::Tensor c_prev, torch::Tensor Wh, torch::Tensor done) {
  for (auto* t : {&dh_out, &acts, &c_prev, &Wh, &done})
    check_gpu_contig(*t, "lstm_seq_train bwd input");
  const int B = acts.size(0), L = acts.size(1), H = acts.size(2) / 4;
  auto fopt = dh_out.options().dtype(torch::kFloat);
  auto dxg = torch::empty({B, L, 4 * H},
                          dh_out.options().dtype(torch::kBFloat16));
  auto dh0 = torch::empty({B, H}, fopt);
  auto dc0 = torch::empty({B, H}, fopt);
  const int lds = H * 4 * H * 2 + 2 * H * 4 + 16;
  hipLaunchKernelGGL(
      drla_lstm_seq_train_bwd, dim3(B), dim3(4 * H), lds, cur_stream(),
      dh_out.data_ptr<float>(),
      dh_fin.has_value() ? dh_fin->data_ptr<float>() : nullptr,
      dc_fin.has_value() ? dc_fin->data_ptr<float>() : nullptr,
      acts.data_ptr<float>(), c_prev.data_ptr<float>(), u16p(Wh),
      reinterpret_cast<const unsigned char*>(done.data_ptr<bool>()),
      u16pm(dxg), dh0.data_ptr<float>(), dc0.data_ptr<float>(), B, L, H);
  return {dxg, dh0, dc0};
}

torch::Tensor sq_norm(torch::Tensor x) {
  check_gpu_contig(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kFloat, "sq_norm wants float32");
  // DRLA_NORM_SLOTS cache-line-spread partials (total = .sum())
  auto out = torch::zeros({DRLA_NORM_SLOTS * 16}, x.options());
  hipLaunchKernelGGL(drla_sq_norm, dim3(drla_grid(x.numel() / 4 + 1)),
                     dim3(DRLA_BLOCK), 0, cur_stream(), x.data_ptr<float>(),
                     out.data_ptr<float>(), (long long)x.numel());
  return out;
}

torch::Tensor sq_norm_bf16(torch::Tensor x) {
  check_gpu_contig(x, "x");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "wants bf16");
  auto out = torch::zeros({DRLA_NORM_SLOTS * 16},
                          x.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(drla_sq_norm_bf16, dim3(drla_grid(x.numel() / 4 + 1)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     reinterpret_cast<const unsigned short*>(x.data_ptr()),
                     out.data_ptr<float>(), (long long)x.numel());
  return out;
}

void rmsprop_step_bf16_t(torch::Tensor p, torch::Tensor g,
                         torch::Tensor master, torch::Tensor ms, double clip,
                         torch::Tensor lr_buf, double rho, double eps,
                         c10::optional<torch::Tensor> norm_ws) {
  for (auto* t : {&p, &g, &master, &ms, &lr_buf})
    check_gpu_contig(*t, "rmsprop bf16 tensor");
  TORCH_CHECK(p.scalar_type() == torch::kBFloat16 &&
                  g.scalar_type() == torch::kBFloat16,
              "params/grads must be bf16");
  const long long n = p.numel();
  torch::Tensor norm_buf;
  if (norm_ws.has_value() && clip > 0) {
    // persistent buffer, zeroed by grad_gather earlier on this stream
    norm_buf = *norm_ws;
    hipLaunchKernelGGL(drla_sq_norm_bf16,
                       dim3(drla_grid(g.numel() / 4 + 1)), dim3(DRLA_BLOCK),
                       0, cur_stream(),
                       reinterpret_cast<const unsigned short*>(g.data_ptr()),
                       norm_buf.data_ptr<float>(), (long long)g.numel());
  } else if (clip > 0) {
    norm_buf = sq_norm_bf16(g);
  } else {
    norm_buf = torch::zeros({DRLA_NORM_SLOTS * 16}, master.options());
  }
  hipLaunchKernelGGL(drla_rmsprop_step_bf16, dim3(drla_grid(n)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     reinterpret_cast<unsigned short*>(p.data_ptr()),
                     reinterpret_cast<const unsigned short*>(g.data_ptr()),
                     master.data_ptr<float>(), ms.data_ptr<float>(),
                     norm_buf.data_ptr<float>(), static_cast<float>(clip),
                     lr_buf.data_ptr<float>(), static_cast<float>(rho),
                     static_cast<float>(eps), n);
}

void adam_step_bf16_t(torch::Tensor p, torch::Tensor g, torch::Tensor master,
                      torch::Tensor m, torch::Tensor v, double clip,
                      torch::Tensor lr_buf, double beta1, double beta2,
                      double eps) {
  for (auto* t : {&p, &g, &master, &m, &v, &lr_buf})
    check_gpu_contig(*t, "adam bf16 tensor");
  const long long n = p.numel();
  torch::Tensor norm_buf;
  if (clip > 0) {
    norm_buf = sq_norm_bf16(g);
  } else {
    norm_buf = torch::zeros({1}, master.options());
  }
  hipLaunchKernelGGL(drla_adam_step_bf16, dim3(drla_grid(n)),
                     dim3(DRLA_BLOCK), 0, cur_stream(),
                     reinterpret_cast<unsigned short*>(p.data_ptr()),
                     reinterpret_cast<const unsigned short*>(g.data_ptr()),
                     master.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), norm_buf.data_ptr<float>(),
                     static_cast<float>(clip), lr_buf.data_ptr<float>(),
                     static_cast<float>(beta1), static_cast<float>(beta2),
                     static_cast<float>(eps), n);
}

void rmsprop_step_t(torch::Tensor p, torch::Tensor g, torch::Tensor ms,
                    double clip, torch::Tensor lr_buf, double rho,
                    double eps) {
  for (auto* t : {&p, &g, &ms, &lr_buf})
    check_gpu_contig(*t, "rmsprop tensor");
  const long long n = p.numel();
  torch::Tensor norm_buf;
  if (clip > 0) {
    norm_buf = sq_norm(g);
  } else {
    norm_buf = torch::zeros({1}, p.options());
  }
  hipLaunchKernelGGL(drla_rmsprop_step, dim3(drla_grid(n)), dim3(DRLA_BLOCK),
                     0, cur_stream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), ms.data_ptr<float>(),
                     norm_buf.data_ptr<float>(), static_cast<float>(clip),
                     lr_buf.data_ptr<float>(), static_cast<float>(rho),
                     static_cast<float>(eps), n);
}

void rmsprop_step(torch::Tensor p, torch::Tensor g, torch::Tensor ms,
                  double clip, double lr, double rho, double eps) {
  auto lr_buf = torch::full({1}, lr, p.options());
  rmsprop_step_t(p, g, ms, clip, lr_buf, rho, eps);
}

void adam_step_t(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, double clip, torch::Tensor lr_buf,
                 double beta1, double beta2, double eps) {
  for (auto* t : {&p, &g, &m, &v, &lr_buf})
    check_gpu_contig(*t, "adam tensor");
  const long long n = p.numel();
  torch::Tensor norm_buf;
  if (clip > 0) {
    norm_buf = sq_norm(g);
  } else {
    norm_buf = torch::zeros({1}, p.options());
  }
  hipLaunchKernelGGL(drla_adam_step, dim3(drla_grid(n)), dim3(DRLA_BLOCK), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     norm_buf.data_ptr<float>(), static_cast<float>(clip),
                     lr_buf.data_ptr<float>(), static_cast<float>(beta1),
                     static_cast<float>(beta2), static_cast<float>(eps), n);
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double clip, double lr_t, double beta1,
               double beta2, double eps) {
  auto lr_buf = torch::full({1}, lr_t, p.options());
  adam_step_t(p, g, m, v, clip, lr_buf, beta1, beta2, eps);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("normalize_frames_f32", &normalize_frames_f32,
        "uint8 frames -> float32/255 (K11)");
  m.def("normalize_frames_bf16", &normalize_frames_bf16,
        "uint8 frames -> bf16/255 (K11)");
  m.def("vtrace_scan", &vtrace_scan, "fused V-trace reverse scan (K5)");
  m.def("vtrace_loss_fwd", &vtrace_loss_fwd,
        "fused IMPALA loss pipeline forward (K5+K7+K13)");
  m.def("vtrace_loss_bwd", &vtrace_loss_bwd,
        "fused IMPALA loss pipeline backward (closed form)");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("conv_fwd", &conv_fwd,
        "MFMA implicit-GEMM conv fwd, fused normalize+bias+ReLU (K1)");
  m.def("conv_wgrad", &conv_wgrad, "MFMA conv weight gradient (K1 bwd)");
  m.def("conv_dgrad", &conv_dgrad, "MFMA conv data gradient (K1 bwd)");
  m.def("relu_mask_bwd", &relu_mask_bwd,
        "fused-ReLU backward mask + bias gradient");
  m.def("dqn_loss_fwd", &dqn_loss_fwd,
        "fused double-DQN target + IS-weighted TD loss (K8)");
  m.def("dqn_loss_bwd", &dqn_loss_bwd, "closed-form K8 backward");
  m.def("a2c_loss_fwd", &a2c_loss_fwd, "fused A2C loss fwd (K6)");
  m.def("a2c_loss_bwd", &a2c_loss_bwd, "fused A2C loss bwd (K6)");
  m.def("r2d2_loss_fwd", &r2d2_loss_fwd,
        "fused R2D2 sequence-TD tail fwd (K9)");
  m.def("dhead_train_fwd", &dhead_train_fwd,
        "grad-carrying dueling head fwd (R2D2 trained window)");
  m.def("dhead_train_bwd", &dhead_train_bwd,
        "grad-carrying dueling head bwd chain + weight grads");
  m.def("dueling_head_fwd", &dueling_head_fwd,
        "no-grad dueling head over the post-burn-in window");
  m.def("r2d2_loss_bwd", &r2d2_loss_bwd,
        "fused R2D2 sequence-TD tail bwd (K9)");
  m.def("per_update", &per_update, "GPU PER segment-tree batched update");
  m.def("per_sample", &per_sample, "GPU PER stratified sample descent");
  m.def("multi_gather", &multi_gather,
        "one-kernel replay batch gather (all fields)");
  m.def("per_rebuild", &per_rebuild,
        "GPU PER interior-sum rebuild (float32 drift repair)");
  m.def("embed_bwd", &embed_bwd,
        "action-embedding table gradient (K2 backward)");
  m.def("lstm_tail_fwd", &lstm_tail_fwd, "fused LSTM gate tail fwd (K3)");
  m.def("lstm_tail_bwd", &lstm_tail_bwd, "fused LSTM gate tail bwd (K3)");
  m.def("mlp_heads_fwd", &mlp_heads_fwd,
        "fused policy+value MLP heads forward (K4)");
  m.def("mlp_heads_bwd", &mlp_heads_bwd,
        "fused heads dgrad chain + ReLU masks + bias grads (K4 bwd)");
  m.def("embed_mlp_fwd", &embed_mlp_fwd,
        "fused action-embedding MLP forward (K2)");
  m.def("embed_mlp_bwd", &embed_mlp_bwd,
        "fused action-embedding MLP backward (K2 bwd)");
  m.def("mlp_heads_pack_wt", &mlp_heads_pack_wt,
        "one-kernel transposed-weight pack for the heads dgrad");
  m.def("mlp_heads_wgrad", &mlp_heads_wgrad,
        "all six head wgrads (dW = dz^T @ act) in one MFMA launch");
  m.def("lstm_seq_train_fwd", &lstm_seq_train_fwd,
        "grad-carrying whole-sequence LSTM forward (K3 seq, R2D2 train)");
  m.def("lstm_seq_train_bwd", &lstm_seq_train_bwd,
        "whole-sequence LSTM backward (one kernel over the recurrence)");
  m.def("lstm_seq_fwd", &lstm_seq_fwd,
        "whole no-grad LSTM unroll in one kernel (K3 seq / burn-in)");
  m.def("sq_norm", &sq_norm, "squared L2 norm of a flat tensor (K12)");
  m.def("grad_gather", &grad_gather,
        "one-kernel scattered-grad -> flat bucket pack (K12b)");
  m.def("rmsprop_step", &rmsprop_step,
        "fused global-norm-clip + TF-RMSProp update (K12)");
  m.def("rmsprop_step_t", &rmsprop_step_t,
        "RMSProp update with device-tensor lr (hipGraph-safe)");
  m.def("adam_step", &adam_step,
        "fused global-norm-clip + TF-Adam update (K12)");
  m.def("adam_step_t", &adam_step_t,
        "Adam update with device-tensor lr_t (hipGraph-safe)");
  m.def("sq_norm_bf16", &sq_norm_bf16, "squared L2 norm of a bf16 tensor");
  m.def("rmsprop_step_bf16_t", &rmsprop_step_bf16_t,
        "bf16-model/fp32-master RMSProp (K12, mixed precision)");
  m.def("adam_step_bf16_t", &adam_step_bf16_t,
        "bf16-model/fp32-master Adam (K12, mixed precision)");
}
