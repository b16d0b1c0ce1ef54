// Host side of the gfx950 HIP extension: tensor checks, stream plumbing,
// pinned-host registration for the shm trajectory arenas, and the pybind
// surface consumed by torched_impala_amd.ops.
//
// Built natively with hipcc (no hipify, no CUDA-compat layer) — see
// torched_impala_amd/build_ext.py.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include <hip/hip_runtime.h>

#include <vector>

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e),    \
                " at " #expr);                                             \
  } while (0)

// launchers defined in vtrace_kernels.hip
extern "C" {
void launch_vtrace_row_fwd(const float*, const float*, const long*, int, int,
                           float*, float*, float*, float*, hipStream_t);
void launch_vtrace_scan(const float*, const float*, const float*, const float*,
                        const float*, int, int, float, float, float, float,
                        float*, float*, hipStream_t);
void launch_vtrace_loss_bwd(const float*, const float*, const long*,
                            const float*, const float*, const float*, float,
                            float, int, int, float*, float*, hipStream_t);
void launch_lstm_seq_fwd(const float*, const void*, const float*,
                         const float*, const float*, int, int, float*,
                         float*, float*, float*, float*, float*, int, int,
                         hipStream_t);
void launch_lstm_seq_bwd(const float*, const float*, const float*,
                         const float*, const float*, const float*,
                         const float*, const void*, int, int, float*,
                         float*, float*, int, hipStream_t);
void launch_grad_sqnorm(const float*, long, const float*, float*,
                        hipStream_t);
void launch_rmsprop_update(float*, const float*, float*, long, const float*,
                           const float*, const float*, float, float, float,
                           float*, hipStream_t);
void launch_shallow_encoder_fwd(const unsigned char*, const float*,
                                const float*, const float*, const float*,
                                const float*, const float*, int, float*,
                                float*, float*, int, hipStream_t);
void launch_shallow_encoder_bwd(const unsigned char*, const float*,
                                const float*, const float*, const float*,
                                const float*, const float*, int, float*,
                                float*, float*, hipStream_t);
int shallow_encoder_slab_elems();
int linear_splitk_factor(int, int, int);
void launch_unpack_time_major(const void*, void*, int, int, long,
                              hipStream_t);
int linear_splitk_factor_bf16(int, int, int);
void launch_linear_mfma_bf16(const void*, const void*, const float*, int,
                             int, int, int, int, void*, float*, int,
                             hipStream_t);
void launch_linear_mfma(const float*, const float*, const float*,
                        const float*, int, int, int, int, float*, float*,
                        int, hipStream_t);
int launch_deep_conv3x3(const void*, const void*, const float*, const void*,
                        void*, int, int, int, int, int, int, hipStream_t);
void launch_deep_maxpool(const void*, void*, void*, int, int, int, int,
                         hipStream_t);
void launch_deep_maxpool_bwd(const void*, const void*, void*, int, int, int,
                             int, hipStream_t);
int launch_deep_conv3x3_dx(const void*, const void*, const void*, void*,
                           int, int, int, int, int, hipStream_t);
int launch_deep_conv3x3_dw(const void*, const void*, float*, int, int, int,
                           int, int, int, hipStream_t);
int deep_dw_nslab();
void launch_deep_dw_reduce(const float*, float*, int, hipStream_t);
void launch_deep_u8_to_nhwc4(const void*, void*, int, int, int, hipStream_t);
void launch_nchw_f32_to_nhwc_bf16(const void*, void*, int, int, int, int,
                                  hipStream_t);
void launch_nhwc_bf16_to_nchw_f32(const void*, void*, int, int, int, int,
                                  hipStream_t);
void launch_shallow_mfma_fwd(const unsigned char*, const void*,
                             const float*, const void*, const float*,
                             const void*, const float*, void*, void*,
                             void*, int, hipStream_t);
int shallow_mfma_slab();
void launch_shallow_mask_convert(const float*, const void*, void*, long,
                                 hipStream_t);
void launch_shallow_bwd_finalize(const float*, const float*, const float*,
                                 const void*, const void*, const void*, int,
                                 float*, float*, float*, float*, float*,
                                 float*, hipStream_t);
void launch_shallow_mfma_bwd(const unsigned char*, const void*, const void*,
                             const void*, const void*, const void*, void*,
                             void*, float*, float*, float*, int,
                             hipStream_t);
}

namespace {

inline void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// torch-on-ROCm reports device type "cuda" (masquerading), which
// c10::hip::HIPGuard rejects — use a plain hipSetDevice scope instead.
struct DeviceScope {
  int old_ = -1;
  explicit DeviceScope(int dev) {
    HIP_CHECK(hipGetDevice(&old_));
    if (old_ != dev) HIP_CHECK(hipSetDevice(dev));
  }
  ~DeviceScope() {
    if (old_ >= 0) (void)hipSetDevice(old_);
  }
};

// ---------------------------------------------------------------------------
// pinned-host registration (the "pinned-host ring buffer" half of the
// transport: arenas are plain shm; the learner page-locks its mapping so
// hipMemcpyAsync drains are real async DMA).
// ---------------------------------------------------------------------------
void host_register(torch::Tensor t) {
  TORCH_CHECK(!t.is_cuda(), "host_register takes a CPU tensor");
  TORCH_CHECK(t.is_contiguous(), "must be contiguous");
  void* ptr = t.data_ptr();
  size_t bytes = t.numel() * t.element_size();
  hipError_t e = hipHostRegister(ptr, bytes, hipHostRegisterDefault);
  if (e == hipErrorHostMemoryAlreadyRegistered) {
    (void)hipGetLastError();
    return;
  }
  TORCH_CHECK(e == hipSuccess, "hipHostRegister failed: ",
              hipGetErrorString(e));
}

void host_unregister(torch::Tensor t) {
  if (t.is_cuda()) return;
  (void)hipHostUnregister(t.data_ptr());
  (void)hipGetLastError();
}

// ---------------------------------------------------------------------------
// V-trace fused forward: row pass + scan.
// Returns (logp, tgt_lp, row_negent, vs, pg_adv).
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> vtrace_loss_fwd(
    torch::Tensor target_logits, torch::Tensor behavior_logits,
    torch::Tensor actions, torch::Tensor rewards, torch::Tensor discounts,
    torch::Tensor values, torch::Tensor bootstrap_value, double clip_rho,
    double clip_c, double lambda_, double clip_pg_rho) {
  check_f32(target_logits, "target_logits");
  check_f32(behavior_logits, "behavior_logits");
  check_f32(rewards, "rewards");
  check_f32(discounts, "discounts");
  check_f32(values, "values");
  check_f32(bootstrap_value, "bootstrap_value");
  TORCH_CHECK(actions.scalar_type() == torch::kInt64 && actions.is_cuda() &&
                  actions.is_contiguous(), "actions must be cuda int64");
  const int T = target_logits.size(0);
  const int B = target_logits.size(1);
  const int A = target_logits.size(2);
  const int N = T * B;
  TORCH_CHECK(A <= 4096, "action dim too large");
  DeviceScope guard(target_logits.get_device());
  auto opts = target_logits.options();
  auto logp = torch::empty({T, B, A}, opts);
  auto tgt_lp = torch::empty({N}, opts);
  auto log_rho = torch::empty({T, B}, opts);
  auto row_negent = torch::empty({N}, opts);
  auto vs = torch::empty({T, B}, opts);
  auto pg_adv = torch::empty({T, B}, opts);
  hipStream_t stream = current_stream();
  launch_vtrace_row_fwd(
      target_logits.data_ptr<float>(), behavior_logits.data_ptr<float>(),
      actions.data_ptr<long>(), N, A, logp.data_ptr<float>(),
      tgt_lp.data_ptr<float>(), log_rho.data_ptr<float>(),
      row_negent.data_ptr<float>(), stream);
  launch_vtrace_scan(
      log_rho.data_ptr<float>(), discounts.data_ptr<float>(),
      rewards.data_ptr<float>(), values.data_ptr<float>(),
      bootstrap_value.data_ptr<float>(), T, B, (float)clip_rho, (float)clip_c,
      (float)lambda_, (float)clip_pg_rho, vs.data_ptr<float>(),
      pg_adv.data_ptr<float>(), stream);
  HIP_CHECK(hipGetLastError());
  return {logp, tgt_lp, row_negent, vs, pg_adv};
}

// standalone scan (parity tests / reuse): takes log_rhos directly
std::vector<torch::Tensor> vtrace_scan(
    torch::Tensor log_rhos, torch::Tensor discounts, torch::Tensor rewards,
    torch::Tensor values, torch::Tensor bootstrap_value, double clip_rho,
    double clip_c, double lambda_, double clip_pg_rho) {
  check_f32(log_rhos, "log_rhos");
  const int T = log_rhos.size(0);
  const int B = log_rhos.size(1);
  DeviceScope guard(log_rhos.get_device());
  auto vs = torch::empty_like(log_rhos);
  auto pg_adv = torch::empty_like(log_rhos);
  launch_vtrace_scan(
      log_rhos.data_ptr<float>(), discounts.data_ptr<float>(),
      rewards.data_ptr<float>(), values.data_ptr<float>(),
      bootstrap_value.data_ptr<float>(), T, B, (float)clip_rho, (float)clip_c,
      (float)lambda_, (float)clip_pg_rho, vs.data_ptr<float>(),
      pg_adv.data_ptr<float>(), current_stream());
  HIP_CHECK(hipGetLastError());
  return {vs, pg_adv};
}

std::vector<torch::Tensor> vtrace_loss_bwd(
    torch::Tensor logp, torch::Tensor row_negent, torch::Tensor actions,
    torch::Tensor pg_adv, torch::Tensor vdiff, torch::Tensor grad_scale,
    double baseline_cost, double entropy_cost) {
  check_f32(logp, "logp");
  check_f32(grad_scale, "grad_scale");
  const int T = logp.size(0);
  const int B = logp.size(1);
  const int A = logp.size(2);
  const int N = T * B;
  DeviceScope guard(logp.get_device());
  auto dlogits = torch::empty({T, B, A}, logp.options());
  auto dvalues = torch::empty({T, B}, logp.options());
  launch_vtrace_loss_bwd(
      logp.data_ptr<float>(), row_negent.data_ptr<float>(),
      actions.data_ptr<long>(), pg_adv.data_ptr<float>(),
      vdiff.data_ptr<float>(), grad_scale.data_ptr<float>(),
      (float)baseline_cost, (float)entropy_cost, N, A,
      dlogits.data_ptr<float>(), dvalues.data_ptr<float>(), current_stream());
  HIP_CHECK(hipGetLastError());
  return {dlogits, dvalues};
}

// ---------------------------------------------------------------------------
// LSTM core sequence (hidden fixed at 256 — the IMPALA core).
// fwd: (Xg (T,B,1024), W_hhT (256,1024), notdone (T,B), h0, c0 (B,256),
//       save_for_bwd) -> (H, act, C, Hp, hT, cT)
// bwd: (dH, dhT, dcT, act, C, c0, notdone, W_hh (1024,256))
//       -> (dG (T,B,1024), dh0, dc0)
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> lstm_seq_fwd(
    torch::Tensor Xg, torch::Tensor W_hhT, torch::Tensor notdone,
    torch::Tensor h0, torch::Tensor c0, bool save_for_bwd) {
  check_f32(Xg, "Xg");
  const bool wbf16 = W_hhT.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(wbf16 || W_hhT.scalar_type() == torch::kFloat32,
              "W_hhT must be fp32 or bf16");
  TORCH_CHECK(W_hhT.is_cuda() && W_hhT.is_contiguous(), "W_hhT layout");
  check_f32(notdone, "notdone");
  check_f32(h0, "h0");
  check_f32(c0, "c0");
  const int T = Xg.size(0);
  const int B = Xg.size(1);
  TORCH_CHECK(Xg.size(2) == 1024 && W_hhT.size(0) == 256 &&
                  W_hhT.size(1) == 1024 && h0.size(1) == 256,
              "lstm kernels are specialized for hidden=256");
  DeviceScope guard(Xg.get_device());
  auto opts = Xg.options();
  auto H = torch::empty({T, B, 256}, opts);
  auto hT = torch::empty({B, 256}, opts);
  auto cT = torch::empty({B, 256}, opts);
  torch::Tensor act, C, Hp;
  if (save_for_bwd) {
    act = torch::empty({T, B, 1024}, opts);
    C = torch::empty({T, B, 256}, opts);
    Hp = torch::empty({T, B, 256}, opts);
  } else {
    act = torch::empty({1}, opts);
    C = torch::empty({1}, opts);
    Hp = torch::empty({1}, opts);
  }
  launch_lstm_seq_fwd(Xg.data_ptr<float>(), W_hhT.data_ptr(),
                      notdone.data_ptr<float>(), h0.data_ptr<float>(),
                      c0.data_ptr<float>(), T, B, H.data_ptr<float>(),
                      act.data_ptr<float>(), C.data_ptr<float>(),
                      Hp.data_ptr<float>(), hT.data_ptr<float>(),
                      cT.data_ptr<float>(), save_for_bwd ? 1 : 0,
                      wbf16 ? 1 : 0, current_stream());
  HIP_CHECK(hipGetLastError());
  return {H, act, C, Hp, hT, cT};
}

std::vector<torch::Tensor> lstm_seq_bwd(
    torch::Tensor dH, torch::Tensor dhT, torch::Tensor dcT,
    torch::Tensor act, torch::Tensor C, torch::Tensor c0,
    torch::Tensor notdone, torch::Tensor W_hh) {
  check_f32(dH, "dH");
  check_f32(act, "act");
  const bool wbf16 = W_hh.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(wbf16 || W_hh.scalar_type() == torch::kFloat32,
              "W_hh must be fp32 or bf16");
  TORCH_CHECK(W_hh.is_cuda() && W_hh.is_contiguous(), "W_hh layout");
  const int T = dH.size(0);
  const int B = dH.size(1);
  TORCH_CHECK(dH.size(2) == 256 && W_hh.size(0) == 1024 && W_hh.size(1) == 256,
              "lstm kernels are specialized for hidden=256");
  DeviceScope guard(dH.get_device());
  auto opts = dH.options();
  auto dG = torch::empty({T, B, 1024}, opts);
  auto dh0 = torch::empty({B, 256}, opts);
  auto dc0 = torch::empty({B, 256}, opts);
  launch_lstm_seq_bwd(dH.data_ptr<float>(), dhT.data_ptr<float>(),
                      dcT.data_ptr<float>(), act.data_ptr<float>(),
                      C.data_ptr<float>(), c0.data_ptr<float>(),
                      notdone.data_ptr<float>(), W_hh.data_ptr(), T, B,
                      dG.data_ptr<float>(), dh0.data_ptr<float>(),
                      dc0.data_ptr<float>(), wbf16 ? 1 : 0,
                      current_stream());
  HIP_CHECK(hipGetLastError());
  return {dG, dh0, dc0};
}

// ---------------------------------------------------------------------------
// fused flat-arena optimizer: clip-by-global-norm + RMSProp in 2 kernels.
// norm_sq must be zeroed by the caller (a cheap zero_ on the stream).
// lr and scale are device scalars so the whole step is graph-capturable.
// ---------------------------------------------------------------------------
void fused_clip_rmsprop(torch::Tensor flat_p, torch::Tensor flat_g,
                        torch::Tensor flat_sqavg, torch::Tensor lr,
                        torch::Tensor norm_sq, torch::Tensor scale,
                        double max_norm, double alpha, double eps,
                        c10::optional<torch::Tensor> skip_count) {
  float* skip_p = nullptr;
  if (skip_count.has_value()) {
    check_f32(*skip_count, "skip_count");
    skip_p = skip_count->data_ptr<float>();
  }
  check_f32(flat_p, "flat_p");
  check_f32(flat_g, "flat_g");
  check_f32(flat_sqavg, "flat_sqavg");
  check_f32(lr, "lr");
  check_f32(norm_sq, "norm_sq");
  check_f32(scale, "scale");
  const long n = flat_p.numel();
  DeviceScope guard(flat_p.get_device());
  hipStream_t stream = current_stream();
  if (max_norm > 0.0) {
    launch_grad_sqnorm(flat_g.data_ptr<float>(), n, scale.data_ptr<float>(),
                       norm_sq.data_ptr<float>(), stream);
  }
  launch_rmsprop_update(flat_p.data_ptr<float>(), flat_g.data_ptr<float>(),
                        flat_sqavg.data_ptr<float>(), n,
                        lr.data_ptr<float>(), norm_sq.data_ptr<float>(),
                        scale.data_ptr<float>(), (float)max_norm,
                        (float)alpha, (float)eps, skip_p, stream);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// fused shallow-encoder convolutions (one workgroup per 84x84x4 image;
// conv1->conv2->conv3 LDS-resident; see conv_kernels.hip).
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> shallow_encoder_fwd(
    torch::Tensor frames, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    bool save_for_bwd) {
  TORCH_CHECK(frames.scalar_type() == torch::kUInt8 && frames.is_cuda() &&
                  frames.is_contiguous(), "frames must be cuda u8 contiguous");
  TORCH_CHECK(frames.size(1) == 4 && frames.size(2) == 84 &&
                  frames.size(3) == 84, "specialized for 4x84x84");
  check_f32(w1, "w1");
  check_f32(w2, "w2");
  check_f32(w3, "w3");
  const int N = frames.size(0);
  DeviceScope guard(frames.get_device());
  auto opts = w1.options();
  torch::Tensor a1, a2;
  if (save_for_bwd) {
    a1 = torch::empty({N, 16, 20, 20}, opts);
    a2 = torch::empty({N, 32, 9, 9}, opts);
  } else {
    a1 = torch::empty({1}, opts);
    a2 = torch::empty({1}, opts);
  }
  auto a3 = torch::empty({N, 1568}, opts);
  launch_shallow_encoder_fwd(
      frames.data_ptr<unsigned char>(), w1.data_ptr<float>(),
      b1.data_ptr<float>(), w2.data_ptr<float>(), b2.data_ptr<float>(),
      w3.data_ptr<float>(), b3.data_ptr<float>(), N, a1.data_ptr<float>(),
      a2.data_ptr<float>(), a3.data_ptr<float>(), save_for_bwd ? 1 : 0,
      current_stream());
  HIP_CHECK(hipGetLastError());
  return {a1, a2, a3};
}

torch::Tensor shallow_encoder_bwd(
    torch::Tensor frames, torch::Tensor a1, torch::Tensor a2,
    torch::Tensor a3, torch::Tensor d_a3, torch::Tensor w2,
    torch::Tensor w3) {
  const int N = frames.size(0);
  check_f32(d_a3, "d_a3");
  DeviceScope guard(frames.get_device());
  auto slab = torch::empty({N, (long)shallow_encoder_slab_elems()},
                           d_a3.options());
  auto d2_tmp = torch::empty({N, 32 * 9 * 9}, d_a3.options());
  auto d1_tmp = torch::empty({N, 16 * 20 * 20}, d_a3.options());
  launch_shallow_encoder_bwd(
      frames.data_ptr<unsigned char>(), a1.data_ptr<float>(),
      a2.data_ptr<float>(), a3.data_ptr<float>(), d_a3.data_ptr<float>(),
      w2.data_ptr<float>(), w3.data_ptr<float>(), N,
      slab.data_ptr<float>(), d2_tmp.data_ptr<float>(),
      d1_tmp.data_ptr<float>(), current_stream());
  HIP_CHECK(hipGetLastError());
  return slab;
}

// ---------------------------------------------------------------------------
// MFMA linear: C = act(A @ B^T + bias). mode 0 plain / 1 relu /
// 2 A-masked-by-(ymask>0) (fused ReLU backward for dX).
// ---------------------------------------------------------------------------
torch::Tensor linear_mfma(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> ymask, int64_t mode) {
  check_f32(A, "A");
  check_f32(B, "B");
  const int M = A.size(0);
  const int K = A.size(1);
  const int N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  DeviceScope guard(A.get_device());
  auto C = torch::empty({M, N}, A.options());
  const float* bias_p = nullptr;
  const float* ymask_p = nullptr;
  if (bias.has_value()) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  if (ymask.has_value()) {
    check_f32(*ymask, "ymask");
    ymask_p = ymask->data_ptr<float>();
  }
  const int S = linear_splitk_factor(M, N, K);
  torch::Tensor ws;
  float* ws_p = nullptr;
  if (S > 1) {
    ws = torch::empty({(int64_t)S, M, N}, A.options());
    ws_p = ws.data_ptr<float>();
  }
  launch_linear_mfma(A.data_ptr<float>(), B.data_ptr<float>(), bias_p,
                     ymask_p, M, N, K, (int)mode, C.data_ptr<float>(),
                     ws_p, S, current_stream());
  HIP_CHECK(hipGetLastError());
  return C;
}

// slot-major staging (K, T1, ...) -> time-major (T1, K*..., ...) as one
// block-permutation kernel (SURVEY.md §2c rollout_pack/unpack).
torch::Tensor unpack_time_major(torch::Tensor src, int64_t lead_keep) {
  // src (K, T1, *rest) contiguous -> out (T1, K*rest[0], *rest[1:])
  // when lead_keep == 1 the first rest dim is merged with K (the
  // envs-per-slot dim n), matching Batcher._to_time_major.
  TORCH_CHECK(src.is_cuda() && src.is_contiguous(), "src layout");
  const int K = src.size(0);
  const int T1 = src.size(1);
  DeviceScope guard(src.get_device());
  std::vector<int64_t> oshape;
  oshape.push_back(T1);
  int64_t inner_elems = 1;
  for (int i = 2; i < src.dim(); ++i) inner_elems *= src.size(i);
  if (lead_keep == 1 && src.dim() >= 3) {
    oshape.push_back((int64_t)K * src.size(2));
    for (int i = 3; i < src.dim(); ++i) oshape.push_back(src.size(i));
  } else {
    oshape.push_back((int64_t)K);
    for (int i = 2; i < src.dim(); ++i) oshape.push_back(src.size(i));
  }
  auto out = torch::empty(oshape, src.options());
  const long inner_bytes = inner_elems * src.element_size();
  launch_unpack_time_major(src.data_ptr(), out.data_ptr(), K, T1,
                           inner_bytes, current_stream());
  HIP_CHECK(hipGetLastError());
  return out;
}

// bf16-input MFMA GEMM (in-tree; see profiles/r2_nan_rootcause.md):
// C = act(A @ B^T + bias), A (M,K) bf16, B (N,K) bf16, C fp32 or bf16.
torch::Tensor linear_bf16(torch::Tensor A, torch::Tensor B,
                          c10::optional<torch::Tensor> bias, int64_t mode,
                          bool out_bf16) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16 &&
                  A.is_contiguous(), "A must be cuda bf16 contiguous");
  TORCH_CHECK(B.is_cuda() && B.scalar_type() == torch::kBFloat16 &&
                  B.is_contiguous(), "B must be cuda bf16 contiguous");
  const int M = A.size(0);
  const int K = A.size(1);
  const int N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "K mismatch");
  DeviceScope guard(A.get_device());
  auto C = torch::empty({M, N}, A.options().dtype(
                                    out_bf16 ? torch::kBFloat16
                                             : torch::kFloat32));
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    check_f32(*bias, "bias");
    bias_p = bias->data_ptr<float>();
  }
  const int S = linear_splitk_factor_bf16(M, N, K);
  torch::Tensor ws;
  float* ws_p = nullptr;
  if (S > 1) {
    ws = torch::empty({(int64_t)S, M, N},
                      A.options().dtype(torch::kFloat32));
    ws_p = ws.data_ptr<float>();
  }
  launch_linear_mfma_bf16(A.data_ptr(), B.data_ptr(), bias_p, M, N, K,
                          (int)mode, out_bf16 ? 1 : 0, C.data_ptr(), ws_p,
                          S, current_stream());
  HIP_CHECK(hipGetLastError());
  return C;
}

// ---------------------------------------------------------------------------
// deep ResNet NHWC bf16 kernels (deep_kernels.hip; validated by deep_probe)
// ---------------------------------------------------------------------------
inline void check_bf16_nhwc(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// x [N][H][W][CIN] bf16, w [COUT][3][3][CIN] bf16, b [COUT] f32,
// res optional [N][H][W][COUT] bf16 -> y [N][H][W][COUT] bf16
torch::Tensor deep_conv3x3(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                           c10::optional<torch::Tensor> res, bool relu_in) {
  check_bf16_nhwc(x, "x");
  check_bf16_nhwc(w, "w");
  check_f32(b, "b");
  const int N = x.size(0), H = x.size(1), W = x.size(2), cin = x.size(3);
  const int cout = w.size(0);
  TORCH_CHECK(w.dim() == 4 && w.size(1) == 3 && w.size(2) == 3 &&
                  w.size(3) == cin, "w must be [cout][3][3][cin]");
  TORCH_CHECK(b.size(0) == cout, "bias size");
  DeviceScope guard(x.get_device());
  auto y = torch::empty({N, H, W, cout}, x.options());
  const void* res_p = nullptr;
  if (res.has_value()) {
    check_bf16_nhwc(*res, "res");
    TORCH_CHECK(res->size(3) == cout, "res channels");
    res_p = res->data_ptr();
  }
  int rc = launch_deep_conv3x3(x.data_ptr(), w.data_ptr(),
                               b.data_ptr<float>(), res_p, y.data_ptr(), N,
                               H, W, cin, cout, relu_in ? 1 : 0,
                               current_stream());
  TORCH_CHECK(rc == 0, "unsupported cin/cout ", cin, "/", cout);
  HIP_CHECK(hipGetLastError());
  return y;
}

std::vector<torch::Tensor> deep_maxpool(torch::Tensor x, bool want_argmax) {
  check_bf16_nhwc(x, "x");
  TORCH_CHECK(x.size(3) % 8 == 0, "maxpool kernels are 8-channel vectorised");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  DeviceScope guard(x.get_device());
  auto y = torch::empty({N, (H + 1) / 2, (W + 1) / 2, C}, x.options());
  torch::Tensor am;
  void* am_p = nullptr;
  if (want_argmax) {
    am = torch::empty({N, (H + 1) / 2, (W + 1) / 2, C},
                      x.options().dtype(torch::kUInt8));
    am_p = am.data_ptr();
  } else {
    am = torch::empty({0}, x.options().dtype(torch::kUInt8));
  }
  launch_deep_maxpool(x.data_ptr(), y.data_ptr(), am_p, N, H, W, C,
                      current_stream());
  HIP_CHECK(hipGetLastError());
  return {y, am};
}

torch::Tensor deep_maxpool_bwd(torch::Tensor dy, torch::Tensor argmax,
                               int64_t H, int64_t W) {
  check_bf16_nhwc(dy, "dy");
  TORCH_CHECK(argmax.scalar_type() == torch::kUInt8 &&
                  argmax.is_contiguous(), "argmax must be u8 contiguous");
  const int N = dy.size(0), C = dy.size(3);
  DeviceScope guard(dy.get_device());
  auto dx = torch::empty({N, H, W, C}, dy.options());
  launch_deep_maxpool_bwd(dy.data_ptr(), argmax.data_ptr(), dx.data_ptr(),
                          N, (int)H, (int)W, C, current_stream());
  HIP_CHECK(hipGetLastError());
  return dx;
}

// dX: wt pre-repacked [cin_fwd][3][3][cout_fwd] (rot180 + transpose);
// xmask (optional) = the forward input, for the fused ReLU backward.
torch::Tensor deep_conv3x3_dx(torch::Tensor dy, torch::Tensor wt,
                              c10::optional<torch::Tensor> xmask) {
  check_bf16_nhwc(dy, "dy");
  check_bf16_nhwc(wt, "wt");
  const int N = dy.size(0), H = dy.size(1), W = dy.size(2);
  const int cout_fwd = dy.size(3), cin_fwd = wt.size(0);
  TORCH_CHECK(wt.size(3) == cout_fwd, "wt must be [cin][3][3][cout]");
  DeviceScope guard(dy.get_device());
  auto dx = torch::empty({N, H, W, cin_fwd}, dy.options());
  const void* mask_p = nullptr;
  if (xmask.has_value()) {
    check_bf16_nhwc(*xmask, "xmask");
    mask_p = xmask->data_ptr();
  }
  int rc = launch_deep_conv3x3_dx(dy.data_ptr(), wt.data_ptr(), mask_p,
                                  dx.data_ptr(), N, H, W, cout_fwd, cin_fwd,
                                  current_stream());
  TORCH_CHECK(rc == 0, "unsupported dx cout/cin ", cout_fwd, "/", cin_fwd);
  HIP_CHECK(hipGetLastError());
  return dx;
}

// dW in the kernel's [cout][3][3][cin] layout, fp32.  The kernel
// scatters into deep_dw_nslab() slab copies (atomic-contention split);
// the sum over slabs happens here as one torch reduction.
torch::Tensor deep_conv3x3_dw(torch::Tensor x, torch::Tensor dy,
                              bool relu_in) {
  check_bf16_nhwc(x, "x");
  check_bf16_nhwc(dy, "dy");
  const int N = x.size(0), H = x.size(1), W = x.size(2), cin = x.size(3);
  const int cout = dy.size(3);
  DeviceScope guard(x.get_device());
  auto slab = torch::zeros({deep_dw_nslab(), cout, 3, 3, cin},
                           x.options().dtype(torch::kFloat32));
  int rc = launch_deep_conv3x3_dw(x.data_ptr(), dy.data_ptr(),
                                  slab.data_ptr<float>(), N, H, W, cin,
                                  cout, relu_in ? 1 : 0, current_stream());
  TORCH_CHECK(rc == 0, "unsupported dw cin/cout ", cin, "/", cout);
  auto out = torch::empty({cout, 3, 3, cin},
                          x.options().dtype(torch::kFloat32));
  launch_deep_dw_reduce(slab.data_ptr<float>(), out.data_ptr<float>(),
                        cout * 9 * cin, current_stream());
  HIP_CHECK(hipGetLastError());
  return out;
}

// u8 [N][3][H][W] -> bf16 [N][H][W][4] with /255 and a zero pad channel
torch::Tensor deep_u8_to_nhwc4(torch::Tensor frames) {
  TORCH_CHECK(frames.is_cuda() && frames.scalar_type() == torch::kUInt8 &&
                  frames.is_contiguous() && frames.size(1) == 3,
              "frames must be cuda u8 [N][3][H][W] contiguous");
  const int N = frames.size(0), H = frames.size(2), W = frames.size(3);
  DeviceScope guard(frames.get_device());
  auto y = torch::empty({N, H, W, 4},
                        frames.options().dtype(torch::kBFloat16));
  launch_deep_u8_to_nhwc4(frames.data_ptr(), y.data_ptr(), N, H, W,
                          current_stream());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor nchw_to_nhwc_bf16(torch::Tensor x) {
  check_f32(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  DeviceScope guard(x.get_device());
  auto y = torch::empty({N, H, W, C}, x.options().dtype(torch::kBFloat16));
  launch_nchw_f32_to_nhwc_bf16(x.data_ptr(), y.data_ptr(), N, C, H, W,
                               current_stream());
  HIP_CHECK(hipGetLastError());
  return y;
}

torch::Tensor nhwc_bf16_to_nchw(torch::Tensor x) {
  check_bf16_nhwc(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  DeviceScope guard(x.get_device());
  auto y = torch::empty({N, C, H, W}, x.options().dtype(torch::kFloat32));
  launch_nhwc_bf16_to_nchw_f32(x.data_ptr(), y.data_ptr(), N, C, H, W,
                               current_stream());
  HIP_CHECK(hipGetLastError());
  return y;
}

// ---------------------------------------------------------------------------
// MFMA bf16 shallow encoder (shallow_mfma.hip / shallow_mfma_bwd.hip)
// ---------------------------------------------------------------------------
std::vector<torch::Tensor> shallow_mfma_fwd(torch::Tensor frames,
                                            torch::Tensor w1p,
                                            torch::Tensor b1,
                                            torch::Tensor w2p,
                                            torch::Tensor b2,
                                            torch::Tensor w3p,
                                            torch::Tensor b3) {
  TORCH_CHECK(frames.scalar_type() == torch::kUInt8 && frames.is_cuda() &&
                  frames.is_contiguous() && frames.size(1) == 4 &&
                  frames.size(2) == 84 && frames.size(3) == 84,
              "frames must be cuda u8 (N,4,84,84)");
  check_bf16_nhwc(w1p, "w1p");
  check_bf16_nhwc(w2p, "w2p");
  check_bf16_nhwc(w3p, "w3p");
  check_f32(b1, "b1");
  check_f32(b2, "b2");
  check_f32(b3, "b3");
  const int N = frames.size(0);
  DeviceScope guard(frames.get_device());
  auto opts = frames.options().dtype(torch::kBFloat16);
  auto a1 = torch::empty({N, 20, 20, 16}, opts);
  auto a2 = torch::empty({N, 9, 9, 32}, opts);
  auto a3 = torch::empty({N, 1568}, opts);
  launch_shallow_mfma_fwd(frames.data_ptr<unsigned char>(), w1p.data_ptr(),
                          b1.data_ptr<float>(), w2p.data_ptr(),
                          b2.data_ptr<float>(), w3p.data_ptr(),
                          b3.data_ptr<float>(), a1.data_ptr(),
                          a2.data_ptr(), a3.data_ptr(), N,
                          current_stream());
  HIP_CHECK(hipGetLastError());
  return {a1, a2, a3};
}

torch::Tensor shallow_mask_convert(torch::Tensor d, torch::Tensor act) {
  check_f32(d, "d");
  check_bf16_nhwc(act, "act");
  DeviceScope guard(d.get_device());
  auto o = torch::empty_like(act);
  launch_shallow_mask_convert(d.data_ptr<float>(), act.data_ptr(),
                              o.data_ptr(), d.numel(), current_stream());
  HIP_CHECK(hipGetLastError());
  return o;
}

// returns (da1, da2, dw1k, dw2k, dw3k): da* bf16 (ReLU-masked), dw*k
// fp32 in k-major [M][F] layout (python reshapes to torch layout).
std::vector<torch::Tensor> shallow_mfma_bwd(torch::Tensor frames,
                                            torch::Tensor a1,
                                            torch::Tensor a2,
                                            torch::Tensor dz3,
                                            torch::Tensor wr3,
                                            torch::Tensor wr2) {
  check_bf16_nhwc(a1, "a1");
  check_bf16_nhwc(a2, "a2");
  check_bf16_nhwc(dz3, "dz3");
  check_bf16_nhwc(wr3, "wr3");
  check_bf16_nhwc(wr2, "wr2");
  const int N = frames.size(0);
  DeviceScope guard(frames.get_device());
  const int SL = shallow_mfma_slab();
  auto fopts = frames.options().dtype(torch::kFloat32);
  auto da1 = torch::empty_like(a1);
  auto da2 = torch::empty_like(a2);
  auto s1 = torch::zeros({SL, 256, 16}, fopts);
  auto s2 = torch::zeros({SL, 256, 32}, fopts);
  auto s3 = torch::zeros({SL, 288, 32}, fopts);
  launch_shallow_mfma_bwd(frames.data_ptr<unsigned char>(), a1.data_ptr(),
                          a2.data_ptr(), dz3.data_ptr(), wr3.data_ptr(),
                          wr2.data_ptr(), da1.data_ptr(), da2.data_ptr(),
                          s1.data_ptr<float>(), s2.data_ptr<float>(),
                          s3.data_ptr<float>(), N, current_stream());
  // fused finalize: slab sums -> torch weight layouts + bias column sums
  // (one launch; replaces ~12 eager ATen ops — profiles/r6 PMC)
  auto dw1 = torch::zeros({16, 4, 8, 8}, fopts);
  auto dw2 = torch::zeros({32, 16, 4, 4}, fopts);
  auto dw3 = torch::zeros({32, 32, 3, 3}, fopts);
  auto db = torch::zeros({80}, fopts);
  float* dbp = db.data_ptr<float>();
  launch_shallow_bwd_finalize(
      s1.data_ptr<float>(), s2.data_ptr<float>(), s3.data_ptr<float>(),
      da1.data_ptr(), da2.data_ptr(), dz3.data_ptr(), N,
      dw1.data_ptr<float>(), dw2.data_ptr<float>(), dw3.data_ptr<float>(),
      dbp, dbp + 16, dbp + 48, current_stream());
  HIP_CHECK(hipGetLastError());
  return {da1, da2, dw1, dw2, dw3, db.narrow(0, 0, 16),
          db.narrow(0, 16, 32), db.narrow(0, 48, 32)};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "torched_impala_amd gfx950 HIP kernels";
  m.def("host_register", &host_register, "page-lock a CPU tensor (hipHostRegister)");
  m.def("host_unregister", &host_unregister);
  m.def("vtrace_loss_fwd", &vtrace_loss_fwd);
  m.def("vtrace_scan", &vtrace_scan);
  m.def("vtrace_loss_bwd", &vtrace_loss_bwd);
  m.def("lstm_seq_fwd", &lstm_seq_fwd);
  m.def("lstm_seq_bwd", &lstm_seq_bwd);
  m.def("fused_clip_rmsprop", &fused_clip_rmsprop, py::arg("flat_p"),
        py::arg("flat_g"), py::arg("flat_sqavg"), py::arg("lr"),
        py::arg("norm_sq"), py::arg("scale"), py::arg("max_norm"),
        py::arg("alpha"), py::arg("eps"),
        py::arg("skip_count") = py::none());
  m.def("shallow_encoder_fwd", &shallow_encoder_fwd);
  m.def("shallow_encoder_bwd", &shallow_encoder_bwd);
  m.def("linear_mfma", &linear_mfma, py::arg("A"), py::arg("B"),
        py::arg("bias") = py::none(), py::arg("ymask") = py::none(),
        py::arg("mode") = 0);
  m.def("unpack_time_major", &unpack_time_major, py::arg("src"),
        py::arg("lead_keep") = 1);
  m.def("linear_bf16", &linear_bf16, py::arg("A"), py::arg("B"),
        py::arg("bias") = py::none(), py::arg("mode") = 0,
        py::arg("out_bf16") = false);
  m.def("deep_conv3x3", &deep_conv3x3, py::arg("x"), py::arg("w"),
        py::arg("b"), py::arg("res") = py::none(),
        py::arg("relu_in") = false);
  m.def("deep_maxpool", &deep_maxpool, py::arg("x"),
        py::arg("want_argmax") = false);
  m.def("deep_maxpool_bwd", &deep_maxpool_bwd);
  m.def("deep_conv3x3_dx", &deep_conv3x3_dx, py::arg("dy"), py::arg("wt"),
        py::arg("xmask") = py::none());
  m.def("deep_conv3x3_dw", &deep_conv3x3_dw, py::arg("x"), py::arg("dy"),
        py::arg("relu_in") = false);
  m.def("deep_u8_to_nhwc4", &deep_u8_to_nhwc4);
  m.def("nchw_to_nhwc_bf16", &nchw_to_nhwc_bf16);
  m.def("nhwc_bf16_to_nchw", &nhwc_bf16_to_nchw);
  m.def("shallow_mfma_fwd", &shallow_mfma_fwd);
  m.def("shallow_mask_convert", &shallow_mask_convert);
  m.def("shallow_mfma_bwd", &shallow_mfma_bwd);
}
