// Host side of the gfx950 HIP extension: tensor checks, stream plumbing,
// pinned-host registration for the shm trajectory arenas, and the pybind
// surface consumed by torched_impala_amd.ops.
//
// Built natively with hipcc (no hipify, no CUDA-compat layer) — see
// torched_impala_amd/build_ext.py.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <c10/hip/HIPGuard.h>
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
                        const float*, int, int, float, float, float, float*,
                        float*, hipStream_t);
void launch_vtrace_loss_bwd(const float*, const float*, const long*,
                            const float*, const float*, const float*, float,
                            float, int, int, float*, float*, hipStream_t);
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
    double clip_c, double lambda_) {
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
  c10::hip::HIPGuard guard(target_logits.device());
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
      (float)lambda_, vs.data_ptr<float>(), pg_adv.data_ptr<float>(), stream);
  HIP_CHECK(hipGetLastError());
  return {logp, tgt_lp, row_negent, vs, pg_adv};
}

// standalone scan (parity tests / reuse): takes log_rhos directly
std::vector<torch::Tensor> vtrace_scan(
    torch::Tensor log_rhos, torch::Tensor discounts, torch::Tensor rewards,
    torch::Tensor values, torch::Tensor bootstrap_value, double clip_rho,
    double clip_c, double lambda_) {
  check_f32(log_rhos, "log_rhos");
  const int T = log_rhos.size(0);
  const int B = log_rhos.size(1);
  c10::hip::HIPGuard guard(log_rhos.device());
  auto vs = torch::empty_like(log_rhos);
  auto pg_adv = torch::empty_like(log_rhos);
  launch_vtrace_scan(
      log_rhos.data_ptr<float>(), discounts.data_ptr<float>(),
      rewards.data_ptr<float>(), values.data_ptr<float>(),
      bootstrap_value.data_ptr<float>(), T, B, (float)clip_rho, (float)clip_c,
      (float)lambda_, vs.data_ptr<float>(), pg_adv.data_ptr<float>(),
      current_stream());
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
  c10::hip::HIPGuard guard(logp.device());
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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "torched_impala_amd gfx950 HIP kernels";
  m.def("host_register", &host_register, "page-lock a CPU tensor (hipHostRegister)");
  m.def("host_unregister", &host_unregister);
  m.def("vtrace_loss_fwd", &vtrace_loss_fwd);
  m.def("vtrace_scan", &vtrace_scan);
  m.def("vtrace_loss_bwd", &vtrace_loss_bwd);
}
