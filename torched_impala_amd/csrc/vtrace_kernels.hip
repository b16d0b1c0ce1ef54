// gfx950 HIP kernels: V-trace forward (row pass + LDS-staged time-major
// reverse scan) and the fused PG/value/entropy loss backward.
//
// BASELINE.json:5: "V-trace clipped-importance return/advantage scan ...
// an LDS-staged time-major scan for V-trace ... fused policy-gradient/
// value/entropy loss" as hand-written CDNA4 kernels. The eager reference
// this replaces is a T-step Python loop of (B,)-sized elementwise torch
// ops (~6T kernel launches, launch-latency-bound — SURVEY.md §3d); here
// the whole V-trace step is 2 kernels and the loss backward is 1.
//
// Shapes are time-major (T, B) / (T*B, A) fp32; actions int64.
// Wave size is 64 (CDNA4); block sizes are multiples of 64.

#include <hip/hip_runtime.h>

#define DEVINL __device__ __forceinline__

namespace impala {

// ---------------------------------------------------------------------------
// Kernel 1: per-(t,b) row pass over the action dimension.
// One thread per row (A <= ~64 actions; rows are N = T*B, grid-stride).
// Computes, per row i:
//   logp[i,:]      log-softmax of target logits
//   tgt_lp[i]      logp[i, a_i]
//   log_rho[i]     logp[i, a_i] - behavior_logp[i, a_i]
//   row_negent[i]  sum_j p_ij * logp_ij      (negative entropy)
// Two passes over A from L2 (rows are tiny and L2/L3-resident right
// after the learner forward writes them).
// ---------------------------------------------------------------------------
__global__ void vtrace_row_fwd(
    const float* __restrict__ tgt, const float* __restrict__ beh,
    const long* __restrict__ actions, int N, int A,
    float* __restrict__ logp_out, float* __restrict__ tgt_lp,
    float* __restrict__ log_rho, float* __restrict__ row_negent) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += gridDim.x * blockDim.x) {
    const float* trow = tgt + (size_t)i * A;
    const float* brow = beh + (size_t)i * A;
    float tmax = -INFINITY, bmax = -INFINITY;
    for (int j = 0; j < A; ++j) {
      tmax = fmaxf(tmax, trow[j]);
      bmax = fmaxf(bmax, brow[j]);
    }
    float tsum = 0.f, bsum = 0.f;
    for (int j = 0; j < A; ++j) {
      tsum += __expf(trow[j] - tmax);
      bsum += __expf(brow[j] - bmax);
    }
    const float tlse = tmax + __logf(tsum);
    const float blse = bmax + __logf(bsum);
    float negent = 0.f;
    float* lrow = logp_out + (size_t)i * A;
    for (int j = 0; j < A; ++j) {
      float lp = trow[j] - tlse;
      lrow[j] = lp;
      negent += __expf(lp) * lp;
    }
    const int a = (int)actions[i];
    const float t_lp = trow[a] - tlse;
    tgt_lp[i] = t_lp;
    log_rho[i] = t_lp - (brow[a] - blse);
    row_negent[i] = negent;
  }
}

// ---------------------------------------------------------------------------
// Kernel 2: the LDS-staged time-major reverse scan.
// One wave (64 lanes) per 64-column stripe of B; each lane owns a batch
// column. All T rows of {log_rho, discount, reward, value} for the
// stripe are staged into LDS with coalesced row loads up-front (every
// load in flight at once), then the serial T-recursion runs out of LDS:
//   A_t = delta_t + disc_t * c_t * A_{t+1};  vs_t = V_t + A_t
//   pg_adv_t = min(rho_t, clip_pg_rho) * (r_t + disc_t * vs_{t+1} - V_t)
// LDS: 4 arrays * T * 64 * 4 B  (T=100 -> 100 KiB; fits 160 KiB/CU).
// ---------------------------------------------------------------------------
__global__ void vtrace_scan_lds(
    const float* __restrict__ log_rho, const float* __restrict__ discounts,
    const float* __restrict__ rewards, const float* __restrict__ values,
    const float* __restrict__ bootstrap, int T, int B,
    float clip_rho, float clip_c, float lambda_, float clip_pg_rho,
    float* __restrict__ vs, float* __restrict__ pg_adv) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* s_rho = smem;              // [T][64]
  float* s_dis = s_rho + (size_t)T * 64;
  float* s_rew = s_dis + (size_t)T * 64;
  float* s_val = s_rew + (size_t)T * 64;

  const int lane = threadIdx.x;       // 64 threads per block
  const int b0 = blockIdx.x * 64;     // column stripe
  const int b = b0 + lane;
  const bool active = b < B;

  // stage: row-major coalesced, all loads issued before any use
  for (int t = 0; t < T; ++t) {
    const size_t idx = (size_t)t * B + b;
    if (active) {
      s_rho[t * 64 + lane] = log_rho[idx];
      s_dis[t * 64 + lane] = discounts[idx];
      s_rew[t * 64 + lane] = rewards[idx];
      s_val[t * 64 + lane] = values[idx];
    }
  }
  __syncthreads();
  if (!active) return;

  float acc = 0.f;
  float vs_next = bootstrap[b];
  float v_next = vs_next;  // V(x_{t+1}); bootstrap doubles as V(x_T)
  for (int t = T - 1; t >= 0; --t) {
    const float rho = __expf(s_rho[t * 64 + lane]);
    const float crho = fminf(rho, clip_rho);
    const float c = lambda_ * fminf(rho, clip_c);
    const float disc = s_dis[t * 64 + lane];
    const float r = s_rew[t * 64 + lane];
    const float v = s_val[t * 64 + lane];
    const float delta = crho * fmaf(disc, v_next, r - v);
    acc = fmaf(disc * c, acc, delta);
    const float vs_t = v + acc;
    const size_t idx = (size_t)t * B + b;
    vs[idx] = vs_t;
    pg_adv[idx] = fminf(rho, clip_pg_rho) * fmaf(disc, vs_next, r - v);
    vs_next = vs_t;
    v_next = v;
  }
}

// Fallback without LDS staging for very long unrolls (T*1024B > LDS).
__global__ void vtrace_scan_global(
    const float* __restrict__ log_rho, const float* __restrict__ discounts,
    const float* __restrict__ rewards, const float* __restrict__ values,
    const float* __restrict__ bootstrap, int T, int B,
    float clip_rho, float clip_c, float lambda_, float clip_pg_rho,
    float* __restrict__ vs, float* __restrict__ pg_adv) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float acc = 0.f;
  float vs_next = bootstrap[b];
  float v_next = vs_next;
  for (int t = T - 1; t >= 0; --t) {
    const size_t idx = (size_t)t * B + b;
    const float rho = __expf(log_rho[idx]);
    const float crho = fminf(rho, clip_rho);
    const float c = lambda_ * fminf(rho, clip_c);
    const float disc = discounts[idx];
    const float r = rewards[idx];
    const float v = values[idx];
    const float delta = crho * fmaf(disc, v_next, r - v);
    acc = fmaf(disc * c, acc, delta);
    const float vs_t = v + acc;
    vs[idx] = vs_t;
    pg_adv[idx] = fminf(rho, clip_pg_rho) * fmaf(disc, vs_next, r - v);
    vs_next = vs_t;
    v_next = v;
  }
}

// ---------------------------------------------------------------------------
// Kernel 3: fused loss backward. One thread per (row, action) element:
//   dlogits[i,j] = gs * ( adv_i * (p_ij - [j==a_i])
//                       + ec * p_ij * (logp_ij - negent_i) )
//   dvalues[i]   = gs * ( -bc * (vs_i - V_i) )         (written at j==0)
// gs = incoming d(total_loss) (0-dim device tensor — no host sync).
// ---------------------------------------------------------------------------
__global__ void vtrace_loss_bwd_kernel(
    const float* __restrict__ logp, const float* __restrict__ row_negent,
    const long* __restrict__ actions, const float* __restrict__ pg_adv,
    const float* __restrict__ vdiff, const float* __restrict__ grad_scale,
    float baseline_cost, float entropy_cost, int N, int A,
    float* __restrict__ dlogits, float* __restrict__ dvalues) {
  const float gs = grad_scale[0];
  const size_t total = (size_t)N * A;
  for (size_t e = (size_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += (size_t)gridDim.x * blockDim.x) {
    const int i = (int)(e / A);
    const int j = (int)(e - (size_t)i * A);
    const float lp = logp[e];
    const float p = __expf(lp);
    const float adv = pg_adv[i];
    const float onehot = (j == (int)actions[i]) ? 1.f : 0.f;
    float d = adv * (p - onehot) + entropy_cost * p * (lp - row_negent[i]);
    dlogits[e] = gs * d;
    if (j == 0) dvalues[i] = gs * (-baseline_cost * vdiff[i]);
  }
}

}  // namespace impala

// -------------------------- launchers (host) -------------------------------

extern "C" {

void launch_vtrace_row_fwd(const float* tgt, const float* beh,
                           const long* actions, int N, int A,
                           float* logp, float* tgt_lp, float* log_rho,
                           float* row_negent, hipStream_t stream) {
  const int block = 256;
  int grid = (N + block - 1) / block;
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(impala::vtrace_row_fwd, dim3(grid), dim3(block), 0,
                     stream, tgt, beh, actions, N, A, logp, tgt_lp, log_rho,
                     row_negent);
}

void launch_vtrace_scan(const float* log_rho, const float* discounts,
                        const float* rewards, const float* values,
                        const float* bootstrap, int T, int B, float clip_rho,
                        float clip_c, float lambda_, float clip_pg_rho,
                        float* vs, float* pg_adv, hipStream_t stream) {
  const size_t lds_bytes = 4ull * T * 64 * sizeof(float);
  if (lds_bytes <= 160 * 1024 - 1024) {
    const int grid = (B + 63) / 64;
    hipLaunchKernelGGL(impala::vtrace_scan_lds, dim3(grid), dim3(64),
                       lds_bytes, stream, log_rho, discounts, rewards, values,
                       bootstrap, T, B, clip_rho, clip_c, lambda_,
                       clip_pg_rho, vs, pg_adv);
  } else {
    const int block = 256;
    const int grid = (B + block - 1) / block;
    hipLaunchKernelGGL(impala::vtrace_scan_global, dim3(grid), dim3(block), 0,
                       stream, log_rho, discounts, rewards, values, bootstrap,
                       T, B, clip_rho, clip_c, lambda_, clip_pg_rho, vs,
                       pg_adv);
  }
}

void launch_vtrace_loss_bwd(const float* logp, const float* row_negent,
                            const long* actions, const float* pg_adv,
                            const float* vdiff, const float* grad_scale,
                            float baseline_cost, float entropy_cost, int N,
                            int A, float* dlogits, float* dvalues,
                            hipStream_t stream) {
  const int block = 256;
  size_t total = (size_t)N * A;
  int grid = (int)((total + block - 1) / block);
  if (grid > 4096) grid = 4096;
  hipLaunchKernelGGL(impala::vtrace_loss_bwd_kernel, dim3(grid), dim3(block),
                     0, stream, logp, row_negent, actions, pg_adv, vdiff,
                     grad_scale, baseline_cost, entropy_cost, N, A, dlogits,
                     dvalues);
}

}  // extern "C"
