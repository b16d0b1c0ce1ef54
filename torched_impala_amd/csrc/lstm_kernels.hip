// gfx950 LSTM-core sequence kernels (BASELINE.json:5 "LSTM cell" kernel).
//
// The IMPALA core is LSTMCell(hidden=256) iterated over the unroll with
// done-mask state resets (SURVEY.md §2c row lstm_cell: "the
// latency-critical kernel" — sequential over T). Eager PyTorch runs
// ~2 kernel launches per step plus Python overhead; here the WHOLE
// unroll is ONE kernel launch per direction:
//
//   * the input-to-gate GEMM for all T steps is hoisted into one batched
//     GEMM outside (Xg = x @ W_ih^T + b_ih + b_hh, torch/rocBLAS),
//   * the sequential recurrence partitions by BATCH ROW: one workgroup
//     owns one batch row for the whole T loop, so there are NO
//     inter-workgroup dependencies, no grid barriers, no weight
//     re-publication — h/c live in LDS across steps and W_hh streams
//     from L2 (1 MiB, read by every workgroup, L2/L3-resident).
//
// Layout choices (CDNA4): the forward reads W_hh TRANSPOSED (256, 1024)
// so that lane j's float2 gate-column reads are coalesced; the backward
// reads W_hh in its native (1024, 256) layout so lane k's dh_prev reads
// are coalesced. 512 threads (8 waves) per workgroup; LDS ~12 KiB.
//
// dW_hh / dW_ih / db are NOT computed here: the backward kernel emits
// dGates (T,B,4H), and the host reduces them with two big GEMMs
// (dW_hh = dG^T @ Hp, dXg = dG feeding autograd) — that work is
// GEMM-shaped and belongs on MFMA via the BLAS path, not in this
// latency-critical loop.

#include <hip/hip_runtime.h>

namespace impala {

constexpr int H = 256;        // hidden size (IMPALA core is fixed at 256)
constexpr int G4 = 4 * H;     // 1024 gate pre-activations
constexpr int THREADS = 1024;  // 16 waves (4/SIMD: the GEMV is
                               // L2-latency-bound; TLP covers it)

// fast activations: libm tanhf lowers to an IEEE division
// (v_div_scale/v_div_fmas/v_div_fixup + divergent special-case branches)
// — measured as the dominant per-step cost. v_rcp_f32 (~1 ulp approx) +
// clamped __expf keeps the whole phase branch-free; |error| ~1e-7 rel,
// well inside the fp32 parity tolerances.
__device__ __forceinline__ float fast_rcp(float x) {
  return __builtin_amdgcn_rcpf(x);
}

__device__ __forceinline__ float sigmoidf(float x) {
  const float xc = fminf(fmaxf(x, -30.f), 30.f);
  return fast_rcp(1.0f + __expf(-xc));
}

__device__ __forceinline__ float fast_tanh(float x) {
  const float xc = fminf(fmaxf(x, -15.f), 15.f);
  const float e = __expf(2.0f * xc);
  return (e - 1.0f) * fast_rcp(e + 1.0f);
}

// templated weight load: fp32 exact path or bf16 (half the L2 bytes —
// the recurrence GEMV is L2-bandwidth-bound per CU: 1 MiB of W per WG
// per timestep; r8 profile 189 us fwd at T=20 B=32).
typedef __bf16 lstm_bf16_t;
template <typename WT>
__device__ __forceinline__ float loadw(const WT* p) {
  return (float)*p;
}

// ---------------------------------------------------------------------------
// forward: grid = (B) blocks x 512 threads
//   Xg      (T, B, 1024)  x-part of gates incl. both biases
//   W_hhT   (256, 1024)   W_hh transposed
//   notdone (T, B)        1 - done, fp32
//   h0, c0  (B, 256)
// outputs:
//   Hout    (T, B, 256)   h_t
//   act     (T, B, 1024)  post-activation i,f,g,o (saved for bwd)
//   Cout    (T, B, 256)   c_t (saved for bwd)
//   Hp      (T, B, 256)   masked h input at step t (saved for bwd + dW_hh)
//   hT, cT  (B, 256)
// ---------------------------------------------------------------------------
// forward runs 512 threads (8 waves), TWO gate columns per thread:
// the per-step cost is barrier-dominated (2 x full-block barriers x T)
// and a narrower block halves the barrier width while the dual-column
// GEMV keeps the same loads in flight per lane.
constexpr int FWD_THREADS = 512;

template <typename WT>
__global__ void __launch_bounds__(FWD_THREADS)
lstm_seq_fwd(const float* __restrict__ Xg, const WT* __restrict__ W_hhT,
             const float* __restrict__ notdone, const float* __restrict__ h0,
             const float* __restrict__ c0, int T, int B,
             float* __restrict__ Hout, float* __restrict__ act,
             float* __restrict__ Cout, float* __restrict__ Hp,
             float* __restrict__ hT, float* __restrict__ cT,
             int save_for_bwd) {
  __shared__ __align__(16) float s_h[H];
  __shared__ __align__(16) float s_c[H];
  __shared__ __align__(16) float s_g[G4];

  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  if (tid < H) {
    s_h[tid] = h0[b * H + tid];
    s_c[tid] = c0[b * H + tid];
  }
  __syncthreads();

  for (int t = 0; t < T; ++t) {
    const float nd = notdone[t * B + b];
    const size_t rowH = ((size_t)t * B + b) * H;
    const size_t rowG = ((size_t)t * B + b) * G4;
    // the done-mask is a SCALAR per (t,b): gates = Xg + (nd*h).W
    //                                            = Xg + nd*(h.W)
    // so no masking phase (and barrier) is needed before the GEMV.
    if (save_for_bwd && tid < H) Hp[rowH + tid] = s_h[tid] * nd;
    // gate GEMV: thread j computes gate columns j and j+512. k chunked
    // by 4 (one ds_read_b128 broadcast of h, eight W loads in flight).
    {
      float acc0 = 0.f, acc1 = 0.f;
      const WT* wc0 = W_hhT + tid;
      const WT* wc1 = W_hhT + tid + 512;
      #pragma unroll 4
      for (int k0 = 0; k0 < H; k0 += 4) {
        const float4 hv = *reinterpret_cast<const float4*>(&s_h[k0]);
        const float a0 = loadw(wc0 + (size_t)(k0 + 0) * G4);
        const float a1 = loadw(wc0 + (size_t)(k0 + 1) * G4);
        const float a2 = loadw(wc0 + (size_t)(k0 + 2) * G4);
        const float a3 = loadw(wc0 + (size_t)(k0 + 3) * G4);
        const float b0 = loadw(wc1 + (size_t)(k0 + 0) * G4);
        const float b1 = loadw(wc1 + (size_t)(k0 + 1) * G4);
        const float b2 = loadw(wc1 + (size_t)(k0 + 2) * G4);
        const float b3 = loadw(wc1 + (size_t)(k0 + 3) * G4);
        acc0 = fmaf(hv.x, a0, acc0);
        acc0 = fmaf(hv.y, a1, acc0);
        acc0 = fmaf(hv.z, a2, acc0);
        acc0 = fmaf(hv.w, a3, acc0);
        acc1 = fmaf(hv.x, b0, acc1);
        acc1 = fmaf(hv.y, b1, acc1);
        acc1 = fmaf(hv.z, b2, acc1);
        acc1 = fmaf(hv.w, b3, acc1);
      }
      s_g[tid] = fmaf(nd, acc0, Xg[rowG + tid]);
      s_g[tid + 512] = fmaf(nd, acc1, Xg[rowG + tid + 512]);
    }
    __syncthreads();
    if (tid < H) {
      const float gi = sigmoidf(s_g[tid]);
      const float gf = sigmoidf(s_g[H + tid]);
      const float gg = fast_tanh(s_g[2 * H + tid]);
      const float go = sigmoidf(s_g[3 * H + tid]);
      const float c = fmaf(gf, s_c[tid] * nd, gi * gg);
      const float h = go * fast_tanh(c);
      if (save_for_bwd) {
        act[rowG + tid] = gi;
        act[rowG + H + tid] = gf;
        act[rowG + 2 * H + tid] = gg;
        act[rowG + 3 * H + tid] = go;
        Cout[rowH + tid] = c;
      }
      Hout[rowH + tid] = h;
      s_h[tid] = h;
      s_c[tid] = c;
    }
    __syncthreads();
  }
  if (tid < H) {
    hT[b * H + tid] = s_h[tid];
    cT[b * H + tid] = s_c[tid];
  }
}

// ---------------------------------------------------------------------------
// backward: grid = (B) blocks x 512 threads, reverse time loop.
// Emits dG (T,B,1024) and dh0/dc0; weight grads are host-side GEMMs.
// ---------------------------------------------------------------------------
template <typename WT>
__global__ void __launch_bounds__(THREADS)
lstm_seq_bwd(const float* __restrict__ dH, const float* __restrict__ dhT,
             const float* __restrict__ dcT, const float* __restrict__ act,
             const float* __restrict__ Cout, const float* __restrict__ c0,
             const float* __restrict__ notdone,
             const WT* __restrict__ W_hh,  // (1024, 256) native layout
             int T, int B, float* __restrict__ dG,
             float* __restrict__ dh0, float* __restrict__ dc0) {
  __shared__ __align__(16) float s_dh[H];
  __shared__ __align__(16) float s_dc[H];
  __shared__ __align__(16) float s_dg[G4];
  __shared__ __align__(16) float s_part[THREADS];

  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  if (tid < H) {
    s_dh[tid] = dhT[b * H + tid];
    s_dc[tid] = dcT[b * H + tid];
  }
  __syncthreads();

  // the per-step combine of the four GEMV quarter-partials is FOLDED
  // into the next iteration's elementwise phase (reading s_part
  // directly), cutting the 16-wave barriers per step from 3 to 2 —
  // the recurrence is barrier-latency-bound, not bandwidth-bound.
  float prev_nd = 1.f;
  bool have_parts = false;
  for (int t = T - 1; t >= 0; --t) {
    const float nd = notdone[t * B + b];
    const size_t rowH = ((size_t)t * B + b) * H;
    const size_t rowG = ((size_t)t * B + b) * G4;
    if (tid < H) {
      const float c_t = Cout[rowH + tid];
      const float tc = fast_tanh(c_t);
      const float cprev =
          (t > 0 ? Cout[rowH + tid - (size_t)B * H] : c0[b * H + tid]) * nd;
      const float gi = act[rowG + tid];
      const float gf = act[rowG + H + tid];
      const float gg = act[rowG + 2 * H + tid];
      const float go = act[rowG + 3 * H + tid];
      const float carried = have_parts
          ? (s_part[tid] + s_part[H + tid] + s_part[2 * H + tid] +
             s_part[3 * H + tid]) * prev_nd
          : s_dh[tid];
      const float dh = dH[rowH + tid] + carried;
      const float dct = s_dc[tid] + dh * go * (1.f - tc * tc);
      const float dgi = (dct * gg) * gi * (1.f - gi);
      const float dgf = (dct * cprev) * gf * (1.f - gf);
      const float dgg = (dct * gi) * (1.f - gg * gg);
      const float dgo = (dh * tc) * go * (1.f - go);
      s_dg[tid] = dgi;
      s_dg[H + tid] = dgf;
      s_dg[2 * H + tid] = dgg;
      s_dg[3 * H + tid] = dgo;
      dG[rowG + tid] = dgi;
      dG[rowG + H + tid] = dgf;
      dG[rowG + 2 * H + tid] = dgg;
      dG[rowG + 3 * H + tid] = dgo;
      s_dc[tid] = dct * gf * nd;  // dc_{t-1} (masked)
    }
    __syncthreads();
    // dh_prev[k] = sum_j s_dg[j] * W_hh[j, k]; j split over four
    // thread quarters, chunked by 4 with batched W row loads.
    {
      const int k = tid & (H - 1);
      const int j0 = (tid >> 8) * (G4 / 4);
      const WT* wcol = W_hh + (size_t)j0 * H + k;
      float acc = 0.f;
      #pragma unroll 4
      for (int j = 0; j < G4 / 4; j += 4) {
        const float4 dg = *reinterpret_cast<const float4*>(&s_dg[j0 + j]);
        const float w0 = loadw(wcol + (size_t)(j + 0) * H);
        const float w1 = loadw(wcol + (size_t)(j + 1) * H);
        const float w2 = loadw(wcol + (size_t)(j + 2) * H);
        const float w3 = loadw(wcol + (size_t)(j + 3) * H);
        acc = fmaf(dg.x, w0, acc);
        acc = fmaf(dg.y, w1, acc);
        acc = fmaf(dg.z, w2, acc);
        acc = fmaf(dg.w, w3, acc);
      }
      s_part[tid] = acc;
    }
    __syncthreads();
    prev_nd = nd;
    have_parts = true;
  }
  if (tid < H) {
    dh0[b * H + tid] = have_parts
        ? (s_part[tid] + s_part[H + tid] + s_part[2 * H + tid] +
           s_part[3 * H + tid]) * prev_nd
        : s_dh[tid];
    dc0[b * H + tid] = s_dc[tid];
  }
}

}  // namespace impala

extern "C" {

void launch_lstm_seq_fwd(const float* Xg, const void* W_hhT,
                         const float* notdone, const float* h0,
                         const float* c0, int T, int B, float* Hout,
                         float* act, float* Cout, float* Hp, float* hT,
                         float* cT, int save_for_bwd, int wbf16,
                         hipStream_t stream) {
  if (wbf16)
    hipLaunchKernelGGL(impala::lstm_seq_fwd<impala::lstm_bf16_t>, dim3(B),
                       dim3(impala::FWD_THREADS), 0, stream, Xg,
                       static_cast<const impala::lstm_bf16_t*>(W_hhT),
                       notdone, h0, c0, T, B, Hout, act, Cout, Hp, hT, cT,
                       save_for_bwd);
  else
    hipLaunchKernelGGL(impala::lstm_seq_fwd<float>, dim3(B),
                       dim3(impala::FWD_THREADS), 0, stream, Xg,
                       static_cast<const float*>(W_hhT), notdone, h0, c0, T,
                       B, Hout, act, Cout, Hp, hT, cT, save_for_bwd);
}

void launch_lstm_seq_bwd(const float* dH, const float* dhT, const float* dcT,
                         const float* act, const float* Cout, const float* c0,
                         const float* notdone, const void* W_hh, int T,
                         int B, float* dG, float* dh0, float* dc0, int wbf16,
                         hipStream_t stream) {
  if (wbf16)
    hipLaunchKernelGGL(impala::lstm_seq_bwd<impala::lstm_bf16_t>, dim3(B),
                       dim3(impala::THREADS), 0, stream, dH, dhT, dcT, act,
                       Cout, c0, notdone,
                       static_cast<const impala::lstm_bf16_t*>(W_hh), T, B,
                       dG, dh0, dc0);
  else
    hipLaunchKernelGGL(impala::lstm_seq_bwd<float>, dim3(B),
                       dim3(impala::THREADS), 0, stream, dH, dhT, dcT, act,
                       Cout, c0, notdone, static_cast<const float*>(W_hh),
                       T, B, dG, dh0, dc0);
}

}  // extern "C"
