// Fused optimizer kernels over the FLAT parameter/gradient arena.
//
// The learner flattens all parameters/grads/optimizer state into single
// contiguous fp32 buffers (ops/optim.py), so global-grad-norm clipping +
// RMSProp becomes 2 kernels regardless of parameter count (vs ~5 kernels
// PER PARAMETER eagerly), every kernel is graph-capturable (lr and the
// DP scale arrive as device scalars), and data-parallel gradient
// reduction is one zero-copy all-reduce of the flat grad buffer
// (BASELINE.json:5 "bigger shards, fewer, larger collectives").

#include <hip/hip_runtime.h>

namespace impala {

// block-reduced sum of squares, ONE atomic per block (a same-word fp32
// atomic costs ~11-13 ns serialized on this chip — per-wave atomics at
// 4k waves measured 54 us; per-block + small grid is ~50x cheaper).
__global__ void grad_sqnorm_kernel(const float* __restrict__ g, long n,
                                   const float* __restrict__ scale,
                                   float* __restrict__ out) {
  __shared__ float s_part[4];  // one partial per wave (256 threads)
  const float s = scale[0];
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float v = g[i] * s;
    acc = fmaf(v, v, acc);
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    acc += __shfl_down(acc, off, 64);
  }
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) s_part[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    atomicAdd(out, s_part[0] + s_part[1] + s_part[2] + s_part[3]);
  }
}

// p -= lr * clip(g*scale) / (sqrt(sqavg') + eps);  sqavg' updated in place.
// clip coef = min(1, max_norm / (sqrt(norm_sq) + 1e-6)) — exactly
// torch.nn.utils.clip_grad_norm_ semantics.
__global__ void rmsprop_update_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ sqavg, long n, const float* __restrict__ lr,
    const float* __restrict__ norm_sq, const float* __restrict__ scale,
    float max_norm, float alpha, float eps,
    float* __restrict__ skip_count) {
  const float s = scale[0];
  // non-finite global norm => at least one grad element is inf/NaN:
  // SKIP the whole update (params and sqavg untouched) instead of
  // poisoning the parameters permanently (0 * inf = NaN even at lr=0).
  // Standard overflow-skip semantics (GradScaler-style); skips are
  // counted in skip_count[0] for observability.
  if (!isfinite(norm_sq[0])) {
    if (blockIdx.x == 0 && threadIdx.x == 0 && skip_count != nullptr)
      skip_count[0] += 1.f;
    return;
  }
  float coef = 1.f;
  if (max_norm > 0.f) {
    const float norm = sqrtf(norm_sq[0]);
    coef = fminf(1.f, max_norm / (norm + 1e-6f));
  }
  const float step = lr[0];
  const float cs = coef * s;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    const float gv = g[i] * cs;
    const float sq = fmaf(1.f - alpha, gv * gv, alpha * sqavg[i]);
    sqavg[i] = sq;
    p[i] -= step * gv / (sqrtf(sq) + eps);
  }
}

}  // namespace impala

extern "C" {

void launch_grad_sqnorm(const float* g, long n, const float* scale,
                        float* out, hipStream_t stream) {
  const int block = 256;
  long grid = (n + block - 1) / block;
  if (grid > 256) grid = 256;
  hipLaunchKernelGGL(impala::grad_sqnorm_kernel, dim3((int)grid), dim3(block),
                     0, stream, g, n, scale, out);
}

void launch_rmsprop_update(float* p, const float* g, float* sqavg, long n,
                           const float* lr, const float* norm_sq,
                           const float* scale, float max_norm, float alpha,
                           float eps, float* skip_count,
                           hipStream_t stream) {
  const int block = 256;
  long grid = (n + block - 1) / block;
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(impala::rmsprop_update_kernel, dim3((int)grid),
                     dim3(block), 0, stream, p, g, sqavg, n, lr, norm_sq,
                     scale, max_norm, alpha, eps, skip_count);
}

}  // extern "C"
