// gfx950 fused shallow-encoder convolution kernels (BASELINE.json:5
// "batched Conv2d encoder" / SURVEY.md §2c conv_encoder_fwd/bwd).
//
// MI355X-native design: the Atari frame is TINY (4x84x84 u8 = 28 KiB),
// so instead of im2col GEMMs with NCHW<->NHWC transposes (the MIOpen
// path: igemm + batched_transpose + elementwise chains per layer), ONE
// workgroup owns ONE image and runs conv1->conv2->conv3 out of LDS:
//
//   u8 frame (28 KiB, read directly — /255 fused, no float frame pass)
//   -> conv1 16x8x8/s4 -> a1 (25.6 KiB LDS) -> conv2 32x4x4/s2
//   -> a2 (10.4 KiB)   -> conv3 32x3x3    -> a3 -> global.
//
// Register blocking (the v1 lesson, measured): weights must be loaded
// once per (c,ky) chunk into registers and reused across an
// ACCUMULATOR ARRAY of output pixels — a w-load per MAC turns the
// whole kernel issue/L1-bound (321 us/step vs ~25 us of math).
// Occupancy pinned at 2 workgroups/CU (launch_bounds + <=64 KiB LDS).
//
// Backward is THREE kernels (dW3+da2 | dW2+da1 | dW1), not one: a
// single kernel's register pressure across phases forced ~200 B/lane
// scratch spills inside the loops (2.4 ms/step measured). Intermediate
// d2/d1 round-trip through global (~24 MB/step, ~8 us at HBM rate).
// Per-image dW/db slabs are reduced by one sum outside (deterministic).
//
// fc / heads GEMMs: csrc/linear_kernels.hip (MFMA) / rocBLAS.

#include <hip/hip_runtime.h>

namespace impala {

// shallow-net shape constants (84x84 Atari; BASELINE.json:2)
constexpr int C0 = 4, H0 = 84, W0 = 84;
constexpr int F1 = 16, K1 = 8, S1 = 4, H1 = 20, W1 = 20;
constexpr int F2 = 32, K2 = 4, S2 = 2, H2 = 9, W2 = 9;
constexpr int F3 = 32, K3 = 3, H3 = 7, W3 = 7;
constexpr int NTHREADS = 256;

constexpr int IMG_BYTES = C0 * H0 * W0;            // 28224
constexpr int A1_N = F1 * H1 * W1;                 // 6400
constexpr int A2_N = F2 * H2 * W2;                 // 2592
constexpr int A3_N = F3 * H3 * W3;                 // 1568
// weight-grad slab layout per image: [dW1 | dW2 | dW3 | db1 | db2 | db3]
constexpr int DW1_N = F1 * C0 * K1 * K1;           // 4096
constexpr int DW2_N = F2 * F1 * K2 * K2;           // 8192
constexpr int DW3_N = F3 * F2 * K3 * K3;           // 9216
constexpr int SLAB_N = DW1_N + DW2_N + DW3_N + F1 + F2 + F3;  // 21584
constexpr int DB_OFF = DW1_N + DW2_N + DW3_N;

__device__ __forceinline__ float relu(float x) { return x > 0.f ? x : 0.f; }

// ---------------------------------------------------------------------------
// forward: grid = (N) blocks x 256 threads
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS, 4)
shallow_encoder_fwd(const unsigned char* __restrict__ img,
                    const float* __restrict__ w1, const float* __restrict__ b1,
                    const float* __restrict__ w2, const float* __restrict__ b2,
                    const float* __restrict__ w3, const float* __restrict__ b3,
                    int N, float* __restrict__ a1g, float* __restrict__ a2g,
                    float* __restrict__ a3g, int save_for_bwd) {
  __shared__ __align__(16) float s_a1[A1_N];
  __shared__ __align__(16) float s_a2[A2_N];

  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  // the frame is read straight from L2 (28 KiB/image, PMC showed the
  // LDS copy bought nothing and halved occupancy)
  const unsigned char* s_img = img + (size_t)n * IMG_BYTES;

  // ---- conv1: thread = (filter f = tid/16, lane = tid%16); lane owns
  //      25 pixels (px = lane, lane+16, ...). Weights per (c,ky): 8
  //      floats in registers, reused across all 25 accumulators. ----
  {
    const float inv255 = 1.0f / 255.0f;
    const int f = tid >> 4;
    const int lane = tid & 15;
    const float* wf = w1 + f * (C0 * K1 * K1);
    float acc[25];
    #pragma unroll
    for (int i = 0; i < 25; ++i) acc[i] = 0.f;
    for (int c = 0; c < C0; ++c) {
      const unsigned char* ic = s_img + c * (H0 * W0);
      const float* wc = wf + c * (K1 * K1);
      #pragma unroll 1
      for (int ky = 0; ky < K1; ++ky) {
        const float4 wlo = *reinterpret_cast<const float4*>(wc + ky * K1);
        const float4 whi = *reinterpret_cast<const float4*>(wc + ky * K1 + 4);
        #pragma unroll
        for (int i = 0; i < 25; ++i) {
          const int px = lane + i * 16;
          const int oy = px / W1, ox = px - oy * W1;
          const unsigned char* row = ic + (S1 * oy + ky) * W0 + S1 * ox;
          const unsigned int lo = *reinterpret_cast<const unsigned int*>(row);
          const unsigned int hi =
              *reinterpret_cast<const unsigned int*>(row + 4);
          float a = acc[i];
          a = fmaf((float)(lo & 0xff), wlo.x, a);
          a = fmaf((float)((lo >> 8) & 0xff), wlo.y, a);
          a = fmaf((float)((lo >> 16) & 0xff), wlo.z, a);
          a = fmaf((float)(lo >> 24), wlo.w, a);
          a = fmaf((float)(hi & 0xff), whi.x, a);
          a = fmaf((float)((hi >> 8) & 0xff), whi.y, a);
          a = fmaf((float)((hi >> 16) & 0xff), whi.z, a);
          a = fmaf((float)(hi >> 24), whi.w, a);
          acc[i] = a;
        }
      }
    }
    const float bias = b1[f];
    #pragma unroll
    for (int i = 0; i < 25; ++i) {
      const int px = lane + i * 16;
      const float v = relu(fmaf(acc[i], inv255, bias));
      const int idx = f * (H1 * W1) + px;
      s_a1[idx] = v;
      if (save_for_bwd) a1g[(size_t)n * A1_N + idx] = v;
    }
  }
  __syncthreads();

  // ---- conv2: thread = (f = tid/8, lane = tid%8); 11 accumulators
  //      (last partially valid). Weights per (c,ky): 4 floats. ----
  {
    const int f = tid >> 3;
    const int lane = tid & 7;
    const float* wf = w2 + f * (F1 * K2 * K2);
    float acc[11];
    #pragma unroll
    for (int i = 0; i < 11; ++i) acc[i] = 0.f;
    for (int c = 0; c < F1; ++c) {
      const float* ic = s_a1 + c * (H1 * W1);
      const float* wc = wf + c * (K2 * K2);
      #pragma unroll 1
      for (int ky = 0; ky < K2; ++ky) {
        const float4 w4 = *reinterpret_cast<const float4*>(wc + ky * K2);
        #pragma unroll
        for (int i = 0; i < 11; ++i) {
          const int px = lane + i * 8;
          if (px < H2 * W2) {
            const int oy = px / W2, ox = px - oy * W2;
            const float* row = ic + (S2 * oy + ky) * W1 + S2 * ox;
            float a = acc[i];
            a = fmaf(row[0], w4.x, a);
            a = fmaf(row[1], w4.y, a);
            a = fmaf(row[2], w4.z, a);
            a = fmaf(row[3], w4.w, a);
            acc[i] = a;
          }
        }
      }
    }
    const float bias = b2[f];
    #pragma unroll
    for (int i = 0; i < 11; ++i) {
      const int px = lane + i * 8;
      if (px < H2 * W2) {
        const float v = relu(acc[i] + bias);
        const int idx = f * (H2 * W2) + px;
        s_a2[idx] = v;
        if (save_for_bwd) a2g[(size_t)n * A2_N + idx] = v;
      }
    }
  }
  __syncthreads();

  // ---- conv3: thread = (f = tid/8, lane = tid%8); 7 accumulators. ----
  {
    const int f = tid >> 3;
    const int lane = tid & 7;
    const float* wf = w3 + f * (F2 * K3 * K3);
    float acc[7];
    #pragma unroll
    for (int i = 0; i < 7; ++i) acc[i] = 0.f;
    for (int c = 0; c < F2; ++c) {
      const float* ic = s_a2 + c * (H2 * W2);
      const float* wc = wf + c * (K3 * K3);
      #pragma unroll 1
      for (int ky = 0; ky < K3; ++ky) {
        const float wa = wc[ky * K3 + 0];
        const float wb = wc[ky * K3 + 1];
        const float wcc = wc[ky * K3 + 2];
        #pragma unroll
        for (int i = 0; i < 7; ++i) {
          const int px = lane + i * 8;
          if (px < H3 * W3) {
            const int oy = px / W3, ox = px - oy * W3;
            const float* row = ic + (oy + ky) * W2 + ox;
            float a = acc[i];
            a = fmaf(row[0], wa, a);
            a = fmaf(row[1], wb, a);
            a = fmaf(row[2], wcc, a);
            acc[i] = a;
          }
        }
      }
    }
    const float bias = b3[f];
    #pragma unroll
    for (int i = 0; i < 7; ++i) {
      const int px = lane + i * 8;
      if (px < H3 * W3)
        a3g[(size_t)n * A3_N + f * (H3 * W3) + px] = relu(acc[i] + bias);
    }
  }
}

// ---------------------------------------------------------------------------
// backward kernel 1: dW3 + db3 + d_a2 (relu-masked) -> d2g (N, A2_N)
// LDS: a2 + d3 + d2 = 27 KiB
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS, 4)
shallow_bwd_k1(const float* __restrict__ a2g, const float* __restrict__ a3g,
               const float* __restrict__ d_a3g, const float* __restrict__ w3,
               int N, float* __restrict__ slab, float* __restrict__ d2g) {
  // grid: (N, ESPLIT) — blockIdx.y picks an element-range slice so the
  // per-image serial loops shrink (the loops are latency-bound; more
  // blocks beat deeper ILP here)
  const int esplit = gridDim.y, eslice = blockIdx.y;
  __shared__ __align__(16) float s_a2[A2_N];
  __shared__ __align__(16) float s_d3[A3_N];
  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  float* my_slab = slab + (size_t)n * SLAB_N;

  #pragma unroll 1
  for (int i = tid; i < A2_N; i += NTHREADS)
    s_a2[i] = a2g[(size_t)n * A2_N + i];
  #pragma unroll 1
  for (int i = tid; i < A3_N; i += NTHREADS) {
    const float m = a3g[(size_t)n * A3_N + i] > 0.f ? 1.f : 0.f;
    s_d3[i] = m * d_a3g[(size_t)n * A3_N + i];
  }
  __syncthreads();

  // dW3[f][c][ky][kx] = sum_px a2[c][oy+ky][ox+kx] * d3[f][px]
  #pragma unroll 1
  for (int e = tid + eslice * NTHREADS; e < DW3_N;
       e += NTHREADS * esplit) {
    const int f = e / (F2 * K3 * K3);
    const int r = e - f * (F2 * K3 * K3);
    const int c = r / (K3 * K3);
    const int k = r - c * (K3 * K3);
    const int ky = k / K3, kx = k - ky * K3;
    const float* dc = s_d3 + f * (H3 * W3);
    const float* ac = s_a2 + c * (H2 * W2);
    float acc = 0.f;
    #pragma unroll 1
    for (int oy = 0; oy < H3; ++oy) {
      const float* arow = ac + (oy + ky) * W2 + kx;
      const float* drow = dc + oy * W3;
      #pragma unroll
      for (int ox = 0; ox < W3; ++ox) acc = fmaf(arow[ox], drow[ox], acc);
    }
    my_slab[DW1_N + DW2_N + e] = acc;
  }
  if (eslice == 0) {
    for (int f = tid; f < F3; f += NTHREADS) {
      float acc = 0.f;
      const float* dc = s_d3 + f * (H3 * W3);
      for (int p = 0; p < H3 * W3; ++p) acc += dc[p];
      my_slab[DB_OFF + F1 + F2 + f] = acc;
    }
  }

  // d_a2[c][y][x]: bounds computed once per element, branch-free inner
  #pragma unroll 1
  for (int e = tid + eslice * NTHREADS; e < A2_N;
       e += NTHREADS * esplit) {
    const int c = e / (H2 * W2);
    const int p = e - c * (H2 * W2);
    const int y = p / W2, x = p - y * W2;
    const int ky_lo = max(0, y - (H3 - 1)), ky_hi = min(K3 - 1, y);
    const int kx_lo = max(0, x - (W3 - 1)), kx_hi = min(K3 - 1, x);
    float acc = 0.f;
    #pragma unroll 4
    for (int f = 0; f < F3; ++f) {
      const float* dc = s_d3 + f * (H3 * W3);
      const float* wc = w3 + (f * F2 + c) * (K3 * K3);
      for (int ky = ky_lo; ky <= ky_hi; ++ky) {
        const float* drow = dc + (y - ky) * W3 + x;
        const float* wrow = wc + ky * K3;
        for (int kx = kx_lo; kx <= kx_hi; ++kx)
          acc = fmaf(drow[-kx], wrow[kx], acc);
      }
    }
    const float v = s_a2[e] > 0.f ? acc : 0.f;
    d2g[(size_t)n * A2_N + e] = v;
  }
}

// ---------------------------------------------------------------------------
// backward kernel 2: dW2 + d_a1 (relu-masked) -> d1g (N, A1_N) + db1
// LDS: a1 + d2 + d1 = 61.6 KiB
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS, 8)
shallow_bwd_k2(const float* __restrict__ a1g, const float* __restrict__ d2g,
               const float* __restrict__ w2, int N,
               float* __restrict__ slab, float* __restrict__ d1g) {
  const int esplit = gridDim.y, eslice = blockIdx.y;
  // a1 reads come straight from L2 (25.6 KiB/image, re-read ~8x, the
  // kernel is latency- not bandwidth-bound): dropping its LDS copy cuts
  // LDS to 10.4 KiB -> 8 blocks/CU, doubling latency coverage
  __shared__ __align__(16) float s_d2[A2_N];
  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  float* my_slab = slab + (size_t)n * SLAB_N;
  const float* s_a1 = a1g + (size_t)n * A1_N;

  #pragma unroll 1
  for (int i = tid; i < A2_N; i += NTHREADS)
    s_d2[i] = d2g[(size_t)n * A2_N + i];
  __syncthreads();

  if (eslice == 0) {
    for (int f = tid; f < F2; f += NTHREADS) {
      float acc = 0.f;
      const float* dc = s_d2 + f * (H2 * W2);
      for (int p = 0; p < H2 * W2; ++p) acc += dc[p];
      my_slab[DB_OFF + F1 + f] = acc;
    }
  }

  // dW2[f][c][ky][kx] = sum_px a1[c][2oy+ky][2ox+kx] * d2[f][px]
  #pragma unroll 1
  for (int e = tid + eslice * NTHREADS; e < DW2_N;
       e += NTHREADS * esplit) {
    const int f = e / (F1 * K2 * K2);
    const int r = e - f * (F1 * K2 * K2);
    const int c = r / (K2 * K2);
    const int k = r - c * (K2 * K2);
    const int ky = k / K2, kx = k - ky * K2;
    const float* dc = s_d2 + f * (H2 * W2);
    const float* ac = s_a1 + c * (H1 * W1);
    float acc = 0.f;
    #pragma unroll 1
    for (int oy = 0; oy < H2; ++oy) {
      const float* arow = ac + (S2 * oy + ky) * W1 + kx;
      const float* drow = dc + oy * W2;
      #pragma unroll
      for (int ox = 0; ox < W2; ++ox)
        acc = fmaf(arow[S2 * ox], drow[ox], acc);
    }
    my_slab[DW1_N + e] = acc;
  }

  // d_a1[c][y][x]: stride-2 transposed conv — per axis exactly two
  // parity-matched kernel offsets; bounds clamp, branch-light inner.
  #pragma unroll 1
  for (int e = tid + eslice * NTHREADS; e < A1_N;
       e += NTHREADS * esplit) {
    const int c = e / (H1 * W1);
    const int p = e - c * (H1 * W1);
    const int y = p / W1, x = p - y * W1;
    const int ky0 = (y & 1), kx0 = (x & 1);
    float acc = 0.f;
    #pragma unroll 4
    for (int f = 0; f < F2; ++f) {
      const float* dc = s_d2 + f * (H2 * W2);
      const float* wc = w2 + (f * F1 + c) * (K2 * K2);
      #pragma unroll
      for (int a = 0; a < 2; ++a) {
        const int ky = ky0 + 2 * a;
        const int oy = (y - ky) >> 1;
        if (oy < 0 || oy >= H2) continue;
        #pragma unroll
        for (int bq = 0; bq < 2; ++bq) {
          const int kx = kx0 + 2 * bq;
          const int ox = (x - kx) >> 1;
          if (ox < 0 || ox >= W2) continue;
          acc = fmaf(dc[oy * W2 + ox], wc[ky * K2 + kx], acc);
        }
      }
    }
    const float v = s_a1[e] > 0.f ? acc : 0.f;
    d1g[(size_t)n * A1_N + e] = v;
  }
}

// ---------------------------------------------------------------------------
// backward kernel 3: dW1 from the u8 frame (fused /255)
// LDS: img + d1 = 53.6 KiB
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS, 4)
shallow_bwd_k3(const unsigned char* __restrict__ img,
               const float* __restrict__ d1g, int N,
               float* __restrict__ slab) {
  const int esplit = gridDim.y, eslice = blockIdx.y;
  __shared__ __align__(16) float s_d1[A1_N];
  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  float* my_slab = slab + (size_t)n * SLAB_N;
  const unsigned char* s_img = img + (size_t)n * IMG_BYTES;

  #pragma unroll 1
  for (int i = tid; i < A1_N; i += NTHREADS)
    s_d1[i] = d1g[(size_t)n * A1_N + i];
  __syncthreads();

  if (eslice == 0) {
    for (int f = tid; f < F1; f += NTHREADS) {
      float acc = 0.f;
      const float* dc = s_d1 + f * (H1 * W1);
      for (int p = 0; p < H1 * W1; ++p) acc += dc[p];
      my_slab[DB_OFF + f] = acc;
    }
  }

  const float inv255 = 1.0f / 255.0f;
  #pragma unroll 1
  for (int e = tid + eslice * NTHREADS; e < DW1_N;
       e += NTHREADS * esplit) {
    const int f = e / (C0 * K1 * K1);
    const int r = e - f * (C0 * K1 * K1);
    const int c = r / (K1 * K1);
    const int k = r - c * (K1 * K1);
    const int ky = k / K1, kx = k - ky * K1;
    const float* dc = s_d1 + f * (H1 * W1);
    const unsigned char* ic = s_img + c * (H0 * W0);
    float acc = 0.f;
    #pragma unroll 1
    for (int oy = 0; oy < H1; ++oy) {
      const unsigned char* irow = ic + (S1 * oy + ky) * W0 + kx;
      const float* drow = dc + oy * W1;
      #pragma unroll 4
      for (int ox = 0; ox < W1; ++ox)
        acc = fmaf((float)irow[S1 * ox], drow[ox], acc);
    }
    my_slab[e] = acc * inv255;
  }
}

}  // namespace impala

extern "C" {

void launch_shallow_encoder_fwd(const unsigned char* img, const float* w1,
                                const float* b1, const float* w2,
                                const float* b2, const float* w3,
                                const float* b3, int N, float* a1,
                                float* a2, float* a3, int save_for_bwd,
                                hipStream_t stream) {
  hipLaunchKernelGGL(impala::shallow_encoder_fwd, dim3(N),
                     dim3(impala::NTHREADS), 0, stream, img, w1, b1, w2, b2,
                     w3, b3, N, a1, a2, a3, save_for_bwd);
}

// d2_tmp: (N, A2_N); d1_tmp: (N, A1_N) — caller-provided workspaces
void launch_shallow_encoder_bwd(const unsigned char* img, const float* a1,
                                const float* a2, const float* a3,
                                const float* d_a3, const float* w2,
                                const float* w3, int N, float* slab,
                                float* d2_tmp, float* d1_tmp,
                                hipStream_t stream) {
  const int ES = 4;  // element-range split: more blocks, shorter loops
  hipLaunchKernelGGL(impala::shallow_bwd_k1, dim3(N, ES),
                     dim3(impala::NTHREADS), 0, stream, a2, a3, d_a3, w3, N,
                     slab, d2_tmp);
  hipLaunchKernelGGL(impala::shallow_bwd_k2, dim3(N, ES),
                     dim3(impala::NTHREADS), 0, stream, a1, d2_tmp, w2, N,
                     slab, d1_tmp);
  hipLaunchKernelGGL(impala::shallow_bwd_k3, dim3(N, ES),
                     dim3(impala::NTHREADS), 0, stream, img, d1_tmp, N, slab);
}

int shallow_encoder_slab_elems() { return impala::SLAB_N; }

}  // extern "C"
