// gfx950 fused shallow-encoder convolution kernels (BASELINE.json:5
// "batched Conv2d encoder" / SURVEY.md §2c conv_encoder_fwd/bwd).
//
// MI355X-native design: the Atari frame is TINY (4x84x84 u8 = 28 KiB),
// so instead of im2col GEMMs with NCHW<->NHWC transposes (what the
// MIOpen path does — rocprof showed igemm + batched_transpose +
// elementwise chains per layer), ONE workgroup owns ONE image and runs
// the whole conv1->conv2->conv3 chain out of LDS:
//
//   u8 frame (28 KiB, read directly — /255 fused, no float frame pass)
//   -> conv1 16x8x8/s4 -> a1 (25.6 KiB LDS) -> conv2 32x4x4/s2
//   -> a2 (10.4 KiB)   -> conv3 32x3x3    -> a3 (6.1 KiB) -> global.
//
// Peak LDS ~64 KiB -> 2 workgroups/CU; grid = N images (672 at the
// headline batch) covers the chip. Weights stream from L2 (shared by
// every workgroup; conv1 row fits L1 per filter). Bias+ReLU fused.
// All math fp32, bit-comparable to the eager fp32 reference (u8->f32
// convert and *(1/255) match torch's x.float()/255).
//
// Backward (same per-image structure, reverse order) emits per-image
// weight-grad slabs (summed over images by one reduction outside) and
// d(flat conv3 output) comes in from the fc backward; no d(frame) is
// needed (frames are inputs). The ReLU masks come from the saved
// activations.
//
// The fc / heads GEMMs stay on the MFMA BLAS path (rocBLAS) — GEMM-
// shaped work at M=672..52k belongs there; see ops/encoder.py.

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

__device__ __forceinline__ float relu(float x) { return x > 0.f ? x : 0.f; }

// ---------------------------------------------------------------------------
// forward: grid = (N) blocks x 256 threads
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NTHREADS, 2)
shallow_encoder_fwd(const unsigned char* __restrict__ img,
                    const float* __restrict__ w1, const float* __restrict__ b1,
                    const float* __restrict__ w2, const float* __restrict__ b2,
                    const float* __restrict__ w3, const float* __restrict__ b3,
                    int N, float* __restrict__ a1g, float* __restrict__ a2g,
                    float* __restrict__ a3g, int save_for_bwd) {
  __shared__ __align__(16) unsigned char s_img[IMG_BYTES];
  __shared__ __align__(16) float s_a1[A1_N];
  __shared__ __align__(16) float s_a2[A2_N];

  const int n = blockIdx.x;
  const int tid = threadIdx.x;

  // stage the u8 frame: 28224 B = 7056 dwords, coalesced
  {
    const unsigned int* src =
        reinterpret_cast<const unsigned int*>(img + (size_t)n * IMG_BYTES);
    unsigned int* dst = reinterpret_cast<unsigned int*>(s_img);
    for (int i = tid; i < IMG_BYTES / 4; i += NTHREADS) dst[i] = src[i];
  }
  __syncthreads();

  // ---- conv1: 16 filters x 20x20, k=8 s=4, u8 input (fused /255) ----
  // thread t: filter f = t/16, pixel-lane p = t%16 walks the 400 pixels.
  {
    const float inv255 = 1.0f / 255.0f;
    const int f = tid >> 4;
    const int lane = tid & 15;
    const float* wf = w1 + f * (C0 * K1 * K1);
    const float bias = b1[f];
    #pragma unroll 1
    for (int px = lane; px < H1 * W1; px += 16) {
      const int oy = px / W1, ox = px - oy * W1;
      float acc = 0.f;
      for (int c = 0; c < C0; ++c) {
        const unsigned char* ic = s_img + c * (H0 * W0);
        const float* wc = wf + c * (K1 * K1);
        #pragma unroll 2
        for (int ky = 0; ky < K1; ++ky) {
          // 8 u8 pixels = 2 aligned dwords (4*ox is dword-aligned)
          const unsigned char* row = ic + (S1 * oy + ky) * W0 + S1 * ox;
          const unsigned int lo = *reinterpret_cast<const unsigned int*>(row);
          const unsigned int hi =
              *reinterpret_cast<const unsigned int*>(row + 4);
          const float* wr = wc + ky * K1;
          acc = fmaf((float)(lo & 0xff), wr[0], acc);
          acc = fmaf((float)((lo >> 8) & 0xff), wr[1], acc);
          acc = fmaf((float)((lo >> 16) & 0xff), wr[2], acc);
          acc = fmaf((float)(lo >> 24), wr[3], acc);
          acc = fmaf((float)(hi & 0xff), wr[4], acc);
          acc = fmaf((float)((hi >> 8) & 0xff), wr[5], acc);
          acc = fmaf((float)((hi >> 16) & 0xff), wr[6], acc);
          acc = fmaf((float)(hi >> 24), wr[7], acc);
        }
      }
      const float v = relu(fmaf(acc, inv255, bias));
      const int idx = f * (H1 * W1) + px;
      s_a1[idx] = v;
      if (save_for_bwd) a1g[(size_t)n * A1_N + idx] = v;
    }
  }
  __syncthreads();

  // ---- conv2: 32 filters x 9x9, k=4 s=2, fp32 from LDS ----
  // thread t: filter f = t/8, pixel-lane p = t%8 walks the 81 pixels.
  {
    const int f = tid >> 3;
    const int lane = tid & 7;
    const float* wf = w2 + f * (F1 * K2 * K2);
    const float bias = b2[f];
    #pragma unroll 1
    for (int px = lane; px < H2 * W2; px += 8) {
      const int oy = px / W2, ox = px - oy * W2;
      float acc = bias;
      for (int c = 0; c < F1; ++c) {
        const float* ic = s_a1 + c * (H1 * W1);
        const float* wc = wf + c * (K2 * K2);
        #pragma unroll
        for (int ky = 0; ky < K2; ++ky) {
          const float* row = ic + (S2 * oy + ky) * W1 + S2 * ox;
          const float* wr = wc + ky * K2;
          acc = fmaf(row[0], wr[0], acc);
          acc = fmaf(row[1], wr[1], acc);
          acc = fmaf(row[2], wr[2], acc);
          acc = fmaf(row[3], wr[3], acc);
        }
      }
      const float v = relu(acc);
      const int idx = f * (H2 * W2) + px;
      s_a2[idx] = v;
      if (save_for_bwd) a2g[(size_t)n * A2_N + idx] = v;
    }
  }
  __syncthreads();

  // ---- conv3: 32 filters x 7x7, k=3 s=1 ----
  {
    const int f = tid >> 3;
    const int lane = tid & 7;
    const float* wf = w3 + f * (F2 * K3 * K3);
    const float bias = b3[f];
    #pragma unroll 1
    for (int px = lane; px < H3 * W3; px += 8) {
      const int oy = px / W3, ox = px - oy * W3;
      float acc = bias;
      for (int c = 0; c < F2; ++c) {
        const float* ic = s_a2 + c * (H2 * W2);
        const float* wc = wf + c * (K3 * K3);
        #pragma unroll
        for (int ky = 0; ky < K3; ++ky) {
          const float* row = ic + (oy + ky) * W2 + ox;
          const float* wr = wc + ky * K3;
          acc = fmaf(row[0], wr[0], acc);
          acc = fmaf(row[1], wr[1], acc);
          acc = fmaf(row[2], wr[2], acc);
        }
      }
      // a3 layout matches torch flatten(1): [f][oy][ox]
      a3g[(size_t)n * A3_N + f * (H3 * W3) + px] = relu(acc);
    }
  }
}

// ---------------------------------------------------------------------------
// backward: grid = (N) blocks x 256 threads.
// In: d_a3 (N,1568) = d(flat conv3 out), saved a1/a2/a3, frame, weights.
// Out: per-image slab [dW1|dW2|dW3|db1|db2|db3] (N, SLAB_N); reduced
// over N outside. No d(frame).
//
// LDS is a phased arena (78.4 KiB -> 2 workgroups/CU): the u8 frame is
// loaded LAST, over the then-dead a2/d3/d2 regions, after db3/db2 are
// banked — a static per-buffer layout needs 106 KiB and halves
// occupancy.
// ---------------------------------------------------------------------------
constexpr int ARENA_F = (A2_N + A3_N + A2_N + A1_N + A1_N);  // floats
__global__ void __launch_bounds__(NTHREADS, 2)
shallow_encoder_bwd(const unsigned char* __restrict__ img,
                    const float* __restrict__ a1g,
                    const float* __restrict__ a2g,
                    const float* __restrict__ a3g,
                    const float* __restrict__ d_a3g,
                    const float* __restrict__ w2, const float* __restrict__ w3,
                    int N, float* __restrict__ slab) {
  __shared__ __align__(16) float s_arena[ARENA_F];
  float* s_a2 = s_arena;                 // A2_N
  float* s_d3 = s_a2 + A2_N;             // A3_N
  float* s_d2 = s_d3 + A3_N;             // A2_N
  float* s_a1 = s_d2 + A2_N;             // A1_N
  float* s_d1 = s_a1 + A1_N;             // A1_N
  // phase-5 overlay: the u8 frame covers [s_a2 .. s_a2+28224B) — the
  // dead a2/d3/d2 (+ start of a1) regions
  unsigned char* s_img = reinterpret_cast<unsigned char*>(s_arena);

  const int n = blockIdx.x;
  const int tid = threadIdx.x;
  float* my_slab = slab + (size_t)n * SLAB_N;
  const int DB_OFF = DW1_N + DW2_N + DW3_N;

  // stage a2, d3 (relu-masked), a1
  for (int i = tid; i < A2_N; i += NTHREADS)
    s_a2[i] = a2g[(size_t)n * A2_N + i];
  for (int i = tid; i < A3_N; i += NTHREADS) {
    const float m = a3g[(size_t)n * A3_N + i] > 0.f ? 1.f : 0.f;
    s_d3[i] = m * d_a3g[(size_t)n * A3_N + i];
  }
  for (int i = tid; i < A1_N; i += NTHREADS)
    s_a1[i] = a1g[(size_t)n * A1_N + i];
  __syncthreads();

  // ---- dW3[f][c][ky][kx] = sum_px s_a2[c][oy+ky][ox+kx] * s_d3[f][px] ----
  #pragma unroll 1
  for (int e = tid; e < DW3_N; e += NTHREADS) {
    const int f = e / (F2 * K3 * K3);
    const int r = e - f * (F2 * K3 * K3);
    const int c = r / (K3 * K3);
    const int k = r - c * (K3 * K3);
    const int ky = k / K3, kx = k - ky * K3;
    const float* dc = s_d3 + f * (H3 * W3);
    const float* ac = s_a2 + c * (H2 * W2);
    float acc = 0.f;
    for (int oy = 0; oy < H3; ++oy) {
      const float* arow = ac + (oy + ky) * W2 + kx;
      const float* drow = dc + oy * W3;
      #pragma unroll
      for (int ox = 0; ox < W3; ++ox) acc = fmaf(arow[ox], drow[ox], acc);
    }
    my_slab[DW1_N + DW2_N + e] = acc;
  }

  // ---- d_a2 (masked by a2>0) ----
  #pragma unroll 1
  for (int e = tid; e < A2_N; e += NTHREADS) {
    const int c = e / (H2 * W2);
    const int p = e - c * (H2 * W2);
    const int y = p / W2, x = p - y * W2;
    float acc = 0.f;
    for (int f = 0; f < F3; ++f) {
      const float* dc = s_d3 + f * (H3 * W3);
      const float* wc = w3 + (f * F2 + c) * (K3 * K3);
      for (int ky = 0; ky < K3; ++ky) {
        const int oy = y - ky;
        if (oy < 0 || oy >= H3) continue;
        for (int kx = 0; kx < K3; ++kx) {
          const int ox = x - kx;
          if (ox < 0 || ox >= W3) continue;
          acc = fmaf(dc[oy * W3 + ox], wc[ky * K3 + kx], acc);
        }
      }
    }
    s_d2[e] = s_a2[e] > 0.f ? acc : 0.f;
  }
  __syncthreads();

  // ---- dW2 ----
  #pragma unroll 1
  for (int e = tid; e < DW2_N; e += NTHREADS) {
    const int f = e / (F1 * K2 * K2);
    const int r = e - f * (F1 * K2 * K2);
    const int c = r / (K2 * K2);
    const int k = r - c * (K2 * K2);
    const int ky = k / K2, kx = k - ky * K2;
    const float* dc = s_d2 + f * (H2 * W2);
    const float* ac = s_a1 + c * (H1 * W1);
    float acc = 0.f;
    for (int oy = 0; oy < H2; ++oy) {
      const float* arow = ac + (S2 * oy + ky) * W1 + kx;
      const float* drow = dc + oy * W2;
      #pragma unroll
      for (int ox = 0; ox < W2; ++ox)
        acc = fmaf(arow[S2 * ox], drow[ox], acc);
    }
    my_slab[DW1_N + e] = acc;
  }

  // ---- d_a1 (masked by a1>0) ----
  #pragma unroll 1
  for (int e = tid; e < A1_N; e += NTHREADS) {
    const int c = e / (H1 * W1);
    const int p = e - c * (H1 * W1);
    const int y = p / W1, x = p - y * W1;
    float acc = 0.f;
    for (int f = 0; f < F2; ++f) {
      const float* dc = s_d2 + f * (H2 * W2);
      const float* wc = w2 + (f * F1 + c) * (K2 * K2);
      for (int ky = 0; ky < K2; ++ky) {
        const int ty = y - ky;
        if (ty < 0 || (ty & 1)) continue;
        const int oy = ty >> 1;
        if (oy >= H2) continue;
        for (int kx = 0; kx < K2; ++kx) {
          const int tx = x - kx;
          if (tx < 0 || (tx & 1)) continue;
          const int ox = tx >> 1;
          if (ox >= W2) continue;
          acc = fmaf(dc[oy * W2 + ox], wc[ky * K2 + kx], acc);
        }
      }
    }
    s_d1[e] = s_a1[e] > 0.f ? acc : 0.f;
  }
  __syncthreads();

  // ---- bank db3/db2/db1 BEFORE the frame overlays their buffers ----
  for (int f = tid; f < F3; f += NTHREADS) {
    float acc = 0.f;
    const float* dc = s_d3 + f * (H3 * W3);
    for (int p = 0; p < H3 * W3; ++p) acc += dc[p];
    my_slab[DB_OFF + F1 + F2 + f] = acc;
  }
  for (int f = tid; f < F2; f += NTHREADS) {
    float acc = 0.f;
    const float* dc = s_d2 + f * (H2 * W2);
    for (int p = 0; p < H2 * W2; ++p) acc += dc[p];
    my_slab[DB_OFF + F1 + f] = acc;
  }
  for (int f = tid; f < F1; f += NTHREADS) {
    float acc = 0.f;
    const float* dc = s_d1 + f * (H1 * W1);
    for (int p = 0; p < H1 * W1; ++p) acc += dc[p];
    my_slab[DB_OFF + f] = acc;
  }
  __syncthreads();

  // ---- overlay: stage the u8 frame over the dead a2/d3/d2 regions ----
  {
    const unsigned int* src =
        reinterpret_cast<const unsigned int*>(img + (size_t)n * IMG_BYTES);
    unsigned int* dst = reinterpret_cast<unsigned int*>(s_img);
    for (int i = tid; i < IMG_BYTES / 4; i += NTHREADS) dst[i] = src[i];
  }
  __syncthreads();

  // ---- dW1 from the u8 frame (fused /255) ----
  {
    const float inv255 = 1.0f / 255.0f;
    #pragma unroll 1
    for (int e = tid; e < DW1_N; e += NTHREADS) {
      const int f = e / (C0 * K1 * K1);
      const int r = e - f * (C0 * K1 * K1);
      const int c = r / (K1 * K1);
      const int k = r - c * (K1 * K1);
      const int ky = k / K1, kx = k - ky * K1;
      const float* dc = s_d1 + f * (H1 * W1);
      const unsigned char* ic = s_img + c * (H0 * W0);
      float acc = 0.f;
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

void launch_shallow_encoder_bwd(const unsigned char* img, const float* a1,
                                const float* a2, const float* a3,
                                const float* d_a3, const float* w2,
                                const float* w3, int N, float* slab,
                                hipStream_t stream) {
  hipLaunchKernelGGL(impala::shallow_encoder_bwd, dim3(N),
                     dim3(impala::NTHREADS), 0, stream, img, a1, a2, a3,
                     d_a3, w2, w3, N, slab);
}

int shallow_encoder_slab_elems() { return impala::SLAB_N; }

}  // extern "C"
