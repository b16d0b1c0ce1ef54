// Shallow IMPALA encoder on bf16 MFMA matrix cores (learner side).
//
// Motivation (ROADMAP re-framing): after the int8 actor chain the system
// is learner-bound; the fp32 LDS-fused encoder (conv_kernels.hip) runs
// at ~70% of the fp32 VECTOR peak, i.e. 1/16 of the bf16 MFMA rate.
// These kernels re-express the three convs as implicit GEMMs on
// v_mfma_f32_16x16x32_bf16 with the fragment mapping hardware-validated
// by deep_probe (profiles/deep_probe_r01.log).
//
// Layouts (all bf16):
//   frames  u8  [N][4][84][84]          (learner batch input, NCHW)
//   a1      bf16 [N][20][20][16]  NHWC
//   a2      bf16 [N][9][9][32]    NHWC
//   a3      bf16 [N][1568]        NCHW-flat (f*49+px) — matches the
//                                 torch fc weight layout directly
// K-packings (torch-natural, no host reorder beyond flatten):
//   conv1 k = c*64 + ky*8 + kx   (w1.flatten(1):   [16][256])
//   conv2 k = c*16 + ky*4 + kx   (w2.flatten(1):   [32][256])
//   conv3 k = tap*32 + c         (w3.permute(0,2,3,1).flatten(1): [32][288])
//
// Correctness-first: single-buffered LDS, no XCD swizzle. Validation:
// deep_probe --shallow cases (round-2 ladder).
#include <hip/hip_runtime.h>

namespace impala_shallow_mfma {

typedef __bf16 bf16_t;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float floatx4;

__device__ __forceinline__ float bf2f(bf16_t v) { return (float)v; }
__device__ __forceinline__ bf16_t f2bf(float v) { return (bf16_t)v; }

constexpr int C0 = 4, H0 = 84, W0 = 84;
constexpr int F1 = 16, K1 = 8, S1 = 4, H1 = 20, W1 = 20;
constexpr int F2 = 32, K2 = 4, S2 = 2, H2 = 9, W2 = 9;
constexpr int F3 = 32, K3 = 3, H3 = 7, W3 = 7;

// ---------------------------------------------------------------------------
// conv1: 4x84x84 u8 -> 16x20x20, K=8 S=4, fused /255 + bias + ReLU.
// ONE workgroup per image (round 2; was 2 y-halves): the full 4x84x84
// image is staged once (55.1 KiB LDS, 2 blocks/CU) instead of two
// 68-row halves (136 rows), and the weights once instead of twice —
// PMC showed the kernel 70% wave-parked on staging.  M-tile = 400 px
// (25 m-frags round-robin over 4 waves), N = 16 (one frag), K = 256
// (8 steps).  A-frag = patch[c][4*oy+ky][4*ox + 0..7], an 8-half run
// at a 4-half-aligned offset -> two b64 reads.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv1_mfma(
    const unsigned char* __restrict__ frames, const bf16_t* __restrict__ w,
    const float* __restrict__ b, bf16_t* __restrict__ a1, int N) {
  constexpr int PY = 84, PX = 84;           // full image (halfs)
  __shared__ bf16_t s_p[C0 * PY * PX];      // 55.1 KiB
  __shared__ bf16_t s_w[F1 * 256];          // 8 KiB

  const int img = blockIdx.x;
  if (img >= N) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  {
    const bf16_t* ws = w;
    for (int i = tid; i < F1 * 256; i += 256) s_w[i] = ws[i];
  }
  {
    // row-wise staging: one div per ROW, uchar4 global loads + bf16x4
    // LDS stores (84 % 4 == 0 keeps both aligned)
    const unsigned char* fr = frames + (size_t)img * C0 * H0 * W0;
    for (int cy = tid; cy < C0 * PY; cy += 256) {
      bf16_t* dst = s_p + (size_t)cy * PX;
      const unsigned char* src = fr + (size_t)cy * W0;
      #pragma unroll
      for (int x = 0; x < PX; x += 4) {
        const uchar4 v = *reinterpret_cast<const uchar4*>(src + x);
        bf16x4 o;
        o[0] = f2bf(v.x * (1.f / 255.f));
        o[1] = f2bf(v.y * (1.f / 255.f));
        o[2] = f2bf(v.z * (1.f / 255.f));
        o[3] = f2bf(v.w * (1.f / 255.f));
        *reinterpret_cast<bf16x4*>(dst + x) = o;
      }
    }
  }
  __syncthreads();

  constexpr int MFRAGS = 25;                 // 400 output px
  constexpr int MPW = 7;                     // ceil(25/4) per wave
  floatx4 acc[MPW];
  #pragma unroll
  for (int m = 0; m < MPW; ++m) acc[m] = floatx4{0.f, 0.f, 0.f, 0.f};
  const int arow = lane & 15;
  const int q = lane >> 4;                   // k-quarter

  #pragma unroll
  for (int s = 0; s < 8; ++s) {
    const int c = s >> 1;
    const int ky = (s & 1) * 4 + q;          // this lane's kernel row
    // B frag: s_w[col][k0..k0+7], 16B aligned
    const bf16x8 bf = *reinterpret_cast<const bf16x8*>(
        s_w + (lane & 15) * 256 + s * 32 + q * 8);
    #pragma unroll
    for (int m = 0; m < MPW; ++m) {
      const int mf = wave + m * 4;
      if (mf >= MFRAGS) break;
      const int p = mf * 16 + arow;               // px in 20x20 out
      const int oy = p / W1, ox = p % W1;
      const bf16_t* src =
          s_p + (c * PY + oy * S1 + ky) * PX + ox * S1;
      bf16x8 af;
      *reinterpret_cast<bf16x4*>(&af) =
          *reinterpret_cast<const bf16x4*>(src);      // 8B aligned
      *(reinterpret_cast<bf16x4*>(&af) + 1) =
          *reinterpret_cast<const bf16x4*>(src + 4);
      acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[m], 0,
                                                       0, 0);
    }
  }

  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
  #pragma unroll
  for (int m = 0; m < MPW; ++m) {
    const int mf = wave + m * 4;
    if (mf >= MFRAGS) break;
    const float bias = b[ccol];
    #pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int p = mf * 16 + crow + reg;         // < 400 always
      const int oy = p / W1, ox = p % W1;
      float v = acc[m][reg] + bias;
      v = v > 0.f ? v : 0.f;
      a1[(((size_t)img * H1 + oy) * W1 + ox) * F1 + ccol] = f2bf(v);
    }
  }
}

// ---------------------------------------------------------------------------
// conv2: 16x20x20 -> 32x9x9, K=4 S=2. One WG per image. M = 96 (6 frags,
// round-robin over 4 waves), N = 32 (2 frags), K = 256 (8 steps).
// Patch: planar LDS [16][20][20] (transpose-on-stage from NHWC a1);
// A-frag = two 4-half runs (rows ky, ky+1), 4B-aligned -> element reads
// via two bf16x4-by-parts; correctness-first.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv2_mfma(
    const bf16_t* __restrict__ a1, const bf16_t* __restrict__ w,
    const float* __restrict__ b, bf16_t* __restrict__ a2, int N) {
  __shared__ bf16_t s_p[F1 * H1 * W1];      // 12.5 KiB planar
  __shared__ bf16_t s_w[F2 * 256];          // 16 KiB

  const int img = blockIdx.x;
  if (img >= N) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  for (int i = tid; i < F2 * 256; i += 256) s_w[i] = w[i];
  {
    const bf16_t* src = a1 + (size_t)img * H1 * W1 * F1;
    for (int i = tid; i < F1 * H1 * W1; i += 256) {
      const int x = i % W1;
      const int y = (i / W1) % H1;
      const int c = i / (W1 * H1);
      s_p[i] = src[((size_t)y * W1 + x) * F1 + c];
    }
  }
  __syncthreads();

  const int arow = lane & 15;
  const int q = lane >> 4;
  for (int mf = wave; mf < 6; mf += 4) {
    floatx4 acc[2];
    acc[0] = floatx4{0.f, 0.f, 0.f, 0.f};
    acc[1] = floatx4{0.f, 0.f, 0.f, 0.f};
    const int p = mf * 16 + arow;
    const int oy = p / W2, ox = p % W2;      // may exceed 80 (masked k)
    const bool live = p < H2 * W2;
    #pragma unroll
    for (int s = 0; s < 8; ++s) {
      // k0 = 32s + 8q: c = k0/16, rem = k0%16 -> ky0 = rem/4 (rem is 0
      // or 8 -> ky0 = 0 or 2), covers rows ky0, ky0+1 (kx 0..3 each)
      const int k0 = s * 32 + q * 8;
      const int c = k0 >> 4;
      const int ky0 = (k0 & 15) >> 2;
      bf16x8 af = {};
      if (live) {
        const bf16_t* r0 =
            s_p + (c * H1 + oy * S2 + ky0) * W1 + ox * S2;
        #pragma unroll
        for (int j = 0; j < 4; ++j) af[j] = r0[j];
        #pragma unroll
        for (int j = 0; j < 4; ++j) af[4 + j] = r0[W1 + j];
      }
      const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          s_w + (lane & 15) * 256 + k0);
      const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          s_w + (16 + (lane & 15)) * 256 + k0);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0, acc[0], 0,
                                                       0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1, acc[1], 0,
                                                       0, 0);
    }
    const int ccol = lane & 15;
    const int crow = (lane >> 4) * 4;
    #pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int f = n * 16 + ccol;
      const float bias = b[f];
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int pp = mf * 16 + crow + reg;
        if (pp < H2 * W2) {
          float v = acc[n][reg] + bias;
          v = v > 0.f ? v : 0.f;
          a2[((size_t)img * H2 * W2 + pp) * F2 + f] = f2bf(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// conv3: 32x9x9 -> 32x7x7 VALID (no pad), K=3 S=1. One WG per image.
// M = 64 (4 frags, 1/wave), N = 32 (2 frags), K = 288 (9 steps,
// k = tap*32 + c, NHWC patch -> contiguous 16B-aligned A reads).
// Output a3 in NCHW-FLAT order (f*49 + px) to match the torch fc.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv3_mfma(
    const bf16_t* __restrict__ a2, const bf16_t* __restrict__ w,
    const float* __restrict__ b, bf16_t* __restrict__ a3, int N) {
  __shared__ bf16_t s_p[H2 * W2 * F2];      // 5.1 KiB NHWC
  __shared__ bf16_t s_w[F3 * 288];          // 18 KiB

  const int img = blockIdx.x;
  if (img >= N) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  for (int i = tid; i < F3 * 288; i += 256) s_w[i] = w[i];
  {
    const bf16_t* src = a2 + (size_t)img * H2 * W2 * F2;
    for (int i = tid; i < H2 * W2 * F2; i += 256) s_p[i] = src[i];
  }
  __syncthreads();

  floatx4 acc[2];
  acc[0] = floatx4{0.f, 0.f, 0.f, 0.f};
  acc[1] = floatx4{0.f, 0.f, 0.f, 0.f};
  const int arow = lane & 15;
  const int kbase = (lane >> 4) * 8;
  const int p = wave * 16 + arow;
  const int oy = p / W3, ox = p % W3;
  const bool live = p < H3 * W3;
  #pragma unroll
  for (int s = 0; s < 9; ++s) {
    const int k0 = s * 32 + kbase;
    const int tap = k0 >> 5, c0 = k0 & 31;
    const int ky = tap / 3, kx = tap % 3;
    bf16x8 af = {};
    if (live)
      af = *reinterpret_cast<const bf16x8*>(
          s_p + (((oy + ky) * W2) + (ox + kx)) * F2 + c0);
    const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
        s_w + (lane & 15) * 288 + k0);
    const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
        s_w + (16 + (lane & 15)) * 288 + k0);
    acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0, acc[0], 0, 0,
                                                     0);
    acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1, acc[1], 0, 0,
                                                     0);
  }
  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
  #pragma unroll
  for (int n = 0; n < 2; ++n) {
    const int f = n * 16 + ccol;
    const float bias = b[f];
    #pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int pp = wave * 16 + crow + reg;
      if (pp < H3 * W3) {
        float v = acc[n][reg] + bias;
        v = v > 0.f ? v : 0.f;
        a3[(size_t)img * (F3 * H3 * W3) + f * (H3 * W3) + pp] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
extern "C" void launch_shallow_mfma_fwd(const unsigned char* frames,
                                        const void* w1, const float* b1,
                                        const void* w2, const float* b2,
                                        const void* w3, const float* b3,
                                        void* a1, void* a2, void* a3, int N,
                                        hipStream_t s) {
  hipLaunchKernelGGL(conv1_mfma, dim3(N), dim3(256), 0, s, frames,
                     static_cast<const bf16_t*>(w1), b1,
                     static_cast<bf16_t*>(a1), N);
  hipLaunchKernelGGL(conv2_mfma, dim3(N), dim3(256), 0, s,
                     static_cast<const bf16_t*>(a1),
                     static_cast<const bf16_t*>(w2), b2,
                     static_cast<bf16_t*>(a2), N);
  hipLaunchKernelGGL(conv3_mfma, dim3(N), dim3(256), 0, s,
                     static_cast<const bf16_t*>(a2),
                     static_cast<const bf16_t*>(w3), b3,
                     static_cast<bf16_t*>(a3), N);
}

}  // namespace impala_shallow_mfma
