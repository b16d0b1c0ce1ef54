// Backward of the MFMA bf16 shallow encoder (shallow_mfma.hip).
//
// Chain (z* = pre-ReLU): d_z3 = d_a3 .* (a3>0)   [masked convert]
//   dW3 = a2 (x) d_z3          d_a2 = conv3^T(d_z3) .* (a2>0)
//   dW2 = a1 (x) d_z2          d_a1 = conv2^T(d_z2) .* (a1>0)
//   dW1 = frames/255 (x) d_z1  (no d_frames)
// ReLU masks for d_a2/d_a1 are fused into the dX epilogues; biases are
// plain sums done with torch on the python side.
//
// dX conv3 (stride 1, VALID 9<-7): full correlation = conv3x3 of the
//   2-halo-padded d_z3 with rot180/transposed weights — same implicit
//   GEMM as forward, K = 9 taps x 32 cout (c-inner, NHWC LDS patch).
// dX conv2 (stride 2, K=4, 20<-9): four (iy&1, ix&1) parity classes,
//   each a regular 10x10 grid with a fixed 2x2 tap subset ->
//   K = 2*2*32 = 128 per class, per-class packed weights.
// dW: per-image outer products M = K_fwd, N = cout, K = px (deep-dw
//   recipe), accumulated into DW_NSLAB slab copies with fp32 atomics.
#include <hip/hip_runtime.h>

namespace impala_shallow_mfma_bwd {

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
// slab count is a HOST-side runtime knob (IMPALA_SHALLOW_SLAB,
// default 128): more slabs = less dW atomic contention (conv1_dw has
// 640 concurrent writers) but more zero-fill + finalize traffic.
// r7 (256 slabs): conv1_dw 48 us; r24 (64): 99 us; 128 = the measured
// balance point candidate.
static inline int shallow_slab_count() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("IMPALA_SHALLOW_SLAB");
    v = e ? atoi(e) : 128;
    if (v < 4) v = 4;
    if (v > 1024) v = 1024;
    v &= ~3;                      // finalize splits the slab dim by 4
  }
  return v;
}
constexpr int DWG = 512;    // dW resident-workgroup cap (2/CU x 256)
// bwd_finalize geometry — grid and kernel MUST agree (single source):
constexpr int FIN_SPLIT = 4;    // slab-dim parallelism per dW chunk
constexpr int FIN_BB = 160;     // db column-sum blocks per tensor
constexpr int FIN_GRID = (16 + 32 + 36) * FIN_SPLIT + 3 * FIN_BB;
// (kernels take the runtime slab count as an argument)

// --------------------------------------------------------------------------
// masked convert: d_z3 = bf16(d_a3_f32) .* (a3 > 0); both NCHW-flat.
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256) void mask_convert(
    const float* __restrict__ d, const bf16_t* __restrict__ act,
    bf16_t* __restrict__ o, long n) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    o[i] = f2bf(bf2f(act[i]) > 0.f ? d[i] : 0.f);
}

// --------------------------------------------------------------------------
// conv3 dX: d_a2[iy][ix][c] = sum_{ky,kx,f} dz3p[iy+ky][ix+kx][f]
//   * wr[c][(ky*3+kx)*32+f], masked by a2>0.  dz3 arrives NCHW-flat
//   [f*49+px]; staged into a 13x13x32 NHWC zero-padded LDS patch.
//   M = 96 (6 frags round-robin), N = 32 (c), K = 288 (9 steps).
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv3_dx(
    const bf16_t* __restrict__ dz3, const bf16_t* __restrict__ wr,
    const bf16_t* __restrict__ a2, bf16_t* __restrict__ da2, int N) {
  constexpr int P = H3 + 2 * (K3 - 1);      // 11? no: 7 + 4 = 11
  __shared__ bf16_t s_p[P * P * F3];        // 11*11*32 = 7.6 KiB
  __shared__ bf16_t s_w[F2 * 288];          // 18 KiB

  const int img = blockIdx.x;
  if (img >= N) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  for (int i = tid; i < F2 * 288; i += 256) s_w[i] = wr[i];
  {
    const bf16_t* src = dz3 + (size_t)img * (F3 * H3 * W3);
    for (int i = tid; i < P * P * F3; i += 256) {
      const int f = i % F3;
      const int x = (i / F3) % P;
      const int y = i / (F3 * P);
      const int sy = y - (K3 - 1), sx = x - (K3 - 1);
      s_p[i] = (sy >= 0 && sy < H3 && sx >= 0 && sx < W3)
                   ? src[f * (H3 * W3) + sy * W3 + sx]
                   : f2bf(0.f);
    }
  }
  __syncthreads();

  const int arow = lane & 15;
  const int kbase = (lane >> 4) * 8;
  for (int mf = wave; mf < 6; mf += 4) {
    floatx4 acc[2];
    acc[0] = floatx4{0.f, 0.f, 0.f, 0.f};
    acc[1] = floatx4{0.f, 0.f, 0.f, 0.f};
    const int p = mf * 16 + arow;
    const int iy = p / W2, ix = p % W2;
    const bool live = p < H2 * W2;
    #pragma unroll
    for (int s = 0; s < 9; ++s) {
      const int k0 = s * 32 + kbase;
      const int tap = k0 >> 5, f0 = k0 & 31;
      const int ky = tap / 3, kx = tap % 3;
      bf16x8 af = {};
      if (live)
        af = *reinterpret_cast<const bf16x8*>(
            s_p + ((iy + ky) * P + (ix + kx)) * F3 + f0);
      const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          s_w + (lane & 15) * 288 + k0);
      const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          s_w + (16 + (lane & 15)) * 288 + k0);
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0, acc[0], 0,
                                                       0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1, acc[1], 0,
                                                       0, 0);
    }
    const int ccol = lane & 15;
    const int crow = (lane >> 4) * 4;
    #pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int c = n * 16 + ccol;
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int pp = mf * 16 + crow + reg;
        if (pp < H2 * W2) {
          const size_t off = ((size_t)img * H2 * W2 + pp) * F2 + c;
          float v = acc[n][reg];
          if (bf2f(a2[off]) <= 0.f) v = 0.f;
          da2[off] = f2bf(v);
        }
      }
    }
  }
}

// --------------------------------------------------------------------------
// conv2 dX: four parity classes (py,px). Class grid 10x10 (100 px, 7
//   frags), K = 128 = (kyi,kxi in 2x2) x 32 f; per-class packed weights
//   wr2[(py*2+px)][c][ (kyi*2+kxi)*32 + f ] with
//   wr2[...] = w2[f][c][py+2kyi][px+2kxi].  dz2 NHWC staged whole
//   (9x9x32); out-of-range oy/ox contribute zeros. Masked by a1>0.
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv2_dx(
    const bf16_t* __restrict__ dz2, const bf16_t* __restrict__ wr2,
    const bf16_t* __restrict__ a1, bf16_t* __restrict__ da1, int N) {
  __shared__ bf16_t s_d[H2 * W2 * F2];      // 5.1 KiB NHWC
  __shared__ bf16_t s_w[4 * F1 * 128];      // 16 KiB

  const int img = blockIdx.x;
  if (img >= N) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  for (int i = tid; i < 4 * F1 * 128; i += 256) s_w[i] = wr2[i];
  {
    const bf16_t* src = dz2 + (size_t)img * H2 * W2 * F2;
    for (int i = tid; i < H2 * W2 * F2; i += 256) s_d[i] = src[i];
  }
  __syncthreads();

  const int arow = lane & 15;
  const int kbase = (lane >> 4) * 8;
  for (int cls = 0; cls < 4; ++cls) {
    const int py = cls >> 1, px = cls & 1;
    const bf16_t* wc = s_w + cls * F1 * 128;
    for (int mf = wave; mf < 7; mf += 4) {
      floatx4 acc = floatx4{0.f, 0.f, 0.f, 0.f};
      const int p = mf * 16 + arow;
      const int yq = p / 10, xq = p % 10;   // class-grid coords
      const bool live = p < 100;
      #pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int k0 = s * 32 + kbase;
        const int t = k0 >> 5;              // (kyi*2 + kxi)
        const int f0 = k0 & 31;
        const int oy = yq - (t >> 1);       // oy = y' - kyi
        const int ox = xq - (t & 1);
        bf16x8 af = {};
        if (live && oy >= 0 && oy < H2 && ox >= 0 && ox < W2)
          af = *reinterpret_cast<const bf16x8*>(
              s_d + ((size_t)oy * W2 + ox) * F2 + f0);
        const bf16x8 bf = *reinterpret_cast<const bf16x8*>(
            wc + (lane & 15) * 128 + k0);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0,
                                                      0);
      }
      const int ccol = lane & 15;
      const int crow = (lane >> 4) * 4;
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int pp = mf * 16 + crow + reg;
        if (pp < 100) {
          const int iy = 2 * (pp / 10) + py, ix = 2 * (pp % 10) + px;
          const size_t off = (((size_t)img * H1 + iy) * W1 + ix) * F1 + ccol;
          float v = acc[reg];
          if (bf2f(a1[off]) <= 0.f) v = 0.f;
          da1[off] = f2bf(v);
        }
      }
    }
  }
}

// --------------------------------------------------------------------------
// dW3: M = 288 (k_fwd = tap*32+c), N = 32 (f), K = 49 px (1 step of 32
//   + 1 step of 17 -> 64 padded).  A[m][px] = a2[(oy+ky)(ox+kx)][c]
//   element gathers from the NHWC a2 LDS; B[px][f] from dz3 staged
//   [f][56] (NCHW-flat rows padded) -> contiguous 8-px reads.
//   Slab-atomic accumulate into dw3_slab[s][m*32+f] (k-major layout;
//   python reshapes).
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv3_dw(
    const bf16_t* __restrict__ a2, const bf16_t* __restrict__ dz3,
    float* __restrict__ dw_slab, int N, int nslab) {
  __shared__ bf16_t s_a[H2 * W2 * F2];      // 5.1 KiB NHWC
  __shared__ bf16_t s_d[F3 * 64];           // 4 KiB; rows padded to the
                                            // k-loop's 2x32 px range

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  float* dw = dw_slab + (size_t)(blockIdx.x % nslab) * (288 * F3);
  constexpr int MPW = 5;                    // ceil(18/4) m-frags per wave
  floatx4 acc[MPW][2];
  #pragma unroll
  for (int m = 0; m < MPW; ++m) {
    acc[m][0] = floatx4{0.f, 0.f, 0.f, 0.f};
    acc[m][1] = floatx4{0.f, 0.f, 0.f, 0.f};
  }
  const int kbase = (lane >> 4) * 8;

  for (int img = blockIdx.x; img < N; img += gridDim.x) {
  __syncthreads();
  {
    const bf16_t* src = a2 + (size_t)img * H2 * W2 * F2;
    for (int i = tid; i < H2 * W2 * F2; i += 256) s_a[i] = src[i];
    const bf16_t* sd = dz3 + (size_t)img * (F3 * H3 * W3);
    for (int i = tid; i < F3 * 64; i += 256) {
      const int px = i % 64, f = i / 64;
      s_d[i] = px < H3 * W3 ? sd[f * (H3 * W3) + px] : f2bf(0.f);
    }
  }
  __syncthreads();

  // 18 m-frags (288/16), round-robin over waves
  #pragma unroll
  for (int mi = 0; mi < MPW; ++mi) {
    const int mf = wave + mi * 4;
    if (mf >= 18) break;
    const int m = mf * 16 + (lane & 15);    // k_fwd = tap*32 + c
    const int tap = m >> 5, c = m & 31;
    const int ky = tap / 3, kx = tap % 3;
    #pragma unroll
    for (int s = 0; s < 2; ++s) {           // px steps (49 -> 64)
      bf16x8 af = {}, b0 = {}, b1 = {};
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int px = s * 32 + kbase + j;
        if (px < H3 * W3) {
          const int oy = px / W3, ox = px % W3;
          af[j] = s_a[((oy + ky) * W2 + (ox + kx)) * F2 + c];
        }
      }
      const int pxb = s * 32 + kbase;
      b0 = *reinterpret_cast<const bf16x8*>(
          s_d + (lane & 15) * 64 + pxb);
      b1 = *reinterpret_cast<const bf16x8*>(
          s_d + (16 + (lane & 15)) * 64 + pxb);
      acc[mi][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0,
                                                           acc[mi][0], 0,
                                                           0, 0);
      acc[mi][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1,
                                                           acc[mi][1], 0,
                                                           0, 0);
    }
  }
  }  // image loop

  {
    const int dcol = lane & 15;
    const int drow = (lane >> 4) * 4;
    #pragma unroll
    for (int mi = 0; mi < MPW; ++mi) {
      const int mf = wave + mi * 4;
      if (mf >= 18) break;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int f = n * 16 + dcol;
        #pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int mm = mf * 16 + drow + reg;
          atomicAdd(&dw[(size_t)mm * F3 + f], acc[mi][n][reg]);
        }
      }
    }
  }
}

// --------------------------------------------------------------------------
// dW2: M = 256 (c*16+tap), N = 32, K = 81 px (3 steps of 32, padded).
//   A from planar a1 LDS; B from dz2 staged TRANSPOSED [f][96].
//   Slab layout dw2_slab[s][m*32+f].
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv2_dw(
    const bf16_t* __restrict__ a1, const bf16_t* __restrict__ dz2,
    float* __restrict__ dw_slab, int N, int nslab) {
  __shared__ bf16_t s_a[F1 * H1 * W1];      // 12.5 KiB planar
  __shared__ bf16_t s_d[F2 * 96];           // 6 KiB transposed

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  float* dw = dw_slab + (size_t)(blockIdx.x % nslab) * (256 * F2);
  constexpr int MPW = 4;                    // 16 m-frags over 4 waves
  floatx4 acc[MPW][2];
  #pragma unroll
  for (int m = 0; m < MPW; ++m) {
    acc[m][0] = floatx4{0.f, 0.f, 0.f, 0.f};
    acc[m][1] = floatx4{0.f, 0.f, 0.f, 0.f};
  }
  const int kbase = (lane >> 4) * 8;

  for (int img = blockIdx.x; img < N; img += gridDim.x) {
  __syncthreads();
  {
    const bf16_t* src = a1 + (size_t)img * H1 * W1 * F1;
    for (int i = tid; i < F1 * H1 * W1; i += 256) {
      const int x = i % W1;
      const int y = (i / W1) % H1;
      const int c = i / (W1 * H1);
      s_a[i] = src[((size_t)y * W1 + x) * F1 + c];
    }
    const bf16_t* sd = dz2 + (size_t)img * H2 * W2 * F2;
    for (int i = tid; i < F2 * 96; i += 256) {
      const int px = i % 96, f = i / 96;
      s_d[i] = px < H2 * W2 ? sd[(size_t)px * F2 + f] : f2bf(0.f);
    }
  }
  __syncthreads();

  #pragma unroll
  for (int mi = 0; mi < MPW; ++mi) {        // 256/16 m-frags
    const int mf = wave + mi * 4;
    const int m = mf * 16 + (lane & 15);    // c*16 + ky*4 + kx
    const int c = m >> 4;
    const int ky = (m & 15) >> 2, kx = m & 3;
    #pragma unroll
    for (int s = 0; s < 3; ++s) {
      bf16x8 af = {};
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int px = s * 32 + kbase + j;
        if (px < H2 * W2) {
          const int oy = px / W2, ox = px % W2;
          af[j] = s_a[(c * H1 + S2 * oy + ky) * W1 + S2 * ox + kx];
        }
      }
      const int pxb = s * 32 + kbase;
      const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          s_d + (lane & 15) * 96 + pxb);
      const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(
          s_d + (16 + (lane & 15)) * 96 + pxb);
      acc[mi][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0,
                                                           acc[mi][0], 0,
                                                           0, 0);
      acc[mi][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b1,
                                                           acc[mi][1], 0,
                                                           0, 0);
    }
  }
  }  // image loop

  {
    const int dcol = lane & 15;
    const int drow = (lane >> 4) * 4;
    #pragma unroll
    for (int mi = 0; mi < MPW; ++mi) {
      const int mf = wave + mi * 4;
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int f = n * 16 + dcol;
        #pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int mm = mf * 16 + drow + reg;
          atomicAdd(&dw[(size_t)mm * F2 + f], acc[mi][n][reg]);
        }
      }
    }
  }
}

// --------------------------------------------------------------------------
// dW1: M = 256 (c*64+tap), N = 16, K = 400 px (13 steps, padded 416).
//   A from the planar u8->bf16 frame patch (whole 4x84x84, 56.4 KiB);
//   B from dz1 staged transposed [16][416].
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void conv1_dw(
    const unsigned char* __restrict__ frames,
    const bf16_t* __restrict__ dz1, float* __restrict__ dw_slab, int N,
    int nslab) {
  __shared__ bf16_t s_a[C0 * H0 * W0];      // 55.1 KiB planar
  __shared__ bf16_t s_d[F1 * 416];          // 13 KiB transposed
  // 68.1 KiB total -> 2 blocks/CU in 160 KiB LDS

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  float* dw = dw_slab + (size_t)(blockIdx.x % nslab) * (256 * F1);
  constexpr int MPW = 4;                    // 16 m-frags over 4 waves
  floatx4 acc[MPW];
  #pragma unroll
  for (int m = 0; m < MPW; ++m) acc[m] = floatx4{0.f, 0.f, 0.f, 0.f};
  const int kbase = (lane >> 4) * 8;

  for (int img = blockIdx.x; img < N; img += gridDim.x) {
  __syncthreads();
  {
    // frames: contiguous u8 -> bf16, uchar4-vectorised (28224 % 4 == 0)
    const unsigned char* fr = frames + (size_t)img * C0 * H0 * W0;
    for (int i = tid * 4; i < C0 * H0 * W0; i += 256 * 4) {
      const uchar4 v = *reinterpret_cast<const uchar4*>(fr + i);
      bf16x4 o;
      o[0] = f2bf(v.x * (1.f / 255.f));
      o[1] = f2bf(v.y * (1.f / 255.f));
      o[2] = f2bf(v.z * (1.f / 255.f));
      o[3] = f2bf(v.w * (1.f / 255.f));
      *reinterpret_cast<bf16x4*>(s_a + i) = o;
    }
    // dz transpose: 16 threads per f-row, no per-element div/mod
    const bf16_t* sd = dz1 + (size_t)img * H1 * W1 * F1;
    const int f = tid >> 4;
    if (f < F1) {
      for (int px = tid & 15; px < 416; px += 16)
        s_d[f * 416 + px] =
            px < H1 * W1 ? sd[(size_t)px * F1 + f] : f2bf(0.f);
    }
  }
  __syncthreads();

  #pragma unroll
  for (int mi = 0; mi < MPW; ++mi) {
    const int mf = wave + mi * 4;
    const int m = mf * 16 + (lane & 15);    // c*64 + ky*8 + kx
    const int c = m >> 6;
    const int ky = (m & 63) >> 3, kx = m & 7;
    #pragma unroll 4
    for (int s = 0; s < 13; ++s) {
      bf16x8 af = {};
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int px = s * 32 + kbase + j;
        if (px < H1 * W1) {
          const int oy = px / W1, ox = px % W1;
          af[j] = s_a[(c * H0 + S1 * oy + ky) * W0 + S1 * ox + kx];
        }
      }
      const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(
          s_d + (lane & 15) * 416 + s * 32 + kbase);
      acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, b0, acc[mi],
                                                        0, 0, 0);
    }
  }
  }  // image loop

  {
    const int dcol = lane & 15;
    const int drow = (lane >> 4) * 4;
    #pragma unroll
    for (int mi = 0; mi < MPW; ++mi) {
      const int mf = wave + mi * 4;
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int mm = mf * 16 + drow + reg;
        atomicAdd(&dw[(size_t)mm * F1 + dcol], acc[mi][reg]);
      }
    }
  }
}

// --------------------------------------------------------------------------
// launchers — slabs are SLAB zeroed copies, k-major [m][f]; the python
// side sums over slabs and reshapes to the torch weight layout.
// --------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// bwd_finalize: fused epilogue producing the torch-layout weight grads
// and the bias column sums in ONE launch.  Replaces ~12 eager ATen ops
// (3x slab .sum(0), 3x permute+contiguous reshapes, 3x channel-sum
// reductions, assorted casts) measured at ~100 us/step of GPU time in
// the round-2 PMC capture (profiles/r6_*).
//   dw1[f][c][ky][kx] = sum_sl s1[sl][m=c*64+ky*8+kx][f]     (16,4,8,8)
//   dw2[f][c][ky][kx] = sum_sl s2[sl][m=c*16+ky*4+kx][f]     (32,16,4,4)
//   dw3[f][c][t/3][t%3] = sum_sl s3[sl][m=t*32+c][f]         (32,32,3,3)
//   db1[f] = sum da1[n][y][x][f];  db2 likewise;  db3[f] = sum dz3[n][f][px]
// db* must be zero-initialised by the caller (atomic partial sums).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void bwd_finalize(
    const float* __restrict__ s1, const float* __restrict__ s2,
    const float* __restrict__ s3, const bf16_t* __restrict__ da1,
    const bf16_t* __restrict__ da2, const bf16_t* __restrict__ dz3, int N,
    int nslab,
    float* __restrict__ dw1, float* __restrict__ dw2,
    float* __restrict__ dw3, float* __restrict__ db1,
    float* __restrict__ db2, float* __restrict__ db3) {
  const int SLC = nslab / FIN_SPLIT;            // slabs per block
  constexpr int SPLIT = FIN_SPLIT;
  constexpr int BW1 = 16 * SPLIT, BW2 = 32 * SPLIT, BW3 = 36 * SPLIT;
  constexpr int BB = FIN_BB;
  const int blk = blockIdx.x;
  const int tid = threadIdx.x;

  if (blk < BW1) {                               // dw1: 4096 outputs
    const int o = (blk / SPLIT) * 256 + tid;
    const int sl0 = (blk % SPLIT) * SLC;
    const int kx = o & 7, ky = (o >> 3) & 7, c = (o >> 6) & 3;
    const int f = o >> 8;
    const int m = c * 64 + ky * 8 + kx;
    float v = 0.f;
    for (int sl = sl0; sl < sl0 + SLC; ++sl)
      v += s1[(size_t)sl * 4096 + m * 16 + f];
    atomicAdd(&dw1[o], v);                       // dw* zero-initialised
    return;
  }
  if (blk < BW1 + BW2) {                         // dw2: 8192 outputs
    const int o = ((blk - BW1) / SPLIT) * 256 + tid;
    const int sl0 = ((blk - BW1) % SPLIT) * SLC;
    const int kx = o & 3, ky = (o >> 2) & 3, c = (o >> 4) & 15;
    const int f = o >> 8;
    const int m = c * 16 + ky * 4 + kx;
    float v = 0.f;
    for (int sl = sl0; sl < sl0 + SLC; ++sl)
      v += s2[(size_t)sl * 8192 + m * 32 + f];
    atomicAdd(&dw2[o], v);
    return;
  }
  if (blk < BW1 + BW2 + BW3) {                   // dw3: 9216 outputs
    const int o = ((blk - BW1 - BW2) / SPLIT) * 256 + tid;
    const int sl0 = ((blk - BW1 - BW2) % SPLIT) * SLC;
    if (o < 9216) {
      const int t = o % 9;                       // tap
      const int c = (o / 9) & 31;
      const int f = o / 288;
      const int m = t * 32 + c;
      float v = 0.f;
      for (int sl = sl0; sl < sl0 + SLC; ++sl)
        v += s3[(size_t)sl * 9216 + m * 32 + f];
      atomicAdd(&dw3[o], v);
    }
    return;
  }
  // db sections: coalesced column partial sums + one atomic per block/col
  __shared__ float red[256];
  const int sec = (blk - BW1 - BW2 - BW3) / BB;  // 0: db1, 1: db2, 2: db3
  const int sb = (blk - BW1 - BW2 - BW3) % BB;
  const bf16_t* src = sec == 0 ? da1 : (sec == 1 ? da2 : dz3);
  float* dst = sec == 0 ? db1 : (sec == 1 ? db2 : db3);
  const int C = sec == 0 ? 16 : 32;
  const size_t total = sec == 0 ? (size_t)N * 400 * 16
                                : (sec == 1 ? (size_t)N * 81 * 32
                                            : (size_t)N * 1568);
  float v = 0.f;
  if (sec < 2) {                                 // NHWC: col = e % C
    const int col = tid % C;
    const int rpt = 256 / C;                     // rows handled per block pass
    for (size_t r = (size_t)sb * rpt + tid / C; r * C + col < total;
         r += (size_t)BB * rpt)
      v += bf2f(src[r * C + col]);
    red[tid] = v;
    __syncthreads();
    // reduce rows with equal col: entries tid, tid+C, ...
    if (tid < (unsigned)C) {
      float acc = 0.f;
      for (int i = tid; i < 256; i += C) acc += red[i];
      atomicAdd(&dst[tid], acc);
    }
  } else {                                       // dz3 [n][32][49]
    (void)red;
    // each thread accumulates per-channel runs, flushing on change
    __shared__ float chan[32];
    if (tid < 32) chan[tid] = 0.f;
    __syncthreads();
    float local = 0.f;
    int last_c = -1;
    for (size_t e = (size_t)sb * 256 + tid; e < total;
         e += (size_t)BB * 256) {
      const int c = (int)((e / 49) & 31);
      if (c != last_c) {
        if (last_c >= 0) atomicAdd(&chan[last_c], local);
        local = 0.f;
        last_c = c;
      }
      local += bf2f(src[e]);
    }
    if (last_c >= 0) atomicAdd(&chan[last_c], local);
    __syncthreads();
    if (tid < 32) atomicAdd(&dst[tid], chan[tid]);
  }
}

extern "C" int shallow_mfma_slab() { return shallow_slab_count(); }

extern "C" void launch_shallow_mask_convert(const float* d, const void* act,
                                            void* o, long n,
                                            hipStream_t s) {
  long blocks = (n + 255) / 256;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(mask_convert, dim3((int)blocks), dim3(256), 0, s, d,
                     static_cast<const bf16_t*>(act),
                     static_cast<bf16_t*>(o), n);
}

// Full backward chain from dz3 (= d_a3 masked by a3>0): the produced
// da2/da1 ARE the masked dz2/dz1 (the dX epilogues fuse the ReLU
// masks), so the dW kernels consume them directly. Same-stream launch
// order gives the dependencies. Python sums the biases from
// dz3/da2/da1 and the slab copies for the weights.
extern "C" void launch_shallow_mfma_bwd(
    const unsigned char* frames, const void* a1, const void* a2,
    const void* dz3, const void* wr3, const void* wr2, void* da1,
    void* da2, float* dw1_slab, float* dw2_slab, float* dw3_slab, int N,
    hipStream_t s) {
  hipLaunchKernelGGL(conv3_dx, dim3(N), dim3(256), 0, s,
                     static_cast<const bf16_t*>(dz3),
                     static_cast<const bf16_t*>(wr3),
                     static_cast<const bf16_t*>(a2),
                     static_cast<bf16_t*>(da2), N);
  const int nslab = shallow_slab_count();
  hipLaunchKernelGGL(conv3_dw, dim3(N < DWG ? N : DWG), dim3(256), 0, s,
                     static_cast<const bf16_t*>(a2),
                     static_cast<const bf16_t*>(dz3), dw3_slab, N, nslab);
  hipLaunchKernelGGL(conv2_dx, dim3(N), dim3(256), 0, s,
                     static_cast<const bf16_t*>(da2),
                     static_cast<const bf16_t*>(wr2),
                     static_cast<const bf16_t*>(a1),
                     static_cast<bf16_t*>(da1), N);
  hipLaunchKernelGGL(conv2_dw, dim3(N < DWG ? N : DWG), dim3(256), 0, s,
                     static_cast<const bf16_t*>(a1),
                     static_cast<const bf16_t*>(da2), dw2_slab, N, nslab);
  // conv1_dw keeps one-image-per-WG (grid N): its 68 KiB staging phase
  // overlaps across queued blocks; the capped grid-stride variant lost
  // that overlap (measured 48 -> 91 us, r8 CSV).
  hipLaunchKernelGGL(conv1_dw, dim3(N), dim3(256), 0, s, frames,
                     static_cast<const bf16_t*>(da1), dw1_slab, N, nslab);
}

}  // namespace impala_shallow_mfma_bwd

extern "C" void launch_shallow_bwd_finalize(
    const float* s1, const float* s2, const float* s3, const void* da1,
    const void* da2, const void* dz3, int N, float* dw1, float* dw2,
    float* dw3, float* db1, float* db2, float* db3, hipStream_t s) {
  using impala_shallow_mfma_bwd::bf16_t;
  // FIN_GRID is the kernel's section geometry — single source of truth
  // (r9/r10 ran a stale 276-block grid: dw3 58% written, db sections
  // never launched — caught by test_shallow_encoder_parity_mfma).
  hipLaunchKernelGGL(impala_shallow_mfma_bwd::bwd_finalize,
                     dim3(impala_shallow_mfma_bwd::FIN_GRID), dim3(256), 0,
                     s, s1, s2,
                     s3, static_cast<const bf16_t*>(da1),
                     static_cast<const bf16_t*>(da2),
                     static_cast<const bf16_t*>(dz3), N,
                     impala_shallow_mfma_bwd::shallow_slab_count(), dw1,
                     dw2, dw3, db1, db2, db3);
}
