// gfx950 MFMA linear kernels (BASELINE.json:5 "MFMA for the conv/linear
// GEMMs"; SURVEY.md §2c linear_fwd/bwd).
//
// f32-input MFMA (v_mfma_f32_16x16x4_f32 — exact f32 at the 157 TF f32
// vector rate, cdna_hip_programming.md §3): the learner is fp32, so the
// f32 "SGEMM"-class instructions are the right matrix-core path (no
// xf32 on gfx950).
//
// One kernel shape serves both directions of the FC layer:
//     C[i][j] = act( sum_t A[i][t] * B[j][t] + bias[j] )
//   fwd:  A = X (M,K),  B = W (N,K),   act = ReLU, bias = b
//   bwdX: A = dY masked by (Y>0) ON THE FLY during staging (fuses the
//         ReLU backward), B = W^T (K,N) pre-transposed once, no bias.
// dW/db are plain reductions over the big M dim — they stay on rocBLAS
// (which consumes the transposed layouts without materialising them).
//
// Tiling: 64x64 block tile, 4 waves of 32x32 (2x2 16x16 fragments),
// K staged in LDS 32 deep, double-buffered, +4-float row pad against
// bank conflicts. Tail M/N guarded.

#include <hip/hip_runtime.h>

namespace impala {

using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 64, BN = 64, BK = 32;
constexpr int LDS_STRIDE = BK + 4;  // pad: avoids b32 column conflicts
constexpr int THREADS = 256;

// mode: 0 = plain, 1 = relu epilogue, 2 = A-staging masked by Ymask>0
// Split-K (round 2): the headline GEMMs are tiny (fc fwd 640x256x1568 =
// a 10x4 grid, 84% of the 256 CUs idle).  With gridDim.z = S > 1 each
// z-block covers K-chunk [z*kchunk, +kchunk) and stores RAW partials to
// P = C + z*M*N (C then sized S*M*N); linear_splitk_reduce applies the
// deterministic S-way sum + bias + activation.  gridDim.z == 1 keeps
// the original fused epilogue.
__global__ void __launch_bounds__(THREADS, 2)
linear_mfma(const float* __restrict__ A, const float* __restrict__ B,
            const float* __restrict__ bias, const float* __restrict__ Ymask,
            int M, int N, int K, int mode, int kchunk,
            float* __restrict__ C) {
  __shared__ __align__(16) float s_a[2][BM * LDS_STRIDE];
  __shared__ __align__(16) float s_b[2][BN * LDS_STRIDE];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int kb0 = blockIdx.z * kchunk;
  const int kend = min(K, kb0 + kchunk);   // this z-block's K range
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // 4 waves
  const int wm = (wave >> 1) * 32;     // wave row offset in tile
  const int wn = (wave & 1) * 32;      // wave col offset in tile

  f32x4 acc[2][2] = {};

  // staging map: 256 threads x 2 rows each; thread handles row r = tid/4,
  // 8 floats at k-offset (tid%4)*8 -> two float4 loads per row-half
  const int srow = tid >> 2;           // 0..63
  const int skoff = (tid & 3) * 8;     // 0,8,16,24

  auto stage = [&](int buf, int k0) {
    // A rows
    {
      const int gr = bm + srow;
      float4 v0 = {0, 0, 0, 0}, v1 = {0, 0, 0, 0};
      if (gr < M) {
        const float* src = A + (size_t)gr * K + k0 + skoff;
        const int krem = kend - (k0 + skoff);
        if (krem >= 8) {
          v0 = *reinterpret_cast<const float4*>(src);
          v1 = *reinterpret_cast<const float4*>(src + 4);
        } else {
          for (int i = 0; i < 8 && i < krem; ++i)
            (i < 4 ? (&v0.x)[i] : (&v1.x)[i - 4]) = src[i];
        }
        if (mode == 2) {  // fused ReLU-backward mask from Y
          const float* ym = Ymask + (size_t)gr * K + k0 + skoff;
          const int mrem = kend - (k0 + skoff);
          for (int i = 0; i < 4; ++i) {
            if (i < mrem && ym[i] <= 0.f) (&v0.x)[i] = 0.f;
            if (i + 4 < mrem && ym[i + 4] <= 0.f) (&v1.x)[i] = 0.f;
          }
        }
      }
      float* dst = s_a[buf] + srow * LDS_STRIDE + skoff;
      *reinterpret_cast<float4*>(dst) = v0;
      *reinterpret_cast<float4*>(dst + 4) = v1;
    }
    // B rows
    {
      const int gr = bn + srow;
      float4 v0 = {0, 0, 0, 0}, v1 = {0, 0, 0, 0};
      if (gr < N) {
        const float* src = B + (size_t)gr * K + k0 + skoff;
        const int krem = kend - (k0 + skoff);
        if (krem >= 8) {
          v0 = *reinterpret_cast<const float4*>(src);
          v1 = *reinterpret_cast<const float4*>(src + 4);
        } else {
          for (int i = 0; i < 8 && i < krem; ++i)
            (i < 4 ? (&v0.x)[i] : (&v1.x)[i - 4]) = src[i];
        }
      }
      float* dst = s_b[buf] + srow * LDS_STRIDE + skoff;
      *reinterpret_cast<float4*>(dst) = v0;
      *reinterpret_cast<float4*>(dst + 4) = v1;
    }
  };

  stage(0, kb0);
  __syncthreads();

  const int nk = (kend - kb0 + BK - 1) / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) stage(cur ^ 1, kb0 + (kt + 1) * BK);
    // MFMA over this K tile: 16x16x4 fragments
    const int ar = lane & 15;          // fragment row
    const int ak = lane >> 4;          // k sub-index (0..3)
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      float a0 = s_a[cur][(wm + ar) * LDS_STRIDE + kk + ak];
      float a1 = s_a[cur][(wm + 16 + ar) * LDS_STRIDE + kk + ak];
      float b0 = s_b[cur][(wn + ar) * LDS_STRIDE + kk + ak];
      float b1 = s_b[cur][(wn + 16 + ar) * LDS_STRIDE + kk + ak];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: C[row][col] with 16x16 C-layout col=lane&15,
  // row=(lane>>4)*4+reg (dtype-independent on gfx950)
  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
  float* __restrict__ out = C + (size_t)blockIdx.z * M * N;
  const bool split = gridDim.z > 1;
  #pragma unroll
  for (int r = 0; r < 2; ++r) {
    #pragma unroll
    for (int c = 0; c < 2; ++c) {
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = bm + wm + r * 16 + crow + reg;
        const int col = bn + wn + c * 16 + ccol;
        if (row < M && col < N) {
          float v = acc[r][c][reg];
          if (!split) {
            if (bias != nullptr) v += bias[col];
            if (mode == 1) v = v > 0.f ? v : 0.f;
          }
          out[(size_t)row * N + col] = v;
        }
      }
    }
  }
}

// deterministic S-way partial sum + bias + activation epilogue
__global__ void __launch_bounds__(256)
linear_splitk_reduce(const float* __restrict__ P,
                     const float* __restrict__ bias, int S, int N,
                     size_t MN, int mode, float* __restrict__ C) {
  for (size_t e = (size_t)blockIdx.x * blockDim.x + threadIdx.x; e < MN;
       e += (size_t)gridDim.x * blockDim.x) {
    float v = 0.f;
    for (int s = 0; s < S; ++s) v += P[(size_t)s * MN + e];
    if (bias != nullptr) v += bias[(int)(e % N)];
    if (mode == 1) v = v > 0.f ? v : 0.f;
    C[e] = v;
  }
}


// ---------------------------------------------------------------------------
// bf16-input GEMM on v_mfma_f32_16x16x32_bf16 (fp32 accumulate): the
// same 64x64 block / 32x32-per-wave / split-K structure as linear_mfma,
// with bf16 LDS staging (BK=64, 2 MFMA k-steps per stage).  Written for
// the fc behind the bf16 MFMA encoder after the hipBLASLt bf16 GEMM
// path was root-caused corrupting gradients under hipGraph replay
// (profiles/r2_nan_rootcause.md) — in-tree bf16 GEMMs are clean.
//   C[i][j] = act( sum_t A[i][t]*B[j][t] + bias[j] )
// mode: 0 plain, 1 bias+relu.  out_bf16: emit C as bf16 (else fp32).
// ---------------------------------------------------------------------------
typedef __bf16 lin_bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 lin_bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 lin_bf16x4;

constexpr int BBK = 64;                 // bf16 K tile
constexpr int BSTR = BBK + 8;           // halfs; 144 B rows (16B aligned)

__global__ void __launch_bounds__(THREADS, 2)
linear_mfma_bf16(const lin_bf16* __restrict__ A,
                 const lin_bf16* __restrict__ B,
                 const float* __restrict__ bias, int M, int N, int K,
                 int mode, int kchunk, int out_bf16,
                 void* __restrict__ C) {
  __shared__ __align__(16) lin_bf16 s_a[2][BM * BSTR];
  __shared__ __align__(16) lin_bf16 s_b[2][BN * BSTR];

  const int bm = blockIdx.x * BM;
  const int bn = blockIdx.y * BN;
  const int kb0 = blockIdx.z * kchunk;
  const int kend = min(K, kb0 + kchunk);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * 32;
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2] = {};

  // staging: 256 threads x 2 halves; thread covers row r = tid/4, 16
  // halfs at k-offset (tid%4)*16 -> one 16B load per half-row chunk
  const int srow = tid >> 2;            // 0..63
  const int skoff = (tid & 3) * 16;     // 0,16,32,48

  auto stage = [&](int buf, int k0) {
    {
      const int gr = bm + srow;
      lin_bf16x8 v0 = {}, v1 = {};
      if (gr < M) {
        const lin_bf16* src = A + (size_t)gr * K + k0 + skoff;
        const int krem = kend - (k0 + skoff);
        if (krem >= 16) {
          v0 = *reinterpret_cast<const lin_bf16x8*>(src);
          v1 = *reinterpret_cast<const lin_bf16x8*>(src + 8);
        } else {
          for (int i = 0; i < 16 && i < krem; ++i) {
            if (i < 8) v0[i] = src[i];
            else v1[i - 8] = src[i];
          }
        }
      }
      lin_bf16* dst = s_a[buf] + srow * BSTR + skoff;
      *reinterpret_cast<lin_bf16x8*>(dst) = v0;
      *reinterpret_cast<lin_bf16x8*>(dst + 8) = v1;
    }
    {
      const int gr = bn + srow;
      lin_bf16x8 v0 = {}, v1 = {};
      if (gr < N) {
        const lin_bf16* src = B + (size_t)gr * K + k0 + skoff;
        const int krem = kend - (k0 + skoff);
        if (krem >= 16) {
          v0 = *reinterpret_cast<const lin_bf16x8*>(src);
          v1 = *reinterpret_cast<const lin_bf16x8*>(src + 8);
        } else {
          for (int i = 0; i < 16 && i < krem; ++i) {
            if (i < 8) v0[i] = src[i];
            else v1[i - 8] = src[i];
          }
        }
      }
      lin_bf16* dst = s_b[buf] + srow * BSTR + skoff;
      *reinterpret_cast<lin_bf16x8*>(dst) = v0;
      *reinterpret_cast<lin_bf16x8*>(dst + 8) = v1;
    }
  };

  stage(0, kb0);
  __syncthreads();

  const int nk = (kend - kb0 + BBK - 1) / BBK;
  const int ar = lane & 15;
  const int ak = (lane >> 4) * 8;       // this lane's k sub-offset
  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) stage(cur ^ 1, kb0 + (kt + 1) * BBK);
    #pragma unroll
    for (int kk = 0; kk < BBK; kk += 32) {
      const lin_bf16x8 a0 = *reinterpret_cast<const lin_bf16x8*>(
          s_a[cur] + (wm + ar) * BSTR + kk + ak);
      const lin_bf16x8 a1 = *reinterpret_cast<const lin_bf16x8*>(
          s_a[cur] + (wm + 16 + ar) * BSTR + kk + ak);
      const lin_bf16x8 b0 = *reinterpret_cast<const lin_bf16x8*>(
          s_b[cur] + (wn + ar) * BSTR + kk + ak);
      const lin_bf16x8 b1 = *reinterpret_cast<const lin_bf16x8*>(
          s_b[cur] + (wn + 16 + ar) * BSTR + kk + ak);
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[0][0],
                                                          0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b1, acc[0][1],
                                                          0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b0, acc[1][0],
                                                          0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[1][1],
                                                          0, 0, 0);
    }
    __syncthreads();
  }

  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
  const bool split = gridDim.z > 1;
  float* outf = reinterpret_cast<float*>(C) + (size_t)blockIdx.z * M * N;
  lin_bf16* outh = reinterpret_cast<lin_bf16*>(C);
  #pragma unroll
  for (int r = 0; r < 2; ++r) {
    #pragma unroll
    for (int c = 0; c < 2; ++c) {
      #pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = bm + wm + r * 16 + crow + reg;
        const int col = bn + wn + c * 16 + ccol;
        if (row < M && col < N) {
          float v = acc[r][c][reg];
          if (!split) {
            if (bias != nullptr) v += bias[col];
            if (mode == 1) v = v > 0.f ? v : 0.f;
            if (out_bf16) {
              outh[(size_t)row * N + col] = (lin_bf16)v;
              continue;
            }
          }
          outf[(size_t)row * N + col] = v;
        }
      }
    }
  }
}

// S-way partial sum + bias + activation, bf16 or fp32 output
__global__ void __launch_bounds__(256)
linear_splitk_reduce_bf16out(const float* __restrict__ P,
                             const float* __restrict__ bias, int S, int N,
                             size_t MN, int mode, int out_bf16,
                             void* __restrict__ C) {
  float* cf = reinterpret_cast<float*>(C);
  lin_bf16* ch = reinterpret_cast<lin_bf16*>(C);
  for (size_t e = (size_t)blockIdx.x * blockDim.x + threadIdx.x; e < MN;
       e += (size_t)gridDim.x * blockDim.x) {
    float v = 0.f;
    for (int s = 0; s < S; ++s) v += P[(size_t)s * MN + e];
    if (bias != nullptr) v += bias[(int)(e % N)];
    if (mode == 1) v = v > 0.f ? v : 0.f;
    if (out_bf16) ch[e] = (lin_bf16)v;
    else cf[e] = v;
  }
}

}  // namespace impala

extern "C" {

// S chosen so the GEMM grid reaches ~2 blocks/CU on the 256-CU chip
// while each K-chunk stays >= 2 LDS stages deep.
int linear_splitk_factor(int M, int N, int K) {
  const int gm = (M + impala::BM - 1) / impala::BM;
  const int gn = (N + impala::BN - 1) / impala::BN;
  int s = (512 + gm * gn - 1) / (gm * gn);
  const int smax = K / (2 * impala::BK) > 0 ? K / (2 * impala::BK) : 1;
  if (s > smax) s = smax;
  if (s > 16) s = 16;
  if (s < 1) s = 1;
  return s;
}

// ws: workspace of S*M*N floats when S = linear_splitk_factor(...) > 1
// (may be null for S == 1).
void launch_linear_mfma(const float* A, const float* B, const float* bias,
                        const float* ymask, int M, int N, int K, int mode,
                        float* C, float* ws, int S, hipStream_t stream) {
  dim3 grid((M + impala::BM - 1) / impala::BM,
            (N + impala::BN - 1) / impala::BN, S);
  const int kchunk = S > 1 ? (K + S - 1) / S : K;
  hipLaunchKernelGGL(impala::linear_mfma, grid, dim3(impala::THREADS), 0,
                     stream, A, B, bias, ymask, M, N, K, mode, kchunk,
                     S > 1 ? ws : C);
  if (S > 1) {
    const size_t MN = (size_t)M * N;
    int blocks = (int)((MN + 255) / 256);
    if (blocks > 2048) blocks = 2048;
    hipLaunchKernelGGL(impala::linear_splitk_reduce, dim3(blocks), dim3(256),
                       0, stream, ws, bias, S, N, MN, mode, C);
  }
}


// bf16-input variant; ws sized S*M*N floats when S>1.
int linear_splitk_factor_bf16(int M, int N, int K) {
  const int gm = (M + impala::BM - 1) / impala::BM;
  const int gn = (N + impala::BN - 1) / impala::BN;
  int s = (512 + gm * gn - 1) / (gm * gn);
  const int smax = K / (2 * impala::BBK) > 0 ? K / (2 * impala::BBK) : 1;
  if (s > smax) s = smax;
  if (s > 16) s = 16;
  if (s < 1) s = 1;
  return s;
}

void launch_linear_mfma_bf16(const void* A, const void* B, const float* bias,
                             int M, int N, int K, int mode, int out_bf16,
                             void* C, float* ws, int S, hipStream_t stream) {
  dim3 grid((M + impala::BM - 1) / impala::BM,
            (N + impala::BN - 1) / impala::BN, S);
  const int kchunk = S > 1 ? (K + S - 1) / S : K;
  hipLaunchKernelGGL(impala::linear_mfma_bf16, grid, dim3(impala::THREADS),
                     0, stream, static_cast<const impala::lin_bf16*>(A),
                     static_cast<const impala::lin_bf16*>(B), bias, M, N, K,
                     mode, kchunk, out_bf16, S > 1 ? (void*)ws : C);
  if (S > 1) {
    const size_t MN = (size_t)M * N;
    int blocks = (int)((MN + 255) / 256);
    if (blocks > 2048) blocks = 2048;
    hipLaunchKernelGGL(impala::linear_splitk_reduce_bf16out, dim3(blocks),
                       dim3(256), 0, stream, ws, bias, S, N, MN, mode,
                       out_bf16, C);
  }
}
}  // extern "C"
