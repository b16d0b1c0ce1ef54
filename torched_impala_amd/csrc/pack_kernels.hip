// Rollout unpack: slot-major device staging -> time-major training
// batch in ONE layout-transform kernel per tensor (SURVEY.md §2c
// rollout_pack/unpack).  The eager path used ATen permute+reshape
// copies inside the captured graph (~80 us/step of elementwise
// kernels, r25 profile); this kernel does the same block permutation
// with uint4 copies at HBM rate.
//
// Mapping (pure block permutation — the inner extent stays contiguous):
//   src (K, T1, inner_bytes)  ->  dst (T1, K, inner_bytes)
//   dst[(t*K + k)*inner + i] = src[(k*T1 + t)*inner + i]
// One workgroup per (k, t) block, grid-stride over inner in 16 B
// vectors (byte tail handled scalar — tiny keys only).

#include <hip/hip_runtime.h>

namespace impala_pack {

typedef unsigned int u32;
typedef __attribute__((ext_vector_type(4))) u32 u32x4;

__global__ __launch_bounds__(256) void unpack_time_major(
    const unsigned char* __restrict__ src, unsigned char* __restrict__ dst,
    int K, int T1, long inner) {
  const int blk = blockIdx.x;
  const int k = blk % K;
  const int t = blk / K;
  if (t >= T1) return;
  const unsigned char* s = src + ((size_t)k * T1 + t) * inner;
  unsigned char* d = dst + ((size_t)t * K + k) * inner;
  const long nvec = inner >> 4;
  const u32x4* sv = reinterpret_cast<const u32x4*>(s);
  u32x4* dv = reinterpret_cast<u32x4*>(d);
  for (long i = threadIdx.x; i < nvec; i += blockDim.x) dv[i] = sv[i];
  const long tail = inner & 15;
  if (tail) {
    const long base = inner - tail;
    for (long i = threadIdx.x; i < tail; i += blockDim.x)
      d[base + i] = s[base + i];
  }
}

}  // namespace impala_pack

extern "C" void launch_unpack_time_major(const void* src, void* dst, int K,
                                         int T1, long inner_bytes,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(impala_pack::unpack_time_major, dim3(K * T1), dim3(256),
                     0, stream,
                     static_cast<const unsigned char*>(src),
                     static_cast<unsigned char*>(dst), K, T1, inner_bytes);
}
