// Shared helpers for genrec_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    if (e_ != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(e_), " at ",        \
                  __FILE__, ":", __LINE__);                                   \
    }                                                                         \
  } while (0)

namespace genrec {

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) {
  return __bfloat162float(x);
}

template <typename T>
__device__ __forceinline__ T from_f32(float x);
template <>
__device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}

// wave-wide reductions (64 lanes)
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

__device__ __forceinline__ float sigmoidf_dev(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// cheap counter-based RNG for dropout masks (Philox-lite hash)
__device__ __forceinline__ unsigned int hash_rng(unsigned int seed,
                                                 unsigned long long idx) {
  unsigned long long z = idx + 0x9E3779B97F4A7C15ULL * (seed + 1);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return (unsigned int)(z ^ (z >> 31));
}

}  // namespace genrec
