// RQ-VAE residual-quantize distance kernel (K8 — SURVEY.md §2.4), gfx950.
//
// dist[b][c] = ||x_b||^2 + ||e_c||^2 - 2 <x_b, e_c>, plus row argmin, in one
// pass. The codebook ([K,d], K=256, d=32 in the shipped configs = 32 KB)
// is staged whole in LDS; each wave owns one x-row: lane c accumulates the
// dot for codes c, c+64, ... with fp32 accumulation.
// Reference computes this as three separate GEMM/reduction ops
// (rqvae.py:186-192) plus a separate argmin.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

template <typename T>
__global__ void sqdist_argmin_kernel(const T* __restrict__ x,       // [B,d]
                                     const T* __restrict__ codebook,  // [K,d]
                                     float* __restrict__ dist,       // [B,K]
                                     int64_t* __restrict__ ids,      // [B]
                                     int B, int K, int d) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* cb = reinterpret_cast<float*>(smem_raw);        // [K][d] transposed: [d][K]
  float* cnorm = cb + d * K;                             // [K]

  const int tid = threadIdx.x;
  for (int idx = tid; idx < K * d; idx += blockDim.x) {
    int c = idx / d, t = idx % d;
    cb[t * K + c] = to_f32(codebook[(int64_t)c * d + t]);
  }
  __syncthreads();
  // code norms once per block
  for (int c = tid; c < K; c += blockDim.x) {
    float s = 0.f;
    for (int t = 0; t < d; ++t) {
      float v = cb[t * K + c];
      s += v * v;
    }
    cnorm[c] = s;
  }
  __syncthreads();

  const int lane = tid & (WAVE - 1);
  const int wid = (blockIdx.x * blockDim.x + tid) / WAVE;
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int row = wid; row < B; row += n_waves) {
    const T* xr = x + (int64_t)row * d;
    // x row into registers (d <= 64 assumed; larger d reread from global)
    float xn = 0.f;
    float best = INFINITY;
    int best_c = 0;
    // ||x||^2 via wave reduction
    for (int t = lane; t < d; t += WAVE) {
      float v = to_f32(xr[t]);
      xn += v * v;
    }
    xn = wave_sum(xn);
    for (int c = lane; c < K; c += WAVE) {
      float dot = 0.f;
      for (int t = 0; t < d; ++t) dot += to_f32(xr[t]) * cb[t * K + c];
      float dd = xn + cnorm[c] - 2.f * dot;
      dist[(int64_t)row * K + c] = dd;
      if (dd < best) { best = dd; best_c = c; }
    }
    // wave argmin
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ob = __shfl_xor(best, off, WAVE);
      int oc = __shfl_xor(best_c, off, WAVE);
      if (ob < best || (ob == best && oc < best_c)) { best = ob; best_c = oc; }
    }
    if (lane == 0) ids[row] = best_c;
  }
}

std::vector<torch::Tensor> sqdist_argmin(torch::Tensor x,
                                         torch::Tensor codebook) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && codebook.dim() == 2);
  const int B = x.size(0), d = x.size(1), K = codebook.size(0);
  TORCH_CHECK((int64_t)K * d * 4 + K * 4 <= 96 * 1024,
              "codebook too large for LDS staging");
  auto dist = torch::empty({B, K}, x.options().dtype(torch::kFloat32));
  auto ids = torch::empty({B}, x.options().dtype(torch::kInt64));
  dim3 block(256);
  dim3 grid(std::min((B + 3) / 4, 2048));
  size_t smem = ((size_t)K * d + K) * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((sqdist_argmin_kernel<float>), grid, block, smem,
                       stream, x.data_ptr<float>(),
                       codebook.contiguous().data_ptr<float>(),
                       dist.data_ptr<float>(), ids.data_ptr<int64_t>(), B, K, d);
  } else if (x.scalar_type() == torch::kBFloat16) {
    auto cbc = codebook.contiguous();
    hipLaunchKernelGGL((sqdist_argmin_kernel<__hip_bfloat16>), grid, block,
                       smem, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(cbc.data_ptr()),
                       dist.data_ptr<float>(), ids.data_ptr<int64_t>(), B, K, d);
  } else {
    TORCH_CHECK(false, "sqdist_argmin: unsupported dtype");
  }
  return {dist, ids};
}

}  // namespace genrec
