// On-device top-k metric kernels (K22 — SURVEY.md §2.4), gfx950.
// topk_hit_ranks: rank of first exact tuple match (metrics.py:40-66 math,
// computed without the [B,K,D] broadcast + host sync of the reference).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

__global__ void topk_hit_ranks_kernel(const int64_t* __restrict__ actual,  // [B,D]
                                      const int64_t* __restrict__ topk,    // [B,K,D]
                                      int64_t* __restrict__ ranks,         // [B]
                                      int B, int K, int D) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int64_t* a = actual + (int64_t)b * D;
  const int64_t* t = topk + (int64_t)b * K * D;
  int64_t r = K;
  for (int k = 0; k < K; ++k) {
    bool all = true;
    for (int dd = 0; dd < D; ++dd) all &= (t[k * D + dd] == a[dd]);
    if (all) { r = k; break; }
  }
  ranks[b] = r;
}

torch::Tensor topk_hit_ranks(torch::Tensor actual, torch::Tensor topk) {
  TORCH_CHECK(actual.is_cuda() && topk.is_cuda());
  auto a = actual.contiguous().to(torch::kInt64);
  auto t = topk.contiguous().to(torch::kInt64);
  const int B = a.size(0), D = a.size(1), K = t.size(1);
  auto ranks = torch::empty({B}, a.options());
  dim3 block(256);
  dim3 grid((B + 255) / 256);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(topk_hit_ranks_kernel, grid, block, 0, stream,
                     a.data_ptr<int64_t>(), t.data_ptr<int64_t>(),
                     ranks.data_ptr<int64_t>(), B, K, D);
  return ranks;
}

}  // namespace genrec
