// Fused softmax cross-entropy (K3/K16 — SURVEY.md §2.4), gfx950.
//
// forward: per-row logsumexp + NLL with ignore_index, mean over valid rows.
// One 256-thread block per chunk of rows, one wave per row, vectorized
// grid-stride over V (SASRec V ~ 12k-60k; TIGER V=769). Saves only the
// per-row lse; backward recomputes softmax(logits) from (logits, lse) in
// one elementwise pass: dlogits = dloss * (softmax - onehot) / n_valid.
// The [N,V] probability tensor is never materialized.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              float* __restrict__ lse,
                              float* __restrict__ row_loss,
                              float* __restrict__ row_valid,
                              int64_t n_rows, int64_t V, int64_t ignore_index) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wid; row < n_rows; row += n_waves) {
    const T* lr = logits + row * V;
    float m = -INFINITY;
    for (int64_t j = lane; j < V; j += WAVE) m = fmaxf(m, to_f32(lr[j]));
    m = wave_max(m);
    float s = 0.f;
    for (int64_t j = lane; j < V; j += WAVE) s += __expf(to_f32(lr[j]) - m);
    s = wave_sum(s);
    float l = m + __logf(s);
    int64_t t = targets[row];
    if (lane == 0) {
      // per-row contributions; reduced by ONE fixed-order ATen sum on the
      // host side — bitwise deterministic (fp32 atomicAdd into a scalar
      // was run-order dependent; caught by tools/race_check.py)
      lse[row] = l;
      bool ok = t != ignore_index;
      row_loss[row] = ok ? (l - to_f32(lr[t])) : 0.f;
      row_valid[row] = ok ? 1.f : 0.f;
    }
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              const float* __restrict__ n_valid_f,
                              T* __restrict__ dlogits,
                              int64_t n_rows, int64_t V, int64_t ignore_index) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  const float g = dloss[0] / fmaxf(n_valid_f[0], 1.f);
  for (int64_t row = wid; row < n_rows; row += n_waves) {
    const T* lr = logits + row * V;
    T* dr = dlogits + row * V;
    int64_t t = targets[row];
    if (t == ignore_index) {
      for (int64_t j = lane; j < V; j += WAVE) dr[j] = from_f32<T>(0.f);
      continue;
    }
    float l = lse[row];
    for (int64_t j = lane; j < V; j += WAVE) {
      float p = __expf(to_f32(lr[j]) - l);
      float d = g * (p - (j == t ? 1.f : 0.f));
      dr[j] = from_f32<T>(d);
    }
  }
}

std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor targets,
                                          int64_t ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  const int64_t n = logits.size(0), V = logits.size(1);
  auto lse = torch::empty({n}, logits.options().dtype(torch::kFloat32));
  auto rowbuf = torch::empty({2, n}, logits.options().dtype(torch::kFloat32));
  dim3 block(256);
  int64_t blocks = std::min<int64_t>((n + 3) / 4, 4096);
  dim3 grid((unsigned)blocks);
  auto stream = at::cuda::getCurrentHIPStream();
  if (logits.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((ce_fwd_kernel<float>), grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<int64_t>(),
                       lse.data_ptr<float>(), rowbuf.data_ptr<float>(),
                       rowbuf.data_ptr<float>() + n, n, V, ignore_index);
  } else if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((ce_fwd_kernel<__hip_bfloat16>), grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
                       targets.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       rowbuf.data_ptr<float>(), rowbuf.data_ptr<float>() + n,
                       n, V, ignore_index);
  } else {
    TORCH_CHECK(false, "softmax_ce: unsupported dtype");
  }
  auto sums = rowbuf.sum(1);  // fixed-order tree reduce: deterministic
  auto n_valid_f = sums[1].reshape({1}).contiguous();
  auto loss_mean = sums[0] / sums[1].clamp_min(1.0);
  return {loss_mean, lse, n_valid_f};
}

torch::Tensor softmax_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor targets, torch::Tensor lse,
                             torch::Tensor n_valid, int64_t ignore_index) {
  const int64_t n = logits.size(0), V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto dloss_f = dloss.to(torch::kFloat32).reshape({1}).contiguous();
  dim3 block(256);
  int64_t blocks = std::min<int64_t>((n + 3) / 4, 4096);
  dim3 grid((unsigned)blocks);
  auto stream = at::cuda::getCurrentHIPStream();
  if (logits.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((ce_bwd_kernel<float>), grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<int64_t>(),
                       lse.data_ptr<float>(), dloss_f.data_ptr<float>(),
                       n_valid.data_ptr<float>(),
                       dlogits.data_ptr<float>(), n, V, ignore_index);
  } else if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((ce_bwd_kernel<__hip_bfloat16>), grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
                       targets.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       dloss_f.data_ptr<float>(), n_valid.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dlogits.data_ptr()),
                       n, V, ignore_index);
  } else {
    TORCH_CHECK(false, "softmax_ce_bwd: unsupported dtype");
  }
  return dlogits;
}

}  // namespace genrec
