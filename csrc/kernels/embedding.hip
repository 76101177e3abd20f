// Embedding gather + scatter-add backward (K2/K15 — SURVEY.md §2.4), gfx950.
//
// Replaces ATen's embedding path inside the models. ATen's dense embedding
// backward on ROCm uses a rocprim radix-sort + partition (decoupled
// lookback) pipeline that (a) is slower than atomic scatter-add at this
// zoo's index counts (15-22k indices onto 256-12k rows) and (b) faults
// under hipGraph replay on ROCm 7.x (lookback scan state) — observed via
// rocm-debug-agent on the captured TIGER train step. This implementation
// is a plain gather forward and fp32 atomic scatter-add backward: fully
// graph-safe, no temp allocations, no sort.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

template <typename T>
__global__ void embedding_fwd_kernel(const T* __restrict__ weight,   // [V,d]
                                     const int64_t* __restrict__ idx,  // [N]
                                     T* __restrict__ out,            // [N,d]
                                     int64_t n, int d, int64_t V) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wid; row < n; row += n_waves) {
    int64_t r = idx[row];
    const T* src = weight + r * d;
    T* dst = out + row * d;
    for (int j = lane; j < d; j += WAVE) dst[j] = src[j];
  }
}

template <typename T>
__global__ void embedding_bwd_kernel(const T* __restrict__ dy,   // [N,d]
                                     const int64_t* __restrict__ idx,
                                     float* __restrict__ dweight,  // [V,d]
                                     int64_t n, int d,
                                     int64_t padding_idx) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wid; row < n; row += n_waves) {
    int64_t r = idx[row];
    if (r == padding_idx) continue;
    const T* src = dy + row * d;
    float* dst = dweight + r * d;
    for (int j = lane; j < d; j += WAVE) {
      atomicAdd(&dst[j], to_f32(src[j]));
    }
  }
}

torch::Tensor embedding_fwd(torch::Tensor weight, torch::Tensor indices) {
  TORCH_CHECK(weight.is_cuda() && weight.dim() == 2 && weight.is_contiguous());
  auto idx = indices.contiguous().reshape(-1);
  const int64_t n = idx.numel();
  const int d = weight.size(1);
  auto sizes = indices.sizes().vec();
  sizes.push_back(d);
  auto out = torch::empty(sizes, weight.options());
  if (n == 0) return out;
  dim3 block(256);
  dim3 grid((unsigned)std::min<int64_t>((n + 3) / 4, 8192));
  auto stream = at::cuda::getCurrentHIPStream();
  if (weight.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((embedding_fwd_kernel<float>), grid, block, 0, stream,
                       weight.data_ptr<float>(), idx.data_ptr<int64_t>(),
                       out.data_ptr<float>(), n, d, weight.size(0));
  } else if (weight.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((embedding_fwd_kernel<__hip_bfloat16>), grid, block, 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(weight.data_ptr()),
                       idx.data_ptr<int64_t>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       n, d, weight.size(0));
  } else {
    TORCH_CHECK(false, "embedding_fwd: unsupported dtype");
  }
  return out;
}

// Small-table variant (V*d fits in LDS): every block privatizes the whole
// dweight table in LDS and flushes once. The zoo's relative-position /
// temporal bias tables (192x1, 64x1, 32x1) take 20k+ indices onto <200
// rows — bucket 0 alone collects thousands of updates, and global fp32
// atomics serialize on those hot lines (~32 us/dispatch measured). LDS
// atomics make the hot-row pileup on-chip (~2 us).
template <typename T>
__global__ void embedding_bwd_small_kernel(const T* __restrict__ dy,
                                           const int64_t* __restrict__ idx,
                                           float* __restrict__ dweight,
                                           int64_t n, int d,
                                           int64_t padding_idx, int vd) {
  extern __shared__ float acc[];
  for (int i = threadIdx.x; i < vd; i += blockDim.x) acc[i] = 0.f;
  __syncthreads();
  const int64_t total = n * d;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = t / d;
    int j = (int)(t - row * d);
    int64_t r = idx[row];
    if (r != padding_idx) atomicAdd(&acc[(int)r * d + j], to_f32(dy[t]));
  }
  __syncthreads();
  for (int i = threadIdx.x; i < vd; i += blockDim.x) {
    if (acc[i] != 0.f) atomicAdd(&dweight[i], acc[i]);
  }
}

torch::Tensor embedding_bwd(torch::Tensor dy, torch::Tensor indices,
                            int64_t num_weights, int64_t padding_idx) {
  auto idx = indices.contiguous().reshape(-1);
  const int64_t n = idx.numel();
  const int d = dy.size(-1);
  auto dy2 = dy.contiguous().reshape({n, d});
  auto dweight = torch::zeros({num_weights, d},
                              dy.options().dtype(torch::kFloat32));
  if (n == 0) return dweight;
  const int64_t vd = num_weights * d;
  const bool small = vd <= 12288;  // 48 KB LDS table
  dim3 block(256);
  dim3 grid(small ? 256u
                  : (unsigned)std::min<int64_t>((n + 3) / 4, 8192));
  size_t smem = small ? (size_t)vd * sizeof(float) : 0;
  auto stream = at::cuda::getCurrentHIPStream();
  if (dy.scalar_type() == torch::kFloat32) {
    if (small) {
      hipLaunchKernelGGL((embedding_bwd_small_kernel<float>), grid, block,
                         smem, stream, dy2.data_ptr<float>(),
                         idx.data_ptr<int64_t>(), dweight.data_ptr<float>(),
                         n, d, padding_idx, (int)vd);
    } else {
      hipLaunchKernelGGL((embedding_bwd_kernel<float>), grid, block, 0,
                         stream, dy2.data_ptr<float>(),
                         idx.data_ptr<int64_t>(), dweight.data_ptr<float>(),
                         n, d, padding_idx);
    }
  } else if (dy.scalar_type() == torch::kBFloat16) {
    if (small) {
      hipLaunchKernelGGL((embedding_bwd_small_kernel<__hip_bfloat16>), grid,
                         block, smem, stream,
                         reinterpret_cast<const __hip_bfloat16*>(dy2.data_ptr()),
                         idx.data_ptr<int64_t>(), dweight.data_ptr<float>(),
                         n, d, padding_idx, (int)vd);
    } else {
      hipLaunchKernelGGL((embedding_bwd_kernel<__hip_bfloat16>), grid, block,
                         0, stream,
                         reinterpret_cast<const __hip_bfloat16*>(dy2.data_ptr()),
                         idx.data_ptr<int64_t>(), dweight.data_ptr<float>(),
                         n, d, padding_idx);
    }
  } else {
    TORCH_CHECK(false, "embedding_bwd: unsupported dtype");
  }
  return dweight;
}

}  // namespace genrec
