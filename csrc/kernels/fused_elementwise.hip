// Fused elementwise kernels: residual + dropout(x) and dropout(relu(x)).
//
// The T5 stack fires ~30 dropout + ~30 add (and 8 relu + 8 dropout)
// elementwise kernels per TIGER step; each is ~4-8 us of pure launch+
// stream time inside the captured graph. Fusing halves the count and the
// bytes moved. Masks are saved as bytes for exact backward replay.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

template <typename T, bool RELU>
__global__ void dropout_fuse_fwd_kernel(const T* __restrict__ x,
                                        const T* __restrict__ residual,
                                        T* __restrict__ out,
                                        unsigned char* __restrict__ mask,
                                        int64_t n, float p, float inv_keep,
                                        unsigned int seed,
                                        const unsigned int* __restrict__ seed_dev) {
  unsigned int s = seed + (seed_dev ? *seed_dev : 0u);
  const unsigned int thresh = (unsigned int)(p * 16777216.0f);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    if (RELU) v = fmaxf(v, 0.f);
    bool keep = (hash_rng(s, i) & 0xFFFFFF) >= thresh;
    mask[i] = keep ? (RELU ? (v > 0.f ? 3 : 1) : 1) : (RELU && v > 0.f ? 2 : 0);
    v = keep ? v * inv_keep : 0.f;
    if (residual) v += to_f32(residual[i]);
    out[i] = from_f32<T>(v);
  }
}

// mask bits: bit0 = dropout keep, bit1 = relu pass (x > 0)
template <typename T, bool RELU>
__global__ void dropout_fuse_bwd_kernel(const T* __restrict__ dy,
                                        const unsigned char* __restrict__ mask,
                                        T* __restrict__ dx,
                                        int64_t n, float inv_keep) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    unsigned char m = mask[i];
    float g = to_f32(dy[i]);
    bool keep = m & 1;
    bool relu_pass = !RELU || (m & 2);
    dx[i] = from_f32<T>((keep && relu_pass) ? g * inv_keep : 0.f);
  }
}

static std::vector<torch::Tensor> dropout_fuse_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> residual, double p,
    int64_t seed, c10::optional<torch::Tensor> seed_dev, bool relu) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int64_t n = x.numel();
  auto out = torch::empty_like(x);
  auto mask = torch::empty({n}, x.options().dtype(torch::kUInt8));
  float inv_keep = p < 1.0 ? 1.0f / (1.0f - (float)p) : 0.f;
  dim3 block(256);
  dim3 grid((unsigned)std::min<int64_t>((n + 255) / 256, 4096));
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_DF(T, RELU)                                                     \
  hipLaunchKernelGGL((dropout_fuse_fwd_kernel<T, RELU>), grid, block, 0,       \
      stream, reinterpret_cast<const T*>(x.data_ptr()),                        \
      residual.has_value() ? reinterpret_cast<const T*>(residual->data_ptr())  \
                           : nullptr,                                          \
      reinterpret_cast<T*>(out.data_ptr()),                                    \
      mask.data_ptr<unsigned char>(), n, (float)p, inv_keep,                   \
      (unsigned int)seed,                                                      \
      seed_dev.has_value()                                                     \
          ? reinterpret_cast<const unsigned int*>(seed_dev->data_ptr())        \
          : nullptr)

  if (x.scalar_type() == torch::kFloat32) {
    if (relu) LAUNCH_DF(float, true);
    else LAUNCH_DF(float, false);
  } else if (x.scalar_type() == torch::kBFloat16) {
    if (relu) LAUNCH_DF(__hip_bfloat16, true);
    else LAUNCH_DF(__hip_bfloat16, false);
  } else {
    TORCH_CHECK(false, "dropout_fuse: unsupported dtype");
  }
#undef LAUNCH_DF
  return {out, mask};
}

std::vector<torch::Tensor> dropout_add_fwd(
    torch::Tensor x, torch::Tensor residual, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev) {
  return dropout_fuse_fwd(x, residual, p, seed, seed_dev, false);
}

std::vector<torch::Tensor> relu_dropout_fwd(
    torch::Tensor x, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev) {
  return dropout_fuse_fwd(x, c10::nullopt, p, seed, seed_dev, true);
}

// plain dropout(x) — ATen native_dropout corrupts under hipGraph replay
// on ROCm 7 (BACKLOG round-2 hazard ledger)
std::vector<torch::Tensor> plain_dropout_fwd(
    torch::Tensor x, double p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev) {
  return dropout_fuse_fwd(x, c10::nullopt, p, seed, seed_dev, false);
}

// Deterministic column sum out[j] = sum_r x[r, j] (fp32 accum), for
// linear BIAS gradients: ATen's outer-dim reduce at these shapes uses a
// semaphore-based two-pass whose state is not hipGraph-replay-safe on
// ROCm 7 (sporadic corrupt elements in bias grads — BACKLOG hazard
// ledger). Two fixed-order stages keep the chip busy and the result
// bit-stable.
template <typename T>
__global__ void colsum1_kernel(const T* __restrict__ x,
                               float* __restrict__ tmp,
                               int64_t rows, int64_t cols, int ch) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= (int64_t)ch * cols) return;
  int64_t j = tid % cols;
  int chunk = (int)(tid / cols);
  float acc = 0.f;
  for (int64_t r = chunk; r < rows; r += ch)
    acc += to_f32(x[r * cols + j]);
  tmp[(int64_t)chunk * cols + j] = acc;
}

// one 64-lane wave per column: lanes stride the ch partials, fixed-order
// shuffle tree finishes (deterministic)
template <typename T>
__global__ void colsum2_kernel(const float* __restrict__ tmp,
                               T* __restrict__ out, int64_t cols, int ch) {
  int64_t j = (int64_t)blockIdx.x * (blockDim.x / WAVE)
      + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (j >= cols) return;
  float acc = 0.f;
  for (int r = lane; r < ch; r += WAVE) acc += tmp[(int64_t)r * cols + j];
  acc = wave_sum(acc);
  if (lane == 0) out[j] = from_f32<T>(acc);
}

torch::Tensor colsum(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const int64_t rows = x.size(0), cols = x.size(1);
  // enough chunks to fill the chip regardless of cols (small-col bias
  // shapes starved at a fixed 32: 46 us at [6400, 256])
  int ch = 32;
  while ((int64_t)ch * cols < 131072 && ch < 512 && ch * 4 < rows) ch *= 2;
  auto tmp = torch::empty({ch, cols}, x.options().dtype(torch::kFloat32));
  auto out = torch::empty({cols}, x.options());
  dim3 block(256);
  dim3 grid1((unsigned)(((int64_t)ch * cols + 255) / 256));
  dim3 grid2((unsigned)((cols + 3) / 4));  // 4 waves/block -> 4 cols
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((colsum1_kernel<float>), grid1, block, 0, stream,
                       x.data_ptr<float>(), tmp.data_ptr<float>(), rows,
                       cols, ch);
    hipLaunchKernelGGL((colsum2_kernel<float>), grid2, block, 0, stream,
                       tmp.data_ptr<float>(), out.data_ptr<float>(), cols,
                       ch);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((colsum1_kernel<__hip_bfloat16>), grid1, block, 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       tmp.data_ptr<float>(), rows, cols, ch);
    hipLaunchKernelGGL((colsum2_kernel<__hip_bfloat16>), grid2, block, 0,
                       stream, tmp.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       cols, ch);
  } else {
    TORCH_CHECK(false, "colsum: unsupported dtype");
  }
  return out;
}

torch::Tensor dropout_fuse_bwd(torch::Tensor dy, torch::Tensor mask,
                               double p, bool relu) {
  const int64_t n = dy.numel();
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(dyc);
  float inv_keep = p < 1.0 ? 1.0f / (1.0f - (float)p) : 0.f;
  dim3 block(256);
  dim3 grid((unsigned)std::min<int64_t>((n + 255) / 256, 4096));
  auto stream = at::cuda::getCurrentHIPStream();
#define LAUNCH_DB(T, RELU)                                                     \
  hipLaunchKernelGGL((dropout_fuse_bwd_kernel<T, RELU>), grid, block, 0,       \
      stream, reinterpret_cast<const T*>(dyc.data_ptr()),                      \
      mask.data_ptr<unsigned char>(),                                          \
      reinterpret_cast<T*>(dx.data_ptr()), n, inv_keep)
  if (dy.scalar_type() == torch::kFloat32) {
    if (relu) LAUNCH_DB(float, true);
    else LAUNCH_DB(float, false);
  } else if (dy.scalar_type() == torch::kBFloat16) {
    if (relu) LAUNCH_DB(__hip_bfloat16, true);
    else LAUNCH_DB(__hip_bfloat16, false);
  } else {
    TORCH_CHECK(false, "dropout_fuse_bwd: unsupported dtype");
  }
#undef LAUNCH_DB
  return dx;
}

}  // namespace genrec
