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

// bf16x4 vectorized variants (guide Guideline 13): 2-byte-per-lane loads
// leave half the memory pipe idle, so the bf16 paths move 4 elements per
// thread as 8-byte words (mask as a 4-byte word). The per-element RNG
// stream (hash_rng(s, i)) and mask encoding are IDENTICAL to the scalar
// kernels, so fwd/bwd stay paired and replays stay deterministic.
// GENREC_SCALAR_ELEMWISE=1 forces the scalar kernels (A/B switch).

union BF16x4 {
  uint2 u;
  __hip_bfloat16 e[4];
};
union U8x4 {
  unsigned int u;
  unsigned char e[4];
};

template <bool RELU, bool HAS_RES>
__global__ void dropout_fuse_fwd_v4_kernel(
    const uint2* __restrict__ x, const uint2* __restrict__ residual,
    uint2* __restrict__ out, unsigned int* __restrict__ mask, int64_t n4,
    float p, float inv_keep, unsigned int seed,
    const unsigned int* __restrict__ seed_dev) {
  unsigned int s = seed + (seed_dev ? *seed_dev : 0u);
  const unsigned int thresh = (unsigned int)(p * 16777216.0f);
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < n4;
       q += (int64_t)gridDim.x * blockDim.x) {
    BF16x4 xi, ri, oo;
    xi.u = x[q];
    if (HAS_RES) ri.u = residual[q];
    U8x4 mm;
    const int64_t base = q * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float v = to_f32(xi.e[k]);
      if (RELU) v = fmaxf(v, 0.f);
      bool keep = (hash_rng(s, base + k) & 0xFFFFFF) >= thresh;
      mm.e[k] = keep ? (RELU ? (v > 0.f ? 3 : 1) : 1)
                     : (RELU && v > 0.f ? 2 : 0);
      v = keep ? v * inv_keep : 0.f;
      if (HAS_RES) v += to_f32(ri.e[k]);
      oo.e[k] = from_f32<__hip_bfloat16>(v);
    }
    out[q] = oo.u;
    mask[q] = mm.u;
  }
}

template <bool RELU>
__global__ void dropout_fuse_bwd_v4_kernel(const uint2* __restrict__ dy,
                                           const unsigned int* __restrict__ mask,
                                           uint2* __restrict__ dx, int64_t n4,
                                           float inv_keep) {
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < n4;
       q += (int64_t)gridDim.x * blockDim.x) {
    BF16x4 g, o;
    g.u = dy[q];
    U8x4 mm;
    mm.u = mask[q];
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      unsigned char m = mm.e[k];
      bool keep = m & 1;
      bool relu_pass = !RELU || (m & 2);
      o.e[k] = from_f32<__hip_bfloat16>(
          (keep && relu_pass) ? to_f32(g.e[k]) * inv_keep : 0.f);
    }
    dx[q] = o.u;
  }
}

static inline bool elemwise_force_scalar() {
  static const bool v = [] {
    const char* e = getenv("GENREC_SCALAR_ELEMWISE");
    return e && e[0] == '1';
  }();
  return v;
}

static inline bool ptr_aligned8(const void* p) {
  return (reinterpret_cast<uintptr_t>(p) & 7) == 0;
}

// Small-row-count column sum out[j] = sum_r x[r, j] with fp32
// accumulation, for the split-K dW partials ([nc<=64, out*in]): ATen's
// bf16 outer-dim reduce at these shapes is a ~6 us latency-bound
// dispatch (26/step on TIGER). One thread per column QUAD (8-byte bf16 /
// 16-byte fp32 loads), fixed-order row loop: deterministic and
// replay-safe by construction. Python guards shape/dtype eligibility.
__global__ void chunk_sum_bf16_kernel(const uint2* __restrict__ x,
                                      uint2* __restrict__ out, int rows,
                                      int64_t cols4) {
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < cols4;
       q += (int64_t)gridDim.x * blockDim.x) {
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    for (int r = 0; r < rows; ++r) {
      BF16x4 v;
      v.u = x[(int64_t)r * cols4 + q];
#pragma unroll
      for (int k = 0; k < 4; ++k) acc[k] += to_f32(v.e[k]);
    }
    BF16x4 o;
#pragma unroll
    for (int k = 0; k < 4; ++k) o.e[k] = from_f32<__hip_bfloat16>(acc[k]);
    out[q] = o.u;
  }
}

__global__ void chunk_sum_f32_kernel(const float4* __restrict__ x,
                                     float4* __restrict__ out, int rows,
                                     int64_t cols4) {
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < cols4;
       q += (int64_t)gridDim.x * blockDim.x) {
    float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int r = 0; r < rows; ++r) {
      float4 v = x[(int64_t)r * cols4 + q];
      acc.x += v.x;
      acc.y += v.y;
      acc.z += v.z;
      acc.w += v.w;
    }
    out[q] = acc;
  }
}

torch::Tensor chunk_sum(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2,
              "chunk_sum: 2D contiguous cuda tensor required");
  const int64_t rows = x.size(0), cols = x.size(1);
  TORCH_CHECK((cols & 3) == 0 && rows <= 65536,
              "chunk_sum: cols %% 4 != 0 or too many rows");
  auto out = torch::empty({cols}, x.options());
  const int64_t cols4 = cols >> 2;
  dim3 block(256);
  dim3 grid((unsigned)std::min<int64_t>((cols4 + 255) / 256, 8192));
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(chunk_sum_bf16_kernel, grid, block, 0, stream,
                       reinterpret_cast<const uint2*>(x.data_ptr()),
                       reinterpret_cast<uint2*>(out.data_ptr()), (int)rows,
                       cols4);
  } else if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(chunk_sum_f32_kernel, grid, block, 0, stream,
                       reinterpret_cast<const float4*>(x.data_ptr()),
                       reinterpret_cast<float4*>(out.data_ptr()), (int)rows,
                       cols4);
  } else {
    TORCH_CHECK(false, "chunk_sum: unsupported dtype");
  }
  return out;
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

  // bf16x4 fast path (see kernel comment)
  if (x.scalar_type() == torch::kBFloat16 && (n & 3) == 0 && n > 0 &&
      ptr_aligned8(x.data_ptr()) && ptr_aligned8(out.data_ptr()) &&
      (!residual.has_value() || ptr_aligned8(residual->data_ptr())) &&
      !elemwise_force_scalar()) {
    const int64_t n4 = n >> 2;
    dim3 grid4((unsigned)std::min<int64_t>((n4 + 255) / 256, 4096));
#define LAUNCH_DF4(RELU, HAS_RES)                                              \
  hipLaunchKernelGGL((dropout_fuse_fwd_v4_kernel<RELU, HAS_RES>), grid4,       \
      block, 0, stream, reinterpret_cast<const uint2*>(x.data_ptr()),          \
      residual.has_value()                                                     \
          ? reinterpret_cast<const uint2*>(residual->data_ptr())               \
          : nullptr,                                                           \
      reinterpret_cast<uint2*>(out.data_ptr()),                                \
      reinterpret_cast<unsigned int*>(mask.data_ptr<unsigned char>()), n4,     \
      (float)p, inv_keep, (unsigned int)seed,                                  \
      seed_dev.has_value()                                                     \
          ? reinterpret_cast<const unsigned int*>(seed_dev->data_ptr())        \
          : nullptr)
    if (relu) {
      if (residual.has_value()) LAUNCH_DF4(true, true);
      else LAUNCH_DF4(true, false);
    } else {
      if (residual.has_value()) LAUNCH_DF4(false, true);
      else LAUNCH_DF4(false, false);
    }
#undef LAUNCH_DF4
    return {out, mask};
  }

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

// bf16x4 stage 1 (column quads, 8-byte loads + float4 partial stores).
// Still deterministic (fixed-order loops); the adaptive chunk count
// differs from the scalar path's, so outputs agree within one bf16 ulp
// rather than bitwise. 2-byte-per-lane loads made the scalar version
// COBRA's single hottest kernel (25 us x 14 bias grads/step).
__global__ void colsum1_v4_kernel(const uint2* __restrict__ x,
                                  float4* __restrict__ tmp,
                                  int64_t rows, int64_t cols4, int ch) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= (int64_t)ch * cols4) return;
  int64_t j4 = tid % cols4;
  int chunk = (int)(tid / cols4);
  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  for (int64_t r = chunk; r < rows; r += ch) {
    BF16x4 v;
    v.u = x[r * cols4 + j4];
#pragma unroll
    for (int k = 0; k < 4; ++k) acc[k] += to_f32(v.e[k]);
  }
  tmp[(int64_t)chunk * cols4 + j4] =
      make_float4(acc[0], acc[1], acc[2], acc[3]);
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
  const bool vec4 = x.scalar_type() == torch::kBFloat16 && (cols & 3) == 0 &&
                    ptr_aligned8(x.data_ptr()) && !elemwise_force_scalar();
  // enough chunks to fill the chip regardless of cols (small-col bias
  // shapes starved at a fixed 32: 46 us at [6400, 256]). The vec4 stage-1
  // kernel has 4x fewer threads per chunk, so it targets the same wave
  // count via a 4x-smaller per-chunk column count.
  const int64_t stage1_cols = vec4 ? cols / 4 : cols;
  int ch = 32;
  while ((int64_t)ch * stage1_cols < 131072 && ch < 512 && ch * 4 < rows)
    ch *= 2;
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
    if (vec4) {
      const int64_t cols4 = cols >> 2;
      dim3 g1v((unsigned)(((int64_t)ch * cols4 + 255) / 256));
      hipLaunchKernelGGL(colsum1_v4_kernel, g1v, block, 0, stream,
                         reinterpret_cast<const uint2*>(x.data_ptr()),
                         reinterpret_cast<float4*>(tmp.data_ptr<float>()),
                         rows, cols4, ch);
    } else {
      hipLaunchKernelGGL((colsum1_kernel<__hip_bfloat16>), grid1, block, 0,
                         stream,
                         reinterpret_cast<const __hip_bfloat16*>(
                             x.data_ptr()),
                         tmp.data_ptr<float>(), rows, cols, ch);
    }
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

  if (dy.scalar_type() == torch::kBFloat16 && (n & 3) == 0 && n > 0 &&
      ptr_aligned8(dyc.data_ptr()) && ptr_aligned8(dx.data_ptr()) &&
      ptr_aligned8(mask.data_ptr()) && !elemwise_force_scalar()) {
    const int64_t n4 = n >> 2;
    dim3 grid4((unsigned)std::min<int64_t>((n4 + 255) / 256, 4096));
#define LAUNCH_DB4(RELU)                                                       \
  hipLaunchKernelGGL((dropout_fuse_bwd_v4_kernel<RELU>), grid4, block, 0,      \
      stream, reinterpret_cast<const uint2*>(dyc.data_ptr()),                  \
      reinterpret_cast<const unsigned int*>(mask.data_ptr<unsigned char>()),   \
      reinterpret_cast<uint2*>(dx.data_ptr()), n4, inv_keep)
    if (relu) LAUNCH_DB4(true);
    else LAUNCH_DB4(false);
#undef LAUNCH_DB4
    return dx;
  }
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
