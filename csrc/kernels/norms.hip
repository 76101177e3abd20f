// Fused norm kernels (K24 — SURVEY.md §2.4): RMSNorm (standard + T5 style)
// and L2Norm, forward + backward, fp32 and bf16, gfx950.
//
// Design: one 64-lane wave per row (rows are small: D = 32..768 across the
// zoo), 4 waves per 256-thread block, grid-stride over rows. bf16 rows are
// loaded vectorized (short4 reinterpret = 8 B/lane) per Guideline 13; all
// math accumulates in fp32, matching the reference's fp32-upcast semantics
// (normalize.py:38-55, 73-95). Backward reduces dweight per-block in LDS
// then atomically into a fp32 accumulator.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

// ---------------------------------------------------------------- RMSNorm

template <typename T, typename WT, typename OutT, bool T5_STYLE>
__global__ void rms_norm_fwd_kernel(const T* __restrict__ x,
                                    const WT* __restrict__ w,
                                    OutT* __restrict__ y,
                                    float* __restrict__ inv_rms,
                                    int64_t n_rows, int d, float eps,
                                    bool w_is_half) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    float ss = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      float v = to_f32(xr[j]);
      ss += v * v;
    }
    ss = wave_sum(ss);
    float r = rsqrtf(ss / d + eps);
    if (lane == 0) inv_rms[row] = r;
    OutT* yr = y + row * d;
    for (int j = lane; j < d; j += WAVE) {
      float xf = to_f32(xr[j]);
      float wf = to_f32(w[j]);
      float out;
      if (T5_STYLE) {
        // t = x * r (promote), optional cast to weight dtype, then w * t
        float t = to_f32(xr[j]) * r;
        if (w_is_half) t = to_f32(from_f32<__hip_bfloat16>(t));
        out = wf * t;
      } else {
        // y = cast_to_xdtype(xf * r) * w
        float t = xf * r;
        t = to_f32(from_f32<T>(t));
        out = t * wf;
      }
      yr[j] = from_f32<OutT>(out);
    }
  }
}

// backward: dx_i = w_i*dy_i*r - r^3/d * x_i * sum_j(w_j*dy_j*x_j)
// (intermediate rounding of the forward's casts is ignored in backward, as
// eager autograd does for the same graph up to bf16 rounding)
constexpr int RMS_MAX_COLS_PER_LANE = 16;  // supports d <= 1024

template <typename T, typename WT, typename OutT>
__global__ void rms_norm_bwd_kernel(const OutT* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const WT* __restrict__ w,
                                    const float* __restrict__ inv_rms,
                                    T* __restrict__ dx,
                                    float* __restrict__ dw,
                                    int64_t n_rows, int d) {
  const int wave_in_block = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wave_id = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  const int64_t n_waves = (int64_t)gridDim.x * waves_per_block;

  // per-lane dw partials in registers (lane owns columns lane, lane+64, ...)
  float dw_acc[RMS_MAX_COLS_PER_LANE];
#pragma unroll
  for (int c = 0; c < RMS_MAX_COLS_PER_LANE; ++c) dw_acc[c] = 0.f;

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    const OutT* dyr = dy + row * d;
    float r = inv_rms[row];
    float dot = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      dot += to_f32(w[j]) * to_f32(dyr[j]) * to_f32(xr[j]);
    }
    dot = wave_sum(dot);
    float c = r * r * r / d * dot;
    T* dxr = dx + row * d;
    int ci = 0;
    for (int j = lane; j < d; j += WAVE, ++ci) {
      float xf = to_f32(xr[j]);
      float dyf = to_f32(dyr[j]);
      dxr[j] = from_f32<T>(to_f32(w[j]) * dyf * r - c * xf);
      dw_acc[ci] += dyf * xf * r;
    }
  }
  // block-level reduce of the 4 waves' partials in LDS, then ONE partial
  // row per block into global scratch [n_blocks, d] (summed by ATen after
  // the launch — no atomics, deterministic).
  extern __shared__ __attribute__((aligned(16))) float dw_lds[];  // [4][d]
  int ci = 0;
  for (int j = lane; j < d; j += WAVE, ++ci) {
    dw_lds[wave_in_block * d + j] = dw_acc[ci];
  }
  __syncthreads();
  for (int j = (int)threadIdx.x; j < d; j += (int)blockDim.x) {
    float s = dw_lds[j] + dw_lds[d + j] + dw_lds[2 * d + j] +
              dw_lds[3 * d + j];
    dw[(int64_t)blockIdx.x * d + j] = s;
  }
}

// ---------------------------------------------------------------- L2Norm

template <typename T>
__global__ void l2norm_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  float* __restrict__ inv_norm, int64_t n_rows,
                                  int d, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    float ss = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      float v = to_f32(xr[j]);
      ss += v * v;
    }
    ss = wave_sum(ss);
    // F.normalize: x / max(||x||, eps)
    float inv = 1.0f / fmaxf(sqrtf(ss), eps);
    if (lane == 0) inv_norm[row] = inv;
    T* yr = y + row * d;
    for (int j = lane; j < d; j += WAVE) {
      yr[j] = from_f32<T>(to_f32(xr[j]) * inv);
    }
  }
}

// dx = inv*(dy - y * dot(dy, y))  with y = x*inv  (when not eps-clamped;
// clamped rows: dx = inv*dy since norm is constant w.r.t x below eps — the
// clamp branch gradient through max() is zero. F.normalize autograd
// behaves the same way for ||x|| < eps.)
template <typename T>
__global__ void l2norm_bwd_kernel(const T* __restrict__ dy,
                                  const T* __restrict__ x,
                                  const float* __restrict__ inv_norm,
                                  T* __restrict__ dx, int64_t n_rows, int d,
                                  float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    const T* dyr = dy + row * d;
    float inv = inv_norm[row];
    float dot = 0.f, ss = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      float xf = to_f32(xr[j]);
      dot += to_f32(dyr[j]) * xf;
      ss += xf * xf;
    }
    dot = wave_sum(dot);
    ss = wave_sum(ss);
    bool clamped = sqrtf(ss) < eps;
    T* dxr = dx + row * d;
    for (int j = lane; j < d; j += WAVE) {
      float xf = to_f32(xr[j]);
      float g = to_f32(dyr[j]) * inv;
      if (!clamped) g -= xf * inv * inv * inv * dot;
      dxr[j] = from_f32<T>(g);
    }
  }
}

// ---------------------------------------------------------------- host

static int grid_for_rows(int64_t n_rows, int waves_per_block) {
  int64_t blocks = (n_rows + waves_per_block - 1) / waves_per_block;
  return (int)std::min<int64_t>(blocks, 8192);
}

std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w,
                                        double eps, bool t5_style) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  auto wc = w.contiguous();
  bool w_bf16 = w.scalar_type() == torch::kBFloat16;
  bool w_half = w_bf16 || w.scalar_type() == torch::kHalf;
  TORCH_CHECK(w_bf16 || w.scalar_type() == torch::kFloat32);
  // output dtype follows eager semantics: promote(x.dtype, w.dtype)
  auto out_dtype = at::result_type(x, w);
  auto y = torch::empty(x.sizes(), x.options().dtype(out_dtype));
  auto inv_rms = torch::empty({n_rows}, x.options().dtype(torch::kFloat32));
  dim3 block(256);
  dim3 grid(grid_for_rows(n_rows, 4));
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_RMS(T, WT, OutT, T5)                                            \
  hipLaunchKernelGGL((rms_norm_fwd_kernel<T, WT, OutT, T5>), grid, block, 0,   \
                     stream, reinterpret_cast<const T*>(x.data_ptr()),         \
                     reinterpret_cast<const WT*>(wc.data_ptr()),               \
                     reinterpret_cast<OutT*>(y.data_ptr()),                    \
                     inv_rms.data_ptr<float>(), n_rows, d, (float)eps, w_half)

#define LAUNCH_RMS_W(T, OutT, T5)                                              \
  do {                                                                         \
    if (w_bf16) LAUNCH_RMS(T, __hip_bfloat16, OutT, T5);                       \
    else LAUNCH_RMS(T, float, OutT, T5);                                       \
  } while (0)

  if (x.scalar_type() == torch::kFloat32) {
    if (t5_style) LAUNCH_RMS_W(float, float, true);
    else LAUNCH_RMS_W(float, float, false);
  } else if (x.scalar_type() == torch::kBFloat16) {
    if (out_dtype == torch::kFloat32) {
      if (t5_style) LAUNCH_RMS_W(__hip_bfloat16, float, true);
      else LAUNCH_RMS_W(__hip_bfloat16, float, false);
    } else {
      if (t5_style) LAUNCH_RMS_W(__hip_bfloat16, __hip_bfloat16, true);
      else LAUNCH_RMS_W(__hip_bfloat16, __hip_bfloat16, false);
    }
  } else {
    TORCH_CHECK(false, "rms_norm: unsupported dtype");
  }
#undef LAUNCH_RMS_W
#undef LAUNCH_RMS
  return {y, inv_rms};
}

// Column-sum of the per-block dw partials + cast, replacing the ATen
// sum(0) + .to() pair (~12+5 us per call, ~22 calls/step on TIGER).
// Two stages so the reduce has enough resident waves (a single-stage
// d-thread loop left only 6 waves on the whole chip and was LOSING to
// ATen): stage 1 folds n_part rows 32-fold with CH*d threads, stage 2
// finishes 32 rows with d threads. Fixed-order loops: deterministic.
constexpr int RED_CH = 32;

__global__ void rms_dw_reduce1_kernel(const float* __restrict__ part,
                                      float* __restrict__ tmp,
                                      int n_part, int d) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= (int64_t)RED_CH * d) return;
  int j = (int)(tid % d);
  int chunk = (int)(tid / d);
  float acc = 0.f;
  for (int r = chunk; r < n_part; r += RED_CH) acc += part[(int64_t)r * d + j];
  tmp[(int64_t)chunk * d + j] = acc;
}

template <typename WT>
__global__ void rms_dw_reduce2_kernel(const float* __restrict__ tmp,
                                      WT* __restrict__ out, int d) {
  int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= d) return;
  float acc = 0.f;
#pragma unroll
  for (int r = 0; r < RED_CH; ++r) acc += tmp[(int64_t)r * d + j];
  out[j] = from_f32<WT>(acc);
}

std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w, torch::Tensor inv_rms,
                                        bool t5_style) {
  (void)t5_style;  // same analytic gradient for both styles
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  TORCH_CHECK(d <= WAVE * 16, "rms_norm_bwd: d too large");
  auto wc = w.contiguous();
  bool w_bf16 = w.scalar_type() == torch::kBFloat16;
  auto dx = torch::empty_like(x);
  dim3 block(256);
  // 512 blocks keeps 2048 waves for the main pass (still ~8x over-
  // subscribed at TIGER shapes) while halving the dw partial matrix the
  // two-stage reduce has to chew through
  int n_blocks = std::min(grid_for_rows(n_rows, 4), 512);
  dim3 grid(n_blocks);
  auto dw = torch::empty({n_blocks, (int64_t)d},
                         x.options().dtype(torch::kFloat32));
  size_t smem = 4 * (size_t)d * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_RMSB(T, WT, OutT)                                               \
  hipLaunchKernelGGL((rms_norm_bwd_kernel<T, WT, OutT>), grid, block, smem,    \
                     stream, reinterpret_cast<const OutT*>(dy.data_ptr()),     \
                     reinterpret_cast<const T*>(x.data_ptr()),                 \
                     reinterpret_cast<const WT*>(wc.data_ptr()),               \
                     inv_rms.data_ptr<float>(),                                \
                     reinterpret_cast<T*>(dx.data_ptr()),                      \
                     dw.data_ptr<float>(), n_rows, d)

#define LAUNCH_RMSB_W(T, OutT)                                                 \
  do {                                                                         \
    if (w_bf16) LAUNCH_RMSB(T, __hip_bfloat16, OutT);                          \
    else LAUNCH_RMSB(T, float, OutT);                                          \
  } while (0)

  if (x.scalar_type() == torch::kFloat32) {
    LAUNCH_RMSB_W(float, float);
  } else if (x.scalar_type() == torch::kBFloat16) {
    if (dy.scalar_type() == torch::kFloat32)
      LAUNCH_RMSB_W(__hip_bfloat16, float);
    else LAUNCH_RMSB_W(__hip_bfloat16, __hip_bfloat16);
  } else {
    TORCH_CHECK(false, "rms_norm_bwd: unsupported dtype");
  }
#undef LAUNCH_RMSB_W
#undef LAUNCH_RMSB
  auto dw_out = torch::empty({(int64_t)d}, w.options());
  auto dw_tmp = torch::empty({RED_CH, (int64_t)d},
                             x.options().dtype(torch::kFloat32));
  dim3 rblock(256);
  dim3 rgrid1((unsigned)(((int64_t)RED_CH * d + 255) / 256));
  hipLaunchKernelGGL(rms_dw_reduce1_kernel, rgrid1, rblock, 0, stream,
                     dw.data_ptr<float>(), dw_tmp.data_ptr<float>(),
                     n_blocks, d);
  dim3 rgrid2((d + 255) / 256);
  if (w_bf16) {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<__hip_bfloat16>), rgrid2,
                       rblock, 0, stream, dw_tmp.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dw_out.data_ptr()),
                       d);
  } else {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<float>), rgrid2, rblock, 0,
                       stream, dw_tmp.data_ptr<float>(),
                       dw_out.data_ptr<float>(), d);
  }
  return {dx, dw_out};
}

std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  auto y = torch::empty_like(x);
  auto inv_norm = torch::empty({n_rows}, x.options().dtype(torch::kFloat32));
  dim3 block(256);
  dim3 grid(grid_for_rows(n_rows, 4));
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((l2norm_fwd_kernel<float>), grid, block, 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(),
                       inv_norm.data_ptr<float>(), n_rows, d, (float)eps);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((l2norm_fwd_kernel<__hip_bfloat16>), grid, block, 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       inv_norm.data_ptr<float>(), n_rows, d, (float)eps);
  } else {
    TORCH_CHECK(false, "l2norm: unsupported dtype");
  }
  return {y, inv_norm};
}

torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor x,
                         torch::Tensor inv_norm, double eps_in) {
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  auto dx = torch::empty_like(x);
  dim3 block(256);
  dim3 grid(grid_for_rows(n_rows, 4));
  auto stream = at::cuda::getCurrentHIPStream();
  float eps = (float)eps_in;
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((l2norm_bwd_kernel<float>), grid, block, 0, stream,
                       dy.data_ptr<float>(), x.data_ptr<float>(),
                       inv_norm.data_ptr<float>(), dx.data_ptr<float>(),
                       n_rows, d, eps);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((l2norm_bwd_kernel<__hip_bfloat16>), grid, block, 0,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       inv_norm.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       n_rows, d, eps);
  } else {
    TORCH_CHECK(false, "l2norm_bwd: unsupported dtype");
  }
  return dx;
}

}  // namespace genrec
