// Fused norm kernels (K24 — SURVEY.md §2.4): RMSNorm (standard + T5 style)
// and L2Norm, forward + backward, fp32 and bf16, gfx950.
//
// Design: one 64-lane wave per row (rows are small: D = 32..768 across the
// zoo), 4 waves per 256-thread block, grid-stride over rows. bf16 rows are
// loaded/stored as column-pair dwords (4 B/lane) per Guideline 13; all
// math accumulates in fp32, matching the reference's fp32-upcast semantics
// (normalize.py:38-55, 73-95). Backward reduces dweight per-block in LDS
// into one partial row per block, finished by a fixed-order two-stage
// column reduce (deterministic, no atomics).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <type_traits>

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

// bf16x2-vectorized variants (guide Guideline 13): 2-byte-per-lane loads
// leave half the memory pipe idle, so the bf16 paths load/store column
// PAIRS (4 B/lane dwords). Math, rounding points and the dw partial
// layout are identical to the scalar kernels; only the fp32 accumulation
// order inside a lane changes (pairs instead of 64-strided singles).
// GENREC_SCALAR_NORMS=1 forces the scalar kernels (A/B switch).

union BF16x2 {
  unsigned int u;
  __hip_bfloat16 e[2];
};

template <typename WT>
__device__ __forceinline__ void load_w2(const WT* __restrict__ w, int j2,
                                        float* wf) {
  if constexpr (std::is_same<WT, float>::value) {
    float2 wv = reinterpret_cast<const float2*>(w)[j2];
    wf[0] = wv.x;
    wf[1] = wv.y;
  } else {
    BF16x2 wv;
    wv.u = reinterpret_cast<const unsigned int*>(w)[j2];
    wf[0] = to_f32(wv.e[0]);
    wf[1] = to_f32(wv.e[1]);
  }
}

template <typename WT, bool T5_STYLE>
__global__ void rms_norm_fwd_v2_kernel(const unsigned int* __restrict__ x,
                                       const WT* __restrict__ w,
                                       unsigned int* __restrict__ y,
                                       float* __restrict__ inv_rms,
                                       int64_t n_rows, int d2, float eps,
                                       bool w_is_half) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  const int d = d2 * 2;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const unsigned int* xr = x + row * d2;
    float ss = 0.f;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 v;
      v.u = xr[j];
      float a = to_f32(v.e[0]), b = to_f32(v.e[1]);
      ss += a * a + b * b;
    }
    ss = wave_sum(ss);
    float r = rsqrtf(ss / d + eps);
    if (lane == 0) inv_rms[row] = r;
    unsigned int* yr = y + row * d2;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 v, o;
      v.u = xr[j];
      float wf[2];
      load_w2(w, j, wf);
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        float xf = to_f32(v.e[k]);
        float out;
        if (T5_STYLE) {
          float t = xf * r;
          if (w_is_half) t = to_f32(from_f32<__hip_bfloat16>(t));
          out = wf[k] * t;
        } else {
          float t = xf * r;
          t = to_f32(from_f32<__hip_bfloat16>(t));
          out = t * wf[k];
        }
        o.e[k] = from_f32<__hip_bfloat16>(out);
      }
      yr[j] = o.u;
    }
  }
}

template <typename WT>
__global__ void rms_norm_bwd_v2_kernel(const unsigned int* __restrict__ dy,
                                       const unsigned int* __restrict__ x,
                                       const WT* __restrict__ w,
                                       const float* __restrict__ inv_rms,
                                       unsigned int* __restrict__ dx,
                                       float* __restrict__ dw,
                                       int64_t n_rows, int d2) {
  const int wave_in_block = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wave_id = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  const int64_t n_waves = (int64_t)gridDim.x * waves_per_block;
  const int d = d2 * 2;

  float dw_acc[RMS_MAX_COLS_PER_LANE];
#pragma unroll
  for (int c = 0; c < RMS_MAX_COLS_PER_LANE; ++c) dw_acc[c] = 0.f;

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const unsigned int* xr = x + row * d2;
    const unsigned int* dyr = dy + row * d2;
    float r = inv_rms[row];
    float dot = 0.f;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 xv, gv;
      xv.u = xr[j];
      gv.u = dyr[j];
      float wf[2];
      load_w2(w, j, wf);
      dot += wf[0] * to_f32(gv.e[0]) * to_f32(xv.e[0]) +
             wf[1] * to_f32(gv.e[1]) * to_f32(xv.e[1]);
    }
    dot = wave_sum(dot);
    float c = r * r * r / d * dot;
    unsigned int* dxr = dx + row * d2;
    int ci = 0;
    for (int j = lane; j < d2; j += WAVE, ci += 2) {
      BF16x2 xv, gv, o;
      xv.u = xr[j];
      gv.u = dyr[j];
      float wf[2];
      load_w2(w, j, wf);
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        float xf = to_f32(xv.e[k]);
        float dyf = to_f32(gv.e[k]);
        o.e[k] = from_f32<__hip_bfloat16>(wf[k] * dyf * r - c * xf);
        dw_acc[ci + k] += dyf * xf * r;
      }
      dxr[j] = o.u;
    }
  }
  extern __shared__ __attribute__((aligned(16))) float dw_lds[];  // [4][d]
  int ci = 0;
  for (int j = lane; j < d2; j += WAVE, ci += 2) {
    dw_lds[wave_in_block * d + 2 * j] = dw_acc[ci];
    dw_lds[wave_in_block * d + 2 * j + 1] = dw_acc[ci + 1];
  }
  __syncthreads();
  for (int j = (int)threadIdx.x; j < d; j += (int)blockDim.x) {
    float s = dw_lds[j] + dw_lds[d + j] + dw_lds[2 * d + j] +
              dw_lds[3 * d + j];
    dw[(int64_t)blockIdx.x * d + j] = s;
  }
}

static inline bool norms_force_scalar() {
  static const bool v = [] {
    const char* e = getenv("GENREC_SCALAR_NORMS");
    return e && e[0] == '1';
  }();
  return v;
}

static inline bool ptr_aligned4(const void* p) {
  return (reinterpret_cast<uintptr_t>(p) & 3) == 0;
}

// -------------------------------------------------------------- LayerNorm
// nn.LayerNorm semantics (elementwise affine, fp32 stats like ATen's
// vectorized_layer_norm_kernel<BFloat16, float>): ATen's backward pair
// (cuComputeGradInput 68 us + cuComputePartGradGammaBeta) was ~7% of the
// COBRA step. Same wave-per-row structure as the RMSNorm kernels; dw/db
// partials go through the adaptive two-stage reduce ([n_blocks, 2d]).

static int grid_for_rows(int64_t n_rows, int waves_per_block);
__global__ void rms_dw_reduce1_kernel(const float* __restrict__ part,
                                      float* __restrict__ tmp, int n_part,
                                      int d, int ch);
template <typename WT>
__global__ void rms_dw_reduce2_kernel(const float* __restrict__ tmp,
                                      WT* __restrict__ out, int d, int ch);

template <typename T, typename WT>
__global__ void layer_norm_fwd_kernel(const T* __restrict__ x,
                                      const WT* __restrict__ w,
                                      const WT* __restrict__ b,
                                      T* __restrict__ y,
                                      float* __restrict__ mean_arr,
                                      float* __restrict__ rstd_arr,
                                      int64_t n_rows, int d, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    float s = 0.f, ss = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      float v = to_f32(xr[j]);
      s += v;
      ss += v * v;
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    float mean = s / d;
    float var = ss / d - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mean_arr[row] = mean;
      rstd_arr[row] = rstd;
    }
    T* yr = y + row * d;
    for (int j = lane; j < d; j += WAVE) {
      float xhat = (to_f32(xr[j]) - mean) * rstd;
      yr[j] = from_f32<T>(xhat * to_f32(w[j]) + to_f32(b[j]));
    }
  }
}

template <typename T, typename WT>
__global__ void layer_norm_bwd_kernel(const T* __restrict__ dy,
                                      const T* __restrict__ x,
                                      const WT* __restrict__ w,
                                      const float* __restrict__ mean_arr,
                                      const float* __restrict__ rstd_arr,
                                      T* __restrict__ dx,
                                      float* __restrict__ dwdb,
                                      int64_t n_rows, int d) {
  const int wave_in_block = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wave_id = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  const int64_t n_waves = (int64_t)gridDim.x * waves_per_block;

  float dw_acc[RMS_MAX_COLS_PER_LANE];
  float db_acc[RMS_MAX_COLS_PER_LANE];
#pragma unroll
  for (int c = 0; c < RMS_MAX_COLS_PER_LANE; ++c) dw_acc[c] = db_acc[c] = 0.f;

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const T* xr = x + row * d;
    const T* dyr = dy + row * d;
    float mean = mean_arr[row];
    float rstd = rstd_arr[row];
    float c1 = 0.f, c2 = 0.f;
    for (int j = lane; j < d; j += WAVE) {
      float g = to_f32(dyr[j]) * to_f32(w[j]);
      float xhat = (to_f32(xr[j]) - mean) * rstd;
      c1 += g;
      c2 += g * xhat;
    }
    c1 = wave_sum(c1) / d;
    c2 = wave_sum(c2) / d;
    T* dxr = dx + row * d;
    int ci = 0;
    for (int j = lane; j < d; j += WAVE, ++ci) {
      float dyf = to_f32(dyr[j]);
      float g = dyf * to_f32(w[j]);
      float xhat = (to_f32(xr[j]) - mean) * rstd;
      dxr[j] = from_f32<T>(rstd * (g - c1 - xhat * c2));
      dw_acc[ci] += dyf * xhat;
      db_acc[ci] += dyf;
    }
  }
  // one [2, d] partial row per block (dw then db), LDS block-reduced
  extern __shared__ __attribute__((aligned(16))) float ln_lds[];  // [4][2d]
  int ci = 0;
  for (int j = lane; j < d; j += WAVE, ++ci) {
    ln_lds[wave_in_block * 2 * d + j] = dw_acc[ci];
    ln_lds[wave_in_block * 2 * d + d + j] = db_acc[ci];
  }
  __syncthreads();
  for (int j = (int)threadIdx.x; j < 2 * d; j += (int)blockDim.x) {
    float sum = ln_lds[j] + ln_lds[2 * d + j] + ln_lds[4 * d + j] +
                ln_lds[6 * d + j];
    dwdb[(int64_t)blockIdx.x * 2 * d + j] = sum;
  }
}

// bf16x2 variants (column-pair dwords, same math/rounding as scalar —
// only the per-lane fp32 accumulation order changes, as for RMSNorm)
__global__ void layer_norm_fwd_v2_kernel(const unsigned int* __restrict__ x,
                                         const unsigned int* __restrict__ w,
                                         const unsigned int* __restrict__ b,
                                         unsigned int* __restrict__ y,
                                         float* __restrict__ mean_arr,
                                         float* __restrict__ rstd_arr,
                                         int64_t n_rows, int d2, float eps) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n_waves = gridDim.x * blockDim.x / WAVE;
  const int d = d2 * 2;
  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const unsigned int* xr = x + row * d2;
    float s = 0.f, ss = 0.f;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 v;
      v.u = xr[j];
      float a = to_f32(v.e[0]), c = to_f32(v.e[1]);
      s += a + c;
      ss += a * a + c * c;
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    float mean = s / d;
    float var = ss / d - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mean_arr[row] = mean;
      rstd_arr[row] = rstd;
    }
    unsigned int* yr = y + row * d2;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 v, wv, bv, o;
      v.u = xr[j];
      wv.u = w[j];
      bv.u = b[j];
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        float xhat = (to_f32(v.e[k]) - mean) * rstd;
        o.e[k] = from_f32<__hip_bfloat16>(
            xhat * to_f32(wv.e[k]) + to_f32(bv.e[k]));
      }
      yr[j] = o.u;
    }
  }
}

__global__ void layer_norm_bwd_v2_kernel(const unsigned int* __restrict__ dy,
                                         const unsigned int* __restrict__ x,
                                         const unsigned int* __restrict__ w,
                                         const float* __restrict__ mean_arr,
                                         const float* __restrict__ rstd_arr,
                                         unsigned int* __restrict__ dx,
                                         float* __restrict__ dwdb,
                                         int64_t n_rows, int d2) {
  const int wave_in_block = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wave_id = (int64_t)blockIdx.x * waves_per_block + wave_in_block;
  const int64_t n_waves = (int64_t)gridDim.x * waves_per_block;
  const int d = d2 * 2;

  float dw_acc[RMS_MAX_COLS_PER_LANE];
  float db_acc[RMS_MAX_COLS_PER_LANE];
#pragma unroll
  for (int c = 0; c < RMS_MAX_COLS_PER_LANE; ++c) dw_acc[c] = db_acc[c] = 0.f;

  for (int64_t row = wave_id; row < n_rows; row += n_waves) {
    const unsigned int* xr = x + row * d2;
    const unsigned int* dyr = dy + row * d2;
    float mean = mean_arr[row];
    float rstd = rstd_arr[row];
    float c1 = 0.f, c2 = 0.f;
    for (int j = lane; j < d2; j += WAVE) {
      BF16x2 xv, gv, wv;
      xv.u = xr[j];
      gv.u = dyr[j];
      wv.u = w[j];
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        float g = to_f32(gv.e[k]) * to_f32(wv.e[k]);
        float xhat = (to_f32(xv.e[k]) - mean) * rstd;
        c1 += g;
        c2 += g * xhat;
      }
    }
    c1 = wave_sum(c1) / d;
    c2 = wave_sum(c2) / d;
    unsigned int* dxr = dx + row * d2;
    int ci = 0;
    for (int j = lane; j < d2; j += WAVE, ci += 2) {
      BF16x2 xv, gv, wv, o;
      xv.u = xr[j];
      gv.u = dyr[j];
      wv.u = w[j];
#pragma unroll
      for (int k = 0; k < 2; ++k) {
        float dyf = to_f32(gv.e[k]);
        float g = dyf * to_f32(wv.e[k]);
        float xhat = (to_f32(xv.e[k]) - mean) * rstd;
        o.e[k] = from_f32<__hip_bfloat16>(rstd * (g - c1 - xhat * c2));
        dw_acc[ci + k] += dyf * xhat;
        db_acc[ci + k] += dyf;
      }
      dxr[j] = o.u;
    }
  }
  extern __shared__ __attribute__((aligned(16))) float ln_lds[];  // [4][2d]
  int ci = 0;
  for (int j = lane; j < d2; j += WAVE, ci += 2) {
    ln_lds[wave_in_block * 2 * d + 2 * j] = dw_acc[ci];
    ln_lds[wave_in_block * 2 * d + 2 * j + 1] = dw_acc[ci + 1];
    ln_lds[wave_in_block * 2 * d + d + 2 * j] = db_acc[ci];
    ln_lds[wave_in_block * 2 * d + d + 2 * j + 1] = db_acc[ci + 1];
  }
  __syncthreads();
  for (int j = (int)threadIdx.x; j < 2 * d; j += (int)blockDim.x) {
    float sum = ln_lds[j] + ln_lds[2 * d + j] + ln_lds[4 * d + j] +
                ln_lds[6 * d + j];
    dwdb[(int64_t)blockIdx.x * 2 * d + j] = sum;
  }
}

std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor w,
                                          torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  TORCH_CHECK(d <= WAVE * 16, "layer_norm: d too large");
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              b.scalar_type() == x.scalar_type(),
              "layer_norm: dtype mismatch");
  auto y = torch::empty_like(x);
  auto mean = torch::empty({n_rows}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty({n_rows}, x.options().dtype(torch::kFloat32));
  dim3 block(256);
  dim3 grid(grid_for_rows(n_rows, 4));
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    if ((d & 1) == 0 && ptr_aligned4(x.data_ptr()) &&
        ptr_aligned4(wc.data_ptr()) && ptr_aligned4(bc.data_ptr()) &&
        !norms_force_scalar()) {
      hipLaunchKernelGGL(layer_norm_fwd_v2_kernel, grid, block, 0, stream,
                         reinterpret_cast<const unsigned int*>(x.data_ptr()),
                         reinterpret_cast<const unsigned int*>(wc.data_ptr()),
                         reinterpret_cast<const unsigned int*>(bc.data_ptr()),
                         reinterpret_cast<unsigned int*>(y.data_ptr()),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         n_rows, d / 2, (float)eps);
      return {y, mean, rstd};
    }
    hipLaunchKernelGGL((layer_norm_fwd_kernel<__hip_bfloat16, __hip_bfloat16>),
                       grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(wc.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(bc.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       n_rows, d, (float)eps);
  } else if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((layer_norm_fwd_kernel<float, float>), grid, block, 0,
                       stream, x.data_ptr<float>(), wc.data_ptr<float>(),
                       bc.data_ptr<float>(), y.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       n_rows, d, (float)eps);
  } else {
    TORCH_CHECK(false, "layer_norm: unsupported dtype");
  }
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w, torch::Tensor mean,
                                          torch::Tensor rstd) {
  const int d = x.size(-1);
  const int64_t n_rows = x.numel() / d;
  TORCH_CHECK(d <= WAVE * 16, "layer_norm_bwd: d too large");
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == x.scalar_type() &&
              w.scalar_type() == x.scalar_type());
  auto wc = w.contiguous();
  auto dx = torch::empty_like(x);
  dim3 block(256);
  static const int cap = [] {
    const char* e = getenv("GENREC_RMS_BWD_CAP");
    int v = e ? atoi(e) : 1024;
    return (v >= 64 && v <= 4096) ? v : 1024;
  }();
  int n_blocks = std::min(grid_for_rows(n_rows, 4), cap);
  dim3 grid(n_blocks);
  auto dwdb = torch::empty({n_blocks, 2 * (int64_t)d},
                           x.options().dtype(torch::kFloat32));
  size_t smem = 8 * (size_t)d * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16 && (d & 1) == 0 &&
      ptr_aligned4(dy.data_ptr()) && ptr_aligned4(x.data_ptr()) &&
      ptr_aligned4(wc.data_ptr()) && !norms_force_scalar()) {
    hipLaunchKernelGGL(layer_norm_bwd_v2_kernel, grid, block, smem, stream,
                       reinterpret_cast<const unsigned int*>(dy.data_ptr()),
                       reinterpret_cast<const unsigned int*>(x.data_ptr()),
                       reinterpret_cast<const unsigned int*>(wc.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<unsigned int*>(dx.data_ptr()),
                       dwdb.data_ptr<float>(), n_rows, d / 2);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((layer_norm_bwd_kernel<__hip_bfloat16, __hip_bfloat16>),
                       grid, block, smem, stream,
                       reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(wc.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       dwdb.data_ptr<float>(), n_rows, d);
  } else if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((layer_norm_bwd_kernel<float, float>), grid, block,
                       smem, stream, dy.data_ptr<float>(),
                       x.data_ptr<float>(), wc.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       dx.data_ptr<float>(), dwdb.data_ptr<float>(), n_rows,
                       d);
  } else {
    TORCH_CHECK(false, "layer_norm_bwd: unsupported dtype");
  }
  // adaptive two-stage column reduce over the [n_blocks, 2d] partials
  auto dwdb_out = torch::empty({2 * (int64_t)d}, w.options());
  const int d2x = 2 * d;
  int ch = 32;
  while ((int64_t)ch * d2x < 131072 && ch < 512 && ch * 4 < n_blocks) ch *= 2;
  auto tmp = torch::empty({ch, (int64_t)d2x},
                          x.options().dtype(torch::kFloat32));
  dim3 rblock(256);
  dim3 rgrid1((unsigned)(((int64_t)ch * d2x + 255) / 256));
  hipLaunchKernelGGL(rms_dw_reduce1_kernel, rgrid1, rblock, 0, stream,
                     dwdb.data_ptr<float>(), tmp.data_ptr<float>(), n_blocks,
                     d2x, ch);
  dim3 rgrid2((unsigned)((d2x + 3) / 4));
  if (w.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<__hip_bfloat16>), rgrid2,
                       rblock, 0, stream, tmp.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dwdb_out.data_ptr()),
                       d2x, ch);
  } else {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<float>), rgrid2, rblock, 0,
                       stream, tmp.data_ptr<float>(),
                       dwdb_out.data_ptr<float>(), d2x, ch);
  }
  auto parts = dwdb_out.split(d, 0);
  return {dx, parts[0], parts[1]};
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

  // bf16x2 fast path: dword loads/stores (see kernel comment)
  if (x.scalar_type() == torch::kBFloat16 && out_dtype == torch::kBFloat16 &&
      (d & 1) == 0 && ptr_aligned4(x.data_ptr()) &&
      ptr_aligned4(y.data_ptr()) &&
      (reinterpret_cast<uintptr_t>(wc.data_ptr()) & (w_bf16 ? 3 : 7)) == 0 &&
      !norms_force_scalar()) {
#define LAUNCH_RMS_V2(WT, T5)                                                  \
  hipLaunchKernelGGL((rms_norm_fwd_v2_kernel<WT, T5>), grid, block, 0,         \
                     stream,                                                   \
                     reinterpret_cast<const unsigned int*>(x.data_ptr()),      \
                     reinterpret_cast<const WT*>(wc.data_ptr()),               \
                     reinterpret_cast<unsigned int*>(y.data_ptr()),            \
                     inv_rms.data_ptr<float>(), n_rows, d / 2, (float)eps,     \
                     w_half)
    if (w_bf16) {
      if (t5_style) LAUNCH_RMS_V2(__hip_bfloat16, true);
      else LAUNCH_RMS_V2(__hip_bfloat16, false);
    } else {
      if (t5_style) LAUNCH_RMS_V2(float, true);
      else LAUNCH_RMS_V2(float, false);
    }
#undef LAUNCH_RMS_V2
    return {y, inv_rms};
  }

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
// ATen). Chunk count is adaptive like colsum's (a fixed 32 chunks left
// only 12k threads at d=384 — 7.3 us for a 768 KB read): stage 1 folds
// n_part rows ch-fold with ch*d threads, stage 2 finishes ch rows with
// one 64-lane wave per column. Fixed-order loops: deterministic.

__global__ void rms_dw_reduce1_kernel(const float* __restrict__ part,
                                      float* __restrict__ tmp,
                                      int n_part, int d, int ch) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= (int64_t)ch * d) return;
  int j = (int)(tid % d);
  int chunk = (int)(tid / d);
  float acc = 0.f;
  for (int r = chunk; r < n_part; r += ch) acc += part[(int64_t)r * d + j];
  tmp[(int64_t)chunk * d + j] = acc;
}

template <typename WT>
__global__ void rms_dw_reduce2_kernel(const float* __restrict__ tmp,
                                      WT* __restrict__ out, int d, int ch) {
  int j = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (j >= d) return;
  float acc = 0.f;
  for (int r = lane; r < ch; r += WAVE) acc += tmp[(int64_t)r * d + j];
  acc = wave_sum(acc);
  if (lane == 0) out[j] = from_f32<WT>(acc);
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
  // Block cap: 1024 blocks = 4096 waves (4 blocks/CU) hide the per-row
  // wave_sum serialization at the big [15616, 384] shapes — same-box A/B
  // vs the old 512 cap: 48.3k -> 49.5k samples/s on the TIGER bench
  // (2048 flat vs 1024). The dw partial matrix grows with the cap but
  // the adaptive two-stage reduce absorbs it. GENREC_RMS_BWD_CAP
  // overrides for A/B.
  static const int cap = [] {
    const char* e = getenv("GENREC_RMS_BWD_CAP");
    int v = e ? atoi(e) : 1024;
    return (v >= 64 && v <= 4096) ? v : 1024;
  }();
  int n_blocks = std::min(grid_for_rows(n_rows, 4), cap);
  dim3 grid(n_blocks);
  auto dw = torch::empty({n_blocks, (int64_t)d},
                         x.options().dtype(torch::kFloat32));
  size_t smem = 4 * (size_t)d * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();

  bool vec2_ok = x.scalar_type() == torch::kBFloat16 &&
                 dy.scalar_type() == torch::kBFloat16 && (d & 1) == 0 &&
                 ptr_aligned4(x.data_ptr()) && ptr_aligned4(dy.data_ptr()) &&
                 ptr_aligned4(dx.data_ptr()) &&
                 (reinterpret_cast<uintptr_t>(wc.data_ptr()) &
                  (w_bf16 ? 3 : 7)) == 0 &&
                 !norms_force_scalar();
  if (vec2_ok) {
#define LAUNCH_RMSB_V2(WT)                                                     \
  hipLaunchKernelGGL((rms_norm_bwd_v2_kernel<WT>), grid, block, smem, stream,  \
                     reinterpret_cast<const unsigned int*>(dy.data_ptr()),     \
                     reinterpret_cast<const unsigned int*>(x.data_ptr()),      \
                     reinterpret_cast<const WT*>(wc.data_ptr()),               \
                     inv_rms.data_ptr<float>(),                                \
                     reinterpret_cast<unsigned int*>(dx.data_ptr()),           \
                     dw.data_ptr<float>(), n_rows, d / 2)
    if (w_bf16) LAUNCH_RMSB_V2(__hip_bfloat16);
    else LAUNCH_RMSB_V2(float);
#undef LAUNCH_RMSB_V2
  }

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

  if (vec2_ok) {
    // already launched above
  } else if (x.scalar_type() == torch::kFloat32) {
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
  int ch = 32;
  while ((int64_t)ch * d < 131072 && ch < 512 && ch * 4 < n_blocks) ch *= 2;
  auto dw_tmp = torch::empty({ch, (int64_t)d},
                             x.options().dtype(torch::kFloat32));
  dim3 rblock(256);
  dim3 rgrid1((unsigned)(((int64_t)ch * d + 255) / 256));
  hipLaunchKernelGGL(rms_dw_reduce1_kernel, rgrid1, rblock, 0, stream,
                     dw.data_ptr<float>(), dw_tmp.data_ptr<float>(),
                     n_blocks, d, ch);
  dim3 rgrid2((unsigned)((d + 3) / 4));  // 4 waves/block, one wave per col
  if (w_bf16) {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<__hip_bfloat16>), rgrid2,
                       rblock, 0, stream, dw_tmp.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dw_out.data_ptr()),
                       d, ch);
  } else {
    hipLaunchKernelGGL((rms_dw_reduce2_kernel<float>), rgrid2, rblock, 0,
                       stream, dw_tmp.data_ptr<float>(),
                       dw_out.data_ptr<float>(), d, ch);
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
