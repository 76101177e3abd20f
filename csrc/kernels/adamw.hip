// Fused flat AdamW step for graph-captured training.
//
// torch's capturable foreach AdamW with a tensor LR falls off the foreach
// fast route (0-dim operands break `can_use_fast_route`), launching ~4
// tiny elementwise kernels PER PARAMETER per step — measured ~185
// DivFunctor + ~130 int-add dispatches and >1.5 ms of a TIGER step.
// Here the optimizer state lives in FLAT buffers (masters fp32, moments
// fp32, params bf16, grads bf16 — the same flat grad buffer the RCCL
// all-reduce uses), so the whole update is ONE bandwidth-bound kernel:
// read grad+master+m+v, write master+m+v+bf16 param (~28 B/element,
// ~150 MB for TIGER => ~20 us at HBM3E rates).
//
// Semantics match torch.optim.AdamW (decoupled wd, bias-corrected
// moments, eps OUTSIDE the bias-corrected sqrt — adamw.py single-tensor
// path). lr / clip-scale / step are DEVICE scalars so LR schedules and
// gradient clipping keep working across hipGraph replays.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

// Element-quad layout (float4 / 8-byte-bf16x4 loads): the scalar version
// moved 28 B/element through seven dword-or-narrower streams and measured
// 70 us/step on TIGER (~3.5x the ~20 us HBM roofline). The last partial
// quad falls back to per-element accesses in the same kernel.
union ADW_BF4 {
  uint2 u;
  __hip_bfloat16 e[4];
};

__global__ void fused_adamw_kernel(
    float* __restrict__ master, const __hip_bfloat16* __restrict__ grad,
    float* __restrict__ m, float* __restrict__ v,
    __hip_bfloat16* __restrict__ out_p,
    const float* __restrict__ lr_p, const float* __restrict__ scale_p,
    const int* __restrict__ step_p,
    float beta1, float beta2, float eps, float wd, int64_t n) {
  const float lr = lr_p[0];
  const float gscale = scale_p ? scale_p[0] : 1.0f;
  const float st = (float)step_p[0];
  const float bc1 = 1.0f - powf(beta1, st);
  const float bc2 = 1.0f - powf(beta2, st);
  const float decay = 1.0f - lr * wd;
  const int64_t nq = (n + 3) >> 2;
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < nq;
       q += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i0 = q * 4;
    if (i0 + 3 < n) {
      ADW_BF4 g4;
      g4.u = reinterpret_cast<const uint2*>(grad)[q];
      float4 p4 = reinterpret_cast<const float4*>(master)[q];
      float4 m4 = reinterpret_cast<const float4*>(m)[q];
      float4 v4 = reinterpret_cast<const float4*>(v)[q];
      float* pp = &p4.x;
      float* mp = &m4.x;
      float* vp = &v4.x;
      ADW_BF4 o4;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float g = gscale * to_f32(g4.e[k]);
        float p = pp[k] * decay;
        float mi = beta1 * mp[k] + (1.0f - beta1) * g;
        float vi = beta2 * vp[k] + (1.0f - beta2) * g * g;
        mp[k] = mi;
        vp[k] = vi;
        p -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
        pp[k] = p;
        o4.e[k] = from_f32<__hip_bfloat16>(p);
      }
      reinterpret_cast<float4*>(m)[q] = m4;
      reinterpret_cast<float4*>(v)[q] = v4;
      reinterpret_cast<float4*>(master)[q] = p4;
      reinterpret_cast<uint2*>(out_p)[q] = o4.u;
    } else {
      for (int64_t i = i0; i < n; ++i) {
        float g = gscale * to_f32(grad[i]);
        float p = master[i] * decay;
        float mi = beta1 * m[i] + (1.0f - beta1) * g;
        float vi = beta2 * v[i] + (1.0f - beta2) * g * g;
        m[i] = mi;
        v[i] = vi;
        p -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
        master[i] = p;
        out_p[i] = from_f32<__hip_bfloat16>(p);
      }
    }
  }
}

__global__ void step_inc_kernel(int* step_p) {
  if (threadIdx.x == 0) step_p[0] += 1;
}

void fused_adamw(torch::Tensor master, torch::Tensor grad, torch::Tensor m,
                 torch::Tensor v, torch::Tensor out_p, torch::Tensor lr,
                 torch::Tensor scale, torch::Tensor step, double beta1,
                 double beta2, double eps, double weight_decay) {
  TORCH_CHECK(master.is_cuda() && master.dtype() == torch::kFloat &&
                  master.is_contiguous(),
              "master must be contiguous fp32 cuda");
  TORCH_CHECK(grad.dtype() == torch::kBFloat16 && grad.is_contiguous());
  TORCH_CHECK(m.dtype() == torch::kFloat && v.dtype() == torch::kFloat);
  TORCH_CHECK(out_p.dtype() == torch::kBFloat16 && out_p.is_contiguous());
  TORCH_CHECK(step.dtype() == torch::kInt && step.numel() == 1);
  TORCH_CHECK(lr.dtype() == torch::kFloat && lr.numel() == 1);
  const int64_t n = master.numel();
  TORCH_CHECK(grad.numel() == n && m.numel() == n && v.numel() == n &&
              out_p.numel() == n);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(step_inc_kernel, dim3(1), dim3(64), 0, stream,
                     step.data_ptr<int>());
  const int threads = 256;
  const int64_t nq = (n + 3) / 4;  // one thread per element quad
  const int blocks =
      (int)std::min<int64_t>(8192, (nq + threads - 1) / threads);
  const float* scale_ptr =
      scale.defined() && scale.numel() ? scale.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(
      fused_adamw_kernel, dim3(blocks), dim3(threads), 0, stream,
      master.data_ptr<float>(),
      reinterpret_cast<const __hip_bfloat16*>(grad.data_ptr()),
      m.data_ptr<float>(), v.data_ptr<float>(),
      reinterpret_cast<__hip_bfloat16*>(out_p.data_ptr()),
      lr.data_ptr<float>(), scale_ptr, step.data_ptr<int>(), (float)beta1,
      (float)beta2, (float)eps, (float)weight_decay, n);
}

}  // namespace genrec
