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
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
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
  const int blocks = (int)std::min<int64_t>(8192, (n + threads - 1) / threads);
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
