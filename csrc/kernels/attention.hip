// Fused attention kernels (K1 SASRec / K4 HSTU / K13 T5 — SURVEY.md §2.4),
// forward + backward, gfx950.
//
// One fused kernel covers the zoo's three attention families via runtime
// flags: score activation (softmax | SiLU), optional pre-mask bias
// ([H,Lq,Lk] or [B,H,Lq,Lk]), key-padding mask at -1e9, causal mask at
// -1e9, additive (-inf style) [Lq,Lk] mask, POST-softmax query mask
// (SASRec's load-bearing quirk, sasrec.py:228-233), and dropout.
//
// Geometry (v1): sequences in this model zoo are tiny (Lq,Lk <= 64:
// TIGER enc 61, SASRec/HSTU 50, TIGER dec 4) — one 256-thread workgroup
// (4 waves) owns one (batch, head) pair and keeps Q, K^T, V, P entirely in
// LDS; scores never touch HBM on the forward pass (the eager reference
// materializes [B,H,L,L] in HBM three times). Each wave processes one
// q-row: lane j owns score column j (Lk<=64 = wave width), so the row
// softmax is a wave-level shuffle reduction; the PV pass re-reads P from
// LDS with lane t owning output column t. All LDS reads are conflict-free
// by construction (per-lane-consecutive or broadcast addresses).
// P (softmax) or raw scores (SiLU) are saved for backward.
//
// Larger Lq is tiled by grid.y; Lk > 64 falls back to the eager composition
// host-side (no model in the zoo needs it yet).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

constexpr int MAXL = 64;   // max Lk per block (wave width)
constexpr int NWAVES = 8;  // waves per workgroup (512 threads)
constexpr int MAXD = 64;   // max head dim
constexpr float NEG_BIG_F = -1e9f;

#define IDX4(b, h, i, j, H, I, J) \
  ((((int64_t)(b) * (H) + (h)) * (I) + (i)) * (J) + (j))

template <typename T>
__global__ void attn_fwd_kernel(
    const T* __restrict__ q,        // [B,H,Lq,D]
    const T* __restrict__ k,        // [B,H,Lk,D]
    const T* __restrict__ v,        // [B,H,Lk,D]
    const float* __restrict__ bias,       // null | [H,Lq,Lk] | [B,H,Lq,Lk]
    const bool* __restrict__ key_pad,     // null | [B,Lk]
    const float* __restrict__ add_mask,   // null | [Lq,Lk]
    const float* __restrict__ query_mask, // null | [B,Lq]
    T* __restrict__ out,            // [B,H,Lq,D]
    float* __restrict__ p_saved,    // [B,H,Lq,Lk] (softmax: P, silu: S)
    unsigned char* __restrict__ drop_mask,  // null | [B,H,Lq,Lk]
    const unsigned int* __restrict__ seed_dev,  // null | device seed counter
    int B, int H, int Lq, int Lk, int D,
    float scale, int bias_dim, bool causal, int act,
    float dropout_p, unsigned int seed, int q_tile) {
  if (seed_dev) seed += *seed_dev;  // hipGraph-replay-varying dropout
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.y * q_tile;
  const int q1 = min(q0 + q_tile, Lq);

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* kt = reinterpret_cast<float*>(smem_raw);          // [D][MAXL+1]
  float* vv = kt + MAXD * (MAXL + 1);                      // [MAXL][MAXD]
  float* qq = vv + MAXL * MAXD;                            // [q_tile][MAXD]
  float* pp = qq + q_tile * MAXD;                          // [NWAVES][MAXL+1] per-wave P rows

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  // stage K^T and V (fp32) — coalesced global reads
  for (int idx = tid; idx < Lk * D; idx += blockDim.x) {
    int j = idx / D, t = idx % D;
    float val = to_f32(k[IDX4(b, h, j, t, H, Lk, D)]);
    kt[t * (MAXL + 1) + j] = val;
    vv[j * MAXD + t] = to_f32(v[IDX4(b, h, j, t, H, Lk, D)]);
  }
  for (int idx = tid; idx < (q1 - q0) * D; idx += blockDim.x) {
    int i = idx / D, t = idx % D;
    qq[i * MAXD + t] = to_f32(q[IDX4(b, h, q0 + i, t, H, Lq, D)]);
  }
  __syncthreads();

  const float inv_keep = dropout_p > 0.f ? 1.0f / (1.0f - dropout_p) : 1.0f;

  for (int i = q0 + wid; i < q1; i += NWAVES) {
    // ----- scores: lane owns column `lane`
    float s = 0.f;
    const bool col_ok = lane < Lk;
    if (col_ok) {
      const float* ktc = kt + lane;  // column `lane`
      const float* qr = qq + (i - q0) * MAXD;
      for (int t = 0; t < D; ++t) s += qr[t] * ktc[t * (MAXL + 1)];
      s *= scale;
      if (bias_dim == 3) s += bias[((int64_t)h * Lq + i) * Lk + lane];
      else if (bias_dim == 4) s += bias[IDX4(b, h, i, lane, H, Lq, Lk)];
      if (causal && lane > i) s = NEG_BIG_F;
      if (key_pad && key_pad[(int64_t)b * Lk + lane]) s = NEG_BIG_F;
      if (add_mask) s += add_mask[(int64_t)i * Lk + lane];
    }

    float p;
    if (act == 0) {  // softmax
      float sm = col_ok ? s : -INFINITY;
      float m = wave_max(sm);
      float e = col_ok ? __expf(sm - m) : 0.f;
      float denom = wave_sum(e);
      p = e / denom;
    } else {  // silu
      p = col_ok ? s * sigmoidf_dev(s) : 0.f;
    }
    if (col_ok) {
      p_saved[IDX4(b, h, i, lane, H, Lq, Lk)] = (act == 0) ? p : s;
    }
    // post-softmax query mask (SASRec) then dropout
    if (query_mask) p *= query_mask[(int64_t)b * Lq + i];
    if (dropout_p > 0.f && col_ok) {
      unsigned long long gidx = IDX4(b, h, i, lane, H, Lq, Lk);
      bool keep = (hash_rng(seed, gidx) & 0xFFFFFF) >=
                  (unsigned int)(dropout_p * 16777216.0f);
      drop_mask[gidx] = keep;
      p = keep ? p * inv_keep : 0.f;
    }
    float* prow = pp + wid * (MAXL + 1);
    if (lane < MAXL) prow[lane] = col_ok ? p : 0.f;
    // no cross-wave sharing of prow; wave-internal visibility is immediate
    __builtin_amdgcn_wave_barrier();

    // ----- PV: lane owns output column `lane`
    if (lane < D) {
      float acc = 0.f;
      const float* vcol = vv + lane;
      for (int j = 0; j < Lk; ++j) acc += prow[j] * vcol[j * MAXD];
      out[IDX4(b, h, i, lane, H, Lq, D)] = from_f32<T>(acc);
    }
  }
}

// Backward. Same geometry: one block per (b,h), q-tiled by grid.y for the
// dP/dS/dQ phases; the dK/dV phase needs all Lq rows' dS, so when Lq fits
// one tile (the common case) everything runs in one pass; otherwise dS is
// read back from the global scratch written in phase 1.
template <typename T>
__global__ void attn_bwd_kernel(
    const T* __restrict__ dout,     // [B,H,Lq,D]
    const T* __restrict__ q,
    const T* __restrict__ k,
    const T* __restrict__ v,
    const float* __restrict__ p_saved,   // P (softmax) or S (silu)
    const float* __restrict__ query_mask,
    const unsigned char* __restrict__ drop_mask,
    T* __restrict__ dq, T* __restrict__ dk, T* __restrict__ dv,
    float* __restrict__ ds_out,     // null | [B,H,Lq,Lk] (dS for bias grad)
    int B, int H, int Lq, int Lk, int D,
    float scale, int act, float dropout_p, unsigned int seed) {
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* vt = reinterpret_cast<float*>(smem_raw);          // [D][MAXL+1]
  float* kk = vt + MAXD * (MAXL + 1);                      // [MAXL][MAXD]
  float* qs = kk + MAXL * MAXD;                            // [MAXL][MAXD]
  float* dos = qs + MAXL * MAXD;                           // [MAXL][MAXD]
  float* ps = dos + MAXL * MAXD;                           // [MAXL][MAXL+1]
  float* dss = ps + MAXL * (MAXL + 1);                     // [MAXL][MAXL+1]
  float* am = dss + MAXL * (MAXL + 1);                     // [MAXL][MAXL+1]

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const float inv_keep = dropout_p > 0.f ? 1.0f / (1.0f - dropout_p) : 1.0f;

  for (int idx = tid; idx < Lk * D; idx += blockDim.x) {
    int j = idx / D, t = idx % D;
    vt[t * (MAXL + 1) + j] = to_f32(v[IDX4(b, h, j, t, H, Lk, D)]);
    kk[j * MAXD + t] = to_f32(k[IDX4(b, h, j, t, H, Lk, D)]);
  }
  for (int idx = tid; idx < Lq * D; idx += blockDim.x) {
    int i = idx / D, t = idx % D;
    qs[i * MAXD + t] = to_f32(q[IDX4(b, h, i, t, H, Lq, D)]);
    dos[i * MAXD + t] = to_f32(dout[IDX4(b, h, i, t, H, Lq, D)]);
  }
  // stage P and the combined post-softmax multiplier (query mask x
  // dropout keep/scale) in one coalesced pass — the inner loops then only
  // touch LDS (a per-element global drop_mask read inside phase 3 was the
  // dominant cost of this kernel: ~500-cycle latency per (i,j) pair).
  const bool has_am = (query_mask != nullptr) || (dropout_p > 0.f);
  for (int idx = tid; idx < Lq * Lk; idx += blockDim.x) {
    int i = idx / Lk, j = idx % Lk;
    ps[i * (MAXL + 1) + j] = p_saved[IDX4(b, h, i, j, H, Lq, Lk)];
    if (has_am) {
      float m = query_mask ? query_mask[(int64_t)b * Lq + i] : 1.f;
      if (dropout_p > 0.f) {
        m *= drop_mask[IDX4(b, h, i, j, H, Lq, Lk)] ? inv_keep : 0.f;
      }
      am[i * (MAXL + 1) + j] = m;
    }
  }
  __syncthreads();

  // ---- phase 1: dS rows (wave per q-row, lane owns column)
  for (int i = wid; i < Lq; i += NWAVES) {
    const bool col_ok = lane < Lk;
    float dp = 0.f;
    if (col_ok) {
      const float* vtc = vt + lane;
      const float* dor = dos + i * MAXD;
      for (int t = 0; t < D; ++t) dp += dor[t] * vtc[t * (MAXL + 1)];
    }
    float ds = 0.f;
    if (act == 0) {
      float pv = col_ok ? ps[i * (MAXL + 1) + lane] : 0.f;  // pre-mask P
      float da = dp;
      if (has_am && col_ok) da *= am[i * (MAXL + 1) + lane];
      float dot = wave_sum(da * pv);
      ds = pv * (da - dot);
    } else {
      // silu: p_saved holds S; silu'(s) = sig(s)*(1 + s*(1-sig(s)))
      float s = col_ok ? ps[i * (MAXL + 1) + lane] : 0.f;
      float sg = sigmoidf_dev(s);
      ds = col_ok ? dp * sg * (1.f + s * (1.f - sg)) : 0.f;
    }
    if (lane < MAXL) dss[i * (MAXL + 1) + lane] = col_ok ? ds : 0.f;
    if (ds_out && col_ok) ds_out[IDX4(b, h, i, lane, H, Lq, Lk)] = ds;
  }
  __syncthreads();

  // ---- phase 2: dQ rows (lane owns head-dim column t)
  for (int i = wid; i < Lq; i += NWAVES) {
    if (lane < D) {
      float acc = 0.f;
      const float* dsr = dss + i * (MAXL + 1);
      const float* kc = kk + lane;
      for (int j = 0; j < Lk; ++j) acc += dsr[j] * kc[j * MAXD];
      dq[IDX4(b, h, i, lane, H, Lq, D)] = from_f32<T>(acc * scale);
    }
  }

  // ---- phase 3: dK, dV rows (wave per k-row j, lane owns column t)
  for (int j = wid; j < Lk; j += NWAVES) {
    if (lane < D) {
      float acc_k = 0.f, acc_v = 0.f;
      for (int i = 0; i < Lq; ++i) {
        float ds = dss[i * (MAXL + 1) + j];
        acc_k += ds * qs[i * MAXD + lane];
        float a = ps[i * (MAXL + 1) + j];
        if (act == 0) {
          if (has_am) a *= am[i * (MAXL + 1) + j];
        } else {
          a = a * sigmoidf_dev(a);  // A = silu(S)
        }
        acc_v += a * dos[i * MAXD + lane];
      }
      dk[IDX4(b, h, j, lane, H, Lk, D)] = from_f32<T>(acc_k * scale);
      dv[IDX4(b, h, j, lane, H, Lk, D)] = from_f32<T>(acc_v);
    }
  }
}

// ---------------------------------------------------------------- host

std::vector<torch::Tensor> attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> key_pad,
    c10::optional<torch::Tensor> add_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, bool causal, int64_t act, double dropout_p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4);
  const int B = q.size(0), H = q.size(1), Lq = q.size(2), D = q.size(3);
  const int Lk = k.size(2);
  TORCH_CHECK(Lk <= MAXL && D <= MAXD,
              "attn kernel v1 supports Lk<=64, D<=64; got Lk=", Lk, " D=", D);
  TORCH_CHECK(!causal || Lq == Lk);

  auto out = torch::empty_like(q);
  auto p_saved = torch::empty({B, H, Lq, Lk},
                              q.options().dtype(torch::kFloat32));
  torch::Tensor dmask;
  if (dropout_p > 0) {
    dmask = torch::empty({B, H, Lq, Lk}, q.options().dtype(torch::kUInt8));
  } else {
    dmask = torch::empty({0}, q.options().dtype(torch::kUInt8));
  }

  torch::Tensor bias_f;
  int bias_dim = 0;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    bias_dim = bias_f.dim();
    TORCH_CHECK(bias_dim == 3 || bias_dim == 4);
  }
  torch::Tensor am_f;
  if (add_mask.has_value()) {
    TORCH_CHECK(add_mask->dim() == 2);
    am_f = add_mask->to(torch::kFloat32).contiguous();
  }
  torch::Tensor qm_f;
  if (query_mask.has_value()) qm_f = query_mask->to(torch::kFloat32).contiguous();

  const int q_tile = std::min(Lq, MAXL);
  dim3 block(64 * NWAVES);
  dim3 grid(B * H, (Lq + q_tile - 1) / q_tile);
  size_t smem = (MAXD * (MAXL + 1) + MAXL * MAXD + q_tile * MAXD +
                 NWAVES * (MAXL + 1)) * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_ATTN_FWD(T)                                                     \
  hipLaunchKernelGGL((attn_fwd_kernel<T>), grid, block, smem, stream,          \
      reinterpret_cast<const T*>(q.data_ptr()),                                \
      reinterpret_cast<const T*>(k.data_ptr()),                                \
      reinterpret_cast<const T*>(v.data_ptr()),                                \
      bias_dim ? bias_f.data_ptr<float>() : nullptr,                           \
      key_pad.has_value() ? key_pad->data_ptr<bool>() : nullptr,               \
      add_mask.has_value() ? am_f.data_ptr<float>() : nullptr,                 \
      query_mask.has_value() ? qm_f.data_ptr<float>() : nullptr,               \
      reinterpret_cast<T*>(out.data_ptr()), p_saved.data_ptr<float>(),         \
      dropout_p > 0 ? dmask.data_ptr<unsigned char>() : nullptr,               \
      seed_dev.has_value()                                                     \
          ? reinterpret_cast<const unsigned int*>(seed_dev->data_ptr())        \
          : nullptr,                                                           \
      B, H, Lq, Lk, D, (float)scale, bias_dim, causal, (int)act,               \
      (float)dropout_p, (unsigned int)seed, q_tile)

  if (q.scalar_type() == torch::kFloat32) LAUNCH_ATTN_FWD(float);
  else if (q.scalar_type() == torch::kBFloat16) LAUNCH_ATTN_FWD(__hip_bfloat16);
  else TORCH_CHECK(false, "attn_fwd: unsupported dtype");
#undef LAUNCH_ATTN_FWD

  // p_saved holds P for softmax, raw scores for silu; dmask is empty
  // unless dropout was active.
  return {out, p_saved, dmask};
}

std::vector<torch::Tensor> attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor p_saved, torch::Tensor drop_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, int64_t act, double dropout_p, int64_t seed,
    bool bias_grad, int64_t bias_dim) {
  TORCH_CHECK(dropout_p <= 0 || drop_mask.numel() > 0,
              "attn_bwd: dropout active but no mask saved");
  const int B = q.size(0), H = q.size(1), Lq = q.size(2), D = q.size(3);
  const int Lk = k.size(2);
  TORCH_CHECK(Lq <= MAXL && Lk <= MAXL && D <= MAXD,
              "attn bwd v1 supports Lq,Lk<=64");

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  // bias gradient: kernel writes per-(b,h) dS to scratch (no atomics —
  // a [H,Lq,Lk] atomic target serializes all B blocks per location); the
  // batch reduction for 3-dim biases is one ATen sum afterwards.
  torch::Tensor ds_scratch;
  float* ds_ptr = nullptr;
  if (bias_grad) {
    ds_scratch = torch::empty({B, H, Lq, Lk},
                              q.options().dtype(torch::kFloat32));
    ds_ptr = ds_scratch.data_ptr<float>();
  }
  torch::Tensor qm_f;
  if (query_mask.has_value()) qm_f = query_mask->to(torch::kFloat32).contiguous();

  dim3 block(64 * NWAVES);
  dim3 grid(B * H);
  size_t smem = (MAXD * (MAXL + 1) + 3 * MAXL * MAXD +
                 3 * MAXL * (MAXL + 1)) * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_ATTN_BWD(T)                                                     \
  hipLaunchKernelGGL((attn_bwd_kernel<T>), grid, block, smem, stream,          \
      reinterpret_cast<const T*>(dout.data_ptr()),                             \
      reinterpret_cast<const T*>(q.data_ptr()),                                \
      reinterpret_cast<const T*>(k.data_ptr()),                                \
      reinterpret_cast<const T*>(v.data_ptr()),                                \
      p_saved.data_ptr<float>(),                                               \
      query_mask.has_value() ? qm_f.data_ptr<float>() : nullptr,               \
      dropout_p > 0 ? drop_mask.data_ptr<unsigned char>() : nullptr,           \
      reinterpret_cast<T*>(dq.data_ptr()),                                     \
      reinterpret_cast<T*>(dk.data_ptr()),                                     \
      reinterpret_cast<T*>(dv.data_ptr()), ds_ptr,                             \
      B, H, Lq, Lk, D, (float)scale, (int)act, (float)dropout_p,               \
      (unsigned int)seed)

  if (q.scalar_type() == torch::kFloat32) LAUNCH_ATTN_BWD(float);
  else if (q.scalar_type() == torch::kBFloat16) LAUNCH_ATTN_BWD(__hip_bfloat16);
  else TORCH_CHECK(false, "attn_bwd: unsupported dtype");
#undef LAUNCH_ATTN_BWD

  torch::Tensor dbias;
  if (bias_grad) {
    dbias = (bias_dim == 3) ? ds_scratch.sum(0) : ds_scratch;
  } else {
    dbias = torch::empty({0}, q.options().dtype(torch::kFloat32));
  }
  return {dq, dk, dv, dbias};
}

}  // namespace genrec
