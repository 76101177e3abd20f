// HSTU fused pointwise attention (K4+K5+K6 — SURVEY.md §2.4), gfx950.
//
// The full HSTU signature op in one MFMA kernel: S = Q K^T + position bias
// + temporal bias, causal/key-pad masks at -1e9, SiLU scores (no softmax),
// O = SiLU(S) V. The two bias terms are computed IN the epilogue — the
// T5-style position bucket table is a precomputed [L,L] index (input-
// independent), the temporal bucket is ln2-bucketed |ts_i - ts_j| computed
// per element — so the reference's [B,H,L,L] bias materialization
// (hstu.py:283-409: ~6 large elementwise + 2 gathers per layer) never
// happens.
//
// Backward: dP = dO V^T (MFMA), dS = dP * SiLU'(S), bias-table gradients
// are reduced per-block into LDS histograms (one head per block, <=
// n_pos + n_time floats) and flushed with one global atomicAdd per bucket,
// then dQ/dK/dV run as MFMA in the same launch.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

constexpr int HTILE = 64;
constexpr float HNEG = -1e9f;

typedef __attribute__((ext_vector_type(8))) short short8vh;
typedef __attribute__((ext_vector_type(4))) float float4vh;

#define HIDX4(b, h, i, j, H, I, J) \
  ((((int64_t)(b) * (H) + (h)) * (I) + (i)) * (J) + (j))

__device__ __forceinline__ int hswz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

// 8x8 in-register block transpose (see attention_mfma.hip::xpose8x8)
__device__ __forceinline__ void hxpose8x8(short (&vals)[8], int g) {
#pragma unroll
  for (int m = 1; m < 8; m <<= 1) {
    short nv[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      int t = __shfl_xor((int)vals[e ^ m], 8 * m, 64);
      nv[e] = ((e & m) != (g & m)) ? (short)t : vals[e];
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) vals[e] = nv[e];
  }
}

__device__ __forceinline__ short8vh hfrag(const char* base, int row0, int k0,
                                          int lane) {
  int row = row0 + (lane & 15);
  int byte = (k0 + ((lane >> 4) << 3)) * 2;
  return *reinterpret_cast<const short8vh*>(base + hswz(row, byte));
}

__device__ __forceinline__ int time_bucket(long long ti, long long tj,
                                           int n_buckets) {
  long long diff = ti - tj;
  if (diff < 0) diff = -diff;
  if (diff < 1) diff = 1;
  int b = (int)(__logf((float)diff) / 0.693f);
  return min(max(b, 0), n_buckets - 1);
}

template <typename BT>
__global__ void __launch_bounds__(256)
hstu_attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q,     // [B,H,L,D]
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const int* __restrict__ pos_bucket,       // [L,L]
    const BT* __restrict__ pos_table,         // [n_pos, H]
    const BT* __restrict__ time_table,        // null | [n_time, H]
    const long long* __restrict__ ts,         // null | [B,L]
    const bool* __restrict__ key_pad,         // null | [B,L]
    __hip_bfloat16* __restrict__ out,
    float* __restrict__ s_saved,              // [B,H,L,L] post-mask scores
    int B, int H, int L, int D, int n_time) {
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* qs = smem;
  char* ks = qs + HTILE * 128;
  char* vt = ks + HTILE * 128;
  char* ps = vt + HTILE * 128;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  for (int idx = tid; idx < HTILE * (HTILE / 8); idx += blockDim.x) {
    int row = idx / (HTILE / 8);
    int d0 = (idx % (HTILE / 8)) * 8;
    short8vh val = {};
    if (row < L && d0 < D) {
      val = *reinterpret_cast<const short8vh*>(
          &q[HIDX4(b, h, row, d0, H, L, D)]);
    }
    *reinterpret_cast<short8vh*>(qs + hswz(row, d0 * 2)) = val;
    short8vh val2 = {};
    if (row < L && d0 < D) {
      val2 = *reinterpret_cast<const short8vh*>(
          &k[HIDX4(b, h, row, d0, H, L, D)]);
    }
    *reinterpret_cast<short8vh*>(ks + hswz(row, d0 * 2)) = val2;
    // V^T: coalesced natural-row load + in-register 8x8 transpose
    short8vh vv = {};
    if (row < L && d0 < D) {
      vv = *reinterpret_cast<const short8vh*>(
          &v[HIDX4(b, h, row, d0, H, L, D)]);
    }
    {
      const int g = (lane >> 3) & 7;
      short tv[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) tv[e] = vv[e];
      hxpose8x8(tv, g);
      short8vh pack;
#pragma unroll
      for (int e = 0; e < 8; ++e) pack[e] = tv[e];
      *reinterpret_cast<short8vh*>(vt + hswz(d0 + g, (row & ~7) * 2)) = pack;
    }
  }
  __syncthreads();

  const int strip = wid * 16;
  float4vh acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    for (int kk = 0; kk < D; kk += 32) {
      acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          hfrag(qs, strip, kk, lane), hfrag(ks, f * 16, kk, lane), acc[f],
          0, 0, 0);
    }
  }

  const int col_base = lane & 15;
  const int row_grp = (lane >> 4) << 2;
#pragma unroll
  for (int f = 0; f < 4; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = strip + row_grp + r;
      int j = f * 16 + col_base;
      float s = acc[f][r];
      float p = 0.f;
      if (i < L && j < L) {
        s += to_f32(pos_table[(int64_t)pos_bucket[i * L + j] * H + h]);
        if (time_table) {
          int tb = time_bucket(ts[(int64_t)b * L + i],
                               ts[(int64_t)b * L + j], n_time);
          s += to_f32(time_table[(int64_t)tb * H + h]);
        }
        if (j > i) s = HNEG;                                     // causal
        if (key_pad && key_pad[(int64_t)b * L + j]) s = HNEG;    // pad
        s_saved[HIDX4(b, h, i, j, H, L, L)] = s;
        p = s * sigmoidf_dev(s);
      }
      *reinterpret_cast<__hip_bfloat16*>(ps + hswz(i, j * 2)) =
          __float2bfloat16(p);
    }
  }
  __builtin_amdgcn_wave_barrier();

  float4vh acc2[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
  const int nfrag_d = (D + 15) / 16;
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    if (f >= nfrag_d) break;
    for (int kk = 0; kk < HTILE; kk += 32) {
      acc2[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          hfrag(ps, strip, kk, lane), hfrag(vt, f * 16, kk, lane), acc2[f],
          0, 0, 0);
    }
  }
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    if (f >= nfrag_d) break;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = strip + row_grp + r;
      int d = f * 16 + col_base;
      if (i < L && d < D) {
        out[HIDX4(b, h, i, d, H, L, D)] = __float2bfloat16(acc2[f][r]);
      }
    }
  }
}

template <typename BT>
__global__ void __launch_bounds__(256)
hstu_attn_bwd_kernel(
    const __hip_bfloat16* __restrict__ dout,
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const float* __restrict__ s_saved,        // [B,H,L,L]
    const int* __restrict__ pos_bucket,       // [L,L]
    const long long* __restrict__ ts,         // null | [B,L]
    __hip_bfloat16* __restrict__ dq_out,
    __hip_bfloat16* __restrict__ dk_out,
    __hip_bfloat16* __restrict__ dv_out,
    float* __restrict__ dpos,                 // [n_pos, H] fp32 accum
    float* __restrict__ dtime,                // null | [n_time, H]
    int B, int H, int L, int D, int n_pos, int n_time) {
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* dos = smem;
  char* vs = dos + HTILE * 128;
  char* kt = vs + HTILE * 128;
  char* qt = kt + HTILE * 128;
  char* dot = qt + HTILE * 128;
  char* dsn = dot + HTILE * 128;
  char* dst = dsn + HTILE * 128;
  char* adt = dst + HTILE * 128;
  float* hist = reinterpret_cast<float*>(adt + HTILE * 128);  // [n_pos+n_time]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n_hist = n_pos + n_time;
  for (int i = tid; i < n_hist; i += blockDim.x) hist[i] = 0.f;

  for (int idx = tid; idx < HTILE * (HTILE / 8); idx += blockDim.x) {
    int row = idx / (HTILE / 8);
    int d0 = (idx % (HTILE / 8)) * 8;
    short8vh val = {};
    if (row < L && d0 < D) {
      val = *reinterpret_cast<const short8vh*>(
          &dout[HIDX4(b, h, row, d0, H, L, D)]);
    }
    *reinterpret_cast<short8vh*>(dos + hswz(row, d0 * 2)) = val;
    short8vh val2 = {};
    if (row < L && d0 < D) {
      val2 = *reinterpret_cast<const short8vh*>(
          &v[HIDX4(b, h, row, d0, H, L, D)]);
    }
    *reinterpret_cast<short8vh*>(vs + hswz(row, d0 * 2)) = val2;
    // K^T/Q^T/dO^T: coalesced natural-row b128 loads + in-register 8x8
    // transpose, one b128 LDS store per tensor
    short8vh kk8 = {}, qq8 = {}, dd8 = {};
    if (row < L && d0 < D) {
      kk8 = *reinterpret_cast<const short8vh*>(
          &k[HIDX4(b, h, row, d0, H, L, D)]);
      qq8 = *reinterpret_cast<const short8vh*>(
          &q[HIDX4(b, h, row, d0, H, L, D)]);
      dd8 = *reinterpret_cast<const short8vh*>(
          &dout[HIDX4(b, h, row, d0, H, L, D)]);
    }
    {
      const int g = (lane >> 3) & 7;
      const int jb = (row & ~7) * 2;
      short tk[8], tq2[8], td2[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        tk[e] = kk8[e]; tq2[e] = qq8[e]; td2[e] = dd8[e];
      }
      hxpose8x8(tk, g);
      hxpose8x8(tq2, g);
      hxpose8x8(td2, g);
      short8vh pk, pq, pd;
#pragma unroll
      for (int e = 0; e < 8; ++e) { pk[e] = tk[e]; pq[e] = tq2[e]; pd[e] = td2[e]; }
      *reinterpret_cast<short8vh*>(kt + hswz(d0 + g, jb)) = pk;
      *reinterpret_cast<short8vh*>(qt + hswz(d0 + g, jb)) = pq;
      *reinterpret_cast<short8vh*>(dot + hswz(d0 + g, jb)) = pd;
    }
  }
  __syncthreads();

  const int strip = wid * 16;
  float4vh acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    for (int kk = 0; kk < D; kk += 32) {
      acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          hfrag(dos, strip, kk, lane), hfrag(vs, f * 16, kk, lane), acc[f],
          0, 0, 0);
    }
  }

  const int col_base = lane & 15;
  const int row_grp = (lane >> 4) << 2;
  // transposed-tile stores pack a lane's 4 r-values (consecutive columns
  // of row j) into ONE aligned 8-byte ds_write — see attention_mfma.hip
  typedef __attribute__((ext_vector_type(4))) short short4vh;
  const int i0 = strip + row_grp;
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    short4vh dpack, apack;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = i0 + r;
      int j = f * 16 + col_base;
      float dsv = 0.f, adv = 0.f;
      if (i < L && j < L) {
        float s = s_saved[HIDX4(b, h, i, j, H, L, L)];
        float sg = sigmoidf_dev(s);
        dsv = acc[f][r] * sg * (1.f + s * (1.f - sg));
        adv = s * sg;  // A = silu(S)
        // bias-table gradients (dS at masked positions is ~0 since
        // SiLU'(-1e9) == 0 — matching autograd's masked_fill)
        if (dsv != 0.f) {
          atomicAdd(&hist[pos_bucket[i * L + j]], dsv);
          if (dtime) {
            int tb = time_bucket(ts[(int64_t)b * L + i],
                                 ts[(int64_t)b * L + j], n_time);
            atomicAdd(&hist[n_pos + tb], dsv);
          }
        }
      }
      __hip_bfloat16 dh = __float2bfloat16(dsv);
      __hip_bfloat16 ah = __float2bfloat16(adv);
      dpack[r] = *reinterpret_cast<short*>(&dh);
      apack[r] = *reinterpret_cast<short*>(&ah);
      *reinterpret_cast<__hip_bfloat16*>(dsn + hswz(i, j * 2)) = dh;
    }
    int j = f * 16 + col_base;
    *reinterpret_cast<short4vh*>(dst + hswz(j, i0 * 2)) = dpack;
    *reinterpret_cast<short4vh*>(adt + hswz(j, i0 * 2)) = apack;
  }
  __builtin_amdgcn_wave_barrier();

  const int nfrag_d = (D + 15) / 16;
  {  // dQ
    float4vh a4[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f >= nfrag_d) break;
      for (int kk = 0; kk < HTILE; kk += 32) {
        a4[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            hfrag(dsn, strip, kk, lane), hfrag(kt, f * 16, kk, lane), a4[f],
            0, 0, 0);
      }
    }
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f >= nfrag_d) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int i = strip + row_grp + r;
        int d = f * 16 + col_base;
        if (i < L && d < D) {
          dq_out[HIDX4(b, h, i, d, H, L, D)] = __float2bfloat16(a4[f][r]);
        }
      }
    }
  }
  __syncthreads();

  {  // dK, dV
    float4vh ak[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
    float4vh av[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f >= nfrag_d) break;
      for (int kk = 0; kk < HTILE; kk += 32) {
        ak[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            hfrag(dst, strip, kk, lane), hfrag(qt, f * 16, kk, lane), ak[f],
            0, 0, 0);
        av[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            hfrag(adt, strip, kk, lane), hfrag(dot, f * 16, kk, lane), av[f],
            0, 0, 0);
      }
    }
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      if (f >= nfrag_d) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int j = strip + row_grp + r;
        int d = f * 16 + col_base;
        if (j < L && d < D) {
          dk_out[HIDX4(b, h, j, d, H, L, D)] = __float2bfloat16(ak[f][r]);
          dv_out[HIDX4(b, h, j, d, H, L, D)] = __float2bfloat16(av[f][r]);
        }
      }
    }
  }

  // flush bias histograms: one global atomicAdd per bucket per block
  __syncthreads();
  for (int i = tid; i < n_hist; i += blockDim.x) {
    float val = hist[i];
    if (val != 0.f) {
      if (i < n_pos) atomicAdd(&dpos[(int64_t)i * H + h], val);
      else atomicAdd(&dtime[(int64_t)(i - n_pos) * H + h], val);
    }
  }
}

// ------------------------------------------------------------------ hosts

std::vector<torch::Tensor> hstu_attn_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor pos_bucket, torch::Tensor pos_table,
    c10::optional<torch::Tensor> time_table,
    c10::optional<torch::Tensor> timestamps,
    c10::optional<torch::Tensor> key_pad) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 && q.dim() == 4);
  const int B = q.size(0), H = q.size(1), L = q.size(2), D = q.size(3);
  TORCH_CHECK(L <= HTILE && D % 32 == 0 && D <= HTILE);
  TORCH_CHECK(pos_table.scalar_type() == torch::kBFloat16);
  auto out = torch::empty_like(q);
  auto s_saved = torch::empty({B, H, L, L},
                              q.options().dtype(torch::kFloat32));
  auto pb = pos_bucket.to(torch::kInt32).contiguous();
  torch::Tensor ts64;
  if (timestamps.has_value())
    ts64 = timestamps->to(torch::kInt64).contiguous();
  int n_time = time_table.has_value() ? time_table->size(0) : 0;
  dim3 block(256);
  dim3 grid(B * H);
  size_t smem = 4 * HTILE * 128;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((hstu_attn_fwd_kernel<__hip_bfloat16>), grid, block,
      smem, stream,
      reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
      pb.data_ptr<int>(),
      reinterpret_cast<const __hip_bfloat16*>(pos_table.data_ptr()),
      time_table.has_value()
          ? reinterpret_cast<const __hip_bfloat16*>(time_table->data_ptr())
          : nullptr,
      timestamps.has_value()
          ? reinterpret_cast<const long long*>(ts64.data_ptr())
          : nullptr,
      key_pad.has_value() ? key_pad->data_ptr<bool>() : nullptr,
      reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
      s_saved.data_ptr<float>(), B, H, L, D, n_time);
  return {out, s_saved};
}

std::vector<torch::Tensor> hstu_attn_bwd(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor s_saved, torch::Tensor pos_bucket,
    c10::optional<torch::Tensor> timestamps,
    int64_t n_pos, int64_t n_time) {
  const int B = q.size(0), H = q.size(1), L = q.size(2), D = q.size(3);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto dpos = torch::zeros({n_pos, H}, q.options().dtype(torch::kFloat32));
  torch::Tensor dtime;
  float* dtime_ptr = nullptr;
  if (n_time > 0) {
    dtime = torch::zeros({n_time, H}, q.options().dtype(torch::kFloat32));
    dtime_ptr = dtime.data_ptr<float>();
  } else {
    dtime = torch::empty({0}, q.options().dtype(torch::kFloat32));
  }
  auto pb = pos_bucket.to(torch::kInt32).contiguous();
  torch::Tensor ts64;
  if (timestamps.has_value())
    ts64 = timestamps->to(torch::kInt64).contiguous();
  auto dc = dout.contiguous();
  dim3 block(256);
  dim3 grid(B * H);
  size_t smem = 8 * HTILE * 128 +
                ((size_t)n_pos + (size_t)n_time) * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL((hstu_attn_bwd_kernel<__hip_bfloat16>), grid, block,
      smem, stream,
      reinterpret_cast<const __hip_bfloat16*>(dc.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
      s_saved.data_ptr<float>(), pb.data_ptr<int>(),
      timestamps.has_value()
          ? reinterpret_cast<const long long*>(ts64.data_ptr())
          : nullptr,
      reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()),
      dpos.data_ptr<float>(), dtime_ptr,
      B, H, L, D, (int)n_pos, (int)n_time);
  return {dq, dk, dv, dpos, dtime};
}

}  // namespace genrec
