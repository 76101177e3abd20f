// MFMA fused attention (bf16) — gfx950 matrix-core path for K1/K4/K13.
//
// Forward: one 256-thread workgroup (4 waves) per (batch, head). Q, K and
// V^T are staged in LDS as bf16 with the T2 XOR swizzle (byte ^= (row&7)<<4)
// so ds_read_b128 fragment loads are conflict-free. Each wave owns a
// 16-row q-strip: S = Q K^T runs as 4 column-fragments x (D/32) K-steps of
// v_mfma_f32_16x16x32_bf16; scale/bias/masks/softmax(or SiLU) are applied
// on the accumulator fragments in-register (the C/D layout places row
// (lane>>4)*4+reg, col lane&15 — a row reduction is an fmax/fadd over the
// 4 column fragments followed by shuffle-xor over the 16-lane group);
// P is written to LDS (bf16, swizzled) and consumed directly as the
// A-operand of the P V MFMA. Scores never touch HBM; P is saved fp32 for
// backward.
//
// Backward: attn_bwd_ds_kernel computes dP = dO V^T with the same MFMA
// geometry, applies the dropout/query-mask multiplier and the softmax
// Jacobian (or SiLU') in-register, stages dS / dS^T / A_d^T in LDS, and
// finishes dQ = dS K, dK = dS^T Q, dV = A_d^T dO as MFMA in the SAME
// launch (no extra batched GEMMs, no extra global round-trips). All
// global tensors are stride-parameterized so [B,L,H,D] transpose views
// flow in/out without permute copies.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

torch::Tensor colsum(torch::Tensor x);  // fused_elementwise.hip

constexpr int TILE = 64;               // max Lq/Lk per block
constexpr float NEG_BIG_MF = -1e9f;

typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

#define IDX4M(b, h, i, j, H, I, J) \
  ((((int64_t)(b) * (H) + (h)) * (I) + (i)) * (J) + (j))

// LDS tile: [64][64] bf16, row stride 128 B, XOR-swizzled byte offset.
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}


// In-register 8x8 bf16 block transpose across the 8 lanes {8g+c : g=0..7}
// (3 butterfly stages of shfl_xor; validated against a host simulation).
// Turns a lane's natural row-chunk into a transposed row-chunk so the
// LDS transpose staging becomes ONE b128 store instead of 8 scattered
// u16 stores.
__device__ __forceinline__ void xpose8x8(short (&vals)[8], int g) {
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

// load one 16x32 A/B fragment (8 bf16 = 16 B per lane) from a swizzled tile
__device__ __forceinline__ short8v frag_load(const char* base, int row0,
                                             int k0, int lane) {
  int row = row0 + (lane & 15);
  int byte = (k0 + ((lane >> 4) << 3)) * 2;
  return *reinterpret_cast<const short8v*>(base + swz(row, byte));
}

template <bool IS_SILU>
__global__ void __launch_bounds__(256)
attn_fwd_mfma_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B,H,Lq,D]
    const __hip_bfloat16* __restrict__ k,   // [B,H,Lk,D]
    const __hip_bfloat16* __restrict__ v,   // [B,H,Lk,D]
    const void* __restrict__ bias,          // null | [H,Lq,Lk] | [B,H,Lq,Lk]
                                            // | table [H*nb] (bucket mode)
    const int* __restrict__ bias_bucket,    // null | [Lq,Lk] table indices
    const bool* __restrict__ key_pad,       // null | [B,Lk]
    const float* __restrict__ add_mask,     // null | [Lq,Lk]
    const float* __restrict__ query_mask,   // null | [B,Lq]
    __hip_bfloat16* __restrict__ out,       // [B,H,Lq,D]
    float* __restrict__ p_saved,            // [B,H,Lq,Lk]
    unsigned char* __restrict__ drop_mask,
    const unsigned int* __restrict__ seed_dev,
    int B, int H, int Lq, int Lk, int D, int n_buckets,
    int64_t q_sb, int64_t q_sh, int64_t q_sl,
    int64_t k_sb, int64_t k_sh, int64_t k_sl,
    int64_t v_sb, int64_t v_sh, int64_t v_sl,
    int64_t o_sb, int64_t o_sh, int64_t o_sl,
    float scale, int bias_dim, bool bias_bf16, bool causal,
    float dropout_p, unsigned int seed, int q_tile) {
  if (seed_dev) seed += *seed_dev;
  const float* bias_f = reinterpret_cast<const float*>(bias);
  const __hip_bfloat16* bias_b = reinterpret_cast<const __hip_bfloat16*>(bias);
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.y * q_tile;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* qs = smem;                 // [64][128B] swizzled bf16
  char* ks = qs + TILE * 128;      // [64][128B]
  char* vt = ks + TILE * 128;      // [64(d)][128B(j)] transposed V
  char* ps = vt + TILE * 128;      // [64][128B] P (bf16)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  // ---- stage: coalesced global rows -> swizzled LDS rows (zero-padded)
  // each thread copies 8 bf16 (16 B); row-major walk
  for (int idx = tid; idx < TILE * (TILE / 8); idx += blockDim.x) {
    int row = idx / (TILE / 8);
    int c8 = idx % (TILE / 8);        // 8-elem chunk
    int d0 = c8 * 8;
    short8v z = {};
    // Q tile
    short8v val = z;
    int qi = q0 + row;
    if (qi < Lq && d0 < D) {
      val = *reinterpret_cast<const short8v*>(
          &q[(int64_t)b * q_sb + h * q_sh + qi * q_sl + d0]);
    }
    *reinterpret_cast<short8v*>(qs + swz(row, d0 * 2)) = val;
    // K tile
    val = z;
    if (row < Lk && d0 < D) {
      val = *reinterpret_cast<const short8v*>(
          &k[(int64_t)b * k_sb + h * k_sh + row * k_sl + d0]);
    }
    *reinterpret_cast<short8v*>(ks + swz(row, d0 * 2)) = val;
    // V^T tile: vt[d][j] = V[j][d]. Read V row-natural (ONE b128
    // coalesced load), transpose the 8x8 block in-register across the
    // lane group, store ONE b128 transposed row-chunk.
    short8v vv = z;
    if (row < Lk && d0 < D) {
      vv = *reinterpret_cast<const short8v*>(
          &v[(int64_t)b * v_sb + h * v_sh + row * v_sl + d0]);
    }
    {
      const int g = (lane >> 3) & 7;
      short tv[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) tv[e] = vv[e];
      xpose8x8(tv, g);
      short8v pack;
#pragma unroll
      for (int e = 0; e < 8; ++e) pack[e] = tv[e];
      // this lane now holds V[j0..j0+7][d0+g] -> vt row d0+g
      *reinterpret_cast<short8v*>(vt + swz(d0 + g, (row & ~7) * 2)) = pack;
    }
  }
  __syncthreads();

  const int strip = wid * 16;            // this wave's q rows [strip, strip+16)
  const float inv_keep = dropout_p > 0.f ? 1.0f / (1.0f - dropout_p) : 1.0f;

  // ---- S = Q K^T : 4 column fragments
  float4v acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    for (int kk = 0; kk < D; kk += 32) {
      short8v a = frag_load(qs, strip, kk, lane);
      short8v bfr = frag_load(ks, f * 16, kk, lane);
      acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc[f], 0, 0, 0);
    }
  }

  // ---- epilogue on fragments: scale + bias + masks
  const int col_base = lane & 15;
  const int row_grp = (lane >> 4) << 2;  // rows row_grp..row_grp+3 (in strip)
  float s_val[4][4];                     // [fragment][reg]
#pragma unroll
  for (int f = 0; f < 4; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = q0 + strip + row_grp + r;   // global q row
      int j = f * 16 + col_base;          // key col
      float s = acc[f][r];
      bool ok = (i < Lq) && (j < Lk);
      if (ok) {
        s *= scale;
        if (bias_dim) {
          int64_t bi = (bias_dim == 3) ? ((int64_t)h * Lq + i) * Lk + j
                                       : IDX4M(b, h, i, j, H, Lq, Lk);
          s += bias_bf16 ? to_f32(bias_b[bi]) : bias_f[bi];
        } else if (bias_bucket) {
          // in-kernel rel-bias table gather (HSTU-style): no
          // materialized [H,Lq,Lk] bias tensor traffic
          s += bias_f[h * n_buckets + bias_bucket[(int64_t)i * Lk + j]];
        }
        if (causal && j > i) s = NEG_BIG_MF;
        if (key_pad && key_pad[(int64_t)b * Lk + j]) s = NEG_BIG_MF;
        if (add_mask) s += add_mask[(int64_t)i * Lk + j];
      } else {
        s = -INFINITY;
      }
      s_val[f][r] = s;
    }
  }

  float p_val[4][4];
  if (IS_SILU) {
#pragma unroll
    for (int f = 0; f < 4; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = s_val[f][r];
        p_val[f][r] = (s == -INFINITY) ? 0.f : s * sigmoidf_dev(s);
      }
  } else {
    // row softmax: combine 4 col fragments per reg, reduce over 16-lane grp
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = fmaxf(fmaxf(s_val[0][r], s_val[1][r]),
                      fmaxf(s_val[2][r], s_val[3][r]));
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
      float sum = 0.f;
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        float e = (s_val[f][r] == -INFINITY) ? 0.f
                                             : __expf(s_val[f][r] - m);
        p_val[f][r] = e;
        sum += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        sum += __shfl_xor(sum, off, 64);
      float inv = (sum > 0.f) ? 1.0f / sum : 0.f;
#pragma unroll
      for (int f = 0; f < 4; ++f) p_val[f][r] *= inv;
    }
  }

  // ---- save P/S, apply post-softmax query mask + dropout, stash P in LDS
#pragma unroll
  for (int f = 0; f < 4; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = q0 + strip + row_grp + r;
      int j = f * 16 + col_base;
      float p = p_val[f][r];
      if (i < Lq && j < Lk) {
        p_saved[IDX4M(b, h, i, j, H, Lq, Lk)] = IS_SILU ? s_val[f][r] : p;
        if (query_mask) p *= query_mask[(int64_t)b * Lq + i];
        if (dropout_p > 0.f) {
          unsigned long long gidx = IDX4M(b, h, i, j, H, Lq, Lk);
          bool keep = (hash_rng(seed, gidx) & 0xFFFFFF) >=
                      (unsigned int)(dropout_p * 16777216.0f);
          drop_mask[gidx] = keep;
          p = keep ? p * inv_keep : 0.f;
        }
      } else {
        p = 0.f;
      }
      // bf16 scatter into swizzled P tile (2 B per element)
      int row = strip + row_grp + r;
      *reinterpret_cast<__hip_bfloat16*>(ps + swz(row, j * 2)) =
          __float2bfloat16(p);
    }
  }
  // wave-local: this wave only reads its own strip rows back
  __builtin_amdgcn_wave_barrier();

  // ---- O = P V : A = P strip rows (k=j), B = vt rows (n=d)
  float4v acc2[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  const int nfrag_d = (D + 15) / 16;
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    if (f >= nfrag_d) break;
    for (int kk = 0; kk < TILE; kk += 32) {
      short8v a = frag_load(ps, strip, kk, lane);
      short8v bfr = frag_load(vt, f * 16, kk, lane);
      acc2[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc2[f],
                                                        0, 0, 0);
    }
  }
#pragma unroll
  for (int f = 0; f < 4; ++f) {
    if (f >= nfrag_d) break;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = q0 + strip + row_grp + r;
      int d = f * 16 + col_base;
      if (i < Lq && d < D) {
        out[(int64_t)b * o_sb + h * o_sh + i * o_sl + d] =
            __float2bfloat16(acc2[f][r]);
      }
    }
  }
}

// ---- backward dS kernel: dP = dO V^T (MFMA), then Jacobian in-register,
// then dQ/dK/dV MFMA in the same launch.
//
// 8-wave occupancy layout (round 2): 512 threads; wave pair p = wid>>1
// owns q/j-strip [16p, 16p+16) and half = wid&1 computes dP column
// fragments {2*half, 2*half+1}. The softmax-Jacobian row dot becomes
// cross-wave: each half reduces its 2-fragment partial over the 16-lane
// group and writes it to a per-half LDS slot (dotbuf[2][64], 512 B — no
// atomics: every slot has exactly one writer); one barrier later both
// halves read the summed dot. dS/dS^T/A_d^T stores stay disjoint per half
// (different column fragments -> different j rows of the transposed
// tiles), and dQ/dK/dV split their OUTPUT d-fragments across the halves
// with full k-loops. LDS grows by only 512 B, so still 2 blocks/CU but
// 16 resident waves — double the latency hiding for the staging-bound
// phases (round-1 PMC: 8 waves, ~49k LDS conflict cycles/dispatch).
template <bool IS_SILU>
__global__ void __launch_bounds__(512)
attn_bwd_ds_kernel(
    const __hip_bfloat16* __restrict__ dout,  // [B,H,Lq,D]
    const __hip_bfloat16* __restrict__ q,     // [B,H,Lq,D]
    const __hip_bfloat16* __restrict__ k,     // [B,H,Lk,D]
    const __hip_bfloat16* __restrict__ v,     // [B,H,Lk,D]
    const float* __restrict__ p_saved,        // [B,H,Lq,Lk] (P or S)
    const float* __restrict__ query_mask,     // null | [B,Lq]
    const unsigned char* __restrict__ drop_mask,
    __hip_bfloat16* __restrict__ dq_out,      // [B,H,Lq,D]
    __hip_bfloat16* __restrict__ dk_out,      // [B,H,Lk,D]
    __hip_bfloat16* __restrict__ dv_out,      // [B,H,Lk,D]
    __hip_bfloat16* __restrict__ ds_saved,    // null | [B,H,Lq,Lk] bias grad
                                              // (bf16: halves the HBM
                                              // round-trip; summed fp32)
    const int* __restrict__ bias_bucket,      // null | [Lq,Lk]
    float* __restrict__ dtab_partial,         // null | [B*H, n_buckets]
    int n_buckets,
    int B, int H, int Lq, int Lk, int D,
    int64_t do_sb, int64_t do_sh, int64_t do_sl,
    int64_t q_sb, int64_t q_sh, int64_t q_sl,
    int64_t k_sb, int64_t k_sh, int64_t k_sl,
    int64_t v_sb, int64_t v_sh, int64_t v_sl,
    int64_t dq_sb, int64_t dq_sh, int64_t dq_sl,
    int64_t dk_sb, int64_t dk_sh, int64_t dk_sl,
    int64_t dv_sb, int64_t dv_sh, int64_t dv_sl,
    float scale, float dropout_p) {
  const int bh = blockIdx.x;
  const int b = bh / H, h = bh % H;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* dos = smem;                 // [64][128B] dO rows (A for dP)
  char* vs = dos + TILE * 128;      // [64][128B] V rows (B for dP)
  char* kt = vs + TILE * 128;       // [64(d)][128B(j)] K^T (B for dQ)
  char* qt = kt + TILE * 128;       // [64(d)][128B(i)] Q^T (B for dK)
  char* dot = qt + TILE * 128;      // [64(d)][128B(i)] dO^T (B for dV)
  char* dsn = dot + TILE * 128;     // [64(i)][128B(j)] dS (A for dQ)
  char* dst = dsn + TILE * 128;     // [64(j)][128B(i)] dS^T (A for dK)
  char* adt = dst + TILE * 128;     // [64(j)][128B(i)] A_d^T (A for dV)
  float* dotbuf = reinterpret_cast<float*>(adt + TILE * 128);  // [2][64]
  float* tab_lds = dotbuf + 2 * TILE;  // [n_buckets] table-grad partials

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 0..7
  const int pair = wid >> 1;         // strip owner
  const int half = wid & 1;          // fragment split
  const int fbase = half * 2;

  if (dtab_partial && tid < n_buckets) tab_lds[tid] = 0.f;

  for (int idx = tid; idx < TILE * (TILE / 8); idx += blockDim.x) {
    int row = idx / (TILE / 8);
    int d0 = (idx % (TILE / 8)) * 8;
    short8v val = {};
    if (row < Lq && d0 < D) {
      val = *reinterpret_cast<const short8v*>(
          &dout[(int64_t)b * do_sb + h * do_sh + row * do_sl + d0]);
    }
    *reinterpret_cast<short8v*>(dos + swz(row, d0 * 2)) = val;
    short8v val2 = {};
    if (row < Lk && d0 < D) {
      val2 = *reinterpret_cast<const short8v*>(
          &v[(int64_t)b * v_sb + h * v_sh + row * v_sl + d0]);
    }
    *reinterpret_cast<short8v*>(vs + swz(row, d0 * 2)) = val2;
    // transposed images (row = d, cols = sequence positions): read each
    // source row-natural — ONE b128 coalesced load per tensor —
    // transpose 8x8 in-register, store ONE b128 per tensor.
    short8v kk8 = {}, qq8 = {}, dd8 = {};
    if (row < Lk && d0 < D) {
      kk8 = *reinterpret_cast<const short8v*>(
          &k[(int64_t)b * k_sb + h * k_sh + row * k_sl + d0]);
    }
    if (row < Lq && d0 < D) {
      qq8 = *reinterpret_cast<const short8v*>(
          &q[(int64_t)b * q_sb + h * q_sh + row * q_sl + d0]);
      dd8 = *reinterpret_cast<const short8v*>(
          &dout[(int64_t)b * do_sb + h * do_sh + row * do_sl + d0]);
    }
    {
      const int g = (lane >> 3) & 7;
      const int jb = (row & ~7) * 2;
      short tk[8], tq2[8], td2[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        tk[e] = kk8[e]; tq2[e] = qq8[e]; td2[e] = dd8[e];
      }
      xpose8x8(tk, g);
      xpose8x8(tq2, g);
      xpose8x8(td2, g);
      short8v pk, pq, pd;
#pragma unroll
      for (int e = 0; e < 8; ++e) { pk[e] = tk[e]; pq[e] = tq2[e]; pd[e] = td2[e]; }
      *reinterpret_cast<short8v*>(kt + swz(d0 + g, jb)) = pk;
      *reinterpret_cast<short8v*>(qt + swz(d0 + g, jb)) = pq;
      *reinterpret_cast<short8v*>(dot + swz(d0 + g, jb)) = pd;
    }
  }
  __syncthreads();

  const int strip = pair * 16;
  const float inv_keep = dropout_p > 0.f ? 1.0f / (1.0f - dropout_p) : 1.0f;

  // dP = dO V^T : A = dO strip rows (k=d), B = V rows (n=j, k=d).
  // This half computes column fragments fbase and fbase+1 only.
  float4v acc[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
    for (int kk = 0; kk < D; kk += 32) {
      short8v a = frag_load(dos, strip, kk, lane);
      short8v bfr = frag_load(vs, (fbase + fi) * 16, kk, lane);
      acc[fi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc[fi],
                                                        0, 0, 0);
    }
  }

  const int col_base = lane & 15;
  const int row_grp = (lane >> 4) << 2;
  float pv[2][4], da[2][4], ad[2][4];
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = strip + row_grp + r;
      int j = (fbase + fi) * 16 + col_base;
      bool ok = (i < Lq) && (j < Lk);
      float p = ok ? p_saved[IDX4M(b, h, i, j, H, Lq, Lk)] : 0.f;
      float m = 1.f;
      if (!IS_SILU && ok) {
        if (query_mask) m *= query_mask[(int64_t)b * Lq + i];
        if (dropout_p > 0.f) {
          m *= drop_mask[IDX4M(b, h, i, j, H, Lq, Lk)] ? inv_keep : 0.f;
        }
      }
      pv[fi][r] = p;          // P (softmax) or S (silu)
      da[fi][r] = ok ? acc[fi][r] * (IS_SILU ? 1.f : m) : 0.f;
      ad[fi][r] = IS_SILU ? (ok ? p * sigmoidf_dev(p) : 0.f) : p * m;
    }
  }

  float ds[2][4];
  if (IS_SILU) {
#pragma unroll
    for (int fi = 0; fi < 2; ++fi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = pv[fi][r];
        float sg = sigmoidf_dev(s);
        ds[fi][r] = da[fi][r] * sg * (1.f + s * (1.f - sg));
      }
  } else {
    // Jacobian row dot across both halves: reduce this half's 32-column
    // partial in-wave, publish to dotbuf[half], barrier, read the sum.
    float part[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float d0 = da[0][r] * pv[0][r] + da[1][r] * pv[1][r];
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        d0 += __shfl_xor(d0, off, 64);
      part[r] = d0;
    }
    if ((lane & 15) == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        dotbuf[half * TILE + strip + row_grp + r] = part[r];
    }
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float dot_r = dotbuf[strip + row_grp + r] +
                    dotbuf[TILE + strip + row_grp + r];
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        ds[fi][r] = pv[fi][r] * (da[fi][r] - dot_r);
    }
  }

  // stash dS (natural + transposed) and A_d^T in LDS; optional global dS.
  // For the transposed tiles a lane's 4 r-values land on consecutive
  // columns of row j, so they pack into ONE aligned 8-byte ds_write
  // (4x fewer LDS stores than per-element bf16 scatter -> fewer bank
  // conflict cycles; rocprofv3 SQ_LDS_BANK_CONFLICT was dominated by
  // these stores). Halves write disjoint j rows / column fragments.
  typedef __attribute__((ext_vector_type(4))) short short4v;
  const int i0 = strip + row_grp;  // multiple of 4 -> byte i0*2 is 8B-aligned
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
    short4v dpack, apack;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int i = i0 + r;
      int j = (fbase + fi) * 16 + col_base;
      float dval = (i < Lq && j < Lk) ? ds[fi][r] : 0.f;
      float aval = (i < Lq && j < Lk) ? ad[fi][r] : 0.f;
      __hip_bfloat16 dh = __float2bfloat16(dval * scale);
      __hip_bfloat16 ah = __float2bfloat16(aval);
      dpack[r] = *reinterpret_cast<short*>(&dh);
      apack[r] = *reinterpret_cast<short*>(&ah);
      *reinterpret_cast<__hip_bfloat16*>(dsn + swz(i, j * 2)) = dh;
      if (ds_saved && i < Lq && j < Lk) {
        ds_saved[IDX4M(b, h, i, j, H, Lq, Lk)] = __float2bfloat16(dval);
      }
      if (dtab_partial && i < Lq && j < Lk) {
        // LDS-privatized table-grad accumulation (32-entry histogram)
        atomicAdd(&tab_lds[bias_bucket[(int64_t)i * Lk + j]], dval);
      }
    }
    int j = (fbase + fi) * 16 + col_base;
    *reinterpret_cast<short4v*>(dst + swz(j, i0 * 2)) = dpack;
    *reinterpret_cast<short4v*>(adt + swz(j, i0 * 2)) = apack;
  }
  // dQ's A-operand (dsn strip rows) now mixes both halves' column
  // fragments, and dK/dV read all pairs' dS^T/A_d^T columns: one
  // workgroup barrier covers every consumer below.
  __syncthreads();
  if (dtab_partial && tid < n_buckets) {
    // one partial row per (b,h) block; reduced deterministically on the
    // host via colsum over [B, H*nb]
    dtab_partial[(int64_t)bh * n_buckets + tid] = tab_lds[tid];
  }

  const int nfrag_d = (D + 15) / 16;
  {  // dQ[strip rows] = (scale*dS) @ K : output d-fragments split by half
    float4v accq[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      if (fbase + fi >= nfrag_d) break;
      for (int kk = 0; kk < TILE; kk += 32) {
        short8v a = frag_load(dsn, strip, kk, lane);
        short8v bfr = frag_load(kt, (fbase + fi) * 16, kk, lane);
        accq[fi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, accq[fi],
                                                           0, 0, 0);
      }
    }
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      if (fbase + fi >= nfrag_d) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int i = strip + row_grp + r;
        int d = (fbase + fi) * 16 + col_base;
        if (i < Lq && d < D) {
          dq_out[(int64_t)b * dq_sb + h * dq_sh + i * dq_sl + d] =
              __float2bfloat16(accq[fi][r]);
        }
      }
    }
  }

  {  // dK[j strip] = (scale*dS)^T @ Q ; dV[j strip] = A_d^T @ dO
    float4v acck[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
    float4v accv[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      if (fbase + fi >= nfrag_d) break;
      for (int kk = 0; kk < TILE; kk += 32) {
        short8v a1 = frag_load(dst, strip, kk, lane);
        short8v b1 = frag_load(qt, (fbase + fi) * 16, kk, lane);
        acck[fi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acck[fi],
                                                           0, 0, 0);
        short8v a2 = frag_load(adt, strip, kk, lane);
        short8v b2 = frag_load(dot, (fbase + fi) * 16, kk, lane);
        accv[fi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, accv[fi],
                                                           0, 0, 0);
      }
    }
#pragma unroll
    for (int fi = 0; fi < 2; ++fi) {
      if (fbase + fi >= nfrag_d) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int j = strip + row_grp + r;
        int d = (fbase + fi) * 16 + col_base;
        if (j < Lk && d < D) {
          dk_out[(int64_t)b * dk_sb + h * dk_sh + j * dk_sl + d] =
              __float2bfloat16(acck[fi][r]);
          dv_out[(int64_t)b * dv_sb + h * dv_sh + j * dv_sl + d] =
              __float2bfloat16(accv[fi][r]);
        }
      }
    }
  }
}

// Batch-reduce of the bf16 bias-grad buffer: out[h,i,j] = sum_b ds[b,h,i,j]
// in fp32 — replaces an ATen sum(0) that cost ~11.7 us/layer (BACKLOG r1
// item 2b). Two stages keep enough resident waves; fixed-order loops:
// deterministic.
constexpr int BS_CH = 16;

__global__ void batch_sum1_kernel(const __hip_bfloat16* __restrict__ in,
                                  float* __restrict__ tmp,
                                  int B, int64_t inner) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= (int64_t)BS_CH * inner) return;
  int64_t j = tid % inner;
  int chunk = (int)(tid / inner);
  float acc = 0.f;
  for (int b = chunk; b < B; b += BS_CH)
    acc += to_f32(in[(int64_t)b * inner + j]);
  tmp[(int64_t)chunk * inner + j] = acc;
}

__global__ void batch_sum2_kernel(const float* __restrict__ tmp,
                                  float* __restrict__ out, int64_t inner) {
  int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= inner) return;
  float acc = 0.f;
#pragma unroll
  for (int r = 0; r < BS_CH; ++r) acc += tmp[(int64_t)r * inner + j];
  out[j] = acc;
}

// ------------------------------------------------------------------ hosts

std::vector<torch::Tensor> attn_fwd_mfma(
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    c10::optional<torch::Tensor> bias, c10::optional<torch::Tensor> key_pad,
    c10::optional<torch::Tensor> add_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, bool causal, int64_t act, double dropout_p, int64_t seed,
    c10::optional<torch::Tensor> seed_dev,
    c10::optional<torch::Tensor> bias_bucket) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  const int B = q.size(0), H = q.size(1), Lq = q.size(2), D = q.size(3);
  const int Lk = k.size(2);
  TORCH_CHECK(Lk <= TILE && D <= TILE && D % 32 == 0);
  // q/k/v may be transpose views (e.g. [B,L,H,D] -> [B,H,L,D]); only the
  // D axis must be unit-stride. Saves the .contiguous() copies around
  // every attention call.
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "attn_fwd_mfma: innermost (D) stride must be 1");
  auto out = torch::empty_like(q);
  auto p_saved = torch::empty({B, H, Lq, Lk},
                              q.options().dtype(torch::kFloat32));
  torch::Tensor dmask;
  if (dropout_p > 0) {
    dmask = torch::empty({B, H, Lq, Lk}, q.options().dtype(torch::kUInt8));
  } else {
    dmask = torch::empty({0}, q.options().dtype(torch::kUInt8));
  }
  torch::Tensor bias_c, bucket_c;
  int bias_dim = 0;
  bool bias_bf16 = false;
  int n_buckets = 0;
  if (bias_bucket.has_value()) {
    // bucket mode: `bias` is the fp32 table [H*nb] (flattened rel-bias
    // embedding weight), `bias_bucket` the [Lq,Lk] int32 index map
    TORCH_CHECK(bias.has_value() &&
                bias->scalar_type() == torch::kFloat32);
    bias_c = bias->contiguous();
    bucket_c = bias_bucket->to(torch::kInt32).contiguous();
    n_buckets = (int)(bias_c.numel() / H);
  } else if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == torch::kFloat32 ||
                bias->scalar_type() == torch::kBFloat16);
    bias_c = bias->contiguous();
    bias_dim = bias_c.dim();
    bias_bf16 = bias_c.scalar_type() == torch::kBFloat16;
  }
  torch::Tensor am_f, qm_f;
  if (add_mask.has_value()) am_f = add_mask->to(torch::kFloat32).contiguous();
  if (query_mask.has_value())
    qm_f = query_mask->to(torch::kFloat32).contiguous();

  const int q_tile = TILE;
  dim3 block(256);
  dim3 grid(B * H, (Lq + q_tile - 1) / q_tile);
  size_t smem = 4 * TILE * 128;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_FWD_MFMA(SILU)                                                  \
  hipLaunchKernelGGL((attn_fwd_mfma_kernel<SILU>), grid, block, smem, stream,  \
      reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),                   \
      reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),                   \
      reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),                   \
      (bias_dim || n_buckets) ? bias_c.data_ptr() : nullptr,                   \
      n_buckets ? bucket_c.data_ptr<int>() : nullptr,                          \
      key_pad.has_value() ? key_pad->data_ptr<bool>() : nullptr,               \
      add_mask.has_value() ? am_f.data_ptr<float>() : nullptr,                 \
      query_mask.has_value() ? qm_f.data_ptr<float>() : nullptr,               \
      reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),                       \
      p_saved.data_ptr<float>(),                                               \
      dropout_p > 0 ? dmask.data_ptr<unsigned char>() : nullptr,               \
      seed_dev.has_value()                                                     \
          ? reinterpret_cast<const unsigned int*>(seed_dev->data_ptr())        \
          : nullptr,                                                           \
      B, H, Lq, Lk, D, n_buckets,                                              \
      q.stride(0), q.stride(1), q.stride(2),                                   \
      k.stride(0), k.stride(1), k.stride(2),                                   \
      v.stride(0), v.stride(1), v.stride(2),                                   \
      out.stride(0), out.stride(1), out.stride(2),                             \
      (float)scale, bias_dim, bias_bf16, causal,                               \
      (float)dropout_p, (unsigned int)seed, q_tile)

  if (act == 0) LAUNCH_FWD_MFMA(false);
  else LAUNCH_FWD_MFMA(true);
#undef LAUNCH_FWD_MFMA
  return {out, p_saved, dmask};
}

std::vector<torch::Tensor> attn_bwd_mfma(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor p_saved, torch::Tensor drop_mask,
    c10::optional<torch::Tensor> query_mask,
    double scale, int64_t act, double dropout_p, int64_t seed,
    bool bias_grad, int64_t bias_dim,
    c10::optional<torch::Tensor> bias_bucket, int64_t n_buckets) {
  (void)seed;
  const int B = q.size(0), H = q.size(1), Lq = q.size(2), D = q.size(3);
  const int Lk = k.size(2);
  TORCH_CHECK(Lq <= TILE && Lk <= TILE && D % 32 == 0);
  TORCH_CHECK(dout.stride(3) == 1 && q.stride(3) == 1 && k.stride(3) == 1 &&
                  v.stride(3) == 1,
              "attn_bwd_mfma: innermost (D) stride must be 1");
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  torch::Tensor ds_saved, bucket_c, dtab_partial;
  __hip_bfloat16* ds_ptr = nullptr;
  const bool table_mode = bias_grad && bias_bucket.has_value();
  if (table_mode) {
    bucket_c = bias_bucket->to(torch::kInt32).contiguous();
    dtab_partial = torch::empty({(int64_t)B * H, n_buckets},
                                q.options().dtype(torch::kFloat32));
  } else if (bias_grad) {
    ds_saved = torch::empty({B, H, Lq, Lk},
                            q.options().dtype(torch::kBFloat16));
    ds_ptr = reinterpret_cast<__hip_bfloat16*>(ds_saved.data_ptr());
  }
  torch::Tensor qm_f;
  if (query_mask.has_value())
    qm_f = query_mask->to(torch::kFloat32).contiguous();
  dim3 block(512);  // 8 waves: pairs own strips, halves split fragments
  dim3 grid(B * H);
  size_t smem = 8 * TILE * 128 + 2 * TILE * sizeof(float)
      + (table_mode ? (size_t)n_buckets * sizeof(float) : 0);
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_BWD_DS(SILU)                                                    \
  hipLaunchKernelGGL((attn_bwd_ds_kernel<SILU>), grid, block, smem, stream,    \
      reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr()),                \
      reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),                   \
      reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),                   \
      reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),                   \
      p_saved.data_ptr<float>(),                                               \
      query_mask.has_value() ? qm_f.data_ptr<float>() : nullptr,               \
      dropout_p > 0 ? drop_mask.data_ptr<unsigned char>() : nullptr,           \
      reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),                        \
      reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),                        \
      reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()), ds_ptr,                \
      table_mode ? bucket_c.data_ptr<int>() : nullptr,                         \
      table_mode ? dtab_partial.data_ptr<float>() : nullptr,                   \
      (int)n_buckets,                                                          \
      B, H, Lq, Lk, D,                                                         \
      dout.stride(0), dout.stride(1), dout.stride(2),                          \
      q.stride(0), q.stride(1), q.stride(2),                                   \
      k.stride(0), k.stride(1), k.stride(2),                                   \
      v.stride(0), v.stride(1), v.stride(2),                                   \
      dq.stride(0), dq.stride(1), dq.stride(2),                                \
      dk.stride(0), dk.stride(1), dk.stride(2),                                \
      dv.stride(0), dv.stride(1), dv.stride(2),                                \
      (float)scale, (float)dropout_p)

  if (act == 0) LAUNCH_BWD_DS(false);
  else LAUNCH_BWD_DS(true);
#undef LAUNCH_BWD_DS

  torch::Tensor dbias;
  if (table_mode) {
    // deterministic reduce of the per-block table-grad partials:
    // [B*H, nb] -> view [B, H*nb] -> replay-safe colsum -> [H*nb]
    dbias = colsum(dtab_partial.view({B, (int64_t)H * n_buckets}));
  } else if (bias_grad) {
    if (bias_dim == 3) {  // bf16 per-element grads, fp32 batch reduce
      const int64_t inner = (int64_t)H * Lq * Lk;
      dbias = torch::empty({H, Lq, Lk},
                           q.options().dtype(torch::kFloat32));
      auto tmp = torch::empty({16, inner},
                              q.options().dtype(torch::kFloat32));
      dim3 rblock(256);
      dim3 rgrid1((unsigned)((16 * inner + 255) / 256));
      hipLaunchKernelGGL(batch_sum1_kernel, rgrid1, rblock, 0, stream,
          reinterpret_cast<const __hip_bfloat16*>(ds_saved.data_ptr()),
          tmp.data_ptr<float>(), B, inner);
      dim3 rgrid2((unsigned)((inner + 255) / 256));
      hipLaunchKernelGGL(batch_sum2_kernel, rgrid2, rblock, 0, stream,
          tmp.data_ptr<float>(), dbias.data_ptr<float>(), inner);
    } else {
      dbias = ds_saved.to(torch::kFloat32);
    }
  } else {
    dbias = torch::empty({0}, q.options().dtype(torch::kFloat32));
  }
  return {dq, dk, dv, dbias};
}

}  // namespace genrec
