// Tall-skinny linear forward GEMM: y[M,N] = x[M,K] @ W[N,K]^T, bf16 in /
// bf16 out, fp32 accumulate.
//
// Shape family (TIGER/SASRec/COBRA linears): M = B*L ~ 6k-16k rows,
// N,K in {128..1024}. hipBLASLt's heuristic picks MT64x64x64 macro-tiles
// here (grid ~1.5k WGs) and lands ~4x off memory speed-of-light
// (11.9 us at M=15616,N=K=384; SOL ~3 us — BACKLOG r1 item 4). This
// kernel instead gives each 256-thread workgroup a 64(M) x 384(N) strip:
// the A tile is read from HBM exactly ONCE (no n-tile re-reads), W
// (288 KB at N=K=384) stays L2-resident and is staged per 64-k slice
// into LDS shared by all 4 waves. Wave w owns 96 consecutive n-columns
// (6 n-frags x 4 m-frags of v_mfma_f32_16x16x32_bf16). LDS: A 8 KB +
// W-slice 48 KB = 56 KB -> 2 blocks/CU.
//
// Grid: ceil(M/64) x ceil(N/384); ragged N handled by fragment guards.
// Requires K % 64 == 0, D-major (row) contiguity on both operands.
//
// Reference parity: the implicit ATen linears behind the reference's
// transformer.py:72-124 / tiger.py:161-207 hot path.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "../core/common.h"

namespace genrec {

namespace {

constexpr int BM = 64;    // M rows per workgroup
constexpr int BN = 384;   // N cols per workgroup (96 per wave)
constexpr int BK = 64;    // K slice per stage

typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

// 128-byte LDS rows with the T2 XOR swizzle (same as attention tiles)
__device__ __forceinline__ int swz128(int row, int byte_in_row) {
  return row * 128 + (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ short8v ld_frag(const char* base, int row0,
                                           int k0, int lane) {
  int row = row0 + (lane & 15);
  int byte = (k0 + ((lane >> 4) << 3)) * 2;
  return *reinterpret_cast<const short8v*>(base + swz128(row, byte));
}

// In-register 8x8 bf16 block transpose across the 8-lane group (same
// butterfly as attention_mfma.hip's V^T staging).
__device__ __forceinline__ void xpose8x8g(short (&vals)[8], int g) {
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

// TRANS_B=false: y[M,N] = x[M,K] @ w[N,K]^T  (linear forward; w row-major
//                in K — B image rows are w rows, staged directly)
// TRANS_B=true:  y[M,N] = x[M,K] @ w[K,N]    (linear dX backward; w
//                row-major in N — B image rows are w COLUMNS, staged via
//                the 8x8 in-register transpose)
// K is always the contraction dim of x's last axis.
template <bool TRANS_B>
__global__ void __launch_bounds__(256)
skinny_gemm_kernel(const __hip_bfloat16* __restrict__ x,  // [M,K]
                   const __hip_bfloat16* __restrict__ w,
                   const __hip_bfloat16* __restrict__ bias,  // null | [N]
                   __hip_bfloat16* __restrict__ y,        // [M,N]
                   int M, int N, int K) {
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* xs = smem;                 // [64][128B]  A tile (one BK slice)
  char* ws = xs + BM * 128;        // [384][128B] W slice

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int nw0 = wid * 96;        // wave's first n within the strip

  float4v acc[6][4];               // [n-frag][m-frag]
#pragma unroll
  for (int nf = 0; nf < 6; ++nf)
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) acc[nf][mf] = float4v{0.f, 0.f, 0.f, 0.f};

  // T14 register-pipelined staging (guide §5): ONE register set holds
  // tile t+1's global loads while the MFMAs consume tile t from LDS;
  // the write pass runs after the read barrier. Synchronous staging
  // measured 2.6x slower on the guide's GEMM — and was why skinny v1
  // lost to hipBLASLt.
  // Per-thread chunks: A 512/256 = 2, W 3072/256 = 12 (16 B each).
  const int a_row = tid >> 3;              // A chunk coords (x2, +32 rows)
  const int a_c8 = (tid & 7) * 8;
  short8v a_reg[2], w_reg[12];
  const int g = (lane >> 3) & 7;           // TN transpose group
  const int t_row = ((lane & 7) << 3) | (lane >> 3);  // TN source k-row

  auto load_regs = [&](int k0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int mi = m0 + a_row + i * 32;
      a_reg[i] = short8v{};
      if (mi < M) {
        a_reg[i] = *reinterpret_cast<const short8v*>(
            &x[(int64_t)mi * K + k0 + a_c8]);
      }
    }
#pragma unroll
    for (int j = 0; j < 12; ++j) {
      w_reg[j] = short8v{};
      if (!TRANS_B) {
        int ni = n0 + a_row + j * 32;      // same row/chunk walk as A
        if (ni < N) {
          w_reg[j] = *reinterpret_cast<const short8v*>(
              &w[(int64_t)ni * K + k0 + a_c8]);
        }
      } else {
        int jn = wid + j * 4;              // n chunk (8 cols)
        int ki = k0 + t_row;
        int nj = n0 + 8 * jn;
        if (ki < K && nj < N) {
          w_reg[j] = *reinterpret_cast<const short8v*>(
              &w[(int64_t)ki * N + nj]);
        }
      }
    }
  };

  auto write_lds = [&]() {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      *reinterpret_cast<short8v*>(
          xs + swz128(a_row + i * 32, a_c8 * 2)) = a_reg[i];
    }
#pragma unroll
    for (int j = 0; j < 12; ++j) {
      if (!TRANS_B) {
        *reinterpret_cast<short8v*>(
            ws + swz128(a_row + j * 32, a_c8 * 2)) = w_reg[j];
      } else {
        short tv[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) tv[e] = w_reg[j][e];
        xpose8x8g(tv, g);
        short8v pack;
#pragma unroll
        for (int e = 0; e < 8; ++e) pack[e] = tv[e];
        int jn = wid + j * 4;
        *reinterpret_cast<short8v*>(
            ws + swz128(8 * jn + g, ((t_row & ~7) * 2))) = pack;
      }
    }
  };

  auto compute = [&]() {
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      short8v a[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) a[mf] = ld_frag(xs, mf * 16, kk, lane);
#pragma unroll
      for (int nf = 0; nf < 6; ++nf) {
        short8v b = ld_frag(ws, nw0 + nf * 16, kk, lane);
#pragma unroll
        for (int mf = 0; mf < 4; ++mf) {
          acc[nf][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mf], b, acc[nf][mf], 0, 0, 0);
        }
      }
    }
  };

  const int n_tiles = (K + BK - 1) / BK;
  load_regs(0);
  write_lds();
  __syncthreads();
  for (int t = 0; t < n_tiles; ++t) {
    if (t + 1 < n_tiles) load_regs((t + 1) * BK);  // overlaps compute(t)
    compute();
    __syncthreads();                 // all reads of tile t done
    if (t + 1 < n_tiles) {
      write_lds();                   // waits the in-flight loads here
      __syncthreads();               // writes visible to every wave
    }
  }

  // epilogue: C layout row=(lane>>4)*4+r (within m-frag), col=lane&15
  const int col_base = lane & 15;
  const int row_grp = (lane >> 4) << 2;
#pragma unroll
  for (int nf = 0; nf < 6; ++nf) {
    int n = n0 + nw0 + nf * 16 + col_base;
    if (n >= N) continue;
    float badd = bias ? to_f32(bias[n]) : 0.f;
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int m = m0 + mf * 16 + row_grp + r;
        if (m < M) {
          y[(int64_t)m * N + n] = __float2bfloat16(acc[nf][mf][r] + badd);
        }
      }
    }
  }
}

}  // namespace

static torch::Tensor launch_skinny(torch::Tensor x, torch::Tensor w,
                                   c10::optional<torch::Tensor> bias,
                                   bool trans_b) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2 && x.is_contiguous() &&
              w.is_contiguous());
  const int M = x.size(0), K = x.size(1);
  const int N = trans_b ? w.size(1) : w.size(0);
  if (trans_b) {
    TORCH_CHECK(w.size(0) == K && N % 8 == 0,
                "skinny_gemm_tn: w must be [K,N], N % 8 == 0");
  } else {
    TORCH_CHECK(w.size(1) == K && K % BK == 0,
                "skinny_gemm: K must be a multiple of 64");
  }
  auto y = torch::empty({M, N}, x.options());
  torch::Tensor bias_c;
  const __hip_bfloat16* bias_p = nullptr;
  if (bias.has_value()) {
    bias_c = bias->to(torch::kBFloat16).contiguous();
    bias_p = reinterpret_cast<const __hip_bfloat16*>(bias_c.data_ptr());
  }
  dim3 block(256);
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  size_t smem = (BM + BN) * 128;
  auto stream = at::cuda::getCurrentHIPStream();
#define LAUNCH_SKINNY(TB)                                                     \
  hipLaunchKernelGGL((skinny_gemm_kernel<TB>), grid, block, smem, stream,     \
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),                  \
      reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),                  \
      bias_p,                                                                 \
      reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),                        \
      M, N, K)
  if (trans_b) LAUNCH_SKINNY(true);
  else LAUNCH_SKINNY(false);
#undef LAUNCH_SKINNY
  return y;
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias) {
  return launch_skinny(x, w, bias, false);
}

// dX backward: y[M,N] = x[M,K] @ w[K,N] (w used un-transposed / B^T)
torch::Tensor skinny_gemm_tn(torch::Tensor x, torch::Tensor w,
                             c10::optional<torch::Tensor> bias) {
  return launch_skinny(x, w, bias, true);
}

}  // namespace genrec
