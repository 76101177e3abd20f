"""Microbench: hand skinny_gemm vs hipBLASLt for the TIGER linear-forward
shape family. Run on GPU: python tools/bench_skinny_gemm.py"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=200, warmup=30):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


def main():
    from genrec_amd import ops

    for (M, N, K) in [(15616, 384, 384), (15616, 768, 384),
                      (15616, 1024, 384), (15616, 384, 1024),
                      (6400, 256, 64), (6400, 64, 64)]:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t_blas = timeit(lambda: x @ w.t())
        t_hand = timeit(lambda: ops.ext().skinny_gemm(x, w, None))
        sol_us = (M * K + N * K + M * N) * 2 / 8e12 * 1e6
        print(f"NN M={M:6d} N={N:5d} K={K:5d}: blaslt {t_blas:7.2f}us  "
              f"hand {t_hand:7.2f}us  ({t_blas / t_hand:4.2f}x, "
              f"SOL~{sol_us:.2f}us)")

    # TN (dX backward) family: dx[M,N] = dy[M,C] @ W[C,N]
    for (M, C, N) in [(15616, 1152, 384), (15616, 1024, 384),
                      (15616, 768, 384), (15616, 384, 384)]:
        dy = torch.randn(M, C, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(C, N, device="cuda", dtype=torch.bfloat16)
        t_blas = timeit(lambda: dy @ w)
        t_hand = timeit(lambda: ops.ext().skinny_gemm_tn(dy, w, None))
        sol_us = (M * C + C * N + M * N) * 2 / 8e12 * 1e6
        print(f"TN M={M:6d} C={C:5d} N={N:5d}: blaslt {t_blas:7.2f}us  "
              f"hand {t_hand:7.2f}us  ({t_blas / t_hand:4.2f}x, "
              f"SOL~{sol_us:.2f}us)")


if __name__ == "__main__":
    main()
