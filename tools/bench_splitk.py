"""Microbench: grad-weight GEMMs (tiny MxN, huge K) — hipBLASLt default vs
chunked-bmm split-K. Informs genrec_amd.ops.linear's backward strategy.

Run on GPU: python tools/bench_splitk.py
"""

import torch


def timeit(fn, iters=200, warmup=20):
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
    dev = "cuda"
    K = 15616
    shapes = [(384, 384), (768, 384), (384, 1024), (1024, 384), (384, 128)]
    for (N, M) in shapes:
        dY = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
        X = torch.randn(K, M, device=dev, dtype=torch.bfloat16)

        def default():
            return dY.t() @ X

        t0 = timeit(default)

        results = [f"dW[{N}x{M}] K={K}: blaslt {t0:8.1f}us"]
        for nchunk in (8, 16, 32, 61):
            if K % nchunk:
                # pad-free chunking requires divisibility; 61 divides 15616
                if K % nchunk != 0:
                    continue
            kc = K // nchunk
            dYc = dY.view(nchunk, kc, N)
            Xc = X.view(nchunk, kc, M)

            def splitk():
                part = torch.bmm(dYc.transpose(1, 2), Xc)
                return part.sum(0)

            t = timeit(splitk)
            results.append(f"splitk{nchunk:3d} {t:8.1f}us")

        # fp32 out variant (precision-preserving sum)
        dYc = dY.view(16, K // 16, N)
        Xc = X.view(16, K // 16, M)

        def splitk_f32sum():
            part = torch.bmm(dYc.transpose(1, 2), Xc)
            return part.sum(0, dtype=torch.float32).to(torch.bfloat16)

        results.append(f"splitk16+f32sum {timeit(splitk_f32sum):8.1f}us")
        print("  ".join(results))

    # correctness spot check
    dY = torch.randn(K, 384, device=dev, dtype=torch.bfloat16)
    X = torch.randn(K, 384, device=dev, dtype=torch.bfloat16)
    ref = (dY.t().float() @ X.float())
    got = torch.bmm(dY.view(16, -1, 384).transpose(1, 2),
                    X.view(16, -1, 384)).sum(0, dtype=torch.float32)
    err = (ref - got).abs().max() / ref.abs().max()
    print(f"splitk rel err vs fp32: {err.item():.2e}")


if __name__ == "__main__":
    main()
