"""Minimal hipGraph repro for the rms_norm kernels."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from genrec_amd import ops  # noqa: E402


def run_case(name, fn, n_replay=6):
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                fn()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        for i in range(n_replay):
            g.replay()
            torch.cuda.synchronize()
        print(f"{name}: OK", flush=True)
    except Exception as e:
        print(f"{name}: FAIL {type(e).__name__}: {e}", flush=True)


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    n_rows, d = 15616, 128

    # case 1: fp32 fwd+bwd
    x32 = torch.randn(n_rows, d, device=dev, requires_grad=True)
    w32 = torch.nn.Parameter(torch.randn(d, device=dev))

    def case1():
        if x32.grad is not None:
            x32.grad.zero_()
        y = ops.rms_norm(x32, w32, 1e-6, t5_style=False)
        y.sum().backward()

    run_case("fp32_std", case1)

    # case 2: bf16 x, fp32 w, t5 style (the TIGER path)
    xb = torch.randn(n_rows, d, device=dev, dtype=torch.bfloat16,
                     requires_grad=True)
    wb = torch.nn.Parameter(torch.randn(d, device=dev))

    def case2():
        if xb.grad is not None:
            xb.grad.zero_()
        y = ops.rms_norm(xb, wb, 1e-6, t5_style=True)
        y.sum().backward()

    run_case("bf16_t5", case2)

    # case 3: chained with GEMM like the model (norm -> matmul -> loss)
    W = torch.randn(d, 384, device=dev, dtype=torch.bfloat16)

    def case3():
        if xb.grad is not None:
            xb.grad.zero_()
        y = ops.rms_norm(xb, wb, 1e-6, t5_style=True)
        z = y.to(torch.bfloat16) @ W
        z.float().sum().backward()

    run_case("bf16_t5_gemm", case3)

    # case 4: many sizes in one step (like 8 layers, two shapes)
    xs = [torch.randn(1024 * (i + 1), d, device=dev, dtype=torch.bfloat16,
                      requires_grad=True) for i in range(6)]

    def case4():
        total = 0
        for xx in xs:
            if xx.grad is not None:
                xx.grad.zero_()
            total = total + ops.rms_norm(xx, wb, 1e-6, True).sum()
        total.backward()

    run_case("multi_shape", case4)

    print("done", flush=True)


if __name__ == "__main__":
    main()
