"""Micro-repros for hipGraph replay corruption: capture single ops and
check outputs/grads stay finite over replays.

Run on GPU: python -u tools/repro_capture_micro.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def capture(fn, warmup=3):
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(warmup):
            fn()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = fn()
    return g, out


def check(name, g, outs, reps=6):
    ok = True
    for r in range(reps):
        g.replay()
        torch.cuda.synchronize()
        for i, t in enumerate(outs):
            if t is not None and not torch.isfinite(t.float()).all():
                n = int((~torch.isfinite(t.float())).sum())
                print(f"BAD {name} replay {r} out{i}: {n} nonfinite",
                      flush=True)
                ok = False
        if not ok:
            break
    if ok:
        print(f"ok  {name}", flush=True)


def main():
    from genrec_amd.ops import eager
    from genrec_amd.ops.attention import advance_dropout_seeds
    from genrec_amd.ops.fused import plain_dropout

    dev = "cuda:0"
    torch.manual_seed(0)

    # 1) plain_dropout alone
    x = torch.randn(672, 8, 16, 16, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)

    def f1():
        advance_dropout_seeds(dev)
        x.grad = None
        y = plain_dropout(x, 0.1, True)
        y.float().sum().backward()
        return y

    g1, y1 = capture(f1)
    check("plain_dropout fwd+bwd", g1, [y1, x.grad])

    # 2) eager fused attention at head_dim 96 (the cobra-768 encoder shape)
    B, H, L, D = 672, 8, 16, 96
    q = torch.randn(B, H, L, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    pad = torch.zeros(B, L, dtype=torch.bool, device=dev)
    pad[5] = True  # one fully-masked row block

    def f2():
        advance_dropout_seeds(dev)
        for t in (q, k, v):
            t.grad = None
        out = eager.fused_attention(q, k, v, scale=0.1, key_pad_mask=pad,
                                    dropout_p=0.1, training=True)
        out.float().sum().backward()
        return out

    g2, o2 = capture(f2)
    check("eager attention hd96 (pad row)", g2, [o2, q.grad, k.grad, v.grad])

    # 3) same without the fully-masked row
    pad2 = torch.zeros(B, L, dtype=torch.bool, device=dev)
    pad2[:, -4:] = True

    def f3():
        advance_dropout_seeds(dev)
        for t in (q, k, v):
            t.grad = None
        out = eager.fused_attention(q, k, v, scale=0.1, key_pad_mask=pad2,
                                    dropout_p=0.1, training=True)
        out.float().sum().backward()
        return out

    g3, o3 = capture(f3)
    check("eager attention hd96 (partial pad)", g3,
          [o3, q.grad, k.grad, v.grad])

    # 4) no dropout
    def f4():
        for t in (q, k, v):
            t.grad = None
        out = eager.fused_attention(q, k, v, scale=0.1, key_pad_mask=pad2,
                                    dropout_p=0.0, training=True)
        out.float().sum().backward()
        return out

    g4, o4 = capture(f4)
    check("eager attention hd96 (no dropout)", g4,
          [o4, q.grad, k.grad, v.grad])
    print("done", flush=True)


if __name__ == "__main__":
    main()
