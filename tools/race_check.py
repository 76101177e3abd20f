#!/usr/bin/env python3
"""Kernel race / nondeterminism checker (SURVEY.md §5.2 sanitizers).

LDS races and missing barriers in HIP kernels typically surface as
run-to-run nondeterminism (wave scheduling varies between launches). This
harness runs every genrec kernel N times on identical inputs — optionally
with a concurrent "scheduling perturber" stream issuing dummy memory
traffic to shake up wave interleaving — and asserts bitwise-identical
outputs. Deterministic-by-construction kernels (no atomics in the output
path) must produce exactly equal bits; the dQ-atomic flash backward is
checked against an fp32 tolerance instead.

Usage (on a GPU box):
    python tools/race_check.py [--iters 20] [--perturb]

Exit code 0 = all kernels deterministic; 1 = mismatch (prints kernel+op).
Also runnable as `pytest tests/test_gpu_sanitize.py` (same checks, fewer
iters).
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def _perturber(stream, n=1 << 20):
    """Issue dummy traffic on a side stream to perturb wave scheduling."""
    with torch.cuda.stream(stream):
        a = torch.randn(n, device="cuda")
        for _ in range(4):
            a = a * 1.0001 + 0.1


def run_case(name, fn, iters=10, perturb=False, atol=0.0):
    torch.manual_seed(0)
    side = torch.cuda.Stream() if perturb else None
    ref = None
    for it in range(iters):
        if side is not None:
            _perturber(side)
        out = fn()
        torch.cuda.synchronize()
        outs = [o.detach().clone() for o in
                (out if isinstance(out, (list, tuple)) else [out])]
        if ref is None:
            ref = outs
            continue
        for i, (a, b) in enumerate(zip(ref, outs)):
            if atol == 0.0:
                if not torch.equal(a, b):
                    print(f"RACE? {name}[out{i}] iter {it}: "
                          f"{(a.float() - b.float()).abs().max().item():.3e}")
                    return False
            else:
                if not torch.allclose(a.float(), b.float(), atol=atol):
                    print(f"RACE? {name}[out{i}] iter {it} (atol {atol})")
                    return False
    print(f"ok   {name} ({iters} iters{', perturbed' if perturb else ''})")
    return True


def build_cases():
    from genrec_amd import ops

    dev = "cuda:0"
    B, H, L, D = 4, 6, 61, 64
    q = torch.randn(B, H, L, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    bias = torch.randn(H, L, L, device=dev)
    dout = torch.randn_like(q)

    def attn_fwd():
        return ops.ext().attn_fwd_mfma(q, k, v, bias, None, None, None,
                                       0.125, False, 0, 0.0, 0, None)[:2]

    fwd = ops.ext().attn_fwd_mfma(q, k, v, bias, None, None, None,
                                  0.125, False, 0, 0.0, 0, None)
    p_saved, dmask = fwd[1], fwd[2]

    def attn_bwd():
        return ops.ext().attn_bwd_mfma(dout, q, k, v, p_saved, dmask, None,
                                       0.125, 0, 0.0, 0, True, 3)

    Lq2, Lk2 = 80, 128
    qf = torch.randn(B, H, Lq2, D, device=dev, dtype=torch.bfloat16)
    kf = torch.randn(B, H, Lk2, D, device=dev, dtype=torch.bfloat16)
    vf = torch.randn_like(kf)
    doutf = torch.randn_like(qf)

    def flash_fwd():
        return ops.ext().attn_fwd_flash(qf, kf, vf, None, None, None, None,
                                        0.125, False, 0.0, 0, None)[:3]

    ffwd = ops.ext().attn_fwd_flash(qf, kf, vf, None, None, None, None,
                                    0.125, False, 0.0, 0, None)
    fo, fs, fml, fdm = ffwd

    def flash_bwd():
        return ops.ext().attn_bwd_flash(doutf, qf, kf, vf, fo, fs, fml, fdm,
                                        None, 0.125, 0.0, False, 0)[:3]

    x = torch.randn(4096, 384, device=dev, dtype=torch.bfloat16)
    w = torch.randn(384, device=dev, dtype=torch.bfloat16)

    def rms():
        return ops.ext().rms_norm_fwd(x, w, 1e-6, True)[0]

    logits = torch.randn(4096, 769, device=dev, dtype=torch.bfloat16)
    targets = torch.randint(0, 769, (4096,), device=dev)

    def ce():
        return ops.ext().softmax_ce_fwd(logits, targets, -100)[:2]

    b = torch.randn(384, device=dev, dtype=torch.bfloat16)

    def ln_fwd():
        return ops.ext().layer_norm_fwd(x, w, b, 1e-5)

    lf = ops.ext().layer_norm_fwd(x, w, b, 1e-5)
    dy_ln = torch.randn_like(x)

    def ln_bwd():
        return ops.ext().layer_norm_bwd(dy_ln, x, w, lf[1], lf[2])

    def csum():
        return ops.ext().colsum(x)

    return [
        ("attn_fwd_mfma", attn_fwd, 0.0),
        ("attn_bwd_ds(8-wave)", attn_bwd, 0.0),
        ("attn_fwd_flash", flash_fwd, 0.0),
        # flash bwd accumulates dQ with fp32 global atomics -> order-
        # dependent rounding; tolerance instead of bitwise
        ("attn_bwd_flash", flash_bwd, 2e-3),
        ("rms_norm", rms, 0.0),
        ("layer_norm_fwd", ln_fwd, 0.0),
        ("layer_norm_bwd", ln_bwd, 0.0),
        ("colsum(vec4)", csum, 0.0),
        ("softmax_ce", ce, 0.0),
    ]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--perturb", action="store_true")
    args = p.parse_args()
    assert torch.cuda.is_available(), "race_check needs a GPU"
    ok = True
    for name, fn, atol in build_cases():
        ok &= run_case(name, fn, iters=args.iters, perturb=args.perturb,
                       atol=atol)
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
