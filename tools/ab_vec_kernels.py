"""A/B the vectorized (bf16x2 / bf16x4) norm+dropout kernels against the
scalar versions on GPU: dropout fwd/bwd must be BITWISE identical (same
hash_rng stream, same rounding); rms_norm may differ by one bf16 ulp
(pairwise fp32 accumulation order inside a lane).

Run on GPU: python tools/ab_vec_kernels.py
Env: the scalar path is forced per-call via GENREC_SCALAR_* read once by
the extension, so this script spawns subprocesses per mode.
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, sys, torch
sys.path.insert(0, %r)
from genrec_amd import ops
dev = "cuda:0"
torch.manual_seed(7)
ext = ops.ext()
out = {}
# dropout_add / relu_dropout / plain dropout fwd+bwd at an odd-ish shape
x = torch.randn(2742, 768, device=dev, dtype=torch.bfloat16)
res = torch.randn_like(x)
seed_dev = torch.zeros(1, device=dev, dtype=torch.int32)
o1, m1 = ext.dropout_add_fwd(x, res, 0.1, 1234, seed_dev)
dx1 = ext.dropout_fuse_bwd(o1, m1, 0.1, False)
o2, m2 = ext.relu_dropout_fwd(x, 0.1, 99, seed_dev)
dx2 = ext.dropout_fuse_bwd(o2, m2, 0.1, True)
# rms norm fwd+bwd
w = torch.randn(768, device=dev, dtype=torch.bfloat16)
y, inv = ext.rms_norm_fwd(x, w, 1e-6, True)
dy = torch.randn_like(y)
dxx, dw = ext.rms_norm_bwd(dy, x, w, inv, True)
wf = torch.randn(768, device=dev, dtype=torch.float32)
y2, inv2 = ext.rms_norm_fwd(x, wf, 1e-6, False)
# fp32 weight + bf16 dy: exercises the vec2<float> weight loads
dx3, dw3 = ext.rms_norm_bwd(dy.to(torch.bfloat16), x, wf, inv2, False)
# colsum: vec4 stage 1 uses a different adaptive chunk count, so partials
# regroup -> compare within bf16 tolerance, not bitwise
cs = ext.colsum(torch.randn(2560, 1536, device=dev, dtype=torch.bfloat16))
torch.save({"o1": o1.cpu(), "m1": m1.cpu(), "dx1": dx1.cpu(),
            "o2": o2.cpu(), "m2": m2.cpu(), "dx2": dx2.cpu(),
            "y": y.cpu(), "inv": inv.cpu(), "dxx": dxx.cpu(),
            "dw": dw.cpu(), "y2": y2.cpu(), "dx3": dx3.cpu(),
            "dw3": dw3.cpu(), "cs": cs.cpu()}, sys.argv[1])
"""


def run(mode_env, path):
    env = dict(os.environ)
    env.update(mode_env)
    code = WORKER % (REPO,)
    subprocess.run([sys.executable, "-c", code, path], check=True, env=env,
                   cwd=REPO)


def main():
    import torch

    run({"GENREC_SCALAR_NORMS": "1", "GENREC_SCALAR_ELEMWISE": "1"},
        "/tmp/ab_scalar.pt")
    run({}, "/tmp/ab_vec.pt")
    a = torch.load("/tmp/ab_scalar.pt")
    b = torch.load("/tmp/ab_vec.pt")
    ok = True
    for k in ("o1", "m1", "dx1", "o2", "m2", "dx2"):
        if not torch.equal(a[k], b[k]):
            print(f"FAIL {k}: dropout scalar vs vec NOT bitwise equal")
            ok = False
    for k in ("y", "inv", "dxx", "dw", "y2", "dx3", "dw3", "cs"):
        d = (a[k].float() - b[k].float()).abs().max().item()
        ref = a[k].float().abs().max().item() + 1e-6
        if d > 2e-2 * ref:
            print(f"FAIL {k}: rms scalar vs vec max diff {d} (ref {ref})")
            ok = False
        else:
            print(f"ok {k}: max abs diff {d:.3e}")
    print("AB_VEC " + ("PASS" if ok else "FAIL"))


if __name__ == "__main__":
    main()
