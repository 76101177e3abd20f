"""Thread-level emulation of attn_bwd_flash_kernel (attention_flash.hip).

Mirrors the backward's exact structure: one block per (b,h,k-tile)
staging V_t / K_t^T once, looping q-tiles (dO / dO^T / Q^T staged with
the 8x8 butterfly), dA = dO V_t^T per wave, the P-reconstruction
epilogue (dS = P*(M*dA - dot_i) with dot_i = rowsum(dO*O)), the
dsn/dst/adt tile stores, and the three MFMAs (dQ per q-tile via
atomics; dK/dV accumulated in registers across q-tiles). Compared
against autograd. Run: python tools/emulate_flash_bwd.py
"""

import numpy as np
import torch

from emulate_flash_fwd import FTILE, frag_load, mfma_16x16x32, xpose8x8


def stage_transposed(src_rows, dst_tile, kt0_or_q0, L, D, b, h, src):
    """Emulate the butterfly staging of a transposed tile: dst[d][local]"""
    for it in range(2):
        for w in range(4):
            by_lane = {}
            for lane in range(64):
                tid = w * 64 + lane
                idx = tid + it * 256
                row = idx // (FTILE // 8)
                d0 = (idx % (FTILE // 8)) * 8
                sj = kt0_or_q0 + row
                vv = src[b, h, sj, d0:d0 + 8] \
                    if (sj < L and d0 < D) else np.zeros(8)
                by_lane[lane] = (row, d0, vv)
            for c in range(8):
                group = [by_lane[8 * g + c][2] for g in range(8)]
                t = xpose8x8(group, 0)
                for g in range(8):
                    row, d0, _ = by_lane[8 * g + c]
                    j0 = row & ~7
                    dst_tile[d0 + g, j0:j0 + 8] = t[g]


def emulate_bwd_block(dout, q, k, v, out, s_saved, ml, scale, b, h, kt0,
                      dq, dk, dv, bias_grad_ds):
    B, H, Lq, D = q.shape
    Lk = k.shape[2]
    vs = np.zeros((FTILE, FTILE))
    kt = np.zeros((FTILE, FTILE))
    dos = np.zeros((FTILE, FTILE))
    qt = np.zeros((FTILE, FTILE))
    dot_t = np.zeros((FTILE, FTILE))
    dsn = np.zeros((FTILE, FTILE))
    dst = np.zeros((FTILE, FTILE))
    adt = np.zeros((FTILE, FTILE))

    # stage V_t natural + K_t^T once
    for tid in range(256):
        for idx in range(tid, FTILE * (FTILE // 8), 256):
            row = idx // (FTILE // 8)
            d0 = (idx % (FTILE // 8)) * 8
            kj = kt0 + row
            vs[row, d0:d0 + 8] = v[b, h, kj, d0:d0 + 8] \
                if (kj < Lk and d0 < D) else 0.0
    stage_transposed(None, kt, kt0, Lk, D, b, h, k)

    nfrag_d = (D + 15) // 16
    acck = np.zeros((4, 64, 4, 4))  # [wave][lane][f][r]
    accv = np.zeros((4, 64, 4, 4))
    dot_row = (dout * out).sum(-1)  # [B,H,Lq]

    for q0 in range(0, Lq, FTILE):
        for tid in range(256):
            for idx in range(tid, FTILE * (FTILE // 8), 256):
                row = idx // (FTILE // 8)
                d0 = (idx % (FTILE // 8)) * 8
                qi = q0 + row
                dos[row, d0:d0 + 8] = dout[b, h, qi, d0:d0 + 8] \
                    if (qi < Lq and d0 < D) else 0.0
        stage_transposed(None, dot_t, q0, Lq, D, b, h, dout)
        stage_transposed(None, qt, q0, Lq, D, b, h, q)

        for w in range(4):
            strip = w * 16
            # dA = dO V_t^T
            acc = [[np.zeros(4) for _ in range(64)] for _ in range(4)]
            for f in range(4):
                fa = [np.zeros(4) for _ in range(64)]
                for kk in range(0, D, 32):
                    a = [frag_load(dos, strip, kk, ln) for ln in range(64)]
                    bb = [frag_load(vs, f * 16, kk, ln) for ln in range(64)]
                    fa = mfma_16x16x32(a, bb, fa)
                for ln in range(64):
                    acc[f][ln] = fa[ln]

            # epilogue: reconstruct P, dS; stash dsn/dst/adt
            for ln in range(64):
                col = ln & 15
                row_grp = (ln >> 4) << 2
                for f in range(4):
                    for r in range(4):
                        i = q0 + strip + row_grp + r
                        j = kt0 + f * 16 + col
                        dval = aval = 0.0
                        if i < Lq and j < Lk:
                            m, l = ml[b, h, i]
                            s = s_saved[b, h, i, j]
                            p = np.exp(s - m) / l if l > 0 else 0.0
                            dp = acc[f][ln][r]  # mult = 1 (no qm/dropout)
                            dval = p * (dp - dot_row[b, h, i])
                            aval = p
                            bias_grad_ds[b, h, i, j] = dval
                        row = strip + row_grp + r
                        jl = f * 16 + col
                        dsn[row, jl] = dval * scale
                        dst[jl, strip + row_grp + r] = dval * scale
                        adt[jl, strip + row_grp + r] = aval

            # dQ (this wave's strip) -> atomic add
            for f in range(nfrag_d):
                fa = [np.zeros(4) for _ in range(64)]
                for kk in range(0, FTILE, 32):
                    a = [frag_load(dsn, strip, kk, ln) for ln in range(64)]
                    bb = [frag_load(kt, f * 16, kk, ln) for ln in range(64)]
                    fa = mfma_16x16x32(a, bb, fa)
                for ln in range(64):
                    col = ln & 15
                    row_grp = (ln >> 4) << 2
                    for r in range(4):
                        i = q0 + strip + row_grp + r
                        d = f * 16 + col
                        if i < Lq and d < D:
                            dq[b, h, i, d] += fa[ln][r]

        # barrier: dst/adt complete across all waves, then dK/dV MFMAs
        for w in range(4):
            strip = w * 16
            for f in range(nfrag_d):
                fk = [acck[w, ln, f, :].copy() for ln in range(64)]
                fv = [accv[w, ln, f, :].copy() for ln in range(64)]
                for kk in range(0, FTILE, 32):
                    a1 = [frag_load(dst, strip, kk, ln) for ln in range(64)]
                    b1 = [frag_load(qt, f * 16, kk, ln) for ln in range(64)]
                    fk = mfma_16x16x32(a1, b1, fk)
                    a2 = [frag_load(adt, strip, kk, ln) for ln in range(64)]
                    b2 = [frag_load(dot_t, f * 16, kk, ln) for ln in range(64)]
                    fv = mfma_16x16x32(a2, b2, fv)
                for ln in range(64):
                    acck[w, ln, f, :] = fk[ln]
                    accv[w, ln, f, :] = fv[ln]

    # write dK/dV rows of this k-tile
    for w in range(4):
        strip = w * 16
        for ln in range(64):
            col = ln & 15
            row_grp = (ln >> 4) << 2
            for f in range(nfrag_d):
                for r in range(4):
                    j = kt0 + strip + row_grp + r
                    d = f * 16 + col
                    if j < Lk and d < D:
                        dk[b, h, j, d] = acck[w, ln, f, r]
                        dv[b, h, j, d] = accv[w, ln, f, r]


def main():
    rng = np.random.default_rng(3)
    B, H, Lq, Lk, D, scale = 1, 1, 80, 128, 32, 0.2
    q = rng.standard_normal((B, H, Lq, D))
    k = rng.standard_normal((B, H, Lk, D))
    v = rng.standard_normal((B, H, Lk, D))
    g = rng.standard_normal((B, H, Lq, D))

    # autograd reference (no masks)
    tq, tk, tv = (torch.tensor(x, requires_grad=True) for x in (q, k, v))
    s = torch.einsum("bhid,bhjd->bhij", tq, tk) * scale
    p = torch.softmax(s, -1)
    o = torch.einsum("bhij,bhjd->bhid", p, tv)
    o.backward(torch.tensor(g))

    out = o.detach().numpy()
    s_np = s.detach().numpy()
    ml = np.stack([s_np.max(-1),
                   np.exp(s_np - s_np.max(-1, keepdims=True)).sum(-1)], -1)

    dq = np.zeros_like(q)
    dk = np.zeros_like(k)
    dv = np.zeros_like(v)
    ds = np.zeros((B, H, Lq, Lk))
    for kt0 in range(0, Lk, FTILE):
        emulate_bwd_block(g, q, k, v, out, s_np, ml, scale, 0, 0, kt0,
                          dq, dk, dv, ds)
    for name, a, b2 in (("dq", dq, tq.grad), ("dk", dk, tk.grad),
                        ("dv", dv, tv.grad)):
        err = np.abs(a - b2.numpy()).max()
        print(f"{name} err {err:.2e}")
        assert err < 1e-9, name
    # bias grad = dS (pre-scale)
    tb = torch.tensor(s_np * 0, requires_grad=True)
    s2 = torch.einsum("bhid,bhjd->bhij",
                      torch.tensor(q), torch.tensor(k)) * scale + tb
    p2 = torch.softmax(s2, -1)
    torch.einsum("bhij,bhjd->bhid", p2, torch.tensor(v)).backward(
        torch.tensor(g))
    err = np.abs(ds - tb.grad.numpy()).max()
    print(f"dbias err {err:.2e}")
    assert err < 1e-9
    print("flash backward kernel logic emulated exactly")


if __name__ == "__main__":
    main()
