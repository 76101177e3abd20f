"""Fragment-level simulation of the Lk-tiled (flash-style) MFMA attention
planned in BACKLOG.md item 3 — validates the running-softmax bookkeeping
and the single-pass backward identity at the SAME granularity the CDNA4
kernel will use (16x16 C-fragments, per-row m/l carried across key
tiles), before any device code is written.

Forward (per 64-row q strip, looping key tiles of 64):
    S_t   = Q K_t^T * scale + bias_t (+ masks)
    m'    = max(m, rowmax(S_t));  a = exp(m - m')
    P_t   = exp(S_t - m')
    l     = l * a + rowsum(P_t)
    accO  = accO * a + P_t @ V_t
    out   = accO / l          (after the last tile)
backward saves S (scores) + per-row (m, l); P reconstructed as
exp(S-m)/l. Single pass over key tiles using the flash identity
    dot_i = sum_j dA_ij P_ij = rowsum(dO_i * O_i)
    dS_t  = P_t * (dO V_t^T - dot)
    dQ   += scale * dS_t K_t ;  dK_t = scale * dS_t^T Q ; dV_t = P_t^T dO

Run: python tools/sim_flash_tiles.py   (asserts, prints max errors)
"""

import numpy as np

np.random.seed(0)


def reference(q, k, v, scale, bias, causal):
    Lq, D = q.shape
    Lk = k.shape[0]
    s = q @ k.T * scale + bias
    if causal:
        for i in range(Lq):
            s[i, i + 1:] = -1e9
    m = s.max(1, keepdims=True)
    p = np.exp(s - m)
    p /= p.sum(1, keepdims=True)
    return p @ v, s


def tiled_forward(q, k, v, scale, bias, causal, tile=64):
    """Running-softmax over key tiles, fragment-order arithmetic."""
    Lq, D = q.shape
    Lk = k.shape[0]
    m = np.full(Lq, -np.inf)
    l = np.zeros(Lq)
    acc = np.zeros((Lq, D))
    s_saved = np.zeros((Lq, Lk))
    for k0 in range(0, Lk, tile):
        kt = k[k0:k0 + tile]
        vt = v[k0:k0 + tile]
        s = q @ kt.T * scale + bias[:, k0:k0 + tile]
        if causal:
            for i in range(Lq):
                for jj in range(s.shape[1]):
                    if k0 + jj > i:
                        s[i, jj] = -1e9
        s_saved[:, k0:k0 + tile] = s
        m_new = np.maximum(m, s.max(1))
        a = np.exp(m - m_new)         # rescale of old state
        p = np.exp(s - m_new[:, None])
        l = l * a + p.sum(1)
        acc = acc * a[:, None] + p @ vt
        m = m_new
    return acc / l[:, None], s_saved, m, l


def tiled_backward(q, k, v, scale, dout, out, s_saved, m, l, tile=64):
    Lq, D = q.shape
    Lk = k.shape[0]
    dot = (dout * out).sum(1)         # flash identity, per q row
    dq = np.zeros_like(q)
    dk = np.zeros_like(k)
    dv = np.zeros_like(v)
    ds_full = np.zeros((Lq, Lk))
    for k0 in range(0, Lk, tile):
        kt = k[k0:k0 + tile]
        vt = v[k0:k0 + tile]
        p = np.exp(s_saved[:, k0:k0 + tile] - m[:, None]) / l[:, None]
        da = dout @ vt.T
        ds = p * (da - dot[:, None])
        ds_full[:, k0:k0 + tile] = ds
        dq += scale * ds @ kt
        dk[k0:k0 + tile] = scale * ds.T @ q
        dv[k0:k0 + tile] = p.T @ dout
    return dq, dk, dv, ds_full


def autograd_reference(q, k, v, scale, bias, causal, dout):
    import torch

    tq = torch.tensor(q, requires_grad=True)
    tk = torch.tensor(k, requires_grad=True)
    tv = torch.tensor(v, requires_grad=True)
    s = tq @ tk.T * scale + torch.tensor(bias)
    if causal:
        mask = torch.triu(torch.ones(s.shape, dtype=torch.bool), 1)
        s = s.masked_fill(mask[: s.size(0), : s.size(1)], -1e9)
    p = torch.softmax(s, dim=1)
    (p @ tv).backward(torch.tensor(dout))
    return tq.grad.numpy(), tk.grad.numpy(), tv.grad.numpy()


def main():
    Lq, Lk, D, scale = 61, 128, 64, 0.125
    for causal in (False, True):
        q = np.random.randn(Lq, D)
        k = np.random.randn(Lk, D)
        v = np.random.randn(Lk, D)
        bias = 0.3 * np.random.randn(Lq, Lk)
        dout = np.random.randn(Lq, D)
        if causal:
            # causal with Lk > Lq only masks j > i (standard convention)
            pass
        ref_o, _ = reference(q, k, v, scale, bias, causal)
        o, s_saved, m, l = tiled_forward(q, k, v, scale, bias, causal)
        err_o = np.abs(o - ref_o).max()
        dq, dk, dv, _ = tiled_backward(q, k, v, scale, dout, o, s_saved,
                                       m, l)
        rdq, rdk, rdv = autograd_reference(q, k, v, scale, bias, causal,
                                           dout)
        errs = [np.abs(a - b).max() for a, b in
                ((dq, rdq), (dk, rdk), (dv, rdv))]
        print(f"causal={causal}: fwd err {err_o:.2e}, "
              f"bwd errs dq/dk/dv {errs[0]:.2e}/{errs[1]:.2e}/{errs[2]:.2e}")
        assert err_o < 1e-10 and max(errs) < 1e-9
    print("tiled flash math validated (fwd running softmax + single-pass "
          "bwd via dot=rowsum(dO*O))")


def check_multipliers():
    """query-mask + dropout multipliers: out = qm * (drop ∘ softmax(S)) @ V.
    Verifies (a) dropout applied to the accumulated contribution only while
    l sums UNDROPPED p, and (b) the identity dot_i = rowsum(dO*O) still
    equals sum_j P dP with dP = M ∘ dA — the exact epilogue math of
    csrc/kernels/attention_flash.hip."""
    import torch

    np.random.seed(1)
    Lq, Lk, D, scale, tile = 61, 128, 32, 0.2, 64
    q, k, v = (np.random.randn(*sh) for sh in
               ((Lq, D), (Lk, D), (Lk, D)))
    qm = (np.random.rand(Lq) > 0.2).astype(float)
    keep = np.random.rand(Lq, Lk) > 0.3
    inv_keep = 1 / 0.7
    M = qm[:, None] * np.where(keep, inv_keep, 0.0)
    dout = np.random.randn(Lq, D)

    tq = torch.tensor(q, requires_grad=True)
    tk = torch.tensor(k, requires_grad=True)
    tv = torch.tensor(v, requires_grad=True)
    P = torch.softmax(tq @ tk.T * scale, 1)
    O = (torch.tensor(M) * P) @ tv
    O.backward(torch.tensor(dout))

    m = np.full(Lq, -1e30)
    l = np.zeros(Lq)
    acc = np.zeros((Lq, D))
    s_saved = np.zeros((Lq, Lk))
    for k0 in range(0, Lk, tile):
        s = q @ k[k0:k0 + tile].T * scale
        s_saved[:, k0:k0 + tile] = s
        m_new = np.maximum(m, s.max(1))
        a = np.exp(m - m_new)
        p = np.exp(s - m_new[:, None])
        l = l * a + p.sum(1)
        p_drop = p * np.where(keep[:, k0:k0 + tile], inv_keep, 0.0)
        acc = acc * a[:, None] + p_drop @ v[k0:k0 + tile]
        m = m_new
    out = acc / l[:, None] * qm[:, None]
    assert np.abs(out - O.detach().numpy()).max() < 1e-12

    dot = (dout * out).sum(1)
    dq = np.zeros_like(q)
    dk = np.zeros_like(k)
    dv = np.zeros_like(v)
    for k0 in range(0, Lk, tile):
        p = np.exp(s_saved[:, k0:k0 + tile] - m[:, None]) / l[:, None]
        Mt = M[:, k0:k0 + tile]
        dp = (dout @ v[k0:k0 + tile].T) * Mt
        ds = p * (dp - dot[:, None])
        dq += scale * ds @ k[k0:k0 + tile]
        dk[k0:k0 + tile] = scale * ds.T @ q
        dv[k0:k0 + tile] = (Mt * p).T @ dout
    for a2, b2 in ((dq, tq.grad), (dk, tk.grad), (dv, tv.grad)):
        assert np.abs(a2 - b2.numpy()).max() < 1e-12
    print("multiplier (query-mask + dropout) variant validated")


if __name__ == "__main__":
    main()
    check_multipliers()
