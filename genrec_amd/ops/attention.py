"""Fused attention dispatch (K1 SASRec, K4 HSTU, K13 T5 — SURVEY.md §2.4).

One generic fused-attention op covers the three attention families of the
model zoo; the CDNA4 kernel (csrc/kernels/attention.hip) is templated on the
score activation (softmax vs SiLU) and fuses the bias adds, the reference's
exact masking semantics, and the PV matmul.

Reference semantics fused here:
  * SASRec (sasrec.py:201-245): scale -> key-mask(-1e9) -> causal(-1e9)
    -> softmax -> POST-softmax query mask -> dropout -> @V
  * T5 (transformer.py:106-159): scale -> +rel-bias -> key-pad-mask(-1e9)
    -> +additive attn mask -> softmax -> dropout -> @V
  * HSTU (hstu.py:232-276): QK^T (no scale) -> +pos bias -> +temporal bias
    -> causal(-1e9) -> key-pad(-1e9) -> SiLU (no softmax!) -> @V
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch import Tensor

from genrec_amd.ops import eager

_ACT_SOFTMAX = 0
_ACT_SILU = 1


def _kernel_available(name: str, *tensors: Tensor) -> bool:
    from genrec_amd import ops

    if not ops.use_hip(*tensors):
        return False
    return hasattr(ops.ext(), name)


_seed_counters = {}
_call_salt = [0]


def _seed_counter(device) -> torch.Tensor:
    """Per-device dropout seed counter read by the fused kernels. Under
    hipGraph capture each CALL SITE bakes a distinct python-side salt into
    its seed, and advance_dropout_seeds() — captured ONCE per step by
    GraphedTrainStep — bumps this counter so masks vary across replays.
    (Per-call device increments cost ~40 tiny int-add kernels per TIGER
    step; one per step is enough for uniqueness.)"""
    t = _seed_counters.get(device)
    if t is None:
        t = torch.zeros(1, dtype=torch.int32, device=device)
        _seed_counters[device] = t
    return t


def advance_dropout_seeds(device) -> None:
    """One captured increment per training step (graph runners call this);
    eager paths get fresh python-random seeds per call and don't need it."""
    _seed_counter(device).add_(1)


def _call_seed() -> int:
    """Per-call seed: python-random when eager; a distinct per-call-site
    salt under capture (the capture-time constant differentiates sites,
    the device counter differentiates replays)."""
    _call_salt[0] = (_call_salt[0] + 1) & 0x7FFFFFFF
    if torch.cuda.is_available() and \
            torch.cuda.is_current_stream_capturing():
        return (12345 + _call_salt[0] * 2654435761) & 0x7FFFFFFF
    return int(torch.randint(0, 2**31 - 1, (1,)).item())


class _FusedAttnFn(torch.autograd.Function):
    """Autograd wrapper over the HIP fused-attention fwd/bwd kernels."""

    @staticmethod
    def forward(ctx, q, k, v, bias, key_pad_mask, additive_mask, query_mask,
                scale, causal, act, dropout_p, training):
        from genrec_amd import ops

        use_mfma = (q.dtype == torch.bfloat16 and q.size(3) % 32 == 0
                    and os.environ.get("GENREC_DISABLE_MFMA", "0") != "1")
        if use_mfma:
            # the MFMA kernels are stride-aware: [B,L,H,D] transpose views
            # (the layout GEMM outputs naturally produce) go in directly —
            # no .contiguous() copies on q/k/v, out, or the grads.
            q = q if q.stride(-1) == 1 else q.contiguous()
            k = k if k.stride(-1) == 1 else k.contiguous()
            v = v if v.stride(-1) == 1 else v.contiguous()
        else:
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        seed_dev = None
        seed = 0
        if dropout_p > 0 and training:
            seed = _call_seed()
            seed_dev = _seed_counter(q.device)
        fwd = ops.ext().attn_fwd_mfma if use_mfma else ops.ext().attn_fwd
        out, probs, dmask = fwd(
            q, k, v, bias, key_pad_mask, additive_mask, query_mask,
            scale, causal, act, dropout_p if training else 0.0, seed,
            seed_dev,
        )
        ctx.use_mfma = use_mfma
        ctx.save_for_backward(q, k, v, probs, dmask,
                              query_mask if query_mask is not None else torch.empty(0))
        ctx.meta = (scale, causal, act, dropout_p if training else 0.0, seed,
                    bias is not None and bias.requires_grad,
                    bias.dim() if bias is not None else 0,
                    bias.dtype if bias is not None else None)
        return out

    @staticmethod
    def backward(ctx, dout):
        from genrec_amd import ops

        q, k, v, probs, dmask, query_mask = ctx.saved_tensors
        (scale, causal, act, dropout_p, seed, bias_grad, bias_dim,
         bias_dtype) = ctx.meta
        bwd = ops.ext().attn_bwd_mfma if ctx.use_mfma else ops.ext().attn_bwd
        if not (ctx.use_mfma and dout.stride(-1) == 1):
            dout = dout.contiguous()
        dq, dk, dv, dbias = bwd(
            dout, q, k, v, probs, dmask,
            query_mask if query_mask.numel() else None,
            scale, act, dropout_p, seed, bias_grad, bias_dim,
        )
        if bias_grad and dbias.dtype != bias_dtype:
            dbias = dbias.to(bias_dtype)
        return (dq, dk, dv, dbias if bias_grad else None,
                None, None, None, None, None, None, None, None)


class _FusedAttnTableBiasFn(torch.autograd.Function):
    """MFMA attention with the rel-bias TABLE gathered in-kernel
    (HSTU-style; round 2): no materialized [H,Lq,Lk] bias tensor in
    forward, and backward accumulates table grads via an LDS histogram
    + deterministic colsum instead of the [B,H,Lq,Lk] ds_saved
    round-trip. Caller guarantees the mfma path fits."""

    @staticmethod
    def forward(ctx, q, k, v, bias_table, bias_bucket, key_pad_mask,
                additive_mask, query_mask, scale, causal, dropout_p,
                training):
        from genrec_amd import ops

        q = q if q.stride(-1) == 1 else q.contiguous()
        k = k if k.stride(-1) == 1 else k.contiguous()
        v = v if v.stride(-1) == 1 else v.contiguous()
        table_f = bias_table.float()  # kernel reads fp32
        seed_dev = None
        seed = 0
        if dropout_p > 0 and training:
            seed = _call_seed()
            seed_dev = _seed_counter(q.device)
        out, probs, dmask = ops.ext().attn_fwd_mfma(
            q, k, v, table_f, key_pad_mask, additive_mask, query_mask,
            scale, causal, 0, dropout_p if training else 0.0, seed,
            seed_dev, bias_bucket)
        ctx.save_for_backward(
            q, k, v, probs, dmask,
            query_mask if query_mask is not None else torch.empty(0),
            bias_bucket)
        ctx.meta = (scale, dropout_p if training else 0.0,
                    bias_table.requires_grad,
                    bias_table.numel() // q.size(1), bias_table.dtype)
        return out

    @staticmethod
    def backward(ctx, dout):
        from genrec_amd import ops

        q, k, v, probs, dmask, query_mask, bucket = ctx.saved_tensors
        scale, dropout_p, bias_grad, n_buckets, tab_dtype = ctx.meta
        if dout.stride(-1) != 1:
            dout = dout.contiguous()
        dq, dk, dv, dtab = ops.ext().attn_bwd_mfma(
            dout, q, k, v, probs, dmask,
            query_mask if query_mask.numel() else None,
            scale, 0, dropout_p, 0, bias_grad, 0, bucket, n_buckets)
        if bias_grad and dtab.dtype != tab_dtype:
            dtab = dtab.to(tab_dtype)
        return (dq, dk, dv, dtab if bias_grad else None, None,
                None, None, None, None, None, None, None)


class _FlashAttnFn(torch.autograd.Function):
    """Flash-tiled attention (Lk/Lq beyond one 64-tile): running-softmax
    forward, single-pass backward via the flash identity (tile math proven
    in tools/sim_flash_tiles.py; GPU-validated round 2 — default for
    Lk>64 softmax attention; GENREC_DISABLE_ATTN_FLASH=1 opts out)."""

    @staticmethod
    def forward(ctx, q, k, v, bias, key_pad_mask, additive_mask, query_mask,
                scale, causal, dropout_p, training):
        from genrec_amd import ops

        # stride-aware: [B,L,H,D] transpose views pass straight through
        # (the host copies only when d isn't innermost / strides are
        # unaligned), killing the q/k/v copy kernels per decoder layer
        seed_dev = None
        seed = 0
        if dropout_p > 0 and training:
            seed = _call_seed()
            seed_dev = _seed_counter(q.device)
        out, s_saved, ml, dmask = ops.ext().attn_fwd_flash(
            q, k, v, bias, key_pad_mask, additive_mask, query_mask,
            scale, causal, dropout_p if training else 0.0, seed, seed_dev)
        ctx.save_for_backward(
            q, k, v, out, s_saved, ml, dmask,
            query_mask if query_mask is not None else torch.empty(0))
        ctx.meta = (scale, dropout_p if training else 0.0,
                    bias is not None and bias.requires_grad,
                    bias.dim() if bias is not None else 0,
                    bias.dtype if bias is not None else None)
        return out

    @staticmethod
    def backward(ctx, dout):
        from genrec_amd import ops

        q, k, v, out, s_saved, ml, dmask, query_mask = ctx.saved_tensors
        scale, dropout_p, bias_grad, bias_dim, bias_dtype = ctx.meta
        dq, dk, dv, dbias = ops.ext().attn_bwd_flash(
            dout, q, k, v, out, s_saved, ml, dmask,
            query_mask if query_mask.numel() else None,
            scale, dropout_p, bias_grad, bias_dim)
        if bias_grad and dbias.dtype != bias_dtype:
            dbias = dbias.to(bias_dtype)
        return (dq, dk, dv, dbias if bias_grad else None,
                None, None, None, None, None, None, None)


def fused_attention(
    q: Tensor,
    k: Tensor,
    v: Tensor,
    *,
    scale: float = 1.0,
    bias: Optional[Tensor] = None,
    bias_table: Optional[Tensor] = None,
    bias_bucket: Optional[Tensor] = None,
    key_pad_mask: Optional[Tensor] = None,
    additive_mask: Optional[Tensor] = None,
    causal: bool = False,
    query_mask: Optional[Tensor] = None,
    score_act: str = "softmax",
    dropout_p: float = 0.0,
    training: bool = False,
) -> Tensor:
    act = _ACT_SOFTMAX if score_act == "softmax" else _ACT_SILU
    fits = (k.size(2) <= 64 and q.size(2) <= 64 and q.size(3) <= 64
            and (additive_mask is None or additive_mask.dim() == 2)
            and os.environ.get("GENREC_DISABLE_ATTN", "0") != "1"
            # experiment knob: route even Lk<=64 through the flash
            # (recompute-P) kernels to trade HBM p_saved traffic for
            # recompute — measured per-config on GPU
            and os.environ.get("GENREC_FORCE_FLASH", "0") != "1")
    if bias_table is not None:
        # rel-bias TABLE mode: gather in-kernel when the mfma path fits
        # (bf16, D % 32 == 0); otherwise materialize the dense bias and
        # fall through to the ordinary paths
        if (fits and act == _ACT_SOFTMAX and q.dtype == torch.bfloat16
                and q.size(3) % 32 == 0
                and os.environ.get("GENREC_DISABLE_MFMA", "0") != "1"
                and _kernel_available("attn_fwd_mfma", q, k, v)):
            kp = key_pad_mask.contiguous() if key_pad_mask is not None \
                else None
            am = additive_mask.contiguous() if additive_mask is not None \
                else None
            qm = query_mask.contiguous() if query_mask is not None else None
            return _FusedAttnTableBiasFn.apply(
                q, k, v, bias_table, bias_bucket.contiguous(), kp, am, qm,
                scale, causal, dropout_p, training)
        h = q.size(1)
        bias = bias_table.view(h, -1)[:, bias_bucket.reshape(-1)] \
            .view(h, bias_bucket.size(0), bias_bucket.size(1))
    if fits and _kernel_available("attn_fwd", q, k, v):
        b = bias.contiguous() if bias is not None else None
        kp = key_pad_mask.contiguous() if key_pad_mask is not None else None
        am = additive_mask.contiguous() if additive_mask is not None else None
        qm = query_mask.contiguous() if query_mask is not None else None
        return _FusedAttnFn.apply(q, k, v, b, kp, am, qm,
                                  scale, causal, act, dropout_p, training)
    flash_ok = (os.environ.get("GENREC_DISABLE_ATTN_FLASH", "0") != "1"
                and act == _ACT_SOFTMAX and q.dtype == torch.bfloat16
                and q.size(3) % 32 == 0 and q.size(3) <= 64
                and (additive_mask is None or additive_mask.dim() == 2)
                and _kernel_available("attn_fwd_flash", q, k, v))
    if flash_ok:
        b = bias.contiguous() if bias is not None else None
        kp = key_pad_mask.contiguous() if key_pad_mask is not None else None
        am = additive_mask.contiguous() if additive_mask is not None else None
        qm = query_mask.contiguous() if query_mask is not None else None
        return _FlashAttnFn.apply(q, k, v, b, kp, am, qm,
                                  scale, causal, dropout_p, training)
    return eager.fused_attention(
        q, k, v, scale=scale, bias=bias, key_pad_mask=key_pad_mask,
        additive_mask=additive_mask, causal=causal, query_mask=query_mask,
        score_act=score_act, dropout_p=dropout_p, training=training,
    )


def sasrec_attention(q, k, v, valid_mask, scale, dropout_p, training):
    """K1: valid_mask [B, L] float (1=valid). Residual stays in the model."""
    key_pad = valid_mask == 0
    return fused_attention(
        q, k, v, scale=scale, key_pad_mask=key_pad, causal=True,
        query_mask=valid_mask, score_act="softmax",
        dropout_p=dropout_p, training=training,
    )


def t5_attention(q, k, v, bias, key_pad_mask, additive_mask, scale,
                 dropout_p, training, bias_table=None, bias_bucket=None):
    """K13: bias [H,Lq,Lk] or None; additive_mask [Lq,Lk] float or None.
    bias_table [H*nb] + bias_bucket [Lq,Lk] select the in-kernel
    rel-bias gather (no materialized bias tensor)."""
    return fused_attention(
        q, k, v, scale=scale, bias=bias, bias_table=bias_table,
        bias_bucket=bias_bucket, key_pad_mask=key_pad_mask,
        additive_mask=additive_mask, score_act="softmax",
        dropout_p=dropout_p, training=training,
    )


def hstu_pointwise_attention(q, k, v, pos_bias, time_bias, key_pad_mask):
    """K4 core: SiLU-score attention with two bias tensors, causal."""
    bias = pos_bias.unsqueeze(0) if pos_bias.dim() == 3 else pos_bias
    if time_bias is not None:
        bias = bias + time_bias
    return fused_attention(
        q, k, v, scale=1.0, bias=bias, key_pad_mask=key_pad_mask,
        causal=True, score_act="silu",
    )


class _HstuAttnFn(torch.autograd.Function):
    """Fully-fused HSTU attention: bias bucketing/gather + SiLU scores +
    table gradients all in-kernel (K4+K5+K6)."""

    @staticmethod
    def forward(ctx, q, k, v, pos_weight, time_weight, pos_bucket,
                timestamps, key_pad):
        from genrec_amd import ops

        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        out, s_saved = ops.ext().hstu_attn_fwd(
            q, k, v, pos_bucket,
            pos_weight.contiguous(),
            time_weight.contiguous() if time_weight is not None else None,
            timestamps, key_pad)
        ctx.save_for_backward(q, k, v, s_saved, pos_bucket,
                              timestamps if timestamps is not None
                              else torch.empty(0))
        ctx.n_pos = pos_weight.size(0)
        ctx.n_time = time_weight.size(0) if time_weight is not None else 0
        ctx.dtypes = (pos_weight.dtype,
                      time_weight.dtype if time_weight is not None else None)
        return out

    @staticmethod
    def backward(ctx, dout):
        from genrec_amd import ops

        q, k, v, s_saved, pos_bucket, ts = ctx.saved_tensors
        dq, dk, dv, dpos, dtime = ops.ext().hstu_attn_bwd(
            dout.contiguous(), q, k, v, s_saved, pos_bucket,
            ts if ts.numel() else None, ctx.n_pos, ctx.n_time)
        dpos = dpos.to(ctx.dtypes[0])
        dtime_out = dtime.to(ctx.dtypes[1]) if ctx.n_time else None
        return dq, dk, dv, dpos, dtime_out, None, None, None


def hstu_fused_attention(q, k, v, pos_bucket, pos_weight, time_weight,
                         timestamps, key_pad_mask):
    """Dispatch for the fully-fused HSTU kernel; falls back to the bias-
    tensor composition (hstu_pointwise_attention) off-GPU / odd shapes."""
    from genrec_amd import ops

    fits = (q.dtype == torch.bfloat16 and q.size(2) <= 64
            and q.size(3) % 32 == 0 and q.size(3) <= 64
            and pos_weight.dtype == torch.bfloat16
            and (time_weight is None or time_weight.dtype == torch.bfloat16)
            and os.environ.get("GENREC_DISABLE_MFMA", "0") != "1")
    if fits and _kernel_available("hstu_attn_fwd", q, k, v):
        ts = timestamps.contiguous() if timestamps is not None else None
        kp = key_pad_mask.contiguous() if key_pad_mask is not None else None
        return _HstuAttnFn.apply(q, k, v, pos_weight, time_weight,
                                 pos_bucket.contiguous(), ts, kp)
    return None  # caller composes the bias-tensor path
