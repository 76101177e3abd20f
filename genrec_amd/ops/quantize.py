"""RQ-VAE residual-quantize ops (K8/K11 — SURVEY.md §2.4).

The hot op is the per-level L2 distance matrix + argmin
(ref rqvae.py:186-199). The HIP kernel computes dist via MFMA tiles and the
row argmin in the same pass; the full dist matrix is still returned because
the Gumbel/Sinkhorn quantize modes consume it (rqvae.py:202-241).
"""

from __future__ import annotations

from torch import Tensor

from genrec_amd.ops import eager


def residual_quantize_step(x: Tensor, codebook: Tensor):
    """Return (dist [B,K], ids [B]) for one quantize level.

    dist carries gradients w.r.t. x and codebook (needed by GUMBEL mode);
    ids = argmin detached.
    """
    from genrec_amd import ops

    if ops.use_hip(x, codebook) and hasattr(ops.ext(), "sqdist_argmin") \
            and not (x.requires_grad or codebook.requires_grad):
        dist, ids = ops.ext().sqdist_argmin(x.contiguous(), codebook.contiguous())
        return dist, ids
    dist = eager.pairwise_sqdist(x, codebook)
    ids = dist.detach().min(dim=1).indices
    return dist, ids
