"""RQ-VAE: residual-quantized VAE producing semantic IDs.

Parity target: /root/reference/genrec/models/rqvae.py (454 LoC). All four
quantize forward modes (GUMBEL_SOFTMAX / STE / ROTATION_TRICK / SINKHORN,
rqvae.py:43-51), L2 and cosine distances (rqvae.py:186-198), kmeans codebook
init on first batch (rqvae.py:165-168), the residual get_semantic_ids loop
(rqvae.py:386-412), and the loss/p_unique_ids computation (rqvae.py:425-446)
are reproduced.

MI355X design notes:
  * the per-level distance+argmin is the K8 kernel (ops.residual_quantize_step)
  * Sinkhorn-Knopp stays in fp64 on device (CDNA4 fp64 is strong; SURVEY.md
    §7.4 item 2) with the reference's 100-iteration schedule
  * no @torch.compile: the training step is captured in a hipGraph by the
    trainer instead (reference uses Inductor reduce-overhead, rqvae.py:414)
"""

from __future__ import annotations

from enum import Enum
from typing import List, NamedTuple

import logging

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from genrec_amd import ops
from genrec_amd.config import ginlite
from genrec_amd.modules.gumbel import gumbel_softmax_sample
from genrec_amd.modules.kmeans import kmeans_init_
from genrec_amd.modules.losses import (
    CategoricalReconstructionLoss, QuantizeLoss, ReconstructionLoss,
)
from genrec_amd.modules.mlp import MLP
from genrec_amd.modules.norms import L2Norm, l2norm

logger = logging.getLogger("genrec_amd")


class QuantizeForwardMode(Enum):
    GUMBEL_SOFTMAX = 1
    STE = 2
    ROTATION_TRICK = 3
    SINKHORN = 4


class QuantizeDistance(Enum):
    L2 = 1
    COSINE = 2


class QuantizeOutput(NamedTuple):
    embeddings: Tensor
    ids: Tensor
    loss: Tensor


def efficient_rotation_trick_transform(u: Tensor, q: Tensor, e: Tensor) -> Tensor:
    """Householder-style rotation trick (arXiv:2410.06424 §4.2;
    ref rqvae.py:71-82)."""
    e = e.unsqueeze(1)  # [B, 1, D]
    w = F.normalize(u + q, p=2, dim=1, eps=1e-6).detach()
    return (
        e
        - 2 * (e @ w.unsqueeze(-1) @ w.unsqueeze(1))
        + 2 * (e @ u.unsqueeze(-1).detach() @ q.unsqueeze(1).detach())
    ).squeeze()


@torch.no_grad()
def sinkhorn_knopp(cost: Tensor, row_marginals: Tensor, col_marginals: Tensor,
                   eps: float = 0.05, max_iter: int = 50) -> Tensor:
    """Entropy-regularized OT row/col scaling (ref rqvae.py:85-110).

    Keeps rocBLAS fp64 gemv for the matvecs: a same-box A/B against a
    broadcast-mul + sum formulation measured gemv FASTER end to end
    (174.0k vs 157.7k RQ-VAE samples/s) — rocprofv3's per-dispatch
    averages had overstated the tiny gemv's cost ~10x (tracer overhead
    dominates microsecond kernels; trust A/B wall-clock, not profiled
    averages, for launch-sized kernels).
    """
    K = torch.exp(-cost / eps)
    u = torch.ones_like(row_marginals)
    v = torch.ones_like(col_marginals)
    for _ in range(max_iter):
        u = row_marginals / (K @ v + 1e-8)
        v = col_marginals / (K.T @ u + 1e-8)
    return u.unsqueeze(1) * K * v.unsqueeze(0)


class Quantize(nn.Module):
    """Single-level vector quantizer (ref rqvae.py:113-254)."""

    def __init__(self, embed_dim: int, n_embed: int,
                 do_kmeans_init: bool = True, codebook_normalize: bool = False,
                 sim_vq: bool = False, commitment_weight: float = 0.25,
                 forward_mode: QuantizeForwardMode = QuantizeForwardMode.GUMBEL_SOFTMAX,
                 distance_mode: QuantizeDistance = QuantizeDistance.L2) -> None:
        super().__init__()
        self.embed_dim = embed_dim
        self.n_embed = n_embed
        self.embedding = nn.Embedding(n_embed, embed_dim)
        self.forward_mode = forward_mode
        self.distance_mode = distance_mode
        self.do_kmeans_init = do_kmeans_init
        self.kmeans_initted = False
        self.out_proj = nn.Sequential(
            nn.Linear(embed_dim, embed_dim, bias=False) if sim_vq else nn.Identity(),
            L2Norm(dim=-1) if codebook_normalize else nn.Identity(),
        )
        self.quantize_loss = QuantizeLoss(commitment_weight)
        nn.init.uniform_(self.embedding.weight)

    @property
    def weight(self) -> Tensor:
        return self.embedding.weight

    @property
    def device(self) -> torch.device:
        return self.embedding.weight.device

    @torch.no_grad()
    def _kmeans_init(self, x: Tensor) -> None:
        kmeans_init_(self.embedding.weight, x=x)
        self.kmeans_initted = True

    def get_item_embeddings(self, item_ids: Tensor) -> Tensor:
        return self.out_proj(self.embedding(item_ids))

    def forward(self, x: Tensor, temperature: float) -> QuantizeOutput:
        assert x.shape[-1] == self.embed_dim
        if self.do_kmeans_init and not self.kmeans_initted:
            self._kmeans_init(x=x)

        codebook = self.out_proj(self.embedding.weight)
        if self.distance_mode == QuantizeDistance.L2:
            dist, ids = ops.residual_quantize_step(x, codebook)
        elif self.distance_mode == QuantizeDistance.COSINE:
            dist = -(
                x / x.norm(dim=1, keepdim=True)
                @ (codebook.T / codebook.T.norm(dim=0, keepdim=True))
            )
            ids = dist.detach().min(dim=1).indices
        else:
            raise ValueError(self.distance_mode)

        if self.training:
            mode = self.forward_mode
            if mode == QuantizeForwardMode.GUMBEL_SOFTMAX:
                weights = gumbel_softmax_sample(-dist, temperature, self.device)
                emb = weights @ codebook
                emb_out = emb
            elif mode == QuantizeForwardMode.STE:
                emb = self.get_item_embeddings(ids)
                emb_out = x + (emb - x).detach()
            elif mode == QuantizeForwardMode.ROTATION_TRICK:
                emb = self.get_item_embeddings(ids)
                emb_out = efficient_rotation_trick_transform(
                    x / (x.norm(dim=-1, keepdim=True) + 1e-8),
                    emb / (emb.norm(dim=-1, keepdim=True) + 1e-8),
                    x,
                )
            elif mode == QuantizeForwardMode.SINKHORN:
                # arXiv:2311.09049; ref rqvae.py:218-241 (fp64, 100 iters)
                b, k = dist.shape
                with torch.no_grad():
                    max_d, min_d = dist.max(), dist.min()
                    mid = (max_d + min_d) / 2
                    amp = max_d - mid + 1e-5
                    p = sinkhorn_knopp(
                        ((dist - mid) / amp).double(),
                        torch.full((b,), 1.0 / b, device=self.device,
                                   dtype=torch.float64),
                        torch.full((k,), 1.0 / k, device=self.device,
                                   dtype=torch.float64),
                        eps=0.003, max_iter=100,
                    )
                    ids = p.argmax(dim=-1)
                emb = self.get_item_embeddings(ids)
                emb_out = x + (emb - x).detach()
            else:
                raise ValueError(mode)
            loss = self.quantize_loss(query=x, value=emb)
        else:
            emb_out = self.get_item_embeddings(ids)
            loss = self.quantize_loss(query=x, value=emb_out)
        return QuantizeOutput(embeddings=emb_out, ids=ids, loss=loss)


class RqVaeOutput(NamedTuple):
    embeddings: Tensor
    residuals: Tensor
    sem_ids: Tensor
    quantize_loss: Tensor


class RqVaeComputedLosses(NamedTuple):
    loss: Tensor
    reconstruction_loss: Tensor
    rqvae_loss: Tensor
    embs_norm: Tensor
    p_unique_ids: Tensor


@ginlite.configurable(name="RqVae")
class RqVae(nn.Module):
    def __init__(self, input_dim: int, embed_dim: int, hidden_dims: List[int],
                 codebook_size: int, codebook_kmeans_init: bool = True,
                 codebook_normalize: bool = False, codebook_sim_vq: bool = False,
                 codebook_mode: QuantizeForwardMode = QuantizeForwardMode.GUMBEL_SOFTMAX,
                 codebook_last_layer_mode: QuantizeForwardMode = QuantizeForwardMode.GUMBEL_SOFTMAX,
                 n_layers: int = 3, commitment_weight: float = 0.25,
                 n_cat_features: int = 18) -> None:
        cfg = {k: v for k, v in locals().items() if k not in ("self", "__class__")}
        super().__init__()
        self._config = cfg
        self.input_dim = input_dim
        self.embed_dim = embed_dim
        self.hidden_dims = list(hidden_dims)
        self.n_layers = n_layers
        self.codebook_size = codebook_size
        self.commitment_weight = commitment_weight
        self.n_cat_feats = n_cat_features

        self.layers = nn.ModuleList([
            Quantize(
                embed_dim=embed_dim, n_embed=codebook_size,
                forward_mode=(codebook_mode if i < n_layers - 1
                              else codebook_last_layer_mode),
                do_kmeans_init=codebook_kmeans_init,
                codebook_normalize=(i == 0 and codebook_normalize),
                sim_vq=codebook_sim_vq,
                commitment_weight=commitment_weight,
                distance_mode=QuantizeDistance.L2,
            )
            for i in range(n_layers)
        ])
        self.encoder = MLP(input_dim=input_dim, hidden_dims=self.hidden_dims,
                           out_dim=embed_dim, normalize=codebook_normalize)
        self.decoder = MLP(input_dim=embed_dim,
                           hidden_dims=self.hidden_dims[::-1],
                           out_dim=input_dim, normalize=True)
        self.reconstruction_loss = (
            CategoricalReconstructionLoss(n_cat_features)
            if n_cat_features != 0 else ReconstructionLoss()
        )

    @property
    def config(self) -> dict:
        return self._config

    @property
    def device(self) -> torch.device:
        return next(self.encoder.parameters()).device

    def load_pretrained(self, path: str) -> None:
        state = torch.load(path, map_location=self.device, weights_only=False)
        self.load_state_dict(state["model"])
        tag = state.get("iter", state.get("epoch", "?"))
        logger.info("Loaded RQVAE checkpoint (step/epoch %s) from %s",
                    tag, path)

    def encode(self, x: Tensor) -> Tensor:
        return self.encoder(x)

    def decode(self, x: Tensor) -> Tensor:
        return self.decoder(x)

    def get_semantic_ids(self, x: Tensor, gumbel_t: float = 0.001) -> RqVaeOutput:
        res = self.encode(x)
        quantize_loss = 0
        embs, residuals, sem_ids = [], [], []
        for layer in self.layers:
            residuals.append(res)
            q = layer(res, temperature=gumbel_t)
            quantize_loss = quantize_loss + q.loss
            res = res - q.embeddings
            embs.append(q.embeddings)
            sem_ids.append(q.ids)
        # layouts match reference rearrange semantics (rqvae.py:405-410):
        # embeddings/residuals: [B, D, n_layers]; sem_ids: [B, n_layers]
        return RqVaeOutput(
            embeddings=torch.stack(embs, dim=-1),
            residuals=torch.stack(residuals, dim=-1),
            sem_ids=torch.stack(sem_ids, dim=-1),
            quantize_loss=quantize_loss,
        )

    def forward(self, batch: Tensor, gumbel_t: float) -> RqVaeComputedLosses:
        x = batch
        quantized = self.get_semantic_ids(x, gumbel_t)
        embs = quantized.embeddings  # [B, D, n_layers]
        x_hat = self.decode(embs.sum(dim=-1))
        if self.n_cat_feats > 0:
            x_hat = torch.cat([
                l2norm(x_hat[..., : x_hat.size(-1) - self.n_cat_feats]),
                x_hat[..., -self.n_cat_feats:],
            ], dim=-1)
        else:
            x_hat = l2norm(x_hat)
        reconstruction_loss = self.reconstruction_loss(x_hat, x)
        rqvae_loss = quantized.quantize_loss
        loss = (reconstruction_loss + rqvae_loss).mean()

        with torch.no_grad():
            embs_norm = embs.norm(dim=1)  # [B, n_layers], norm over D
            ids = quantized.sem_ids  # [B, n_layers]
            same = (ids.unsqueeze(1) == ids.unsqueeze(0)).all(dim=-1)
            dup = torch.triu(same, diagonal=1).any(dim=0)
            p_unique_ids = (~dup).sum() / ids.size(0)

        return RqVaeComputedLosses(
            loss=loss,
            reconstruction_loss=reconstruction_loss.mean(),
            rqvae_loss=rqvae_loss.mean(),
            embs_norm=embs_norm,
            p_unique_ids=p_unique_ids,
        )
