"""Loss modules (parity: /root/reference/genrec/modules/loss.py)."""

from __future__ import annotations

from torch import Tensor, nn
import torch.nn.functional as F


class ReconstructionLoss(nn.Module):
    """Sum-squared-error per row (ref loss.py:8-23)."""

    def forward(self, x_hat: Tensor, x: Tensor) -> Tensor:
        return ((x_hat - x) ** 2).sum(dim=-1)


class CategoricalReconstructionLoss(nn.Module):
    """SSE on continuous dims + BCE-with-logits on trailing categorical dims
    (ref loss.py:26-54)."""

    def __init__(self, n_cat_feats: int) -> None:
        super().__init__()
        self.n_cat_feats = n_cat_feats
        self.sse = ReconstructionLoss()

    def forward(self, x_hat: Tensor, x: Tensor) -> Tensor:
        n = self.n_cat_feats
        out = self.sse(x_hat[:, : x_hat.size(1) - n], x[:, : x.size(1) - n])
        if n > 0:
            out = out + F.binary_cross_entropy_with_logits(
                x_hat[:, -n:], x[:, -n:], reduction="none"
            ).sum(dim=-1)
        return out


class QuantizeLoss(nn.Module):
    """VQ commitment loss ||sg(q)-v||^2 + beta*||q-sg(v)||^2 (ref loss.py:57-77)."""

    def __init__(self, commitment_weight: float = 1.0) -> None:
        super().__init__()
        self.commitment_weight = commitment_weight

    def forward(self, query: Tensor, value: Tensor) -> Tensor:
        emb_loss = ((query.detach() - value) ** 2).sum(dim=-1)
        commit_loss = ((query - value.detach()) ** 2).sum(dim=-1)
        return emb_loss + self.commitment_weight * commit_loss
