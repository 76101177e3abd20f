"""MLP encoder/decoder stack used by RQ-VAE (parity: encoder.py:380-420).

Bias-free Linear + SiLU between hidden layers, optional L2Norm tail.
State-dict keys mirror the reference (`mlp.{i}.weight`) for checkpoint
interop.
"""

from __future__ import annotations

from typing import List

from torch import Tensor, nn

from genrec_amd.ops.linear import SplitKLinear

from genrec_amd.modules.norms import L2Norm


class MLP(nn.Module):
    def __init__(self, input_dim: int, hidden_dims: List[int], out_dim: int,
                 dropout: float = 0.0, normalize: bool = False) -> None:
        super().__init__()
        self.input_dim = input_dim
        self.hidden_dims = list(hidden_dims)
        self.out_dim = out_dim
        dims = [input_dim] + self.hidden_dims + [out_dim]
        layers: list[nn.Module] = []
        for i, (d_in, d_out) in enumerate(zip(dims[:-1], dims[1:])):
            layers.append(SplitKLinear(d_in, d_out, bias=False))
            if i != len(dims) - 2:
                layers.append(nn.SiLU())
                if dropout != 0:
                    layers.append(nn.Dropout(dropout))
        layers.append(L2Norm() if normalize else nn.Identity())
        self.mlp = nn.Sequential(*layers)

    def forward(self, x: Tensor) -> Tensor:
        assert x.shape[-1] == self.input_dim
        return self.mlp(x)
