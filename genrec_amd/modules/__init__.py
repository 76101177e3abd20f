"""Reusable building blocks (parity: reference genrec/modules/__init__.py:35-51)."""

from genrec_amd.modules.norms import (
    l2norm, L2Norm, RMSNorm, T5RMSNorm, RootMeanSquareLayerNorm, SwishLayerNorm,
)
from genrec_amd.modules.losses import (
    ReconstructionLoss, CategoricalReconstructionLoss, QuantizeLoss,
)
from genrec_amd.modules.metrics import TopKAccumulator
from genrec_amd.modules.embeddings import SemIdEmbedding, UserIdEmbedding
from genrec_amd.modules.kmeans import Kmeans, KmeansOutput, kmeans_init_
from genrec_amd.modules.gumbel import sample_gumbel, gumbel_softmax_sample
from genrec_amd.modules.mlp import MLP
from genrec_amd.modules.schedulers import (
    InverseSquareRootScheduler,
    get_linear_schedule_with_warmup,
    get_cosine_schedule_with_warmup,
)
from genrec_amd.modules.transformer import (
    T5Attention, FeedForward, TransformerBlock, TransformerEncoder,
    TransformerDecoder, TransformerEncoderDecoder, relative_position_bucket,
)

__all__ = [
    "l2norm", "L2Norm", "RMSNorm", "T5RMSNorm", "RootMeanSquareLayerNorm",
    "SwishLayerNorm", "ReconstructionLoss", "CategoricalReconstructionLoss",
    "QuantizeLoss", "TopKAccumulator", "SemIdEmbedding", "UserIdEmbedding",
    "Kmeans", "KmeansOutput", "kmeans_init_", "sample_gumbel",
    "gumbel_softmax_sample", "MLP", "InverseSquareRootScheduler",
    "get_linear_schedule_with_warmup", "get_cosine_schedule_with_warmup",
    "T5Attention", "FeedForward", "TransformerBlock", "TransformerEncoder",
    "TransformerDecoder", "TransformerEncoderDecoder",
    "relative_position_bucket",
]
