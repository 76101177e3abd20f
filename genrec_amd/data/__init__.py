from genrec_amd.data.schemas import SeqData, SeqBatch, TokenizedSeqBatch
from genrec_amd.data.synthetic import (
    SyntheticSASRecDataset, SyntheticHSTUDataset, SyntheticItemDataset,
    SyntheticSemIdSeqDataset,
)
from genrec_amd.data.collate import (
    sasrec_collate_fn, sasrec_eval_collate_fn, hstu_collate_fn,
    hstu_eval_collate_fn, tiger_pad_collate,
)
from genrec_amd.data.utils import cycle

__all__ = [
    "SeqData", "SeqBatch", "TokenizedSeqBatch",
    "SyntheticSASRecDataset", "SyntheticHSTUDataset", "SyntheticItemDataset",
    "SyntheticSemIdSeqDataset", "sasrec_collate_fn", "sasrec_eval_collate_fn",
    "hstu_collate_fn", "hstu_eval_collate_fn", "tiger_pad_collate",
    "cycle",
]
