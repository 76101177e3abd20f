"""Model zoo (parity: reference genrec/models/__init__.py:18-33)."""

from genrec_amd.models.sasrec import SASRec
from genrec_amd.models.hstu import HSTU
from genrec_amd.models.rqvae import (
    RqVae, Quantize, QuantizeForwardMode, QuantizeDistance,
)
from genrec_amd.models.tiger import Tiger, TigerOutput, TigerGenerationOutput, DeviceTrie

__all__ = [
    "SASRec", "HSTU", "RqVae", "Quantize", "QuantizeForwardMode",
    "QuantizeDistance", "Tiger", "TigerOutput", "TigerGenerationOutput",
    "DeviceTrie",
]


def __getattr__(name):
    # LCRec / COBRA / NoteLLM import transformers lazily
    if name == "LCRec":
        from genrec_amd.models.lcrec import LCRec
        return LCRec
    if name == "Cobra":
        from genrec_amd.models.cobra import Cobra
        return Cobra
    if name == "Query2Embedding":
        from genrec_amd.models.notellm import Query2Embedding
        return Query2Embedding
    raise AttributeError(name)
