"""genrec_amd — MI355X-native generative-recommendation framework.

A from-scratch rebuild of the capabilities of `phonism/genrec` designed
MI355X-first: PyTorch-ROCm orchestration, hand-written CDNA4 (gfx950) HIP
kernels for the hot ops (MFMA + LDS-staged tiles), and RCCL over xGMI for
data-parallel training.

Layout:
    genrec_amd.config    gin-compatible config engine + parse_config CLI
    genrec_amd.data      dataset schemas, synthetic + Amazon pipelines
    genrec_amd.models    SASRec, HSTU, RQ-VAE, TIGER, LCRec, COBRA, NoteLLM
    genrec_amd.modules   norms, losses, metrics, embeddings, transformer, ...
    genrec_amd.ops       HIP-kernel dispatch layer (eager CPU reference path)
    genrec_amd.parallel  RCCL/xGMI distributed engine
    genrec_amd.trainers  one gin-configurable train() per model
"""

__version__ = "0.1.0"
