from genrec_amd.parallel.ddp import (
    DistributedContext, GradReducer, init_distributed, reduce_scalars,
)

__all__ = ["DistributedContext", "GradReducer", "init_distributed",
           "reduce_scalars"]
