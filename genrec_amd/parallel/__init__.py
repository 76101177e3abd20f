from genrec_amd.parallel.ddp import (
    DistributedContext, GradReducer, init_distributed, reduce_scalars,
)
from genrec_amd.parallel.graph_runner import GraphedTrainStep

__all__ = ["DistributedContext", "GradReducer", "GraphedTrainStep",
           "init_distributed", "reduce_scalars"]
