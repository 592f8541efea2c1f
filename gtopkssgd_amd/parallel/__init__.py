from .allreducer import (  # noqa: F401
    GradReducer,
    dense_allreduce,
    gtopk_sparse_allreduce,
    topk_sparse_allreduce,
)
from .optimizer import DistributedOptimizer, rank, size  # noqa: F401
